#!/usr/bin/env python3
"""Graph-replay microbench: sk2 vs hipBLASLt at the decode GEMM shapes.

Eager timing is launch-bound (~19 us/call regardless of size — the round-2
skinny.txt trap); inside a replayed graph the kernel execution time is what
matters, so each op is captured 20x in one graph and timed by replay."""
import sys, pathlib, time
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import torch
import torch.nn.functional as F
import dalle_pytorch_amd._hip as ext
from dalle_pytorch_amd.engine.decode import FastDecoder


def gtime(fn, inner=20, reps=50):
    fn(); torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        for _ in range(inner):
            fn()
    g.replay(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        g.replay()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps / inner * 1e6


def main():
    torch.manual_seed(0)
    rows = int(sys.argv[1]) if len(sys.argv) > 1 else 64
    shapes = [('qkv', 1024, 3072, 0), ('out', 1024, 1024, 0),
              ('ff1+geglu', 1024, 8192, 1), ('ff2', 4096, 1024, 0),
              ('head', 1024, 8192, 2)]
    for name, K, N, mode in shapes:
        x = (torch.randn(rows, K, device='cuda') * 0.3).bfloat16()
        w = (torch.randn(N, K, device='cuda') * 0.05).bfloat16()
        b32 = torch.randn(N, device='cuda')
        b16 = b32.bfloat16()
        wt = w.t().contiguous()
        pk = FastDecoder._sk2_pack(w)
        us_sk = gtime(lambda: ext.sk2(x, pk, b32, N, K, mode))
        if mode == 1:
            from dalle_pytorch_amd.ops import geglu
            us_bl = gtime(lambda: geglu(torch.addmm(b16, x, wt)))
        elif mode == 2:
            us_bl = gtime(lambda: torch.addmm(b16, x, wt).float())
        else:
            us_bl = gtime(lambda: torch.addmm(b16, x, wt))
        wbytes = N * K * 2
        print(f'M{rows} {name:10s} K{K} N{N}: sk2 {us_sk:6.2f}us '
              f'({wbytes / us_sk / 1e3:5.2f} TB/s)  '
              f'hipblaslt(+epi) {us_bl:6.2f}us ({wbytes / us_bl / 1e3:5.2f} TB/s)')


if __name__ == '__main__':
    main()


def fp8_compare():
    torch.manual_seed(0)
    from dalle_pytorch_amd.engine.decode import FastDecoder as FD
    for rows, K, N, mode in [(64, 1024, 3072, 0), (64, 1024, 8192, 1),
                             (64, 4096, 1024, 0)]:
        x = (torch.randn(rows, K, device='cuda') * 0.3).bfloat16()
        w = (torch.randn(N, K, device='cuda') * 0.05).bfloat16()
        b32 = torch.randn(N, device='cuda')
        pk = FD._sk2_pack(w)
        pk8, ws = FD._sk2_pack_fp8(w)
        us_bf = gtime(lambda: ext.sk2(x, pk, b32, N, K, mode))
        us_f8 = gtime(lambda: ext.sk2(x, pk8, b32, N, K, mode, ws))
        print(f'M{rows} K{K} N{N} mode{mode}: bf16 {us_bf:6.2f}us  '
              f'fp8 {us_f8:6.2f}us')
