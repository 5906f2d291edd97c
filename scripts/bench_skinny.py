"""Microbench: skinny_gemm vs F.linear at decode shapes; plus one generate-step profile."""
import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import torch, time
import dalle_pytorch_amd._hip as ext
import torch.nn.functional as F

def timeit(fn, iters=200):
    for _ in range(20): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

for M in (64, 128):
    for (K, N) in ((1024, 3072), (1024, 1024), (1024, 8192), (4096, 1024), (1024, 8192)):
        x = torch.randn(M, K, device='cuda').bfloat16()
        w = torch.randn(N, K, device='cuda').bfloat16()
        b = torch.randn(N, device='cuda').float()
        bb = b.bfloat16()
        us_s = timeit(lambda: ext.skinny_gemm(x, w, b))
        us_l = timeit(lambda: F.linear(x, w, bb))
        wbytes = N*K*2
        print(f'M{M} K{K} N{N}: skinny {us_s:7.1f}us ({wbytes/us_s/1e3:6.2f} TB/s)  linear {us_l:7.1f}us ({wbytes/us_l/1e3:6.2f} TB/s)')
