import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import torch, time

def timeit(fn, iters=30):
    for _ in range(8): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6

M = 81920
for (K, N, tag) in ((8192, 1024, 'ff1-dgrad'), (1024, 10256, 'txt-head'),
                    (3072, 1024, 'qkv-dgrad')):
    dy = torch.randn(M if tag != 'txt-head' else 16448, K, device='cuda').bfloat16()
    w = torch.randn(K, N, device='cuda').bfloat16()          # [K, N] row-major
    wt = w.t().contiguous()                                  # [N, K]
    flops = 2 * dy.shape[0] * K * N
    a = timeit(lambda: dy @ w)                                # B row-major
    b = timeit(lambda: dy @ wt.t())                           # B = (N,K)^T view
    c = timeit(lambda: (wt @ dy.t()).t().contiguous())        # swapped
    print(f'{tag} [{dy.shape[0]}x{K}]x[{K}x{N}]: B-rowmajor {a:7.0f}us ({flops/a/1e6:5.0f} TF) '
          f'B-transposed-view {b:7.0f}us ({flops/b/1e6:5.0f} TF) swapped+copy {c:7.0f}us')
