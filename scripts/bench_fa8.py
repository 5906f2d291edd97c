import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import os, time, torch
import dalle_pytorch_amd._hip as ext

def timeit(fn, iters=100):
    for _ in range(15): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0)/iters*1e6

b, h, n, d = 64, 16, 1280, 64
q = torch.randn(b, h, n, d, device='cuda').bfloat16()
k = torch.randn(b, h, n, d, device='cuda').bfloat16()
v = torch.randn(b, h, n, d, device='cuda').bfloat16()
flops = b*h*n*n*d*2*2/2   # causal half
name = 'fa16' if os.environ.get('DALLE_AMD_FA8') == '0' else 'fa8'
us = timeit(lambda: ext.fa_fwd(q, k, v, 0.125, True, None, None, None, True))
print(f'{name} causal dense fwd: {us:8.1f} us  {flops/us/1e6:6.0f} TF/s')
t, S = 257, 32
us = timeit(lambda: ext.fa_fwd(q, k, v, 0.125, True, None, None, None, True, t, S, 0))
print(f'{name}-axial axis0: {us:8.1f} us')
us = timeit(lambda: ext.fa_fwd(q, k, v, 0.125, True, None, None, None, True, t, S, 1))
print(f'{name}-axial axis1: {us:8.1f} us')
