import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import os, torch, time
os.environ['DALLE_AMD_FP8'] = '1'
from dalle_pytorch_amd.ops import fp8 as fp8_mod

def timeit(fn, iters=50):
    for _ in range(10): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6

M = 81920
for (K, N, bias) in ((1024, 8192, True), (1024, 3072, False), (4096, 1024, True)):
    lin = torch.nn.Linear(K, N, bias=bias).cuda().bfloat16()
    x = torch.randn(M, K, device='cuda').bfloat16()
    us_b = timeit(lambda: lin(x))
    with torch.no_grad():
        us_8 = timeit(lambda: fp8_mod.fp8_linear(lin, x))
    ext = __import__('dalle_pytorch_amd._hip', fromlist=['x'])
    us_amax = timeit(lambda: ext.amax_bf16(x))
    sc = ext.amax_bf16(x)
    us_q = timeit(lambda: ext.quant_fp8(x, sc))
    xq = ext.quant_fp8(x, sc); scs = sc.squeeze()
    wq = ext.quant_fp8(lin.weight.detach().contiguous(), ext.amax_bf16(lin.weight.detach().contiguous()))
    wscs = ext.amax_bf16(lin.weight.detach().contiguous()).squeeze()
    b8 = None if not bias else lin.bias.detach()
    us_mm = timeit(lambda: torch._scaled_mm(xq, wq.t(), scale_a=scs, scale_b=wscs, bias=b8, out_dtype=torch.bfloat16))
    print(f'K{K} N{N} bias={bias}: bf16 {us_b:7.0f}  fp8_linear {us_8:7.0f} (amax {us_amax:5.0f} quant {us_q:5.0f} mm {us_mm:6.0f})')
