import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import torch, time
print('dtypes:', hasattr(torch, 'float8_e4m3fn'), hasattr(torch, 'float8_e4m3fnuz'))
dev = 'cuda'
M, K, N = 64 * 1280, 1024, 8192
x = torch.randn(M, K, device=dev).bfloat16()
w = torch.randn(N, K, device=dev).bfloat16() * 0.02

def timeit(fn, iters=50):
    for _ in range(10): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0)/iters*1e6

flops = 2 * M * K * N
us = timeit(lambda: torch.nn.functional.linear(x, w))
print(f'bf16 linear: {us:.0f} us {flops/us/1e6:.0f} TF/s')

for dt_name in ('float8_e4m3fn', 'float8_e4m3fnuz'):
    try:
        dt = getattr(torch, dt_name)
        sx = x.abs().amax().float() / 448.0
        sw = w.abs().amax().float() / 448.0
        xq = (x.float() / sx).clamp(-448, 448).to(dt)
        wq = (w.float() / sw).clamp(-448, 448).to(dt)
        out = torch._scaled_mm(xq, wq.t(), scale_a=sx, scale_b=sw, out_dtype=torch.bfloat16)
        ref = torch.nn.functional.linear(x.float(), w.float())
        rel = (out.float() - ref).abs().mean().item() / ref.abs().mean().item()
        us = timeit(lambda: torch._scaled_mm(xq, wq.t(), scale_a=sx, scale_b=sw, out_dtype=torch.bfloat16))
        print(f'{dt_name}: OK, mean rel err {rel:.4f}, {us:.0f} us {flops/us/1e6:.0f} TF/s')
        # incl. quantization cost
        def full():
            sxl = x.abs().amax().float() / 448.0
            xql = (x.float() / sxl).clamp(-448, 448).to(dt)
            return torch._scaled_mm(xql, wq.t(), scale_a=sxl, scale_b=sw, out_dtype=torch.bfloat16)
        us2 = timeit(full)
        print(f'{dt_name} incl act-quant: {us2:.0f} us {flops/us2/1e6:.0f} TF/s')
    except Exception as e:
        print(f'{dt_name}: FAIL {type(e).__name__} {str(e)[:120]}')
