"""Isolated decode-attention kernel timing at flagship shapes."""
import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import torch, time
import dalle_pytorch_amd._hip as ext
from dalle_pytorch_amd.models.attention import axial_mask

b, h, N, S, t = 64, 16, 1281, 32, 257
dev = 'cuda'
torch.manual_seed(0)
qkv = torch.randn(b, 3*h*64, device=dev).bfloat16()
kc = torch.randn(b, h, N, 64, device=dev).bfloat16()
vc = torch.randn(b, h, N, 64, device=dev).bfloat16()
cos = torch.randn(N, 60, device=dev).float().contiguous()
sin = torch.randn(N, 60, device=dev).float().contiguous()
off = torch.tensor([1100], device=dev)

am = axial_mask(N, t, S, 0).cuda()
ar = torch.arange(N, device=dev)
causal = ar[None] <= ar[:, None]
allow = causal & am
cnt = allow.sum(1, dtype=torch.int32)
lmax = int(cnt.max())
idx = torch.where(allow, ar[None].expand_as(allow), N)
idx, _ = idx.sort(1)
live = torch.where(idx[:, :lmax] == N, 0, idx[:, :lmax]).int().contiguous()
cntc = cnt.contiguous()
print('lmax', lmax, 'live@1100', int(cnt[1100]))

def timeit(fn, iters=300):
    # graph-replay timing: eager launches cost ~19 us of host overhead and
    # mask the kernel (the skinny.txt lesson); capture 20 calls, replay
    fn(); torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        for _ in range(20):
            fn()
    g.replay(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters // 10):
        g.replay()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / (iters // 10) / 20 * 1e6

off_t = torch.tensor([1100], device=dev)
for clip in (32, 64, 128, 224, 269):
    cnt_c = cnt.clamp(max=clip).contiguous()
    us = timeit(lambda: ext.fa_decode(qkv, kc, vc, cos, sin, off_t, None, 0.125, live, cnt_c))
    traffic = b*h*min(clip, int(cnt[1100]))*256
    print(f'live={min(clip,int(cnt[1100])):4d}: {us:6.1f} us  eff {traffic/us/1e6:.2f} TB/s')
