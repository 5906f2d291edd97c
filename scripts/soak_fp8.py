import sys; sys.path.insert(0, '/root/repo')
import torch, time
sys.argv = ['bench.py', '--config', 'c', '--steps', '0', '--warmup', '0']
# 200-step soak watching memory + loss finiteness (cache-leak insurance for
# the fp8/wt/cast per-step caches)
from dalle_pytorch_amd import DALLE, DiscreteVAE
from dalle_pytorch_amd.ops.fp8 import set_fp8_enabled, fp8_mark_step
set_fp8_enabled(True)
vae = DiscreteVAE(image_size=256, num_layers=3, num_tokens=8192, codebook_dim=512, hidden_dim=64)
d = DALLE(dim=1024, vae=vae, num_text_tokens=10000, text_seq_len=256, depth=12,
          heads=16, dim_head=64, attn_types=('axial_row','axial_col'),
          reversible=True, shift_tokens=True).cuda()
opt = torch.optim.Adam([p for p in d.parameters() if p.requires_grad], lr=3e-4, fused=True)
text = torch.randint(1, 10000, (32, 256), device='cuda')
imgs = torch.rand(32, 3, 256, 256, device='cuda')
mems = []
for i in range(200):
    with torch.autocast('cuda', dtype=torch.bfloat16):
        loss = d(text, imgs, return_loss=True)
    loss.backward()
    opt.step(); opt.zero_grad(); fp8_mark_step()
    if i % 40 == 0 or i == 199:
        torch.cuda.synchronize()
        m = torch.cuda.memory_allocated() / 2**30
        mems.append(m)
        print(f'step {i}: loss {loss.item():.4f} mem {m:.2f} GB', flush=True)
        assert torch.isfinite(loss)
assert mems[-1] - mems[1] < 0.5, f'memory grew: {mems}'
print('soak ok: no leak, loss finite throughout')
