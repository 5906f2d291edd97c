"""15-batch generation soak: graph recapture + out-buffer allocation must
not grow memory across generate() calls."""
import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import torch
from dalle_pytorch_amd import DALLE, DiscreteVAE
from dalle_pytorch_amd.engine import FastDecoder

vae = DiscreteVAE(image_size=256, num_layers=3, num_tokens=8192,
                  codebook_dim=512, hidden_dim=64)
d = DALLE(dim=1024, vae=vae, num_text_tokens=10000, text_seq_len=256,
          depth=12, heads=16, dim_head=64,
          attn_types=('axial_row', 'axial_col'), reversible=True,
          shift_tokens=True).cuda().eval()
dec = FastDecoder(d, batch_size=16, use_graph=True)
text = torch.randint(1, 10000, (16, 256), device='cuda')
mems = []
for i in range(15):
    imgs = dec.generate(text, filter_thres=0.9)
    torch.cuda.synchronize()
    m = torch.cuda.memory_allocated() / 2**30
    mems.append(m)
    if i % 5 == 0 or i == 14:
        print(f'batch {i}: mem {m:.2f} GB', flush=True)
    assert torch.isfinite(imgs).all()
assert mems[-1] - mems[2] < 0.25, f'memory grew across generates: {mems}'
print('generate soak ok')
