#!/usr/bin/env python3
"""Decode-path timing breakdown: prefill, eager step, graph replay, sampling."""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from dalle_pytorch_amd import DALLE, DiscreteVAE
from dalle_pytorch_amd.engine import FastDecoder
from dalle_pytorch_amd.models.dalle import top_k, gumbel_sample


def timeit(fn, iters=50, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    b = int(sys.argv[1]) if len(sys.argv) > 1 else 16
    torch.manual_seed(0)
    vae = DiscreteVAE(image_size=256, num_layers=3, num_tokens=8192,
                      codebook_dim=512, hidden_dim=64)
    d = DALLE(dim=1024, vae=vae, num_text_tokens=10000, text_seq_len=256,
              depth=12, heads=16, dim_head=64,
              attn_types=('axial_row', 'axial_col'), reversible=True,
              shift_tokens=True, rotary_emb=True).cuda().eval()
    text = torch.randint(1, 10000, (b, 256), device='cuda')
    token = torch.randint(0, 8192, (b,), device='cuda')

    dec = FastDecoder(d, batch_size=b, use_graph=False)
    with torch.no_grad():
        logits = dec.prefill(text)
        t_eager = timeit(lambda: dec.step(token), iters=30)
        print(f'eager step: {t_eager:.3f} ms')

        dec2 = FastDecoder(d, batch_size=b, use_graph=True)
        dec2.prefill(text)
        dec2._graph_step(token)   # capture
        t_graph = timeit(lambda: dec2._graph_step(token), iters=50)
        print(f'graph replay step: {t_graph:.3f} ms')

        def sample():
            f = top_k(logits.float(), thres=0.9)
            s = gumbel_sample(f)
            return s

        t_sample = timeit(sample, iters=50)
        print(f'sampling: {t_sample:.3f} ms')

        # fp32 engine for comparison (no autocast weight casts in the loop)
        dec3 = FastDecoder(d, batch_size=b, dtype=torch.float32, use_graph=True)
        dec3.prefill(text)
        dec3._graph_step(token)
        t_g32 = timeit(lambda: dec3._graph_step(token), iters=50)
        print(f'graph replay step (fp32): {t_g32:.3f} ms')

        import dalle_pytorch_amd
        t_pre = timeit(lambda: dec.prefill(text), iters=3, warmup=1)
        print(f'prefill: {t_pre:.3f} ms')

    est = t_graph + t_sample
    print(f'-> est per-token {est:.3f} ms; 1024 tokens = {est * 1.024:.2f} s '
          f'for batch {b} = {b / (est * 1.024):.2f} imgs/s')


if __name__ == '__main__':
    main()
