#!/usr/bin/env python3
"""End-to-end synthetic workload: the reference notebook's de-facto
integration test (examples/rainbow_dalle.ipynb) as a script.

Generates compositional colored-shape images with fully-descriptive
captions, trains a DiscreteVAE, trains a DALLE on the (caption, image)
pairs, then measures image-token reconstruction accuracy of generation
against re-encoding — the reference's correctness signal (train full-string
accuracy approaches 1.0 when run to convergence).

CPU-runnable at the default tiny scale; pass --gpu-scale on an MI355X for
the notebook-sized run.
"""

import argparse
import sys
from pathlib import Path

import torch
from torch.utils.data import DataLoader

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from dalle_pytorch_amd import DALLE, DiscreteVAE
from dalle_pytorch_amd.utils.loader import RainbowDataset
from dalle_pytorch_amd.utils.tokenizer import SimpleTokenizer


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument('--samples', type=int, default=256)
    ap.add_argument('--image_size', type=int, default=32)
    ap.add_argument('--vae_steps', type=int, default=150)
    ap.add_argument('--dalle_steps', type=int, default=300)
    ap.add_argument('--dim', type=int, default=128)
    ap.add_argument('--depth', type=int, default=2)
    ap.add_argument('--batch_size', type=int, default=16)
    ap.add_argument('--gpu-scale', action='store_true',
                    help='notebook-sized run (dim 1024, depth 12)')
    args = ap.parse_args(argv)
    if args.gpu_scale:
        args.dim, args.depth, args.samples = 1024, 12, 4096
        args.vae_steps, args.dalle_steps = 1000, 3000

    device = torch.device('cuda:0') if torch.cuda.is_available() else torch.device('cpu')
    torch.manual_seed(0)
    tok = SimpleTokenizer()
    text_len = 16
    ds = RainbowDataset(length=args.samples, image_size=args.image_size,
                        text_len=text_len, tokenizer=tok)
    dl = DataLoader(ds, batch_size=args.batch_size, shuffle=True, drop_last=True)

    # ---- stage 1: dVAE
    vae = DiscreteVAE(image_size=args.image_size, num_layers=2, num_tokens=64,
                      codebook_dim=64, hidden_dim=32,
                      straight_through=False, temperature=0.9).to(device)
    opt = torch.optim.Adam(vae.parameters(), lr=1e-3)
    step = 0
    while step < args.vae_steps:
        for _, images in dl:
            images = images.to(device)
            loss = vae(images, return_loss=True)
            loss.backward()
            opt.step()
            opt.zero_grad()
            step += 1
            if step % 50 == 0:
                print(f'[vae] step {step} loss {loss.item():.4f}')
            if step >= args.vae_steps:
                break

    # ---- stage 2: DALLE
    dalle = DALLE(dim=args.dim, vae=vae, num_text_tokens=tok.vocab_size,
                  text_seq_len=text_len, depth=args.depth, heads=4, dim_head=64,
                  attn_types=('full',), shift_tokens=True,
                  rotary_emb=True).to(device)
    opt = torch.optim.Adam((p for p in dalle.parameters() if p.requires_grad),
                           lr=3e-4)
    step = 0
    first_loss = last_loss = None
    while step < args.dalle_steps:
        for text, images in dl:
            text, images = text.to(device), images.to(device)
            loss = dalle(text, images, return_loss=True)
            loss.backward()
            opt.step()
            opt.zero_grad()
            if first_loss is None:
                first_loss = loss.item()
            last_loss = loss.item()
            step += 1
            if step % 100 == 0:
                print(f'[dalle] step {step} loss {last_loss:.4f}')
            if step >= args.dalle_steps:
                break

    # ---- evaluation: per-position image-token accuracy on training prompts
    dalle.eval()
    text, images = next(iter(dl))
    text, images = text[:8].to(device), images[:8].to(device)
    with torch.no_grad():
        target_codes = vae.get_codebook_indices(images)
        gen = dalle.generate_images(text, use_cache=True, filter_thres=0.99,
                                    temperature=1e-4)
        gen_codes = vae.get_codebook_indices(gen)
    acc = (gen_codes == target_codes).float().mean().item()
    print(f'loss {first_loss:.3f} -> {last_loss:.3f}; '
          f'per-position image-token accuracy {acc:.3f}')
    return first_loss, last_loss, acc


if __name__ == '__main__':
    main()
