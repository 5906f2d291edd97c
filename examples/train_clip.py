#!/usr/bin/env python3
"""Train CLIP on (caption, image) pairs — the re-ranker for generate.py
(reference README.md:268-293 API, as a runnable script). Defaults train on
the rainbow synthetic set so the pipeline runs without any dataset."""

import argparse
import sys
from pathlib import Path

import torch
from torch.utils.data import DataLoader

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from dalle_pytorch_amd import CLIP
from dalle_pytorch_amd.utils.checkpoint import save_clip_checkpoint
from dalle_pytorch_amd.utils.loader import RainbowDataset, TextImageDataset
from dalle_pytorch_amd.utils.tokenizer import SimpleTokenizer


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument('--image_text_folder', default=None)
    ap.add_argument('--image_size', type=int, default=32)
    ap.add_argument('--text_seq_len', type=int, default=16)
    ap.add_argument('--dim', type=int, default=128)
    ap.add_argument('--depth', type=int, default=2)
    ap.add_argument('--steps', type=int, default=300)
    ap.add_argument('--batch_size', type=int, default=32)
    ap.add_argument('--out', default='clip.pt')
    args = ap.parse_args(argv)

    device = torch.device('cuda:0') if torch.cuda.is_available() else torch.device('cpu')
    tok = SimpleTokenizer()
    if args.image_text_folder:
        ds = TextImageDataset(args.image_text_folder, text_len=args.text_seq_len,
                              image_size=args.image_size, tokenizer=tok,
                              truncate_captions=True, shuffle=True)
    else:
        ds = RainbowDataset(length=1024, image_size=args.image_size,
                            text_len=args.text_seq_len, tokenizer=tok)
    dl = DataLoader(ds, args.batch_size, shuffle=True, drop_last=True)

    clip_params = dict(
        dim_text=args.dim, dim_image=args.dim, dim_latent=args.dim,
        num_text_tokens=tok.vocab_size, text_enc_depth=args.depth,
        text_seq_len=args.text_seq_len, text_heads=4,
        visual_enc_depth=args.depth, visual_heads=4,
        visual_image_size=args.image_size,
        visual_patch_size=max(args.image_size // 4, 8))
    clip = CLIP(**clip_params).to(device)
    opt = torch.optim.Adam(clip.parameters(), lr=3e-4)

    step = 0
    first = last = None
    while step < args.steps:
        for text, images in dl:
            text, images = text.to(device), images.to(device)
            loss = clip(text, images, return_loss=True)
            loss.backward()
            opt.step()
            opt.zero_grad()
            first = first if first is not None else loss.item()
            last = loss.item()
            step += 1
            if step % 50 == 0:
                print(f'[clip] step {step} loss {last:.4f}')
            if step >= args.steps:
                break
    save_clip_checkpoint(args.out, clip, clip_params)
    print(f'loss {first:.3f} -> {last:.3f}; saved {args.out}')
    return first, last


if __name__ == '__main__':
    main()
