#!/usr/bin/env python3
"""Generate images from a trained DALL-E checkpoint (reference generate.py
parity): load checkpoint -> rebuild VAE+DALLE -> autoregressive sampling
(cached decode by default on GPU) -> save PNGs; optional text completion
first (--gentxt) and CLIP re-ranking.
"""

import argparse
from pathlib import Path

import torch

from dalle_pytorch_amd.utils.checkpoint import (
    load_dalle_checkpoint, build_dalle_from_checkpoint)
from dalle_pytorch_amd.utils import tokenizer as tokenizer_mod


def parse_args(argv=None):
    p = argparse.ArgumentParser(description='DALL-E image generation (MI355X-native)')
    p.add_argument('--dalle_path', type=str, required=True)
    p.add_argument('--vqgan_model_path', type=str, default=None)
    p.add_argument('--vqgan_config_path', type=str, default=None)
    p.add_argument('--text', type=str, required=True,
                   help='prompt(s); separate multiple with |')
    p.add_argument('--num_images', type=int, default=128)
    p.add_argument('--batch_size', type=int, default=4)
    p.add_argument('--top_k', type=float, default=0.9)
    p.add_argument('--temperature', type=float, default=1.)
    p.add_argument('--outputs_dir', type=str, default='./outputs')
    p.add_argument('--bpe_path', type=str, default=None)
    p.add_argument('--hug', action='store_true')
    p.add_argument('--chinese', action='store_true')
    p.add_argument('--taming', action='store_true',
                   help='use the native VQGAN (with --vqgan_model_path/'
                        '--vqgan_config_path) as the decoder VAE')
    p.add_argument('--gentxt', action='store_true',
                   help='complete the prompt with the model before generating')
    p.add_argument('--no_cache', action='store_true')
    p.add_argument('--no_fast', action='store_true',
                   help='disable the static-shape decode engine (HIP graphs)')
    p.add_argument('--cond_scale', type=float, default=1.0)
    p.add_argument('--clip_path', type=str, default=None,
                   help='trained CLIP checkpoint: re-rank generations and '
                        'keep the best --num_images of an oversampled batch')
    p.add_argument('--clip_oversample', type=int, default=2)
    return p.parse_args(argv)


def get_tokenizer(args):
    if args.chinese:
        return tokenizer_mod.ChineseTokenizer()
    if args.hug:
        return tokenizer_mod.HugTokenizer(args.bpe_path)
    if args.bpe_path is not None:
        suffix = Path(args.bpe_path).suffix
        if suffix == '.json':
            return tokenizer_mod.HugTokenizer(args.bpe_path)
        if suffix == '.model':
            return tokenizer_mod.YttmTokenizer(args.bpe_path)
        return tokenizer_mod.SimpleTokenizer(args.bpe_path)
    return tokenizer_mod.tokenizer


def main(argv=None):
    args = parse_args(argv)
    device = torch.device('cuda:0') if torch.cuda.is_available() else torch.device('cpu')
    tok = get_tokenizer(args)

    clip = None
    if args.clip_path:
        from dalle_pytorch_amd.utils.checkpoint import load_clip_checkpoint
        clip, _ = load_clip_checkpoint(args.clip_path)
        clip = clip.to(device).eval()

    ckpt = load_dalle_checkpoint(args.dalle_path)
    vae = None
    if args.taming or (ckpt.get('vae_class_name') == 'VQGanVAE'
                       and (args.vqgan_model_path or args.vqgan_config_path)):
        from dalle_pytorch_amd import VQGanVAE
        vae = VQGanVAE(args.vqgan_model_path, args.vqgan_config_path)
    dalle, vae = build_dalle_from_checkpoint(ckpt, vae=vae)
    dalle = dalle.to(device).eval()

    texts = args.text.split('|')
    out_root = Path(args.outputs_dir)

    for raw_text in texts:
        if args.gentxt:
            text_tokens, gen_texts = dalle.generate_texts(tok, text=raw_text)
            raw_text = gen_texts[0]
            print(f'completed text: {raw_text}')
            text_tokens = text_tokens.repeat(args.num_images, 1)
        else:
            text_tokens = tok.tokenize([raw_text], dalle.text_seq_len,
                                       truncate_text=True).to(device)
            text_tokens = text_tokens.repeat(args.num_images, 1)

        # the fast decoder covers guided generation (cond_scale != 1) with a
        # doubled-batch stream and gentxt (text completion happens above,
        # image decode is the same); only explicit --no_fast falls back
        guided = args.cond_scale != 1.0
        use_fast = not args.no_fast
        decoder = None
        if use_fast:
            try:
                from dalle_pytorch_amd.engine import FastDecoder
                decoder = FastDecoder(
                    dalle,
                    batch_size=args.batch_size * (2 if guided else 1),
                    use_graph=device.type == 'cuda')
            except (ValueError, AssertionError) as e:
                print(f'fast decoder unavailable ({e}); using cached decode')

        images = []
        for i in range(0, text_tokens.shape[0], args.batch_size):
            chunk = text_tokens[i:i + args.batch_size]
            if decoder is not None and chunk.shape[0] == args.batch_size:
                images.append(decoder.generate(
                    chunk, filter_thres=args.top_k,
                    temperature=args.temperature, cond_scale=args.cond_scale))
            else:
                images.append(dalle.generate_images(
                    chunk, filter_thres=args.top_k, temperature=args.temperature,
                    use_cache=not args.no_cache, cond_scale=args.cond_scale))
        images = torch.cat(images, dim=0)

        if clip is not None:
            # CLIP re-rank (reference generate path: dalle_pytorch.py:558-560)
            with torch.no_grad():
                scores = clip(text_tokens[:images.shape[0], :clip.text_pos_emb.num_embeddings],
                              images, return_loss=False)
            keep = scores.argsort(descending=True)[:args.num_images]
            images = images[keep]

        subdir = out_root / raw_text.replace(' ', '_')[:100]
        subdir.mkdir(parents=True, exist_ok=True)
        from dalle_pytorch_amd.utils.vision import save_image
        for j, img in enumerate(images):
            save_image(img.float().cpu(), subdir / f'{j}.png')
        print(f'created {images.shape[0]} images at "{subdir}"')


if __name__ == '__main__':
    main()
