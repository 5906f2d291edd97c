#!/usr/bin/env python3
"""Convert OpenAI's released dVAE weights to a dalle_pytorch_amd checkpoint.

The published encoder.pkl / decoder.pkl are pickled torch *modules* pinned
to torch<1.11 (reference vae.py:114 — incompatible with ROCm torch 2.10).
This script ingests either those pickles (if loadable) or state_dict files
extracted elsewhere, maps the dall_e key schema onto
:class:`dalle_pytorch_amd.models.vae_adapters.OpenAIDiscreteVAE`, and writes
a plain state_dict .pt.

dall_e layout -> ours:
  enc blocks.input.{w,b}                      -> encoder.input.{weight,bias}
  enc blocks.group_G.block_J.id_path.{w,b}    -> encoder.group_G.<J-1>.id_path.*
  enc blocks.group_G.block_J.res_path.conv_I  -> encoder.group_G.<J-1>.res_path.<2I-1>
  enc blocks.output.conv.{w,b}                -> encoder.output.1.*
  dec blocks.input.w [n_init, vocab, 1, 1]    -> codebook.weight (transposed)
  dec blocks.group_G... / output.conv          -> decoder.group_G... / decoder.output.1

Usage:
  python scripts/convert_openai_dvae.py --encoder enc.pkl --decoder dec.pkl \
      --out openai_dvae_amd.pt
"""

import argparse
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


def load_state(path):
    obj = torch.load(path, map_location='cpu', weights_only=False)
    if hasattr(obj, 'state_dict'):
        obj = obj.state_dict()
    return dict(obj)


def map_block_keys(src, src_prefix, dst_prefix, out):
    """One dall_e {Encoder,Decoder}Block -> _OAIEnc/DecBlock."""
    for name, tensor in src.items():
        if not name.startswith(src_prefix):
            continue
        tail = name[len(src_prefix):]
        if tail.startswith('id_path.'):
            param = 'weight' if tail.endswith('.w') else 'bias'
            out[f'{dst_prefix}.id_path.{param}'] = tensor
        elif tail.startswith('res_path.conv_'):
            idx = int(tail.split('conv_')[1].split('.')[0])
            param = 'weight' if tail.endswith('.w') else 'bias'
            out[f'{dst_prefix}.res_path.{2 * idx - 1}.{param}'] = tensor


def convert(enc_state, dec_state, blocks_per_group=2):
    out = {}
    # encoder
    out['encoder.input.weight'] = enc_state['blocks.input.w']
    out['encoder.input.bias'] = enc_state['blocks.input.b'].reshape(-1)
    for g in range(1, 5):
        for j in range(1, blocks_per_group + 1):
            map_block_keys(enc_state, f'blocks.group_{g}.block_{j}.',
                           f'encoder.group_{g}.{j - 1}', out)
    out['encoder.output.1.weight'] = enc_state['blocks.output.conv.w']
    out['encoder.output.1.bias'] = enc_state['blocks.output.conv.b'].reshape(-1)

    # decoder: input conv == the codebook, transposed
    w = dec_state['blocks.input.w']            # [n_init, vocab, 1, 1]
    out['codebook.weight'] = w[:, :, 0, 0].t().contiguous()
    for g in range(1, 5):
        for j in range(1, blocks_per_group + 1):
            map_block_keys(dec_state, f'blocks.group_{g}.block_{j}.',
                           f'decoder.group_{g}.{j - 1}', out)
    out['decoder.output.1.weight'] = dec_state['blocks.output.conv.w']
    out['decoder.output.1.bias'] = dec_state['blocks.output.conv.b'].reshape(-1)
    # dall_e stores biases as [1, C, 1, 1]; conv biases must be flat
    for k, v in list(out.items()):
        if k.endswith('.bias') and v.dim() > 1:
            out[k] = v.reshape(-1)
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--encoder', required=True)
    ap.add_argument('--decoder', required=True)
    ap.add_argument('--out', default='openai_dvae_amd.pt')
    args = ap.parse_args()

    from dalle_pytorch_amd import OpenAIDiscreteVAE
    enc, dec = load_state(args.encoder), load_state(args.decoder)
    state = convert(enc, dec)
    vae = OpenAIDiscreteVAE()
    missing, unexpected = vae.load_state_dict(state, strict=False)
    print(f'missing: {len(missing)}, unexpected: {len(unexpected)}')
    if missing:
        print('first missing:', missing[:5])
    torch.save(state, args.out)
    print(f'wrote {args.out}')


if __name__ == '__main__':
    main()
