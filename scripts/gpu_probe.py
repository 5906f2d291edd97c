#!/usr/bin/env python3
"""Diagnostic: MFMA layout probe + flash-attention error report.
Run on a GPU box; prints everything needed to fix operand layouts offline."""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

import dalle_pytorch_amd._hip as ext


def main():
    torch.manual_seed(0)
    # ramp matrices make layout transpositions obvious
    A = (torch.arange(16 * 32).reshape(16, 32).float() / 100).bfloat16().cuda()
    B = (torch.arange(32 * 16).reshape(32, 16).float() / 100 + 0.37).bfloat16().cuda()
    C = ext.mfma_probe(A, B)
    ref = A.float() @ B.float()
    print('probe err', (C - ref).abs().max().item())
    print('C^T err', (C - ref.t()).abs().max().item())
    print('C[0,:4]', C[0, :4].tolist())
    print('ref[0,:4]', ref[0, :4].tolist())
    print('ref^T[0,:4]', ref.t()[0, :4].tolist())

    # flash fwd small
    q = torch.randn(1, 1, 64, 64, device='cuda').bfloat16()
    k = torch.randn(1, 1, 64, 64, device='cuda').bfloat16()
    v = torch.randn(1, 1, 64, 64, device='cuda').bfloat16()
    out, lse = ext.fa_fwd(q, k, v, 0.125, True, None, None)
    d = (q.float() * 0.125) @ k.float().transpose(-1, -2)
    cm = torch.ones(64, 64, dtype=torch.bool, device='cuda').triu_(1)
    d = d.masked_fill(cm, float('-inf'))
    ref_o = d.softmax(-1) @ v.float()
    print('fa err', (out.float() - ref_o).abs().max().item())
    print('fa lse err', (lse - d.logsumexp(-1)).abs().max().item())
    print('fa out[0,0,0,:4]', out[0, 0, 0, :4].float().tolist())
    print('ref out[0,0,0,:4]', ref_o[0, 0, 0, :4].tolist())
    print('fa out[0,0,17,:4]', out[0, 0, 17, :4].float().tolist())
    print('ref out[0,0,17,:4]', ref_o[0, 0, 17, :4].tolist())


if __name__ == '__main__':
    main()
