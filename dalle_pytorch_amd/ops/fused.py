"""Fused elementwise ops (memory-bound; reference kernel rows K6/K7).

On gfx950 these are single HBM-bound passes (vectorized bf16x8 loads per the
CDNA4 guide); on CPU the eager equivalents run.
"""

import torch
import torch.nn.functional as F

from dalle_pytorch_amd.ops.dispatch import hip_module, using_eager_fallback


class _GegluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = hip_module()
        x = x.contiguous()
        out = ext.geglu_fwd(x)
        ctx.save_for_backward(x)
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = hip_module()
        (x,) = ctx.saved_tensors
        return ext.geglu_bwd(x, dout.contiguous())


def geglu(x: torch.Tensor) -> torch.Tensor:
    """GEGLU gate: split the last dim in half, return value * gelu(gate).

    Matches reference transformer.py:106-109 (erf-based exact gelu).
    """
    if using_eager_fallback(x) or x.shape[-1] % 16:
        a, b = x.chunk(2, dim=-1)
        return a * F.gelu(b)
    return _GegluFn.apply(x)
