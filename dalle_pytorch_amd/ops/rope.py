"""Fused QKV split + rotary application (autograd wrapper).

One HIP sweep replaces the eager chain chunk -> rearrange -> cos/sin ->
rotate-half mul/add -> cat for q, k AND v (reference applies rotary to all
three: attention.py:32-35,67). Gradient is the transpose rotation fused the
same way.
"""

import torch

from dalle_pytorch_amd.ops.dispatch import hip_module

def trig_tables(angles: torch.Tensor):
    """cos/sin fp32 [N, rot] for a rotary angle table [1, N, rot]; cached on
    the tensor object itself (the table is a fixed model buffer, so the
    cache's lifetime tracks the model's)."""
    hit = getattr(angles, '_dalle_amd_trig', None)
    if hit is None:
        a = angles.squeeze(0).float()
        hit = (a.cos().contiguous(), a.sin().contiguous())
        angles._dalle_amd_trig = hit
    return hit


class _RopeSplit(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, heads, cos, sin):
        ext = hip_module()
        q, k, v = ext.rope_split_fwd(qkv.contiguous(), heads, cos, sin)
        ctx.cos, ctx.sin = cos, sin
        return q, k, v

    @staticmethod
    def backward(ctx, dq, dk, dv):
        ext = hip_module()
        dqkv = ext.rope_split_bwd(dq, dk, dv, ctx.cos, ctx.sin)
        return dqkv, None, None, None


def rope_split(qkv, heads, cos=None, sin=None):
    """qkv [b, n, 3*h*64] bf16 -> (q, k, v) [b, h, n, 64] with rotary applied
    to the first cos.shape[-1] channels (cos/sin None = pure split)."""
    return _RopeSplit.apply(qkv, heads, cos, sin)


def rope_split_supported(qkv, dim_head):
    return (qkv.is_cuda and qkv.dtype == torch.bfloat16 and dim_head == 64
            and hip_module() is not None)
