"""Fused causal attention core (reference kernels K3-K5, attention.py:78-96).

One entry point serves every softmax-attention site in the framework:

* full causal self-attention (reference ``Attention``),
* the static-mask simulation of axial attention (``optimize_for_inference``),
* non-causal CLIP attention,
* the cached single-query decode step.

The math is plain scaled-dot-product with *finite* masking (the reference
fills masked scores with ``-finfo.max`` rather than -inf, which makes a
fully-masked row degrade to a uniform distribution instead of NaN — we keep
that behavior). The reference's ``stable_softmax`` (attention.py:27-30) is
algebraically identical to max-subtracted softmax in forward AND backward, so
the single fused kernel covers stable and non-stable models alike.

GPU path: gfx950 flash-style HIP kernel (MFMA bf16, LDS-tiled K/V, online
softmax), fp32 accumulation, O(n) memory. CPU path: eager oracle below.
"""

import warnings

import torch

from dalle_pytorch_amd.ops.dispatch import hip_module, using_eager_fallback

_SUPPORTED_HEAD_DIMS = (64,)


def _eager_attention(q, k, v, scale, causal, key_mask, static_mask):
    dots = torch.matmul(q * scale, k.transpose(-1, -2))
    big_neg = -torch.finfo(dots.dtype).max
    if key_mask is not None:
        dots = dots.masked_fill(~key_mask[:, None, None, :], big_neg)
    if causal:
        i, j = dots.shape[-2:]
        cm = torch.ones(i, j, dtype=torch.bool, device=dots.device).triu_(j - i + 1)
        dots = dots.masked_fill(cm, big_neg)
    if static_mask is not None:
        dots = dots.masked_fill(~static_mask, big_neg)
    attn = dots.softmax(dim=-1)
    return torch.matmul(attn, v)


class _FlashAttention(torch.autograd.Function):
    """Binds the gfx950 flash-attention kernels (fwd saves out + logsumexp;
    bwd recomputes probabilities tile-by-tile — no n x n matrix ever hits
    HBM)."""

    @staticmethod
    def forward(ctx, q, k, v, scale, causal, key_mask, static_mask):
        ext = hip_module()
        q, k, v = (t.contiguous() for t in (q, k, v))
        out, lse = ext.fa_fwd(q, k, v, scale, causal,
                              key_mask, static_mask)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.scale, ctx.causal = scale, causal
        ctx.key_mask, ctx.static_mask = key_mask, static_mask
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = hip_module()
        q, k, v, out, lse = ctx.saved_tensors
        dq, dk, dv = ext.fa_bwd(q, k, v, out, lse, dout.contiguous(),
                                ctx.scale, ctx.causal,
                                ctx.key_mask, ctx.static_mask)
        return dq, dk, dv, None, None, None, None


def _hip_supported(q, k, causal, key_mask):
    if q.shape[-1] not in _SUPPORTED_HEAD_DIMS:
        return False
    if q.dtype not in (torch.bfloat16, torch.float16):
        return False
    return True


_warned_shapes = set()


def attention_core(q, k, v, scale, causal=True, key_mask=None, static_mask=None):
    """Scaled-dot-product attention with the reference's masking semantics.

    q: [b, h, nq, d] (unscaled), k/v: [b, h, nk, d],
    key_mask: optional bool [b, nk] (True = attend),
    static_mask: optional bool [nq, nk] (True = attend), already sliced for
    any cache offset (reference attention.py:91-92).
    Returns [b, h, nq, d].
    """
    if using_eager_fallback(q):
        return _eager_attention(q, k, v, scale, causal, key_mask, static_mask)
    if not _hip_supported(q, k, causal, key_mask):
        key = (q.shape[-1], str(q.dtype))
        if key not in _warned_shapes:
            _warned_shapes.add(key)
            warnings.warn(f'attention_core: shape/dtype {key} not covered by '
                          'the HIP kernel yet; using eager path on GPU')
        return _eager_attention(q, k, v, scale, causal, key_mask, static_mask)
    if static_mask is not None:
        static_mask = static_mask.contiguous()
    if key_mask is not None:
        key_mask = key_mask.contiguous()
    return _FlashAttention.apply(q, k, v, scale, causal, key_mask, static_mask)
