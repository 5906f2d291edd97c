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


def _eager_attention(q, k, v, scale, causal, key_mask, static_mask,
                     return_lse=False):
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
    out = torch.matmul(attn, v)
    if return_lse:
        return out, torch.logsumexp(dots.float(), dim=-1)
    return out


def _flash_bwd_composite(q, k, v, out, lse, dout, scale, causal, key_mask,
                         static_mask, q_chunk=128, grad_lse=None):
    """Flash-style backward at the torch level: recompute P per q-chunk from
    the saved logsumexp, never materializing the full n x n matrix. All
    GEMMs run in bf16 on rocBLAS MFMA paths (fp32 accumulation inside);
    the P/dS elementwise math is fp32. The hand-written fa_bwd kernels are
    the production path; this stays as the oracle-adjacent fallback for
    extensions built without them."""
    b, h, nq, d = q.shape
    nk = k.shape[2]
    diag = nk - nq
    dq = torch.empty_like(q)
    dk = torch.zeros(b, h, nk, d, dtype=torch.float32, device=q.device)
    dv = torch.zeros_like(dk)
    kT = k.transpose(-1, -2)
    arange_k = torch.arange(nk, device=q.device)

    for c0 in range(0, nq, q_chunk):
        c1 = min(c0 + q_chunk, nq)
        qc, oc, doc = q[:, :, c0:c1], out[:, :, c0:c1], dout[:, :, c0:c1]
        lsec = lse[:, :, c0:c1]

        s = torch.matmul(qc, kT).float() * scale              # [b,h,C,nk]
        if causal:
            cm = arange_k[None, :] > (torch.arange(c0, c1, device=q.device)[:, None] + diag)
            s = s.masked_fill(cm, float('-inf'))
        if key_mask is not None:
            s = s.masked_fill(~key_mask[:, None, None, :], float('-inf'))
        if static_mask is not None:
            s = s.masked_fill(~static_mask[c0:c1], float('-inf'))

        p = torch.exp(s - lsec.unsqueeze(-1))
        p = torch.nan_to_num(p, nan=0.0)                      # -inf - -inf rows
        pb = p.to(q.dtype)

        dv += torch.matmul(pb.transpose(-1, -2), doc).float()
        dp = torch.matmul(doc, v.transpose(-1, -2)).float()
        Dc = (doc.float() * oc.float()).sum(dim=-1, keepdim=True)
        if grad_lse is not None:
            Dc = Dc - grad_lse[:, :, c0:c1].unsqueeze(-1)
        ds = (p * (dp - Dc) * scale).to(q.dtype)
        dq[:, :, c0:c1] = torch.matmul(ds, k)
        dk += torch.matmul(ds.transpose(-1, -2), qc).float()

    return dq, dk.to(q.dtype), dv.to(q.dtype)


class _FlashAttention(torch.autograd.Function):
    """Binds the gfx950 flash-attention forward kernel (saves out +
    logsumexp); backward recomputes probabilities chunk-by-chunk — no
    n x n matrix is ever stored across the fwd/bwd boundary. With
    ``fold_heads`` the kernel epilogue writes [b, n, h, d] directly (no
    head-merge permute before the output projection)."""

    @staticmethod
    def forward(ctx, q, k, v, scale, causal, key_mask, static_mask,
                tile_map, tile_map_t, fold_heads):
        ext = hip_module()
        q, k, v = (t.contiguous() for t in (q, k, v))
        out, lse = ext.fa_fwd(q, k, v, scale, causal,
                              key_mask, static_mask, tile_map, fold_heads)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.scale, ctx.causal = scale, causal
        ctx.key_mask, ctx.static_mask = key_mask, static_mask
        ctx.tile_map, ctx.tile_map_t = tile_map, tile_map_t
        ctx.fold_heads = fold_heads
        return out, lse

    @staticmethod
    def backward(ctx, dout, dlse):
        q, k, v, out, lse = ctx.saved_tensors
        ext = hip_module()
        if dlse is not None:
            dlse = dlse.contiguous().float()
        if hasattr(ext, 'fa_bwd'):
            # with fold_heads, out/dout stay in their contiguous [b,n,h,d]
            # layout — the bwd kernels read them with bnhd strides directly
            dq, dk, dv = ext.fa_bwd(
                q, k, v, out, lse, dout.contiguous(),
                ctx.scale, ctx.causal, ctx.key_mask, ctx.static_mask,
                ctx.tile_map, ctx.tile_map_t, ctx.fold_heads, dlse)
        else:
            if ctx.fold_heads:
                out = out.permute(0, 2, 1, 3)
                dout = dout.permute(0, 2, 1, 3)
            dq, dk, dv = _flash_bwd_composite(
                q, k, v, out, lse, dout.contiguous(), ctx.scale, ctx.causal,
                ctx.key_mask, ctx.static_mask, grad_lse=dlse)
        return dq, dk, dv, None, None, None, None, None, None, None


def _hip_supported(q, k, causal, key_mask):
    if q.shape[-1] not in _SUPPORTED_HEAD_DIMS:
        return False
    if q.dtype != torch.bfloat16:
        return False
    if q.shape[-2] < 32:
        # the model's dict-cache decode path: rocBLAS batched GEMV is fine
        # there (the production decode is FastDecoder's fa_decode kernels);
        # not a training-path fallback
        return False
    return True


_warned_shapes = set()


def _fold(out):
    b, h, n, d = out.shape
    return out.permute(0, 2, 1, 3).reshape(b, n, h * d)


def attention_core(q, k, v, scale, causal=True, key_mask=None, static_mask=None,
                   static_tiles=None, static_tiles_t=None, fold_heads=False,
                   return_lse=False):
    """Scaled-dot-product attention with the reference's masking semantics.

    q: [b, h, nq, d] (unscaled), k/v: [b, h, nk, d],
    key_mask: optional bool [b, nk] (True = attend),
    static_mask: optional bool [nq, nk] (True = attend), already sliced for
    any cache offset (reference attention.py:91-92),
    static_tiles: optional uint8 [ceil(nq/64), ceil(nk/32)] block map of
    static_mask — fully-zero tiles are skipped by the kernel (this is how
    axial/conv/block-sparse patterns become truly sparse),
    fold_heads: return [b, nq, h*d] (GPU kernel writes it directly),
    return_lse: also return the per-row logsumexp [b, h, nq] (fp32),
    DIFFERENTIABLE — partial attentions combined by lse-merge backprop
    correctly through both outputs (the axial decomposition path).
    Returns [b, h, nq, d] or [b, nq, h*d] (+ lse).
    """
    if using_eager_fallback(q) or not _hip_supported(q, k, causal, key_mask):
        if q.is_cuda and q.shape[-2] >= 32:
            key = (q.shape[-1], str(q.dtype))
            if key not in _warned_shapes:
                _warned_shapes.add(key)
                warnings.warn(f'attention_core: shape/dtype {key} not covered '
                              'by the HIP kernel yet; using eager path on GPU')
        res = _eager_attention(q, k, v, scale, causal, key_mask, static_mask,
                               return_lse=return_lse)
        out, lse = res if return_lse else (res, None)
        out = _fold(out) if fold_heads else out
        return (out, lse) if return_lse else out
    if static_mask is not None:
        static_mask = static_mask.contiguous()
    if key_mask is not None:
        key_mask = key_mask.contiguous()
    out, lse = _FlashAttention.apply(q, k, v, scale, causal, key_mask,
                                     static_mask, static_tiles, static_tiles_t,
                                     fold_heads)
    if fold_heads:
        b, n, h, d = out.shape
        out = out.view(b, n, h * d)
    return (out, lse) if return_lse else out


def build_tile_map(static_mask, causal=False):
    """uint8 [ceil(nq/64), ceil(nk/32)] block map of the static mask:
    0 = no entry set (kernel skips the tile), 1 = partially set (kernel
    applies the exact element mask), 2 = fully set (kernel can skip the
    per-element mask loop entirely when causality also allows it)."""
    nq, nk = static_mask.shape
    tq, tk = (nq + 63) // 64, (nk + 31) // 32
    padded = torch.zeros(tq * 64, tk * 32, dtype=torch.bool,
                         device=static_mask.device)
    padded[:nq, :nk] = static_mask
    blocks = padded.reshape(tq, 64, tk, 32)
    any_set = blocks.any(dim=3).any(dim=1)
    all_set = blocks.all(dim=3).all(dim=1)
    return (any_set.to(torch.uint8) + all_set.to(torch.uint8)).contiguous()
