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
                tile_map, tile_map_t, fold_heads, axial=None):
        ext = hip_module()
        q, k, v = (t.contiguous() for t in (q, k, v))
        ax_t, ax_s, ax_axis = axial if axial is not None else (0, 0, -1)
        out, lse = ext.fa_fwd(q, k, v, scale, causal,
                              key_mask, static_mask, tile_map, fold_heads,
                              ax_t, ax_s, ax_axis)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.scale, ctx.causal = scale, causal
        ctx.key_mask, ctx.static_mask = key_mask, static_mask
        ctx.tile_map, ctx.tile_map_t = tile_map, tile_map_t
        ctx.fold_heads = fold_heads
        ctx.axial = (ax_t, ax_s, ax_axis)
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
                ctx.tile_map, ctx.tile_map_t, ctx.fold_heads, dlse,
                *ctx.axial)
        else:
            if ctx.axial[2] >= 0:
                raise RuntimeError('axial-mode backward needs the fa_bwd '
                                   'kernel (extension built without it)')
            if ctx.fold_heads:
                out = out.permute(0, 2, 1, 3)
                dout = dout.permute(0, 2, 1, 3)
            dq, dk, dv = _flash_bwd_composite(
                q, k, v, out, lse, dout.contiguous(), ctx.scale, ctx.causal,
                ctx.key_mask, ctx.static_mask, grad_lse=dlse)
        return (dq, dk, dv) + (None,) * 8


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


def axial_attention(q, k, v, scale, text_len, image_size, axis, key_mask=None):
    """Axial (row/col) causal attention over [text prefix + image grid]
    (reference attention.py:225-335 semantics). Returns folded [b, n, h*d].

    GPU + extension: ONE fused kernel call in axial mode — the pattern is
    evaluated arithmetically in virtual (column-major for axis 1) coords; no
    mask tensors, no tile maps, no transposes.
    Fallback (CPU / unsupported shapes): lse-merge decomposition into three
    dense attentions (pinned by tests/test_attention.py).
    """
    b, h, n, d = q.shape
    t, S = text_len, image_size
    fused = (not using_eager_fallback(q) and _hip_supported(q, k, True, None)
             and S & (S - 1) == 0 and n > t)
    if fused:
        if key_mask is not None:
            key_mask = key_mask.contiguous()
        out, _ = _FlashAttention.apply(q, k, v, scale, True, key_mask, None,
                                       None, None, True, (t, S, 0 if axis == 0 else 1))
        return out.view(b, n, h * d)

    # ---- decomposition fallback: text-causal + img->text dense + grid-local
    km_t = key_mask[:, :t].contiguous() if key_mask is not None else None
    kt, vt = k[:, :, :t], v[:, :, :t]
    out_text = attention_core(q[:, :, :t], kt, vt, scale, causal=True,
                              key_mask=km_t, fold_heads=True)
    if n <= t:
        return out_text
    n_img = n - t
    o1, l1 = attention_core(q[:, :, t:], kt, vt, scale, causal=False,
                            key_mask=km_t, return_lse=True)
    pad = S * S - n_img

    def grid(z):
        zi = z[:, :, t:]
        if pad:
            zi = torch.nn.functional.pad(zi, (0, 0, 0, pad))
        g = zi.reshape(b, h, S, S, -1)
        if axis == 1:
            g = g.transpose(2, 3)
        return g.reshape(b, h * S, S, -1).contiguous()

    o2g, l2g = attention_core(grid(q), grid(k), grid(v), scale, causal=True,
                              return_lse=True)

    def ungrid(z):
        g = z.reshape(b, h, S, S, *z.shape[3:])
        if axis == 1:
            g = g.transpose(2, 3)
        return g.reshape(b, h, S * S, *z.shape[3:])[:, :, :n_img]

    o2 = ungrid(o2g)
    l2 = ungrid(l2g.unsqueeze(-1)).squeeze(-1)
    mx = torch.maximum(l1, l2)
    w1 = (l1 - mx).exp().unsqueeze(-1)
    w2 = (l2 - mx).exp().unsqueeze(-1)
    out_img = (o1.float() * w1 + o2.float() * w2) / (w1 + w2)
    out_img = out_img.to(q.dtype).permute(0, 2, 1, 3).reshape(b, n_img, -1)
    return torch.cat((out_text, out_img), dim=1)


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
