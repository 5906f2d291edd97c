"""Attention variants for the DALL-E transformer stack.

Four interchangeable modules with the same contract as the reference
(attention.py:39-398): full causal, axial row/col sparse, conv-like sparse,
and block-sparse. All softmax cores route through
:func:`dalle_pytorch_amd.ops.attention_core`, the single fused CDNA4 flash
kernel, so every variant hits MFMA+LDS on GPU; the sparse variants express
their patterns as masks / gathered key sets around that core.

Semantics notes carried over from the reference (kept bit-for-bit):
* masked scores are filled with finite ``-finfo.max`` (not -inf),
* ``stable_softmax`` (attention.py:27-30) == plain max-subtracted softmax in
  fwd and bwd, so ``stable`` changes nothing inside the fused core,
* rotary tables rotate q, k AND v (attention.py:35,67),
* the k/v inference cache concatenates along the sequence dim and causality
  is implicit once ``offset > 0`` (attention.py:71-76,86).
"""

from math import ceil

import torch
import torch.nn.functional as F
from torch import nn

from dalle_pytorch_amd.models.positional import apply_rotary_to_qkv
from dalle_pytorch_amd.ops import attention_core


def _split_heads(t, h):
    b, n, hd = t.shape
    return t.reshape(b, n, h, hd // h).permute(0, 2, 1, 3)


def _merge_heads(t):
    b, h, n, d = t.shape
    return t.permute(0, 2, 1, 3).reshape(b, n, h * d)


class Attention(nn.Module):
    """Full (optionally causal) multi-head self-attention with k/v cache.

    Parity target: reference attention.py:39-99. ``static_mask`` lets this
    class also serve as the cache-friendly simulation of axial attention
    (reference transformer.py:252-260,333-350).
    """

    def __init__(self, dim, seq_len, causal=True, heads=8, dim_head=64,
                 dropout=0., stable=False, static_mask=None):
        super().__init__()
        self.heads = heads
        self.dim_head = dim_head
        self.seq_len = seq_len
        self.scale = dim_head ** -0.5
        self.causal = causal
        self.stable = stable  # numerically equivalent either way; kept for hparams parity
        self.register_buffer('static_mask', static_mask, persistent=False)
        inner = heads * dim_head
        self.to_qkv = nn.Linear(dim, inner * 3, bias=False)
        self.to_out = nn.Sequential(nn.Linear(inner, dim), nn.Dropout(dropout))

    def forward(self, x, mask=None, rotary_pos_emb=None, cache=None, cache_key=None):
        h = self.heads
        offset = cache.get('offset', 0) if cache is not None else 0

        qkv = self.to_qkv(x).chunk(3, dim=-1)
        q, k, v = (_split_heads(t, h) for t in qkv)

        if rotary_pos_emb is not None:
            q, k, v = apply_rotary_to_qkv(rotary_pos_emb[..., offset:, :], (q, k, v))

        if offset > 0:
            k_prev, v_prev = cache[cache_key]
            k = torch.cat((k_prev, k), dim=-2)
            v = torch.cat((v_prev, v), dim=-2)
        if cache is not None:
            cache[cache_key] = (k, v)

        n_q, n_k = q.shape[-2], k.shape[-2]
        static = None
        if self.static_mask is not None:
            static = self.static_mask[offset:offset + n_q, :n_k]

        out = attention_core(
            q, k, v, self.scale,
            causal=self.causal and offset == 0,
            key_mask=mask, static_mask=static)
        return self.to_out(_merge_heads(out))


class SparseAxialCausalAttention(nn.Module):
    """Axial (row or column) causal attention over the image grid; the text
    prefix gets full causal attention and every image token attends to all
    text. Parity target: reference attention.py:225-335.

    axis=0: attend within the image row; axis=1: within the column.
    """

    def __init__(self, dim, seq_len, image_size=32, axis=0, heads=8, dim_head=64,
                 dropout=0., stable=False, causal=True, **kwargs):
        super().__init__()
        assert axis in (0, 1)
        self.axis = axis
        self.heads = heads
        self.seq_len = seq_len
        self.scale = dim_head ** -0.5
        self.image_size = image_size
        self.stable = stable
        inner = heads * dim_head
        self.to_qkv = nn.Linear(dim, inner * 3, bias=False)
        self.to_out = nn.Sequential(nn.Linear(inner, dim), nn.Dropout(dropout))

    def forward(self, x, mask=None, rotary_pos_emb=None):
        b, n, _ = x.shape
        h, S = self.heads, self.image_size
        img_len = S * S
        text_len = self.seq_len + 1 - img_len
        pad = self.seq_len - n + 1

        if mask is None:
            mask = torch.ones(b, text_len, dtype=torch.bool, device=x.device)
        mask = mask[:, :text_len]

        x = F.pad(x, (0, 0, 0, pad))
        qkv = self.to_qkv(x).chunk(3, dim=-1)
        q, k, v = (_split_heads(t, h) for t in qkv)  # [b, h, seq+1, d]

        if rotary_pos_emb is not None:
            # reference flattens (b h) before rotary; table broadcasts the same
            q, k, v = apply_rotary_to_qkv(rotary_pos_emb, (q, k, v))

        (q_t, q_i), (k_t, k_i), (v_t, v_i) = (
            (t[..., :text_len, :], t[..., text_len:, :]) for t in (q, k, v))

        # --- text prefix: full causal attention, with the padding mask
        out_text = attention_core(q_t, k_t, v_t, self.scale,
                                  causal=True, key_mask=mask)

        # --- image: per-row (axis 0) or per-column (axis 1) causal + all text
        d = q_i.shape[-1]
        if self.axis == 0:
            to_grid = lambda t: t.reshape(b, h, S, S, d)
        else:
            to_grid = lambda t: t.reshape(b, h, S, S, d).transpose(2, 3)
        qg, kg, vg = to_grid(q_i), to_grid(k_i), to_grid(v_i)  # [b,h,S(x),S(i),d]

        # fold the axial stripe into the batch dim: [b*h*S, S, d]
        fold = lambda t: t.reshape(b * h * S, S, d).unsqueeze(1)
        # broadcast text keys across the S stripes: [b*h*S, text_len, d]
        fold_text = lambda t: (t.unsqueeze(2).expand(b, h, S, text_len, d)
                               .reshape(b * h * S, text_len, d).unsqueeze(1))

        k_cat = torch.cat((fold_text(k_t), fold(kg)), dim=-2)
        v_cat = torch.cat((fold_text(v_t), fold(vg)), dim=-2)

        # static mask: text cols always on; image cols causal within stripe
        sm = torch.ones(S, text_len + S, dtype=torch.bool, device=x.device)
        sm[:, text_len:] = ~torch.ones(S, S, dtype=torch.bool, device=x.device).triu_(1)
        # padding mask on the text keys, broadcast over heads & stripes
        km = torch.cat((mask, torch.ones(b, S, dtype=torch.bool, device=x.device)), dim=1)
        km = km.unsqueeze(1).unsqueeze(2).expand(b, h, S, text_len + S)
        km = km.reshape(b * h * S, text_len + S)

        out_img = attention_core(fold(qg), k_cat, v_cat, self.scale,
                                 causal=False, key_mask=km, static_mask=sm)
        out_img = out_img.squeeze(1).reshape(b, h, S, S, d)
        if self.axis == 1:
            out_img = out_img.transpose(2, 3)
        out_img = out_img.reshape(b, h, img_len, d)

        out = torch.cat((out_text, out_img), dim=-2)
        out = self.to_out(_merge_heads(out))
        return out[:, :n]


class SparseConvCausalAttention(nn.Module):
    """Conv-like sparse attention: each image token attends to a causally
    padded k x k dilated neighborhood plus all text. Parity target:
    reference attention.py:103-221.
    """

    def __init__(self, dim, seq_len, image_size=32, kernel_size=5, dilation=1,
                 heads=8, dim_head=64, dropout=0., stable=False, causal=True, **kwargs):
        super().__init__()
        assert kernel_size % 2 == 1, 'kernel size must be odd'
        self.heads = heads
        self.seq_len = seq_len
        self.scale = dim_head ** -0.5
        self.image_size = image_size
        self.kernel_size = kernel_size
        self.dilation = dilation
        self.stable = stable
        inner = heads * dim_head
        self.to_qkv = nn.Linear(dim, inner * 3, bias=False)
        self.to_out = nn.Sequential(nn.Linear(inner, dim), nn.Dropout(dropout))

    def forward(self, x, mask=None, rotary_pos_emb=None):
        b, n, _ = x.shape
        h, S, ks, dil = self.heads, self.image_size, self.kernel_size, self.dilation
        img_len = S * S
        text_len = self.seq_len + 1 - img_len
        pad = self.seq_len - n + 1

        if mask is None:
            mask = torch.ones(b, text_len, dtype=torch.bool, device=x.device)
        mask = mask[:, :text_len]

        x = F.pad(x, (0, 0, 0, pad))
        qkv = self.to_qkv(x).chunk(3, dim=-1)
        q, k, v = (_split_heads(t, h) for t in qkv)

        if rotary_pos_emb is not None:
            q, k, v = apply_rotary_to_qkv(rotary_pos_emb, (q, k, v))

        d = q.shape[-1]
        (q_t, q_i), (k_t, k_i), (v_t, v_i) = (
            (t[..., :text_len, :], t[..., text_len:, :]) for t in (q, k, v))

        # text prefix: full causal
        out_text = attention_core(q_t, k_t, v_t, self.scale, causal=True, key_mask=mask)

        # image neighborhoods via causal unfold (pad top/left by the full window)
        eff = (ks - 1) * dil + 1
        half = eff // 2
        causal_pad = (2 * half, 0, 2 * half, 0)

        kn = _conv_neigh(k_i, b, h, S, d, ks, dil, causal_pad)
        vn = _conv_neigh(v_i, b, h, S, d, ks, dil, causal_pad)

        qf = (q_i * self.scale).reshape(b * h, img_len, d)
        dots_img = torch.einsum('bid,bijd->bij', qf, kn)           # [bh, img, k*k]
        dots_txt = torch.einsum('bid,bjd->bij', qf,
                                k_t.reshape(b * h, text_len, d))   # [bh, img, text]

        # neighborhood validity mask from unfolding a ones-grid
        ones = torch.ones(1, 1, S, S, device=x.device)
        ones = F.pad(ones, causal_pad)
        ones = F.unfold(ones, ks, dilation=dil)                    # [1, k*k, img]
        valid = (ones > 0).permute(0, 2, 1)                        # [1, img, k*k]

        big_neg = -torch.finfo(dots_img.dtype).max
        dots = torch.cat((dots_txt, dots_img), dim=-1)
        tmask = mask.unsqueeze(1).unsqueeze(1).expand(b, h, img_len, text_len) \
                    .reshape(b * h, img_len, text_len)
        full_mask = torch.cat((tmask, valid.expand(b * h, -1, -1)), dim=-1)
        dots = dots.masked_fill(~full_mask, big_neg)

        attn = dots.softmax(dim=-1)
        a_txt, a_img = attn[..., :text_len], attn[..., text_len:]
        out_img = torch.einsum('bij,bijd->bid', a_img, vn)
        out_img = out_img + torch.einsum('bij,bjd->bid', a_txt,
                                         v_t.reshape(b * h, text_len, d))
        out_img = out_img.reshape(b, h, img_len, d)

        out = torch.cat((out_text, out_img), dim=-2)
        out = self.to_out(_merge_heads(out))
        return out[:, :n]


def _conv_neigh(t, b, h, S, d, ks, dil, causal_pad):
    """[b,h,S*S,d] -> [b*h, S*S, ks*ks, d] causal dilated neighborhoods."""
    t = t.reshape(b * h, S, S, d).permute(0, 3, 1, 2)
    t = F.pad(t, causal_pad)
    t = F.unfold(t, ks, dilation=dil)
    t = t.reshape(b * h, d, ks * ks, S * S)
    return t.permute(0, 3, 2, 1)


class SparseAttention(Attention):
    """Variable block-sparse causal attention (native CDNA4 replacement for
    the DeepSpeed/Triton kernel stack, reference attention.py:339-398).

    The layout follows the reference's ``VariableSparsityConfig`` call
    (block=16, global blocks = the text prefix, num_random_blocks =
    seq/block/4, unidirectional): per block-row, the local window, the global
    text blocks, the diagonal, and a per-row random set are attended; all
    clipped to the causal triangle. The random choices are drawn once at
    construction from a fixed seed so checkpoints reproduce the layout.
    """

    def __init__(self, *args, block_size=16, text_seq_len=256,
                 num_random_blocks=None, num_local_blocks=4, layout_seed=0, **kwargs):
        super().__init__(*args, **kwargs)
        self.block_size = block_size
        nb = ceil(self.seq_len / block_size)
        if num_random_blocks is None:
            num_random_blocks = self.seq_len // block_size // 4
        n_global = ceil(text_seq_len / block_size)

        g = torch.Generator().manual_seed(layout_seed)
        layout = torch.zeros(nb, nb, dtype=torch.bool)
        layout[:, :n_global] = True                       # global text blocks
        for r in range(nb):
            lo = max(0, (r // num_local_blocks) * num_local_blocks)
            layout[r, lo:r + 1] = True                    # local window (causal)
            layout[r, r] = True                           # diagonal
            if num_random_blocks and r > 0:
                cols = torch.randint(0, r + 1, (num_random_blocks,), generator=g)
                layout[r, cols] = True
        tri = torch.ones(nb, nb, dtype=torch.bool).tril_()
        layout &= tri
        dense = layout.repeat_interleave(block_size, 0).repeat_interleave(block_size, 1)
        dense = dense[: self.seq_len * 2, : self.seq_len * 2]  # generous; sliced at use
        self.register_buffer('block_mask', dense[:self.seq_len + block_size,
                                                 :self.seq_len + block_size],
                             persistent=False)

    def forward(self, x, mask=None, rotary_pos_emb=None, cache=None, cache_key=None):
        b, n, _ = x.shape
        rem = n % self.block_size
        pad = (self.block_size - rem) % self.block_size
        if mask is None:
            mask = torch.ones(b, n, dtype=torch.bool, device=x.device)

        h = self.heads
        qkv = self.to_qkv(x).chunk(3, dim=-1)
        q, k, v = (_split_heads(t, h) for t in qkv)
        if rotary_pos_emb is not None:
            q, k, v = apply_rotary_to_qkv(rotary_pos_emb, (q, k, v))
        if pad:
            # rotary applied pre-pad (the table covers seq_len+1 positions);
            # padded keys are masked out below, padded queries sliced at return
            q, k, v = (F.pad(t, (0, 0, 0, pad)) for t in (q, k, v))
            mask = F.pad(mask, (0, pad), value=False)

        m = n + pad
        static = self.block_mask[:m, :m].to(x.device)
        out = attention_core(q, k, v, self.scale, causal=self.causal,
                             key_mask=mask, static_mask=static)
        out = self.to_out(_merge_heads(out))
        return out[:, :n]
