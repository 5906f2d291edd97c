"""Attention variants for the DALL-E transformer stack.

Four interchangeable modules with the same contract as the reference
(attention.py:39-398): full causal, axial row/col sparse, conv-like sparse,
and block-sparse. All of them run on ONE fused CDNA4 flash kernel
(:func:`dalle_pytorch_amd.ops.attention_core`): the sparse variants express
their pattern as a cached static mask plus a (64x32)-tile block map the
kernel uses to skip fully-masked K/V tiles — so "sparse" means skipped MFMA
tiles, not eager gather/scatter composition. The QKV projection output is
split + rotary-rotated in a single fused sweep (ops/rope.py), and the kernel
epilogue emits the [b, n, h*d] layout the output projection consumes.

Semantics notes carried over from the reference (kept bit-for-bit):
* masked scores are filled so masked keys get zero probability,
* ``stable_softmax`` (attention.py:27-30) == plain max-subtracted softmax in
  fwd and bwd, so ``stable`` changes nothing inside the fused core,
* rotary tables rotate q, k AND v (attention.py:35,67),
* the k/v inference cache concatenates along the sequence dim and causality
  is implicit once ``offset > 0`` (attention.py:71-76,86),
* axial semantics == the reference's static-mask formulation, which it
  proves equivalent to the gather-based form (transformer.py:252-260);
  equality is pinned by tests/test_attention.py against independent oracles.
"""

import os
from math import ceil

import torch
from torch import nn

from dalle_pytorch_amd.models.positional import apply_rotary_to_qkv
from dalle_pytorch_amd.ops import attention_core, axial_attention
from dalle_pytorch_amd.ops.fp8 import fp8_linear
from dalle_pytorch_amd.ops.attention import build_tile_map
from dalle_pytorch_amd.ops.rope import rope_split, rope_split_supported, trig_tables


def _split_heads(t, h):
    b, n, hd = t.shape
    return t.reshape(b, n, h, hd // h).permute(0, 2, 1, 3)


def _qkv_heads(x, to_qkv, h, dim_head, rotary_pos_emb, offset=0):
    """Project to q/k/v [b,h,n,d] with rotary applied to q,k,v — fused on
    GPU (rope_split), eager elsewhere. rotary_pos_emb is the angle table
    [1, N, rot], consumed from position ``offset``."""
    qkv = fp8_linear(to_qkv, x)
    n = x.shape[1]
    if rope_split_supported(qkv, dim_head):
        cos = sin = None
        if rotary_pos_emb is not None:
            cos_f, sin_f = trig_tables(rotary_pos_emb)
            cos = cos_f[offset:offset + n]
            sin = sin_f[offset:offset + n]
        return rope_split(qkv, h, cos, sin)
    q, k, v = (t.reshape(x.shape[0], n, h, dim_head).permute(0, 2, 1, 3)
               for t in qkv.chunk(3, dim=-1))
    if rotary_pos_emb is not None:
        q, k, v = apply_rotary_to_qkv(
            rotary_pos_emb[..., offset:offset + n, :], (q, k, v))
    return q, k, v


# --------------------------------------------------------------------- masks

_MASK_CACHE = {}


def _cached_mask(key, device, build):
    """(mask bool, tiles uint8 [nq/64,nk/32], tiles_t uint8 [nk/64,nq/32])
    on device, built once per pattern. tiles serves the forward + dQ kernels;
    tiles_t the dK/dV kernel (keys-major blocking)."""
    full_key = (*key, str(device))
    hit = _MASK_CACHE.get(full_key)
    if hit is None:
        mask = build().to(device)
        hit = (mask, build_tile_map(mask), build_tile_map(mask.t()))
        _MASK_CACHE[full_key] = hit
    return hit


def axial_mask(seq_len, text_len, image_size, axis):
    """Text keys always on; image keys within the same grid row (axis 0) or
    column (axis 1). Combined with the kernel's causal mask this reproduces
    SparseAxialCausalAttention exactly (reference attention.py:225-335 ==
    transformer.py:333-350)."""
    S = image_size
    m = torch.zeros(seq_len, seq_len, dtype=torch.bool)
    m[:, :text_len] = True
    if axis == 0:
        for row in range(S):
            lo = text_len + row * S
            m[lo:lo + S, lo:lo + S] = True
    else:
        for col in range(S):
            lo = text_len + col
            m[lo::S, lo::S] = True
    return m


def conv_mask(seq_len, text_len, image_size, kernel_size, dilation):
    """Text keys always on; image query (r, c) sees the causally-padded
    kernel_size^2 dilated window ending at (r, c) (reference
    attention.py:166-191: pad by 2*half top/left then unfold)."""
    S = image_size
    m = torch.zeros(seq_len, seq_len, dtype=torch.bool)
    m[:, :text_len] = True
    half = ((kernel_size - 1) * dilation + 1) // 2
    for qi in range(S * S):
        qr, qc = divmod(qi, S)
        q_idx = text_len + qi
        if q_idx >= seq_len:
            break
        for di in range(kernel_size):
            for dj in range(kernel_size):
                kr = qr - 2 * half + di * dilation
                kc = qc - 2 * half + dj * dilation
                if 0 <= kr < S and 0 <= kc < S:
                    k_idx = text_len + kr * S + kc
                    if k_idx < seq_len:
                        m[q_idx, k_idx] = True
    return m


def block_sparse_mask(seq_len, block_size, text_seq_len, num_random_blocks,
                      num_local_blocks, layout_seed):
    """Variable block-sparse layout (native replacement for the DeepSpeed
    VariableSparsityConfig call, reference attention.py:352-365): per block
    row — the local window, the global text blocks, the diagonal, and a
    seeded random set; clipped causal."""
    nb = ceil(seq_len / block_size)
    n_global = ceil(text_seq_len / block_size)
    g = torch.Generator().manual_seed(layout_seed)
    layout = torch.zeros(nb, nb, dtype=torch.bool)
    layout[:, :n_global] = True
    for r in range(nb):
        lo = max(0, (r // num_local_blocks) * num_local_blocks)
        layout[r, lo:r + 1] = True
        layout[r, r] = True
        if num_random_blocks and r > 0:
            cols = torch.randint(0, r + 1, (num_random_blocks,), generator=g)
            layout[r, cols] = True
    layout &= torch.ones(nb, nb, dtype=torch.bool).tril_()
    dense = layout.repeat_interleave(block_size, 0).repeat_interleave(block_size, 1)
    return dense


# ------------------------------------------------------------------- modules

class Attention(nn.Module):
    """Full (optionally causal) multi-head self-attention with k/v cache.

    Parity target: reference attention.py:39-99. ``static_mask`` lets this
    class also serve the cache-friendly simulation of axial attention
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

    def _masks(self, offset, n_q, n_k, device):
        if self.static_mask is None:
            return None, None, None
        sm = self.static_mask[offset:offset + n_q, :n_k]
        tiles = tiles_t = None
        if sm.is_cuda and offset == 0 and n_k == self.static_mask.shape[1]:
            hit = getattr(self.static_mask, '_dalle_amd_tiles', None)
            if hit is None or hit[0] != (n_q, n_k):
                hit = ((n_q, n_k), build_tile_map(sm),
                       build_tile_map(sm.t()))
                self.static_mask._dalle_amd_tiles = hit
            _, tiles, tiles_t = hit
        return sm, tiles, tiles_t

    def forward(self, x, mask=None, rotary_pos_emb=None, cache=None, cache_key=None):
        h = self.heads
        offset = cache.get('offset', 0) if cache is not None else 0
        q, k, v = _qkv_heads(x, self.to_qkv, h, self.dim_head,
                             rotary_pos_emb, offset)

        if offset > 0:
            k_prev, v_prev = cache[cache_key]
            k = torch.cat((k_prev, k), dim=-2)
            v = torch.cat((v_prev, v), dim=-2)
        if cache is not None:
            cache[cache_key] = (k, v)

        n_q, n_k = q.shape[-2], k.shape[-2]
        static, tiles, tiles_t = self._masks(offset, n_q, n_k, x.device)
        out = attention_core(
            q, k, v, self.scale,
            causal=self.causal and offset == 0,
            key_mask=mask, static_mask=static, static_tiles=tiles,
            static_tiles_t=tiles_t, fold_heads=True)
        return self.to_out[1](fp8_linear(self.to_out[0], out))


class _StaticMaskSparseAttention(nn.Module):
    """Shared machinery for the pattern-masked variants: one fused kernel
    call over the full sequence with a cached pattern mask + tile map. The
    kernel skips fully-masked (64q, 32k) tiles, so compute scales with the
    pattern's live area, not n^2."""

    def __init__(self, dim, seq_len, heads, dim_head, dropout, stable, causal=True):
        super().__init__()
        self.heads = heads
        self.dim_head = dim_head
        self.seq_len = seq_len
        self.scale = dim_head ** -0.5
        self.causal = causal
        self.stable = stable
        inner = heads * dim_head
        self.to_qkv = nn.Linear(dim, inner * 3, bias=False)
        self.to_out = nn.Sequential(nn.Linear(inner, dim), nn.Dropout(dropout))

    # subclasses define: _mask_key(), _build_mask() -> [seq_len+1?, ...] bool
    def _pattern(self, n, device):
        mask, tiles, tiles_t = _cached_mask(self._mask_key(), device,
                                            self._build_mask)
        if n == mask.shape[0]:
            return mask, tiles, tiles_t
        sm = mask[:n, :n]
        _, t, tt = _cached_mask((*self._mask_key(), 'n', n), device, lambda: sm)
        return sm, t, tt

    def _key_mask(self, mask, b, n, text_len, device):
        if mask is None:
            return None
        km = torch.ones(b, n, dtype=torch.bool, device=device)
        t = min(text_len, mask.shape[1], n)
        km[:, :t] = mask[:, :t]
        return km

    def forward(self, x, mask=None, rotary_pos_emb=None):
        b, n, _ = x.shape
        q, k, v = _qkv_heads(x, self.to_qkv, self.heads, self.dim_head,
                             rotary_pos_emb, 0)
        static, tiles, tiles_t = self._pattern(n, x.device)
        km = self._key_mask(mask, b, n, getattr(self, 'text_len', n), x.device)
        out = attention_core(q, k, v, self.scale, causal=self.causal,
                             key_mask=km, static_mask=static,
                             static_tiles=tiles, static_tiles_t=tiles_t,
                             fold_heads=True)
        return self.to_out[1](fp8_linear(self.to_out[0], out))


class SparseAxialCausalAttention(_StaticMaskSparseAttention):
    """Axial (row or column) causal attention over the image grid; the text
    prefix gets full causal attention and every image token attends to all
    text. Parity target: reference attention.py:225-335 (verified equal to
    the static-mask form by tests/test_attention.py)."""

    def __init__(self, dim, seq_len, image_size=32, axis=0, heads=8, dim_head=64,
                 dropout=0., stable=False, causal=True, **kwargs):
        super().__init__(dim, seq_len, heads, dim_head, dropout, stable, causal)
        assert axis in (0, 1)
        self.axis = axis
        self.image_size = image_size
        self.text_len = seq_len - image_size ** 2 + 1

    def _mask_key(self):
        return ('axial', self.seq_len, self.image_size, self.axis)

    def _build_mask(self):
        return axial_mask(self.seq_len, self.text_len, self.image_size, self.axis)

    # NOTE (measured, round 1): re-ordering the image tokens column-major so
    # the axis=1 pattern tiles contiguously was tried in two forms —
    # index_select (backward scatter-add atomics) and grid-transpose copies.
    # Both REGRESSED the flagship step (-3% / -5%): three extra tensor
    # passes per layer-visit cost more than the skipped partial tiles save.
    # Round 2 replaces the masked formulation entirely with the lse-merge
    # decomposition below (NOTES_ROUND2.md ladder option 3, pinned by
    # tests/test_attention.py::test_axial_attention_lse_decomposition).

    def forward(self, x, mask=None, rotary_pos_emb=None):
        """Axial attention without pattern masks: on GPU, ONE fused kernel
        call in axial mode (liveness evaluated arithmetically, virtual
        column-major coordinates for axis 1 — no transposes, no mask/tile-map
        traffic, no partial tiles beyond the grid lines themselves).
        Elsewhere, the lse-merge decomposition into three dense attentions
        (ops.attention.axial_attention). Set DALLE_AMD_AXIAL_MASKED=1 to
        force the round-1 masked-dense path for A/B comparison."""
        t = self.text_len
        b, n, _ = x.shape
        if (n <= t or not self.causal
                or os.environ.get('DALLE_AMD_AXIAL_MASKED', '0') == '1'):
            return super().forward(x, mask=mask, rotary_pos_emb=rotary_pos_emb)

        q, k, v = _qkv_heads(x, self.to_qkv, self.heads, self.dim_head,
                             rotary_pos_emb, 0)
        km = self._key_mask(mask, b, n, t, x.device)
        out = axial_attention(q, k, v, self.scale, t, self.image_size,
                              self.axis, key_mask=km)
        return self.to_out[1](fp8_linear(self.to_out[0], out))


class SparseConvCausalAttention(_StaticMaskSparseAttention):
    """Conv-like sparse attention: each image token attends to a causally
    padded k x k dilated neighborhood plus all text. Parity target:
    reference attention.py:103-221 (verified against an unfold-based oracle
    in tests/test_attention.py)."""

    def __init__(self, dim, seq_len, image_size=32, kernel_size=5, dilation=1,
                 heads=8, dim_head=64, dropout=0., stable=False, causal=True,
                 **kwargs):
        assert kernel_size % 2 == 1, 'kernel size must be odd'
        super().__init__(dim, seq_len, heads, dim_head, dropout, stable, causal)
        self.image_size = image_size
        self.kernel_size = kernel_size
        self.dilation = dilation
        self.text_len = seq_len - image_size ** 2 + 1

    def _mask_key(self):
        return ('conv', self.seq_len, self.image_size, self.kernel_size,
                self.dilation)

    def _build_mask(self):
        return conv_mask(self.seq_len, self.text_len, self.image_size,
                         self.kernel_size, self.dilation)


class SparseAttention(_StaticMaskSparseAttention):
    """Variable block-sparse causal attention (native CDNA4 replacement for
    the DeepSpeed/Triton kernel stack, reference attention.py:339-398).

    Layout parameters mirror the reference's VariableSparsityConfig call
    (block=16, global blocks = the text prefix, num_random_blocks =
    seq/block/4, unidirectional); random choices are drawn once from a fixed
    seed so the layout is reproducible across ranks and resumes.

    NOTE: the random-block choices use this module's own seeded generator,
    NOT DeepSpeed's — a model trained with DeepSpeed's sparse attention has
    a different (framework-internal) random layout, so such checkpoints
    load but attend through a different random-block pattern. This is a
    deliberate native redesign, not an oversight.
    """

    def __init__(self, dim, seq_len, causal=True, heads=8, dim_head=64,
                 dropout=0., stable=False, static_mask=None, block_size=16,
                 text_seq_len=256, num_random_blocks=None, num_local_blocks=4,
                 layout_seed=0, **kwargs):
        super().__init__(dim, seq_len, heads, dim_head, dropout, stable, causal)
        self.block_size = block_size
        self.text_seq_len = text_seq_len
        if num_random_blocks is None:
            num_random_blocks = seq_len // block_size // 4
        self.num_random_blocks = num_random_blocks
        self.num_local_blocks = num_local_blocks
        self.layout_seed = layout_seed
        self.text_len = text_seq_len + 1

    def _mask_key(self):
        return ('bsparse', self.seq_len, self.block_size, self.text_seq_len,
                self.num_random_blocks, self.num_local_blocks, self.layout_seed)

    def _build_mask(self):
        n = ceil((self.seq_len + 1) / self.block_size) * self.block_size
        return block_sparse_mask(n, self.block_size, self.text_seq_len,
                                 self.num_random_blocks, self.num_local_blocks,
                                 self.layout_seed)[:self.seq_len + 1,
                                                   :self.seq_len + 1]

    @property
    def block_mask(self):
        return self._build_mask()

    def forward(self, x, mask=None, rotary_pos_emb=None, cache=None, cache_key=None):
        return super().forward(x, mask=mask, rotary_pos_emb=rotary_pos_emb)
