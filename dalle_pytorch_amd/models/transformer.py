"""Transformer assembly: wrapper chain, feed-forward, token shift, rotary.

Parity target: reference transformer.py (layer cycling :204-328, wrappers
:38-102, GEGLU/FF :106-122, PreShiftToken :126-200, static axial masks
:333-350). The module *nesting and attribute names* deliberately reproduce
the reference checkpoint key schema so state dicts are
interchangeable; the compute inside routes through the fused CDNA4 ops.
"""

from collections import deque
from collections.abc import Iterable
from functools import partial
from itertools import islice, cycle

import torch
import torch.nn.functional as F
from torch import nn

from dalle_pytorch_amd.models.reversible import ReversibleSequence, SequentialSequence
from dalle_pytorch_amd.models.attention import (
    Attention, SparseAttention, SparseConvCausalAttention, SparseAxialCausalAttention)
from dalle_pytorch_amd.models.positional import build_dalle_rotary_table
from dalle_pytorch_amd.ops import geglu
from dalle_pytorch_amd.ops.fused import (token_shift, token_shift_supported,
                                          layer_norm)


def _as_tuple(val, depth=1):
    return val if isinstance(val, Iterable) else (val,) * depth


class DivideMax(nn.Module):
    """x / max(x) with a detached max (reference transformer.py:29-36)."""

    def __init__(self, dim):
        super().__init__()
        self.dim = dim

    def forward(self, x):
        maxes = x.amax(dim=self.dim, keepdim=True).detach()
        return x / maxes


class NonCached(nn.Module):
    """Rebuilds the full sequence for layers without native cache support,
    then returns only the suffix (reference transformer.py:38-58).

    When no cache_key is supplied (the PreShiftToken wrapper does not thread
    one through, reference transformer.py:153,200 — under which the
    reference's sparse layers would all share ``cache[None]`` and corrupt
    each other during cached decode) a per-instance key is used instead.
    """

    def __init__(self, fn):
        super().__init__()
        self.fn = fn

    def forward(self, x, *, cache=None, cache_key=None, **kwargs):
        n = x.shape[-2]
        if cache is not None:
            if cache_key is None:
                cache_key = ('noncached', id(self))
            if cache_key in cache:
                x = torch.cat([cache[cache_key], x], dim=-2)
            cache[cache_key] = x
        return self.fn(x, **kwargs)[:, -n:]


class CachedAs(nn.Module):
    """Names the cache slot for the wrapped layer (reference transformer.py:60-71)."""

    def __init__(self, cache_key, fn):
        super().__init__()
        self.cache_key = cache_key
        self.fn = fn

    def forward(self, x, *, cache=None, **kwargs):
        return self.fn(x, cache=cache, cache_key=self.cache_key, **kwargs)


class LayerScale(nn.Module):
    """Per-channel learned residual scale, depth-tiered init
    (https://arxiv.org/abs/2103.17239; reference transformer.py:74-88)."""

    def __init__(self, dim, depth, fn):
        super().__init__()
        if depth <= 18:
            init_eps = 0.1
        elif depth <= 24:
            init_eps = 1e-5
        else:
            init_eps = 1e-6
        self.scale = nn.Parameter(torch.full((1, 1, dim), init_eps))
        self.fn = fn

    def forward(self, x, **kwargs):
        out = self.fn(x, **kwargs)
        # multiply in the stream dtype: a fp32 scale would silently promote
        # the bf16 residual stream (2x elementwise traffic downstream)
        return out * self.scale.to(out.dtype)


class PreNorm(nn.Module):
    """Pre-LN (optionally sandwich) around a layer (reference
    transformer.py:92-102); the norm runs through the fused bf16 kernel on
    GPU (ops/fused.py)."""

    def __init__(self, dim, fn, sandwich=False):
        super().__init__()
        self.norm = nn.LayerNorm(dim)
        self.norm_out = nn.LayerNorm(dim) if sandwich else nn.Identity()
        self.fn = fn

    def forward(self, x, **kwargs):
        y = layer_norm(x, self.norm.weight, self.norm.bias, self.norm.eps)
        y = self.fn(y, **kwargs)
        if isinstance(self.norm_out, nn.LayerNorm):
            y = layer_norm(y, self.norm_out.weight, self.norm_out.bias,
                           self.norm_out.eps)
        return y


class GEGLU(nn.Module):
    def forward(self, x):
        return geglu(x)


class FeedForward(nn.Module):
    """Linear -> GEGLU -> Dropout -> Linear (reference transformer.py:111-122)."""

    def __init__(self, dim, dropout=0., mult=4.):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(dim, int(dim * mult * 2)),
            GEGLU(),
            nn.Dropout(dropout),
            nn.Linear(int(dim * mult), dim),
        )

    def forward(self, x, cache=None, cache_key=None):
        return self.net(x)


class PreShiftToken(nn.Module):
    """Token-shift preprocessor (reference transformer.py:126-200).

    Text positions receive half their channels from the previous token;
    image positions receive a quarter from the grid-row above and a quarter
    from the left neighbor. The cached decode path keeps a deque of the last
    image row so single-token steps reproduce the training-time shift.
    """

    def __init__(self, fn, image_size, seq_len):
        super().__init__()
        self.fn = fn
        self.image_size = image_size
        self.seq_len = seq_len
        self.img_seq_len = image_size ** 2
        self.text_len = seq_len - self.img_seq_len + 1

    def forward(self, x, cache=None, cache_key=None, **kwargs):
        seq_len, image_size, text_len = self.seq_len, self.image_size, self.text_len

        if cache is not None and cache_key in cache:
            offset = cache['offset']
            assert offset >= text_len, 'cached inference for text is not supported'
            q = cache[cache_key]
            assert isinstance(q, deque) and len(q) == image_size

            x_top, x_left, *x_pass = x[:, -1].chunk(4, dim=-1)
            q.append((x_top, x_left))
            x_top = q.popleft()[0]
            x_left = q[-2][1]
            if (offset - text_len) % image_size == 0:
                x_left = torch.zeros_like(x_left)
            x = torch.cat((x_top, x_left, *x_pass), dim=-1)
            return self.fn(x[:, None], cache=cache, **kwargs)

        n = x.shape[1]
        padding = seq_len - n + 1
        if n < text_len:
            return self.fn(x, **kwargs)

        if cache is None and token_shift_supported(x):
            # fused gather kernel (no pad/cat/fill chain) — training hot path
            x = token_shift(x, text_len, image_size)
            return self.fn(x, **kwargs)

        x_text, x_img = x[:, :text_len], x[:, text_len:]
        x_img = F.pad(x_img, (0, 0, 0, padding))
        b = x_img.shape[0]
        x_img = x_img.reshape(b, image_size, image_size, -1)

        # text: shift half the channels one token to the right
        x_text_shift, x_text_pass = x_text.chunk(2, dim=-1)
        x_text_shift = F.pad(x_text_shift, (0, 0, 1, -1))
        x_text = torch.cat((x_text_shift, x_text_pass), dim=-1)

        # image: quarter from the row above, quarter from the left neighbor
        x_top, x_left, *x_pass = x_img.chunk(4, dim=-1)
        x_left = F.pad(x_left, (0, 0, 1, -1))
        x_top = F.pad(x_top, (0, 0, 0, 0, 1, -1))
        x_img = torch.cat((x_top, x_left, *x_pass), dim=-1)

        x_img = x_img.reshape(b, -1, x_img.shape[-1])
        x_img = x_img[:, :-padding]
        x = torch.cat((x_text, x_img), dim=1)

        if cache is not None:
            d_top, d_left, *_ = x[:, -1].chunk(4, dim=-1)
            d_top, d_left = torch.zeros_like(d_top), torch.zeros_like(d_left)
            q = deque()
            last_row = x_img[:, -image_size:]
            for _ in range(image_size - last_row.shape[1]):
                q.append((d_top, d_left))
            for i in range(last_row.shape[1]):
                q.append(last_row[:, i].chunk(4, dim=-1)[:2])
            cache[cache_key] = q

        return self.fn(x, cache=cache, **kwargs)


class Transformer(nn.Module):
    """depth x (attn, ff) stack with attention-type cycling, layer sharing,
    LayerScale, optional sandwich norm, token shifting, reversibility, and a
    host-precomputed rotary table (reference transformer.py:204-331)."""

    def __init__(
        self,
        *,
        dim,
        depth,
        seq_len,
        reversible=False,
        causal=True,
        heads=8,
        dim_head=64,
        ff_mult=4,
        attn_dropout=0.,
        ff_dropout=0.,
        attn_types=None,
        image_fmap_size=None,
        sparse_attn=False,
        stable=False,
        sandwich_norm=False,
        shift_tokens=False,
        rotary_emb=True,
        shared_attn_ids=None,
        shared_ff_ids=None,
        optimize_for_inference=False,
    ):
        super().__init__()
        self.seq_len = seq_len
        self.image_fmap_size = image_fmap_size

        attn_types = _as_tuple(attn_types if attn_types is not None else ('full',))
        attn_type_iter = islice(cycle(attn_types), depth)
        sparse_layer = _as_tuple(sparse_attn, depth)
        shared_attn_ids = cycle(shared_attn_ids if shared_attn_ids is not None else range(depth))
        shared_ff_ids = cycle(shared_ff_ids if shared_ff_ids is not None else range(depth))
        shared_attn = {}
        shared_ff = {}

        layers = nn.ModuleList([])
        for ind, _sparse, attn_type, attn_id, ff_id in zip(
                range(depth), sparse_layer, attn_type_iter, shared_attn_ids, shared_ff_ids):
            if attn_type == 'full':
                make_attn = partial(Attention, stable=stable)
            elif attn_type == 'sparse':
                make_attn = SparseAttention
            elif attn_type in ('axial_row', 'axial_col'):
                if optimize_for_inference:
                    make_attn = partial(Attention, stable=stable,
                                        static_mask=self._get_attention_mask(attn_type))
                else:
                    axis = 0 if attn_type == 'axial_row' else 1
                    make_attn = partial(SparseAxialCausalAttention, seq_len=seq_len,
                                        axis=axis, image_size=image_fmap_size, stable=stable)
            elif attn_type == 'conv_like':
                make_attn = partial(SparseConvCausalAttention, seq_len=seq_len,
                                    image_size=image_fmap_size, stable=stable)
            else:
                raise ValueError(f'attention type "{attn_type}" is not valid')

            attn, reused_type = shared_attn.get(attn_id, (None, None))
            if attn is None:
                attn = make_attn(dim, causal=causal, seq_len=seq_len, heads=heads,
                                 dim_head=dim_head, dropout=attn_dropout)
                shared_attn[attn_id] = (attn, attn_type)
            elif attn_type != reused_type:
                raise ValueError(
                    f'attn_types do not match shared_attn_ids (ind={ind}, '
                    f'attn_type="{attn_type}", reused="{reused_type}")')

            ff = shared_ff.get(ff_id)
            if ff is None:
                ff = FeedForward(dim, mult=ff_mult, dropout=ff_dropout)
                shared_ff[ff_id] = ff

            if isinstance(attn, Attention):
                attn = CachedAs(f'attn_{ind}', attn)
            else:
                attn = NonCached(attn)

            if shift_tokens:
                attn = CachedAs(f'preshift_attn_{ind}',
                                PreShiftToken(attn, image_size=image_fmap_size, seq_len=seq_len))
                ff = CachedAs(f'preshift_ff_{ind}',
                              PreShiftToken(ff, image_size=image_fmap_size, seq_len=seq_len))

            layers.append(nn.ModuleList([
                LayerScale(dim, ind + 1, PreNorm(dim, attn, sandwich=sandwich_norm)),
                LayerScale(dim, ind + 1, PreNorm(dim, ff, sandwich=sandwich_norm)),
            ]))

        execute_type = ReversibleSequence if reversible else SequentialSequence
        route_attn = ((True, False),) * depth
        route_all = ((True, True),) * depth
        attn_route_map = {'mask': route_attn, 'rotary_pos_emb': route_attn,
                          'cache': route_all}
        self.layers = execute_type(layers, args_route=attn_route_map)

        pos_emb = None
        if rotary_emb:
            img_seq_len = image_fmap_size ** 2
            text_len = seq_len - img_seq_len + 1
            pos_emb = build_dalle_rotary_table(dim_head, text_len, image_fmap_size)
        self.register_buffer('pos_emb', pos_emb)

    def forward(self, x, **kwargs):
        return self.layers(x, rotary_pos_emb=self.pos_emb, **kwargs)

    def _get_attention_mask(self, attn_type):
        """Dense bool mask reproducing axial attention (reference
        transformer.py:333-350): text keys always visible; image keys only
        within the same grid row (axial_row) or column (axial_col)."""
        img_seq_len = self.image_fmap_size ** 2
        text_len = self.seq_len + 1 - img_seq_len
        S = self.image_fmap_size
        m = torch.zeros(self.seq_len, self.seq_len, dtype=torch.bool)
        m[:, :text_len] = True
        if attn_type == 'axial_row':
            for row in range(S):
                lo = text_len + row * S
                hi = text_len + (row + 1) * S
                m[lo:hi, lo:hi] = True
        elif attn_type == 'axial_col':
            for col in range(S):
                lo = text_len + col
                m[lo::S, lo::S] = True
        else:
            raise ValueError(f'attention type "{attn_type}" cannot be simulated '
                             'with a static mask')
        return m
