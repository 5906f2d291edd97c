"""Positional embeddings: rotary (GPT-J interleaved style) and axial.

Native re-implementations of the two external packages the reference pulls in:
``rotary_embedding_torch`` (used at reference transformer.py:14,304-328 and
attention.py:9,32-35) and ``axial_positional_embedding`` (reference
dalle_pytorch.py:7,389). Semantics verified against the reference survey
(SURVEY.md §2.4 N11/N12, §2.5 K2): rot tables are built once on the host
(per the CDNA4 guide: trig on-device turns memory-bound ops VALU-bound) and
applied to q, k AND v — the reference's quirk (attention.py:35,67) is kept.
"""

from math import pi

import torch
from torch import nn


def rotary_freqs(dim: int, kind: str = 'lang', theta: float = 10000.0,
                 max_freq: float = 10.0) -> torch.Tensor:
    """Base frequency vector, length dim//2."""
    if kind == 'lang':
        return 1.0 / (theta ** (torch.arange(0, dim, 2)[: dim // 2].float() / dim))
    if kind == 'pixel':
        return torch.linspace(1.0, max_freq / 2, dim // 2) * pi
    raise ValueError(f'unknown rotary freq kind {kind!r}')


def rotary_angles(positions: torch.Tensor, freqs: torch.Tensor) -> torch.Tensor:
    """Angle table for given scalar positions: [..., n] x [f] -> [..., n, 2f].

    Each base frequency is duplicated into adjacent slots (interleaved /
    GPT-J convention), matching rotate-half pairing below.
    """
    ang = positions.float().unsqueeze(-1) * freqs  # [..., n, f]
    return ang.repeat_interleave(2, dim=-1)        # [..., n, 2f]


def _rotate_every_two(x: torch.Tensor) -> torch.Tensor:
    # (x0, x1, x2, x3, ...) -> (-x1, x0, -x3, x2, ...)
    x = x.reshape(*x.shape[:-1], -1, 2)
    a, b = x.unbind(dim=-1)
    return torch.stack((-b, a), dim=-1).reshape(*x.shape[:-2], -1)


def apply_rotary(angles: torch.Tensor, t: torch.Tensor) -> torch.Tensor:
    """Rotate the first ``angles.shape[-1]`` channels of ``t``; pass the rest.

    ``angles`` broadcasts against ``t[..., :rot]``. Matches
    rotary_embedding_torch.apply_rotary_emb with start_index=0. cos/sin are
    cast to ``t``'s dtype so a bf16 autocast stream stays bf16 (keeping the
    fused bf16 attention kernel on the hot path).
    """
    rot = angles.shape[-1]
    head, tail = t[..., :rot], t[..., rot:]
    cos = angles.cos().to(t.dtype)
    sin = angles.sin().to(t.dtype)
    head = head * cos + _rotate_every_two(head) * sin
    return torch.cat((head, tail), dim=-1) if tail.shape[-1] else head


def apply_rotary_to_qkv(angles, qkv):
    """Apply the rotary table to each of q, k, v — all three, per the
    reference quirk (attention.py:32-35: v is rotated too). ``angles`` is
    sliced to the query length of the tensors."""
    n = qkv[0].shape[-2]
    a = angles[..., :n, :]
    return tuple(apply_rotary(a, t) for t in qkv)


def build_dalle_rotary_table(dim_head: int, text_len: int, image_fmap_size: int) -> torch.Tensor:
    """The [1, seq_len+1, 3*(2*(dim_head//3//2))] angle table for DALLE.

    Reference construction (transformer.py:304-328): a 1-D 'lang' branch over
    text positions (image positions pinned at 8192), plus a 2-D 'pixel' axial
    branch over the image grid (text positions pinned at -10 on both axes).
    For dim_head=64 this yields 60 rotated channels; the last 4 pass through.
    """
    rot_dim = dim_head // 3
    img_seq_len = image_fmap_size ** 2

    lang = rotary_freqs(rot_dim, 'lang')
    pixel = rotary_freqs(rot_dim, 'pixel')

    # 1-D text branch
    text_1d = rotary_angles(torch.arange(text_len), lang)                    # [T, 2f]
    img_1d = rotary_angles(torch.full((img_seq_len,), 8192.0), lang)         # [I, 2f]
    branch_1d = torch.cat((text_1d, img_1d), dim=0)                          # [T+I, 2f]

    # 2-D axial image branch
    ax = rotary_angles(torch.linspace(-1, 1, steps=image_fmap_size), pixel)  # [S, 2f]
    grid = torch.cat(
        (ax.unsqueeze(1).expand(-1, image_fmap_size, -1),
         ax.unsqueeze(0).expand(image_fmap_size, -1, -1)), dim=-1)           # [S, S, 4f]
    grid = grid.reshape(img_seq_len, -1)                                     # [I, 4f]
    text_ax = rotary_angles(torch.full((text_len,), -10.0), pixel)
    text_ax = torch.cat((text_ax, text_ax), dim=-1)                          # [T, 4f]
    branch_2d = torch.cat((text_ax, grid), dim=0)                            # [T+I, 4f]

    table = torch.cat((branch_1d, branch_2d), dim=-1)                        # [T+I, 6f]
    return table.unsqueeze(0)                                                # [1, T+I, 6f]


class AxialPositionalEmbedding(nn.Module):
    """Learned factorized position embedding over an (h, w) token grid.

    State-dict compatible with lucidrains/axial-positional-embedding as used
    by the reference (dalle_pytorch.py:389): parameters land at
    ``weights.0`` [1, h, 1, dim] and ``weights.1`` [1, 1, w, dim].
    """

    def __init__(self, dim: int, axial_shape):
        super().__init__()
        self.dim = dim
        self.shape = tuple(axial_shape)
        self.max_seq_len = self.shape[0] * self.shape[1]
        h, w = self.shape
        self.weights = nn.ParameterList([
            nn.Parameter(torch.randn(1, h, 1, dim)),
            nn.Parameter(torch.randn(1, 1, w, dim)),
        ])

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b, n = x.shape[0], x.shape[1]
        assert n <= self.max_seq_len, \
            f'sequence length {n} exceeds axial capacity {self.max_seq_len}'
        emb = (self.weights[0] + self.weights[1]).reshape(1, self.max_seq_len, self.dim)
        return emb[:, :n].to(x.dtype)
