"""Transformer assembly semantics (wrappers, GEGLU, token shift, LayerScale)."""

import torch
import torch.nn.functional as F

from dalle_pytorch_amd.models.transformer import (
    Transformer, DivideMax, LayerScale, PreShiftToken, FeedForward, GEGLU)
from dalle_pytorch_amd.ops import geglu

torch.manual_seed(0)


def test_geglu_matches_reference_formula():
    x = torch.randn(3, 5, 8, requires_grad=True)
    out = geglu(x)
    a, g = x.chunk(2, dim=-1)
    assert torch.allclose(out, a * F.gelu(g), atol=1e-6)
    out.sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()


def test_divide_max():
    x = torch.randn(2, 3, 4)
    out = DivideMax(dim=-1)(x)
    assert torch.allclose(out, x / x.amax(-1, keepdim=True))


def test_layerscale_init_tiers():
    # depth tiers 0.1 / 1e-5 / 1e-6 (reference transformer.py:77-82)
    for depth, eps in ((1, 0.1), (18, 0.1), (19, 1e-5), (24, 1e-5), (25, 1e-6)):
        ls = LayerScale(8, depth, torch.nn.Identity())
        assert torch.allclose(ls.scale, torch.full((1, 1, 8), eps)), depth


def test_preshift_token_text_and_image():
    """Text: half channels shifted right by one token. Image: quarter from
    row above, quarter from left neighbor (reference transformer.py:165-186)."""
    S = 2
    seq_len = 3 + S * S - 1   # text_len 3
    captured = {}

    class Capture(torch.nn.Module):
        def forward(self, x, **kw):
            captured['x'] = x
            return x

    shift = PreShiftToken(Capture(), image_size=S, seq_len=seq_len)
    x = torch.arange(seq_len * 8, dtype=torch.float32).reshape(1, seq_len, 8)
    shift(x)
    y = captured['x']
    text_len = 3
    # text token 0: first half zeros (shift from nothing), second half original
    assert (y[0, 0, :4] == 0).all() and (y[0, 0, 4:] == x[0, 0, 4:]).all()
    # text token 2: first half = token 1's first half
    assert (y[0, 2, :4] == x[0, 1, :4]).all()
    # image grid position (1,0) (= seq index text_len + S): top quarter comes
    # from grid (0,0) = seq index text_len
    assert (y[0, text_len + S, :2] == x[0, text_len, :2]).all()
    # grid (0,0): no row above, no left -> first half zeros
    assert (y[0, text_len, :4] == 0).all()
    # grid (0,1): left neighbor (0,0) provides the second quarter
    assert (y[0, text_len + 1, 2:4] == x[0, text_len, 2:4]).all()
    # pass-through half untouched everywhere
    assert (y[0, :, 4:] == x[0, :, 4:]).all()


def test_transformer_attention_type_cycling_and_sharing():
    tr = Transformer(dim=32, depth=4, seq_len=18, heads=2, dim_head=16,
                     attn_types=('full', 'axial_row'), image_fmap_size=4,
                     shared_attn_ids=(0, 1, 0, 1), shared_ff_ids=(0, 0, 1, 1))
    x = torch.randn(1, 18, 32)
    out = tr(x)
    assert out.shape == x.shape
    # shared attn: layers 0 and 2 use the same leaf module
    leaf = lambda i: tr.layers.layers[i][0].fn.fn
    assert leaf(0).fn is leaf(2).fn


def test_transformer_static_mask_optimize_for_inference():
    tr = Transformer(dim=32, depth=2, seq_len=18, heads=2, dim_head=16,
                     attn_types=('axial_row', 'axial_col'), image_fmap_size=4,
                     optimize_for_inference=True)
    x = torch.randn(1, 18, 32)
    assert tr(x).shape == x.shape
    m = tr._get_attention_mask('axial_row')
    assert m.shape == (18, 18)
    assert m[:, :3].all()          # text keys always on


def test_feedforward_mult():
    ff = FeedForward(16, mult=4)
    assert ff.net[0].out_features == 128   # dim*mult*2
    assert ff.net[3].in_features == 64     # dim*mult
    x = torch.randn(2, 3, 16)
    assert ff(x).shape == x.shape


def test_sandwich_norm():
    tr = Transformer(dim=16, depth=1, seq_len=8, heads=2, dim_head=8,
                     image_fmap_size=2, sandwich_norm=True, rotary_emb=False)
    assert isinstance(tr.layers.layers[0][0].fn.norm_out, torch.nn.LayerNorm)


def test_add_scaled_matches_eager():
    from dalle_pytorch_amd.ops.fused import add_scaled
    torch.manual_seed(4)
    x = torch.randn(2, 6, 64, requires_grad=True)
    y = torch.randn(2, 6, 64, requires_grad=True)
    g = torch.full((1, 1, 64), 0.1, requires_grad=True)
    out = add_scaled(x, y, g)
    ref = x + y * g
    assert torch.allclose(out, ref, atol=1e-6)
    gx, gy, gg = torch.autograd.grad(out.square().sum(), (x, y, g))
    rx, ry, rg = torch.autograd.grad(ref.square().sum(), (x, y, g))
    assert torch.allclose(gx, rx) and torch.allclose(gy, ry) and torch.allclose(gg, rg)


def test_layerscale_residual_path_equals_plain():
    """LayerScale(residual=x) == x + LayerScale()(x) — the executors rely on
    this equivalence for the fused residual path."""
    from dalle_pytorch_amd.models.transformer import LayerScale
    torch.manual_seed(5)
    ls = LayerScale(32, 3, torch.nn.Linear(32, 32))
    x = torch.randn(2, 4, 32)
    assert torch.allclose(ls(x, residual=x), x + ls(x), atol=1e-6)


def test_shift_full_matches_reference_formulation():
    """PreShiftToken._shift_full vs the reference's pad/chunk/cat chain
    (reference transformer.py:165-186), all shapes incl. partial last row."""
    import torch.nn.functional as F
    from dalle_pytorch_amd.models.transformer import PreShiftToken
    torch.manual_seed(6)
    S, t = 4, 3
    seq_len = t + S * S - 1
    mod = PreShiftToken(torch.nn.Identity(), image_size=S, seq_len=seq_len)

    def reference_shift(x):
        n = x.shape[1]
        padding = seq_len - n + 1
        x_text, x_img = x[:, :t], x[:, t:]
        x_img = F.pad(x_img, (0, 0, 0, padding))
        b = x_img.shape[0]
        x_img = x_img.reshape(b, S, S, -1)
        x_text_shift, x_text_pass = x_text.chunk(2, dim=-1)
        x_text_shift = F.pad(x_text_shift, (0, 0, 1, -1))
        x_text = torch.cat((x_text_shift, x_text_pass), dim=-1)
        xt, xl, *xp = x_img.chunk(4, dim=-1)
        xl = F.pad(xl, (0, 0, 1, -1))
        xt = F.pad(xt, (0, 0, 0, 0, 1, -1))
        x_img = torch.cat((xt, xl, *xp), dim=-1)
        x_img = x_img.reshape(b, -1, x_img.shape[-1])[:, :-padding]
        return torch.cat((x_text, x_img), dim=1)

    for n in (t, t + 1, t + S, seq_len - 1, seq_len):
        x = torch.randn(2, n, 16)
        assert torch.allclose(mod._shift_full(x), reference_shift(x)), n
