"""Attention variants vs independent oracles (reference semantics from
SURVEY.md §2.5 K2-K4/K9-K11)."""

import math

import pytest
import torch

from dalle_pytorch_amd.models.attention import (
    Attention, SparseAxialCausalAttention, SparseConvCausalAttention,
    SparseAttention)
from dalle_pytorch_amd.models.positional import (
    build_dalle_rotary_table, apply_rotary, rotary_freqs, rotary_angles)
from dalle_pytorch_amd.ops.attention import attention_core

torch.manual_seed(0)


def dense_oracle(q, k, v, scale, causal=True, key_mask=None, static_mask=None):
    dots = (q * scale) @ k.transpose(-1, -2)
    big = -torch.finfo(dots.dtype).max
    if key_mask is not None:
        dots = dots.masked_fill(~key_mask[:, None, None, :], big)
    if causal:
        i, j = dots.shape[-2:]
        dots = dots.masked_fill(torch.ones(i, j, dtype=torch.bool).triu_(j - i + 1), big)
    if static_mask is not None:
        dots = dots.masked_fill(~static_mask, big)
    return dots.softmax(-1) @ v


def test_attention_core_matches_oracle():
    q, k, v = torch.randn(3, 2, 4, 10, 16), torch.randn(2, 4, 10, 16), None
    q = torch.randn(2, 4, 10, 16)
    v = torch.randn(2, 4, 10, 16)
    out = attention_core(q, k, v, 0.25, causal=True)
    assert torch.allclose(out, dense_oracle(q, k, v, 0.25), atol=1e-6)


def test_attention_core_masks():
    q = torch.randn(2, 2, 8, 16)
    k = torch.randn(2, 2, 8, 16)
    v = torch.randn(2, 2, 8, 16)
    km = torch.rand(2, 8) > 0.3
    km[:, 0] = True
    sm = torch.rand(8, 8) > 0.3
    sm.fill_diagonal_(True)
    out = attention_core(q, k, v, 0.25, causal=True, key_mask=km, static_mask=sm)
    ref = dense_oracle(q, k, v, 0.25, causal=True, key_mask=km, static_mask=sm)
    assert torch.allclose(out, ref, atol=1e-6)


def test_cached_attention_matches_full():
    attn = Attention(dim=32, seq_len=12, heads=2, dim_head=16).eval()
    x = torch.randn(1, 8, 32)
    full = attn(x)
    cache = {}
    outs = []
    for i in range(8):
        cache['offset'] = i
        outs.append(attn(x[:, i:i + 1], cache=cache, cache_key='a'))
    step = torch.cat(outs, dim=1)
    assert torch.allclose(full, step, atol=1e-5)


def test_rotary_table_shape_and_values():
    # dim_head=64: 60 rotated channels
    t = build_dalle_rotary_table(64, 257, 32)
    assert t.shape == (1, 257 + 1024, 60)
    # text 1-D branch at position p rotates with lang freqs
    lang = rotary_freqs(64 // 3, 'lang')
    ang5 = rotary_angles(torch.tensor([5.0]), lang)[0]
    assert torch.allclose(t[0, 5, :20], ang5)
    # image rows all share the pinned 8192 position in the 1-D branch
    ang_far = rotary_angles(torch.tensor([8192.0]), lang)[0]
    assert torch.allclose(t[0, 300, :20], ang_far)


def test_apply_rotary_rotates_pairs():
    ang = torch.tensor([[math.pi / 2, math.pi / 2]])  # one pair, 90 degrees
    x = torch.tensor([[1.0, 2.0, 7.0]])
    out = apply_rotary(ang, x)
    # (x0,x1) -> (x0 c - x1 s, x1 c + x0 s) = (-2, 1); x2 passes through
    assert torch.allclose(out, torch.tensor([[-2.0, 1.0, 7.0]]), atol=1e-6)


def test_axial_attention_matches_static_mask_dense():
    """SparseAxialCausalAttention == dense attention under the axial static
    mask (the reference's optimize_for_inference equivalence,
    transformer.py:252-260)."""
    torch.manual_seed(1)
    S, text_len = 4, 3
    seq_len = text_len + S * S - 1   # 18
    for axis in (0, 1):
        sparse = SparseAxialCausalAttention(
            dim=32, seq_len=seq_len, image_size=S, axis=axis, heads=2,
            dim_head=16)
        # dense mask replicating transformer._get_attention_mask
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
        dense = Attention(dim=32, seq_len=seq_len, heads=2, dim_head=16,
                          causal=True, static_mask=m)
        dense.to_qkv.weight.data = sparse.to_qkv.weight.data.clone()
        dense.to_out[0].weight.data = sparse.to_out[0].weight.data.clone()
        dense.to_out[0].bias.data = sparse.to_out[0].bias.data.clone()

        x = torch.randn(2, seq_len, 32)
        assert torch.allclose(sparse(x), dense(x), atol=1e-5), f'axis={axis}'


def test_axial_lse_merge_gradients_match_masked_path(monkeypatch):
    """The decomposed (lse-merge) axial forward must produce the same
    gradients as the masked-dense formulation — exercises the logsumexp
    gradient path through attention_core's backward (Dv - grad_lse)."""
    torch.manual_seed(2)
    S, text_len = 4, 3
    seq_len = text_len + S * S - 1
    for axis in (0, 1):
        sparse = SparseAxialCausalAttention(
            dim=32, seq_len=seq_len, image_size=S, axis=axis, heads=2,
            dim_head=16)
        x = torch.randn(2, seq_len, 32, requires_grad=True)
        mask = torch.ones(2, text_len, dtype=torch.bool)
        mask[1, -1] = False

        monkeypatch.setenv('DALLE_AMD_AXIAL_MASKED', '1')
        ref = sparse(x, mask=mask)
        gref = torch.autograd.grad(ref.square().sum(), (x,) + tuple(
            sparse.parameters()), retain_graph=False)

        monkeypatch.setenv('DALLE_AMD_AXIAL_MASKED', '0')
        out = sparse(x, mask=mask)
        gnew = torch.autograd.grad(out.square().sum(), (x,) + tuple(
            sparse.parameters()))

        assert torch.allclose(out, ref, atol=1e-5), f'axis={axis}'
        for a, b_ in zip(gnew, gref):
            assert torch.allclose(a, b_, atol=1e-4), f'axis={axis}'


def test_conv_attention_matches_dense_mask():
    """conv_like == dense attention under the unfolded-neighborhood mask."""
    torch.manual_seed(2)
    S, text_len, ks = 4, 3, 3
    seq_len = text_len + S * S - 1
    sparse = SparseConvCausalAttention(dim=32, seq_len=seq_len, image_size=S,
                                       kernel_size=ks, heads=2, dim_head=16)
    # build the dense mask: text causal; image attends text fully + its
    # causally-padded ks x ks neighborhood
    n = seq_len + 1
    m = torch.zeros(n, n, dtype=torch.bool)
    tri = torch.ones(text_len, text_len, dtype=torch.bool).tril_()
    m[:text_len, :text_len] = tri
    m[text_len:, :text_len] = True
    half = ks // 2
    for qi in range(S * S):
        qr, qc = divmod(qi, S)
        for di in range(ks):
            for dj in range(ks):
                kr, kc = qr - 2 * half + di, qc - 2 * half + dj
                if 0 <= kr < S and 0 <= kc < S:
                    m[text_len + qi, text_len + kr * S + kc] = True

    x = torch.randn(2, seq_len, 32)
    h, dh, scale = 2, 16, 16 ** -0.5
    xp = torch.nn.functional.pad(x, (0, 0, 0, 1))
    qkv = sparse.to_qkv(xp).chunk(3, -1)
    q, k, v = (t.reshape(2, n, h, dh).permute(0, 2, 1, 3) for t in qkv)
    ref = dense_oracle(q, k, v, scale, causal=False, static_mask=m)
    ref = ref.permute(0, 2, 1, 3).reshape(2, n, h * dh)
    ref = sparse.to_out(ref)[:, :seq_len]
    assert torch.allclose(sparse(x), ref, atol=1e-5)


def test_block_sparse_layout_properties():
    attn = SparseAttention(32, 64, heads=2, dim_head=16, block_size=8,
                           text_seq_len=16, num_random_blocks=1)
    x = torch.randn(1, 64, 32)
    out = attn(x)
    assert out.shape == (1, 64, 32)
    bm = attn.block_mask
    nb = 64 // 8
    blocks = bm[:64, :64].reshape(nb, 8, nb, 8).any(dim=(1, 3))
    assert blocks.diagonal().all()                 # diagonal present
    assert not blocks.triu(1).any()                # strictly causal
    for r in range(nb):                            # global text blocks (causal-clipped)
        assert blocks[r, :min(r + 1, 2)].all()


def test_rotary_applied_to_v_quirk():
    """The reference rotates v as well as q/k (attention.py:35,67); outputs
    must differ from a no-rotary run even when q/k rotation cancels."""
    attn = Attention(dim=32, seq_len=8, heads=2, dim_head=16).eval()
    x = torch.randn(1, 8, 32)
    table = build_dalle_rotary_table(16, 4, 2)[..., :8, :]
    out_rot = attn(x, rotary_pos_emb=table)
    out_plain = attn(x)
    assert not torch.allclose(out_rot, out_plain, atol=1e-4)


def _dense_attn_with_lse(q, k, v, scale, causal):
    """Plain attention returning (out_unnormalized_by_partition, lse):
    out = softmax(qk^T)v and lse = logsumexp of the scores row."""
    s = torch.matmul(q * scale, k.transpose(-1, -2))
    if causal:
        i, j = s.shape[-2:]
        cm = torch.ones(i, j, dtype=torch.bool).triu_(j - i + 1)
        s = s.masked_fill(cm, float('-inf'))
    lse = torch.logsumexp(s, dim=-1)
    p = torch.exp(s - lse.unsqueeze(-1))
    return torch.matmul(p, v), lse


def test_axial_attention_lse_decomposition():
    """Axial attention == two DENSE flash-style attentions merged by lse:
    (a) image queries over the text prefix (non-causal: every text key is
    allowed), (b) image queries over their own grid row/col, causal.
    Pins the round-2 decomposition (NOTES_ROUND2.md, ladder option 3) that
    removes masks and partial tiles entirely."""
    torch.manual_seed(4)
    S, t = 4, 3
    n_img_full = S * S
    for axis in (0, 1):
        for n_img in (n_img_full, n_img_full - 1):
            n = t + n_img
            b, h, d = 2, 2, 16
            scale = d ** -0.5
            q = torch.randn(b, h, n, d)
            k = torch.randn(b, h, n, d)
            v = torch.randn(b, h, n, d)

            # oracle: dense attention under the axial mask + causality
            from dalle_pytorch_amd.models.attention import axial_mask
            m = axial_mask(n, t, S, axis)
            m &= torch.ones(n, n, dtype=torch.bool).tril_()
            s = torch.matmul(q * scale, k.transpose(-1, -2))
            s = s.masked_fill(~m, -torch.finfo(s.dtype).max)
            ref = torch.matmul(s.softmax(-1), v)

            # text queries: plain causal over the text prefix
            out_text, _ = _dense_attn_with_lse(
                q[:, :, :t], k[:, :, :t], v[:, :, :t], scale, causal=True)

            # image queries, part 1: all text keys (non-causal)
            qi = q[:, :, t:]
            o1, l1 = _dense_attn_with_lse(qi, k[:, :, :t], v[:, :, :t],
                                          scale, causal=False)

            # image queries, part 2: row-(or col-)local causal attention.
            # pad the grid to S*S (the pad token is a key only for queries
            # after it — none — and is sliced off as a query)
            pad = n_img_full - n_img
            def grid(z):
                zi = torch.nn.functional.pad(z[:, :, t:], (0, 0, 0, pad))
                g = zi.reshape(b, h, S, S, d)
                if axis == 1:
                    g = g.transpose(2, 3)
                return g.reshape(b, h * S, S, d)
            qg, kg, vg = grid(q), grid(k), grid(v)
            o2g, l2g = _dense_attn_with_lse(qg, kg, vg, scale, causal=True)
            def ungrid(z, last):
                g = z.reshape(b, h, S, S, *z.shape[3:])
                if axis == 1:
                    g = g.transpose(2, 3)
                return g.reshape(b, h, n_img_full, *z.shape[3:])[:, :, :last]
            o2 = ungrid(o2g, n_img)
            l2 = ungrid(l2g.unsqueeze(-1), n_img).squeeze(-1)

            # lse merge
            mx = torch.maximum(l1, l2)
            w1 = torch.exp(l1 - mx).unsqueeze(-1)
            w2 = torch.exp(l2 - mx).unsqueeze(-1)
            out_img = (o1 * w1 + o2 * w2) / (w1 + w2)

            out = torch.cat((out_text, out_img), dim=2)
            assert torch.allclose(out, ref, atol=1e-5), \
                (axis, n_img, (out - ref).abs().max())
