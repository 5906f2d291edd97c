"""FastDecoder parity: engine generation must equal the model's dict-cache
generation (which is itself pinned bitwise-equal to uncached recompute)."""

import pytest
import torch

from dalle_pytorch_amd import DALLE, DiscreteVAE
from dalle_pytorch_amd.engine import FastDecoder

torch.manual_seed(0)


def tiny_dalle(**kw):
    vae = DiscreteVAE(image_size=64, num_layers=3, num_tokens=64,
                      codebook_dim=32, hidden_dim=8)
    args = dict(dim=32, num_text_tokens=50, text_seq_len=8, depth=2, heads=2,
                dim_head=16, attn_types=('full',), shift_tokens=True)
    args.update(kw)
    return DALLE(vae=vae, **args)


@pytest.mark.parametrize('kw', [
    dict(attn_types=('full',)),
    dict(attn_types=('full',), shift_tokens=False),
    dict(attn_types=('axial_row', 'axial_col')),
    dict(attn_types=('conv_like',)),
    dict(attn_types=('sparse',)),
    dict(attn_types=('full',), stable=True),
    dict(attn_types=('full', 'axial_row'), reversible=True),
    dict(attn_types=('full',), rotary_emb=False),
])
def test_engine_matches_model_generation(kw):
    torch.manual_seed(3)
    d = tiny_dalle(**kw).eval()
    text = torch.randint(1, 50, (2, 8))

    torch.manual_seed(11)
    ref = d.generate_images(text, use_cache=True, temperature=1e-8,
                            filter_thres=0.99)
    dec = FastDecoder(d, batch_size=2)
    torch.manual_seed(11)
    got = dec.generate(text, temperature=1e-8, filter_thres=0.99)
    assert torch.allclose(got, ref, atol=1e-4), \
        (got - ref).abs().max().item()


def test_engine_guided_matches_model_guided():
    """cond_scale != 1 on the fast path: the doubled-batch (cond + null)
    stream must reproduce the model's guided generation (which is itself
    pinned cached == uncached by test_dalle.py)."""
    torch.manual_seed(6)
    d = tiny_dalle(attn_types=('full', 'axial_row')).eval()
    text = torch.randint(1, 50, (2, 8))

    torch.manual_seed(23)
    ref = d.generate_images(text, use_cache=True, temperature=1e-8,
                            filter_thres=0.99, cond_scale=2.0)
    dec = FastDecoder(d, batch_size=4)   # 2*b for the guided stream pair
    torch.manual_seed(23)
    got = dec.generate(text, temperature=1e-8, filter_thres=0.99,
                       cond_scale=2.0)
    assert got.shape == ref.shape
    assert torch.allclose(got, ref, atol=1e-4), \
        (got - ref).abs().max().item()


def test_engine_step_logits_match_model_cache():
    """Per-step logits comparison (tighter than end-to-end argmax parity)."""
    torch.manual_seed(4)
    d = tiny_dalle(attn_types=('axial_row', 'axial_col')).eval()
    text = torch.randint(1, 50, (1, 8))

    cache = {}
    with torch.no_grad():
        ref_logits = d(text, None, cache=cache)[:, -1]
    dec = FastDecoder(d, batch_size=1)
    with torch.no_grad():
        got_logits = dec.prefill(text)
    valid = ref_logits > -1e30
    assert torch.allclose(got_logits[valid], ref_logits[valid], atol=1e-4)

    token = torch.tensor([5])
    with torch.no_grad():
        step_ref = d(text, token.unsqueeze(0), cache=cache)[:, -1]
        step_got = dec.step(token)
    valid = step_ref > -1e30
    assert torch.allclose(step_got[valid], step_ref[valid], atol=1e-4), \
        (step_got[valid] - step_ref[valid]).abs().max().item()


@pytest.mark.gpu
def test_engine_gpu_graph_mode():
    torch.manual_seed(5)
    d = tiny_dalle(attn_types=('axial_row', 'axial_col'), depth=2).cuda().eval()
    text = torch.randint(1, 50, (2, 8), device='cuda')
    torch.manual_seed(7)
    ref = d.generate_images(text, use_cache=True, temperature=1e-8,
                            filter_thres=0.99)
    dec = FastDecoder(d, batch_size=2, dtype=torch.float32, use_graph=True)
    torch.manual_seed(7)
    got = dec.generate(text, temperature=1e-8, filter_thres=0.99)
    assert torch.allclose(got, ref, atol=1e-3), (got - ref).abs().max().item()


@pytest.mark.gpu
def test_fused_decode_kernels_match_torch_path():
    """bf16 fused fa_decode/shift_decode vs the torch-op decode path."""
    torch.manual_seed(6)
    d = tiny_dalle(attn_types=('axial_row', 'full'), depth=2, dim=256,
                   heads=4, dim_head=64).cuda().eval()
    text = torch.randint(1, 50, (2, 8), device='cuda')
    token = torch.randint(0, 64, (2,), device='cuda')

    dec_f = FastDecoder(d, batch_size=2, dtype=torch.bfloat16)
    assert dec_f._fused_decode
    dec_t = FastDecoder(d, batch_size=2, dtype=torch.bfloat16)
    dec_t._fused_decode = False

    with torch.no_grad():
        a = dec_f.prefill(text)
        b = dec_t.prefill(text)
        for _ in range(5):
            a = dec_f.step(token)
            b = dec_t.step(token)
    valid = b > -1e30
    rel = (a[valid].float() - b[valid].float()).abs().max() / \
        b[valid].float().abs().max()
    assert rel < 5e-2, rel.item()


def test_sk2_pack_layout():
    """The sk2 packed-weight layout must match the kernel's index math:
    element (lane, e) of chunk (nt, kc) is w[nt*16 + (lane&15),
    kc*32 + (lane>>4)*8 + e] (hip_ops.hip sk2_kernel). Pure-CPU check of
    the reshape/permute formula used by FastDecoder._sk2_pack."""
    N, K = 64, 1024
    w = torch.arange(N * K, dtype=torch.float32).reshape(N, K)
    pk = (w.reshape(N // 16, 16, K // 32, 4, 8)
           .permute(0, 2, 3, 1, 4).contiguous())    # [nt][kc][kg][col][e]
    flat = pk.reshape(N // 16, K // 32, 64, 8)      # lane = kg*16 + col
    for nt, kc, lane, e in [(0, 0, 0, 0), (1, 3, 17, 5), (3, 31, 63, 7),
                            (2, 10, 48, 2)]:
        col, kg = lane & 15, lane >> 4
        assert flat[nt, kc, lane, e] == w[nt * 16 + col, kc * 32 + kg * 8 + e]


def test_sk2_pack_gates():
    """_sk2_pack returns None for shapes the kernel contract rejects
    (K % 1024, N % 32) and for non-bf16/non-cuda tensors (CPU here)."""
    from dalle_pytorch_amd.engine.decode import FastDecoder
    assert FastDecoder._sk2_pack(torch.zeros(64, 512)) is None     # K % 1024
    assert FastDecoder._sk2_pack(torch.zeros(48, 1024)) is None    # N % 32
    assert FastDecoder._sk2_pack(torch.zeros(64, 1024)) is None    # cpu/fp32


@pytest.mark.gpu
def test_sk2_decode_path_matches_unpacked():
    """In-situ sk2 coverage: at batch 16 and dim 1024 every sk2 call site
    engages (qkv/out mode 0, geglu-fused ff1 mode 1, fp32 head mode 2).
    Parity against the same engine with sk2 disabled."""
    torch.manual_seed(9)
    d = tiny_dalle(attn_types=('axial_row', 'axial_col'), depth=2, dim=1024,
                   heads=16, dim_head=64).cuda().eval()
    text = torch.randint(1, 50, (16, 8), device='cuda')
    token = torch.randint(0, 64, (16,), device='cuda')

    dec_s = FastDecoder(d, batch_size=16, dtype=torch.bfloat16)
    assert dec_s._fused_decode and dec_s._sk2_on
    assert dec_s.states[0].w.get('qkv_pk') is not None
    assert dec_s.head_w.get('w_img_pk') is not None
    dec_u = FastDecoder(d, batch_size=16, dtype=torch.bfloat16)
    dec_u._sk2_on = False

    with torch.no_grad():
        a = dec_s.prefill(text)
        b = dec_u.prefill(text)
        for _ in range(4):
            a = dec_s.step(token)
            b = dec_u.step(token)
    valid = b > -1e30
    rel = (a[valid].float() - b[valid].float()).abs().max() / \
        b[valid].float().abs().max()
    assert rel < 5e-2, rel.item()
