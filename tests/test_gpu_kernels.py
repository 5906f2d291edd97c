"""GPU numerics: HIP kernels vs plain-PyTorch fp32 oracles (on an MI355X
box — every test here is @pytest.mark.gpu)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope='module')
def ext():
    import dalle_pytorch_amd._hip as m
    return m


def fp32_oracle(q, k, v, scale, causal=True, key_mask=None, static_mask=None):
    q, k, v = q.float(), k.float(), v.float()
    dots = (q * scale) @ k.transpose(-1, -2)
    if key_mask is not None:
        dots = dots.masked_fill(~key_mask[:, None, None, :], float('-inf'))
    if causal:
        i, j = dots.shape[-2:]
        cm = torch.ones(i, j, dtype=torch.bool, device=q.device).triu_(j - i + 1)
        dots = dots.masked_fill(cm, float('-inf'))
    if static_mask is not None:
        dots = dots.masked_fill(~static_mask, float('-inf'))
    return dots.softmax(-1) @ v


def test_mfma_layout_probe(ext):
    """Asymmetric A,B (guide G9): catches any operand/output layout swap."""
    torch.manual_seed(0)
    A = (torch.randn(16, 32) * 0.5).bfloat16().cuda()
    B = (torch.randn(32, 16) * 0.5).bfloat16().cuda()
    C = ext.mfma_probe(A, B)
    ref = (A.float() @ B.float())
    assert torch.allclose(C, ref, atol=2e-2, rtol=2e-2), \
        f'max err {(C - ref).abs().max().item()}'


@pytest.mark.parametrize('b,h,nq,nk,causal', [
    (2, 4, 128, 128, True),
    (2, 4, 96, 96, True),        # non-multiple of 64
    (1, 16, 1280, 1280, True),   # flagship shape
    (2, 4, 64, 64, False),
    (1, 2, 64, 160, False),      # nq != nk (cross/cached-style)
])
def test_fa_fwd_vs_oracle(ext, b, h, nq, nk, causal):
    torch.manual_seed(1)
    q = torch.randn(b, h, nq, 64, device='cuda').bfloat16()
    k = torch.randn(b, h, nk, 64, device='cuda').bfloat16()
    v = torch.randn(b, h, nk, 64, device='cuda').bfloat16()
    scale = 64 ** -0.5
    out, lse = ext.fa_fwd(q, k, v, scale, causal, None, None, None, False)
    ref = fp32_oracle(q, k, v, scale, causal)
    err = (out.float() - ref).abs().max().item()
    assert err < 2e-2, f'max err {err}'
    # lse must reproduce the softmax denominator
    dots = (q.float() * scale) @ k.float().transpose(-1, -2)
    if causal:
        cm = torch.ones(nq, nk, dtype=torch.bool, device='cuda').triu_(nk - nq + 1)
        dots = dots.masked_fill(cm, float('-inf'))
    ref_lse = dots.logsumexp(-1)
    assert (lse - ref_lse).abs().max().item() < 1e-2


def test_fa_fwd_masks(ext):
    torch.manual_seed(2)
    b, h, n = 2, 2, 128
    q = torch.randn(b, h, n, 64, device='cuda').bfloat16()
    k = torch.randn(b, h, n, 64, device='cuda').bfloat16()
    v = torch.randn(b, h, n, 64, device='cuda').bfloat16()
    km = (torch.rand(b, n, device='cuda') > 0.2)
    km[:, 0] = True
    sm = (torch.rand(n, n, device='cuda') > 0.2)
    sm.fill_diagonal_(True)
    out, _ = ext.fa_fwd(q, k, v, 0.125, True, km, sm, None, False)
    ref = fp32_oracle(q, k, v, 0.125, True, km, sm)
    assert (out.float() - ref).abs().max().item() < 2e-2


def test_fa_backward_vs_oracle():
    """attention_core autograd on GPU vs fp32 eager autograd."""
    from dalle_pytorch_amd.ops import attention_core
    torch.manual_seed(3)
    b, h, n = 2, 4, 256
    q0 = torch.randn(b, h, n, 64, device='cuda')
    k0 = torch.randn(b, h, n, 64, device='cuda')
    v0 = torch.randn(b, h, n, 64, device='cuda')
    dout = torch.randn(b, h, n, 64, device='cuda')
    scale = 0.125

    q = q0.bfloat16().requires_grad_()
    k = k0.bfloat16().requires_grad_()
    v = v0.bfloat16().requires_grad_()
    out = attention_core(q, k, v, scale, causal=True)
    out.backward(dout.bfloat16())

    qr = q0.clone().requires_grad_()
    kr = k0.clone().requires_grad_()
    vr = v0.clone().requires_grad_()
    ref = fp32_oracle(qr, kr, vr, scale, causal=True)
    ref.backward(dout)

    for got, want, name in ((q.grad, qr.grad, 'dq'), (k.grad, kr.grad, 'dk'),
                            (v.grad, vr.grad, 'dv')):
        err = (got.float() - want).abs().max().item()
        rel = err / want.abs().max().item()
        assert rel < 5e-2, f'{name} rel err {rel}'


def test_fa_backward_with_lse_gradient_vs_oracle():
    """return_lse=True: gradients flowing through BOTH out and lse must match
    fp32 autograd (the axial lse-merge path: Dv - grad_lse in fa_bwd)."""
    from dalle_pytorch_amd.ops import attention_core
    torch.manual_seed(9)
    b, h, n = 2, 4, 192
    q0 = torch.randn(b, h, n, 64, device='cuda')
    k0 = torch.randn(b, h, n, 64, device='cuda')
    v0 = torch.randn(b, h, n, 64, device='cuda')
    dout = torch.randn(b, h, n, 64, device='cuda')
    dlse = torch.randn(b, h, n, device='cuda')
    scale = 0.125

    q, k, v = (t.bfloat16().requires_grad_() for t in (q0, k0, v0))
    out, lse = attention_core(q, k, v, scale, causal=True, return_lse=True)
    (out.float() * dout).sum().add_((lse * dlse).sum()).backward()

    qr, kr, vr = (t.clone().requires_grad_() for t in (q0, k0, v0))
    s = torch.matmul(qr * scale, kr.transpose(-1, -2))
    cm = torch.ones(n, n, dtype=torch.bool, device='cuda').triu_(1)
    s = s.masked_fill(cm, float('-inf'))
    ref_lse = torch.logsumexp(s, dim=-1)
    ref_out = torch.matmul(s.softmax(-1), vr)
    (ref_out * dout).sum().add_((ref_lse * dlse).sum()).backward()

    for got, want, name in ((q.grad, qr.grad, 'dq'), (k.grad, kr.grad, 'dk'),
                            (v.grad, vr.grad, 'dv')):
        rel = (got.float() - want).abs().max().item() / want.abs().max().item()
        assert rel < 5e-2, f'{name} rel err {rel}'


def test_axial_lse_merge_gpu_matches_masked(monkeypatch):
    """Decomposed axial (lse-merge, dense kernels only) vs the masked-dense
    kernel path, at the flagship grid size (S=32, t=257, n=1280) — forward
    and gradients, both axes."""
    from dalle_pytorch_amd.models.attention import SparseAxialCausalAttention
    torch.manual_seed(10)
    S, t = 32, 257
    n = t + S * S - 1   # 1280
    for axis in (0, 1):
        m = SparseAxialCausalAttention(
            dim=128, seq_len=n, image_size=S, axis=axis, heads=2,
            dim_head=64).cuda().bfloat16()
        x = torch.randn(2, n, 128, device='cuda', dtype=torch.bfloat16,
                        requires_grad=True)

        monkeypatch.setenv('DALLE_AMD_AXIAL_MASKED', '1')
        ref = m(x)
        gref = torch.autograd.grad(ref.float().square().sum(), (x,))[0]

        monkeypatch.setenv('DALLE_AMD_AXIAL_MASKED', '0')
        out = m(x)
        gnew = torch.autograd.grad(out.float().square().sum(), (x,))[0]

        err = (out.float() - ref.float()).abs().max().item()
        assert err < 0.05, f'axis={axis} fwd err {err}'
        rel = (gnew.float() - gref.float()).abs().max().item() / \
            max(gref.float().abs().max().item(), 1e-6)
        assert rel < 0.1, f'axis={axis} grad rel err {rel}'


def test_fa_backward_masked_vs_oracle():
    """Backward kernels with static mask + tile maps vs fp32 autograd."""
    from dalle_pytorch_amd.ops import attention_core
    from dalle_pytorch_amd.ops.attention import build_tile_map
    torch.manual_seed(8)
    b, h, n = 2, 2, 192
    sm = torch.zeros(n, n, dtype=torch.bool, device='cuda')
    sm[:, :33] = True
    for r in range(0, n, 32):
        sm[r:r + 32, r:r + 32] = True
    tiles = build_tile_map(sm)
    tiles_t = build_tile_map(sm.t())
    km = torch.ones(b, n, dtype=torch.bool, device='cuda')
    km[:, 10:20] = False

    q0 = torch.randn(b, h, n, 64, device='cuda')
    k0 = torch.randn(b, h, n, 64, device='cuda')
    v0 = torch.randn(b, h, n, 64, device='cuda')
    dout = torch.randn(b, h, n, 64, device='cuda')

    q = q0.bfloat16().requires_grad_()
    k = k0.bfloat16().requires_grad_()
    v = v0.bfloat16().requires_grad_()
    out = attention_core(q, k, v, 0.125, causal=True, key_mask=km,
                         static_mask=sm, static_tiles=tiles,
                         static_tiles_t=tiles_t)
    out.backward(dout.bfloat16())

    qr, kr, vr = (t.clone().requires_grad_() for t in (q0, k0, v0))
    ref = fp32_oracle(qr, kr, vr, 0.125, True, km, sm)
    ref = torch.nan_to_num(ref)   # fully-masked rows
    ref.backward(torch.nan_to_num(dout))
    for got, want, name in ((q.grad, qr.grad, 'dq'), (k.grad, kr.grad, 'dk'),
                            (v.grad, vr.grad, 'dv')):
        want = torch.nan_to_num(want)
        rel = (got.float() - want).abs().max().item() / max(want.abs().max().item(), 1e-6)
        assert rel < 5e-2, f'{name} rel err {rel}'


def test_attention_core_uses_hip_kernel():
    """The training path must run the native kernel, not eager."""
    from dalle_pytorch_amd.ops import attention_core, hip_available
    from dalle_pytorch_amd.ops import attention as attn_mod
    assert hip_available()
    called = {}
    orig = attn_mod._FlashAttention.apply

    def spy(*a):
        called['yes'] = True
        return orig(*a)

    attn_mod._FlashAttention.apply = spy
    try:
        q = torch.randn(1, 2, 64, 64, device='cuda').bfloat16()
        attention_core(q, q, q, 0.125, causal=True)
    finally:
        attn_mod._FlashAttention.apply = orig
    assert called.get('yes'), 'eager fallback ran on GPU'


def test_rope_split_vs_eager(ext):
    """Fused qkv split+rotary == eager chunk/rearrange/rotate (fwd + bwd)."""
    from dalle_pytorch_amd.models.positional import (
        build_dalle_rotary_table, apply_rotary_to_qkv)
    torch.manual_seed(5)
    b, n, h, d = 2, 68, 4, 64
    table = build_dalle_rotary_table(d, 4, 8).cuda()[..., :n, :]
    qkv0 = torch.randn(b, n, 3 * h * d, device='cuda')

    qkv = qkv0.bfloat16().requires_grad_()
    ang = table.squeeze(0).float()
    q, k, v = ext.rope_split_fwd(qkv, h, ang.cos().contiguous(),
                                 ang.sin().contiguous())

    qkv_ref = qkv0.bfloat16().requires_grad_()
    parts = qkv_ref.chunk(3, -1)
    qe, ke, ve = (t.reshape(b, n, h, d).permute(0, 2, 1, 3) for t in parts)
    qe, ke, ve = apply_rotary_to_qkv(table, (qe, ke, ve))
    for got, want in ((q, qe), (k, ke), (v, ve)):
        err = (got.float() - want.float()).abs().max().item()
        rel = err / want.float().abs().max().item()
        assert rel < 2e-2, f'rope fwd rel err {rel}'

    # backward through the autograd wrapper
    from dalle_pytorch_amd.ops.rope import rope_split
    qkv2 = qkv0.bfloat16().requires_grad_()
    q2, k2, v2 = rope_split(qkv2, h, ang.cos().contiguous(), ang.sin().contiguous())
    (q2.square().sum() + 2 * k2.square().sum() + 3 * v2.square().sum()).backward()
    (qe.square().sum() + 2 * ke.square().sum() + 3 * ve.square().sum()).backward()
    err = (qkv2.grad.float() - qkv_ref.grad.float()).abs().max().item()
    rel = err / qkv_ref.grad.float().abs().max().item()
    assert rel < 2e-2, f'rope bwd rel err {rel}'


def test_tile_map_skipping_correct(ext):
    """Block-sparse tile skipping must not change results vs dense mask."""
    from dalle_pytorch_amd.ops.attention import build_tile_map
    torch.manual_seed(6)
    b, h, n = 1, 2, 256
    q = torch.randn(b, h, n, 64, device='cuda').bfloat16()
    k = torch.randn(b, h, n, 64, device='cuda').bfloat16()
    v = torch.randn(b, h, n, 64, device='cuda').bfloat16()
    sm = torch.zeros(n, n, dtype=torch.bool, device='cuda')
    sm[:, :40] = True                       # "text" prefix
    for r in range(0, n, 32):               # diagonal stripes
        sm[r:r + 32, r:r + 32] = True
    tiles = build_tile_map(sm)
    assert (tiles == 0).any()               # something actually skips
    o1, l1 = ext.fa_fwd(q, k, v, 0.125, True, None, sm, tiles, False)
    o2, l2 = ext.fa_fwd(q, k, v, 0.125, True, None, sm, None, False)
    assert torch.equal(o1, o2) and torch.equal(l1, l2)
    ref = fp32_oracle(q, k, v, 0.125, True, None, sm)
    assert (o1.float() - ref).abs().max().item() < 2e-2


def test_fold_heads_layout(ext):
    torch.manual_seed(7)
    b, h, n = 2, 4, 128
    q = torch.randn(b, h, n, 64, device='cuda').bfloat16()
    o_std, _ = ext.fa_fwd(q, q, q, 0.125, True, None, None, None, False)
    o_fold, _ = ext.fa_fwd(q, q, q, 0.125, True, None, None, None, True)
    assert o_fold.shape == (b, n, h, 64)
    assert torch.equal(o_fold.permute(0, 2, 1, 3), o_std)


def test_autocast_rotary_stays_on_hip_path():
    """Rotary application must not promote q/k/v to fp32 under autocast —
    that silently pushes training attention onto the eager path."""
    from dalle_pytorch_amd.models.attention import Attention
    from dalle_pytorch_amd.models.positional import build_dalle_rotary_table
    from dalle_pytorch_amd.ops import attention as attn_mod
    attn = Attention(dim=128, seq_len=68, heads=2, dim_head=64).cuda()
    table = build_dalle_rotary_table(64, 4, 8).cuda()
    x = torch.randn(1, 68, 128, device='cuda')
    called = {}
    orig = attn_mod._FlashAttention.apply

    def spy(*a):
        called['yes'] = True
        return orig(*a)

    attn_mod._FlashAttention.apply = spy
    try:
        with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
            attn(x, rotary_pos_emb=table[..., :68, :])
    finally:
        attn_mod._FlashAttention.apply = orig
    assert called.get('yes'), 'rotary path fell back to eager attention'


def test_token_shift_vs_eager(ext):
    """Fused token shift == the eager pad/cat formulation (fwd + adjoint)."""
    from dalle_pytorch_amd.models.transformer import PreShiftToken

    class Capture(torch.nn.Module):
        def forward(self, x, **kw):
            return x

    S, text_len = 8, 5
    seq_len = text_len + S * S - 1
    shift = PreShiftToken(Capture(), image_size=S, seq_len=seq_len)
    torch.manual_seed(9)
    for n in (seq_len, seq_len - 7):
        x0 = torch.randn(2, n, 64, device='cuda')
        # eager reference on CPU path semantics (run the module on fp32 CPU)
        ref = shift(x0.cpu()).cuda()
        got = ext.token_shift(x0.contiguous(), text_len, S, False)
        assert torch.allclose(got, ref, atol=1e-6), n

    # adjoint check: <shift(x), y> == <x, shift^T(y)>
    x = torch.randn(2, seq_len, 64, device='cuda')
    y = torch.randn(2, seq_len, 64, device='cuda')
    lhs = (ext.token_shift(x, text_len, S, False) * y).sum()
    rhs = (x * ext.token_shift(y, text_len, S, True)).sum()
    assert torch.allclose(lhs, rhs, rtol=1e-4)


def test_geglu_vs_oracle(ext):
    torch.manual_seed(4)
    x0 = torch.randn(4, 96, 512, device='cuda')
    x = x0.bfloat16().requires_grad_()
    from dalle_pytorch_amd.ops import geglu
    out = geglu(x)
    a, g = x0.chunk(2, -1)
    ref = a * torch.nn.functional.gelu(g)
    err = (out.float() - ref).abs().max().item() / ref.abs().max().item()
    assert err < 1e-2, f'fwd rel err {err}'   # bf16 storage rounding

    dout = torch.randn_like(out, dtype=torch.float32)
    out.backward(dout.bfloat16())
    xr = x0.clone().requires_grad_()
    ar, gr = xr.chunk(2, -1)
    (ar * torch.nn.functional.gelu(gr)).backward(dout)
    gerr = (x.grad.float() - xr.grad).abs().max().item() / xr.grad.abs().max().item()
    assert gerr < 1e-2, f'bwd rel err {gerr}'


def test_axial_col_module_gpu_matches_cpu():
    """axis=1 full-length forward on the HIP kernel (bf16, tile-skipped):
    output and input grads must match the CPU fp32 module."""
    from dalle_pytorch_amd.models.attention import SparseAxialCausalAttention
    torch.manual_seed(11)
    S, text_len = 16, 64
    seq_len = text_len + S * S - 1
    mod = SparseAxialCausalAttention(dim=128, seq_len=seq_len, image_size=S,
                                     axis=1, heads=2, dim_head=64)
    x = torch.randn(2, seq_len, 128)

    x_cpu = x.clone().requires_grad_(True)
    ref = mod(x_cpu)
    ref.square().sum().backward()

    mod_gpu = SparseAxialCausalAttention(dim=128, seq_len=seq_len,
                                         image_size=S, axis=1, heads=2,
                                         dim_head=64)
    mod_gpu.load_state_dict(mod.state_dict())
    mod_gpu = mod_gpu.cuda()
    x_gpu = x.cuda().requires_grad_(True)
    with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
        out = mod_gpu(x_gpu)
    out.square().sum().backward()
    torch.cuda.synchronize()

    scale = ref.abs().max().item()
    assert (out.float().cpu() - ref).abs().max().item() < 3e-2 * scale
    gscale = x_cpu.grad.abs().max().item()
    assert (x_gpu.grad.float().cpu() - x_cpu.grad).abs().max().item() < 5e-2 * gscale


def _poison_allocator(mb=512):
    """Fill a chunk of the caching allocator with NaNs so freshly 'empty'
    kernel outputs start poisoned — un-written output elements then surface
    as NaNs instead of silently reading stale zeros."""
    t = torch.full((mb << 18,), float('nan'), device='cuda')  # mb MiB of fp32
    del t
    torch.cuda.synchronize()


def test_fa_fwd_poisoned_output_fully_written(ext):
    torch.manual_seed(13)
    b, h, n = 2, 3, 193   # ragged: partial q tiles and k tiles
    q = torch.randn(b, h, n, 64, device='cuda').bfloat16()
    k = torch.randn(b, h, n, 64, device='cuda').bfloat16()
    v = torch.randn(b, h, n, 64, device='cuda').bfloat16()
    _poison_allocator()
    out, lse = ext.fa_fwd(q, k, v, 0.125, True, None, None, None, False)
    assert torch.isfinite(out.float()).all(), 'unwritten/NaN forward output'
    assert torch.isfinite(lse).all()
    _poison_allocator()
    dout = torch.randn_like(out)
    dq, dk, dv = ext.fa_bwd(q, k, v, out, lse, dout, 0.125, True,
                            None, None, None, None, False, None)
    for t_, name in ((dq, 'dq'), (dk, 'dk'), (dv, 'dv')):
        assert torch.isfinite(t_.float()).all(), f'unwritten/NaN {name}'


def test_fa_fully_masked_rows_no_nan(ext):
    """A batch element whose key_mask masks EVERY key: forward must produce
    zero output and -inf lse for it, backward must produce zero (not NaN)
    gradients everywhere it touches."""
    from dalle_pytorch_amd.ops import attention_core
    torch.manual_seed(14)
    b, h, n = 2, 2, 96
    km = torch.ones(b, n, dtype=torch.bool, device='cuda')
    km[1] = False   # batch 1: nothing attendable
    q = torch.randn(b, h, n, 64, device='cuda').bfloat16().requires_grad_()
    k = torch.randn(b, h, n, 64, device='cuda').bfloat16().requires_grad_()
    v = torch.randn(b, h, n, 64, device='cuda').bfloat16().requires_grad_()
    out, lse = attention_core(q, k, v, 0.125, causal=True, key_mask=km,
                              return_lse=True)
    assert (out[1].float() == 0).all()
    assert torch.isinf(lse[1]).all() and (lse[1] < 0).all()
    assert torch.isfinite(out[0].float()).all()
    out.float().square().sum().backward()
    for g in (q.grad, k.grad, v.grad):
        assert torch.isfinite(g.float()).all(), 'NaN grads from masked rows'
        assert (g[1].float() == 0).all(), 'masked batch leaked gradient'


@pytest.mark.parametrize('variant', ['conv_like', 'sparse'])
def test_conv_and_blocksparse_modules_gpu_at_real_shapes(ext, variant):
    """conv_like (kernel_size=5) and variable block-sparse (block=16) at the
    flagship shape (S=32, t=257, n=1280) on the HIP tile-map kernels vs the
    CPU eager oracle of the same module — fwd and input grads (VERDICT
    weak #4: these patterns had never run on hardware)."""
    from dalle_pytorch_amd.models.attention import (
        SparseConvCausalAttention, SparseAttention)
    torch.manual_seed(33)
    S, t = 32, 257
    n = t + S * S - 1   # 1280
    if variant == 'conv_like':
        m = SparseConvCausalAttention(dim=128, seq_len=n, image_size=S,
                                      kernel_size=5, heads=2, dim_head=64)
    else:
        m = SparseAttention(dim=128, seq_len=n, block_size=16,
                            text_seq_len=t - 1, num_random_blocks=n // 16 // 4,
                            heads=2, dim_head=64)
    x0 = torch.randn(2, n, 128) * 0.5

    x_cpu = x0.clone().requires_grad_()
    ref = m(x_cpu)
    gref = torch.autograd.grad(ref.square().sum(), x_cpu)[0]

    mg = m.cuda().bfloat16()
    x = x0.cuda().bfloat16().requires_grad_()
    out = mg(x)
    g = torch.autograd.grad(out.float().square().sum(), x)[0]

    err = (out.float().cpu() - ref).abs().max().item()
    assert err < 0.06, (variant, err)
    rel = (g.float().cpu() - gref).abs().max().item() / \
        max(gref.abs().max().item(), 1e-6)
    assert rel < 0.1, (variant, rel)


def test_sample_topk_gumbel_vs_torch(ext):
    """Fused decode sampler vs the eager top-k+gumbel chain with identical
    noise — exact match away from threshold ties (measure-zero for
    continuous logits)."""
    torch.manual_seed(41)
    for rows, V, k, temp in ((64, 8192, 5786, 0.7), (16, 8192, 1, 1.0),
                             (8, 1024, 1024, 0.3)):
        logits = torch.randn(rows, V, device='cuda')
        noise = torch.rand(rows, V, device='cuda')
        got = ext.sample_topk_gumbel(logits, noise, k, temp)
        val, ind = torch.topk(logits, k)
        filt = torch.full_like(logits, float('-inf')).scatter_(1, ind, val)
        g = -torch.log((-torch.log(noise.clamp(min=1e-20))).clamp(min=1e-20))
        want = (filt / temp + g).argmax(-1)
        assert torch.equal(got, want), (rows, V, k, temp,
                                        (got != want).sum().item())


def test_fp8_linear_vs_bf16(ext, monkeypatch):
    """fp8 forward path: numerics close to bf16, gradients match the bf16
    master-weight gradients in direction, and a short training run's loss
    curve tracks the bf16 run (VERDICT item 6 sanity bar)."""
    from dalle_pytorch_amd.ops import fp8 as fp8_mod
    torch.manual_seed(31)
    lin = torch.nn.Linear(1024, 4096).cuda().bfloat16()
    x = torch.randn(512, 1024, device='cuda', dtype=torch.bfloat16,
                    requires_grad=True)
    ref = lin(x)
    gref = torch.autograd.grad(ref.float().square().sum(), (x, lin.weight))

    monkeypatch.setenv('DALLE_AMD_FP8', '1')
    out = fp8_mod.fp8_linear(lin, x)
    g = torch.autograd.grad(out.float().square().sum(), (x, lin.weight))
    rel = (out.float() - ref.float()).abs().mean().item() / \
        ref.float().abs().mean().item()
    assert rel < 0.08, rel   # e4m3 quantization noise bound
    for a, b in zip(g, gref):
        cos = torch.nn.functional.cosine_similarity(
            a.float().flatten(), b.float().flatten(), dim=0).item()
        assert cos > 0.98, cos

    # loss-curve sanity: tiny flagship-shaped model, 8 Adam steps
    from dalle_pytorch_amd import DALLE, DiscreteVAE

    def run_losses():
        torch.manual_seed(5)
        vae = DiscreteVAE(image_size=64, num_layers=2, num_tokens=64,
                          codebook_dim=64, hidden_dim=16).cuda()
        d = DALLE(dim=1024, vae=vae, num_text_tokens=200, text_seq_len=16,
                  depth=2, heads=16, dim_head=64,
                  attn_types=('axial_row',), shift_tokens=True).cuda()
        opt = torch.optim.Adam([p for p in d.parameters() if p.requires_grad],
                               lr=3e-4)
        torch.manual_seed(7)
        text = torch.randint(1, 200, (16, 16), device='cuda')
        imgs = torch.rand(16, 3, 64, 64, device='cuda')
        losses = []
        for _ in range(8):
            with torch.autocast('cuda', dtype=torch.bfloat16):
                loss = d(text, imgs, return_loss=True)
            opt.zero_grad()
            loss.backward()
            opt.step()
            losses.append(loss.item())
        return losses

    fp8_losses = run_losses()
    monkeypatch.setenv('DALLE_AMD_FP8', '0')
    bf16_losses = run_losses()
    # same trajectory within quantization noise; both must actually descend
    assert fp8_losses[-1] < fp8_losses[0]
    for a, b in zip(fp8_losses, bf16_losses):
        assert abs(a - b) / abs(b) < 0.05, (fp8_losses, bf16_losses)


def test_fa8_ladder_vs_oracle(ext, monkeypatch):
    """Opt-in 8-wave 32x32 ladder forward vs the fp32 oracle (dense causal,
    non-causal, key-masked, axial) — kept correct although the 16x16 kernel
    is faster at D=64 (see the dispatch comment in hip_ops.hip)."""
    monkeypatch.setenv('DALLE_AMD_FA8', '1')
    torch.manual_seed(21)
    for nq, nk, causal in ((256, 256, True), (1280, 1280, True),
                           (192, 320, False), (300, 300, True)):
        q = torch.randn(2, 3, nq, 64, device='cuda').bfloat16()
        k = torch.randn(2, 3, nk, 64, device='cuda').bfloat16()
        v = torch.randn(2, 3, nk, 64, device='cuda').bfloat16()
        out, lse = ext.fa_fwd(q, k, v, 0.125, causal, None, None, None, False)
        ref = fp32_oracle(q.float(), k.float(), v.float(), 0.125, causal)
        err = (out.float() - ref).abs().max().item()
        assert err < 0.02, (nq, nk, causal, err)
    # key mask
    km = torch.ones(2, 320, dtype=torch.bool, device='cuda')
    km[:, 5:40] = False
    q = torch.randn(2, 2, 320, 64, device='cuda').bfloat16()
    k = torch.randn(2, 2, 320, 64, device='cuda').bfloat16()
    v = torch.randn(2, 2, 320, 64, device='cuda').bfloat16()
    out, _ = ext.fa_fwd(q, k, v, 0.125, True, km, None, None, False)
    ref = fp32_oracle(q.float(), k.float(), v.float(), 0.125, True, km)
    assert (out.float() - ref).abs().max().item() < 0.02
    # axial mode (S=16, t=65, n=321)
    from dalle_pytorch_amd.models.attention import axial_mask
    S, t = 16, 65
    n = t + S * S
    q = torch.randn(1, 2, n, 64, device='cuda').bfloat16()
    k = torch.randn(1, 2, n, 64, device='cuda').bfloat16()
    v = torch.randn(1, 2, n, 64, device='cuda').bfloat16()
    for axis in (0, 1):
        out, _ = ext.fa_fwd(q, k, v, 0.125, True, None, None, None, False,
                            t, S, axis)
        sm = axial_mask(n, t, S, axis).cuda()
        ref = fp32_oracle(q.float(), k.float(), v.float(), 0.125, True,
                          None, sm)
        assert (out.float() - ref).abs().max().item() < 0.02, axis


def test_permlane_semantics(ext):
    """Pin v_permlane32_swap_b32: with (a, b) the results must be
    r0 = [a_lo | b_lo-from-partner...] — concretely, the fa8 kernel's
    partner_u32(w) = (half ? r0 : r1) of swap(w, w) must yield w[lane^32]."""
    a = torch.arange(64, dtype=torch.int32, device='cuda')
    b = a + 1000
    r0, r1 = ext.permlane_probe(a, b)
    lanes = torch.arange(64)
    partner = torch.where(lanes < 32, lanes + 32, lanes - 32)
    # the kernel's assumption: swap(w, w) gives partner value in r1 for
    # lanes < 32 and in r0 for lanes >= 32
    s0, s1 = ext.permlane_probe(a, a)
    got = torch.where(lanes.cuda() < 32, s1, s0)
    assert torch.equal(got.cpu(), a.cpu()[partner]), \
        (r0.cpu().tolist(), r1.cpu().tolist())


def test_skinny_gemm_vs_oracle(ext):
    """Weights-streaming skinny-M GEMM (decode projections) vs rocBLAS at
    the real decode shapes, with and without bias, M 64 and 128."""
    torch.manual_seed(15)
    for M in (64, 128):
        for (K, N) in ((1024, 3072), (1024, 1024), (1024, 8192),
                       (4096, 1024), (1024, 8192 + 4)):
            if N % 4:
                continue
            x = (torch.randn(M, K, device='cuda') * 0.5).bfloat16()
            w = (torch.randn(N, K, device='cuda') * 0.1).bfloat16()
            bias = torch.randn(N, device='cuda').bfloat16()
            got = ext.skinny_gemm(x, w, bias)
            want = torch.nn.functional.linear(x.float(), w.float(), bias.float())
            rel = (got.float() - want).abs().max().item() / want.abs().max().item()
            assert rel < 2e-2, (M, K, N, rel)
            got_nb = ext.skinny_gemm(x, w, None)
            want_nb = x.float() @ w.float().t()
            rel = (got_nb.float() - want_nb).abs().max().item() / want_nb.abs().max().item()
            assert rel < 2e-2, (M, K, N, 'nobias', rel)


def test_add_scaled_vs_oracle(ext):
    """Fused residual+LayerScale kernel (resls_fwd/bwd) vs fp32 autograd."""
    from dalle_pytorch_amd.ops.fused import add_scaled
    torch.manual_seed(12)
    for dim in (512, 1024, 2048):
        x0 = torch.randn(3, 40, dim, device='cuda')
        y0 = torch.randn(3, 40, dim, device='cuda')
        g0 = torch.randn(1, 1, dim, device='cuda') * 0.1
        x = x0.bfloat16().requires_grad_()
        y = y0.bfloat16().requires_grad_()
        g = g0.clone().requires_grad_()
        out = add_scaled(x, y, g)
        out.float().square().sum().backward()

        xr, yr, gr = (t.clone().requires_grad_() for t in (x0, y0, g0))
        ref = xr + yr * gr
        ref.square().sum().backward()
        assert (out.float() - ref).abs().max() < 0.05
        for got, want in ((x.grad.float(), xr.grad), (y.grad.float(), yr.grad),
                          (g.grad.float(), gr.grad)):
            rel = (got - want).abs().max().item() / want.abs().max().item()
            assert rel < 5e-2, (dim, rel)


def test_dalle_train_step_gpu():
    """One full flagship-shaped training step on GPU, bf16, HIP path."""
    from dalle_pytorch_amd import DALLE, DiscreteVAE
    torch.manual_seed(0)
    vae = DiscreteVAE(image_size=256, num_layers=3, num_tokens=512,
                      codebook_dim=64, hidden_dim=32)
    d = DALLE(dim=512, vae=vae, num_text_tokens=1000, text_seq_len=256,
              depth=2, heads=8, dim_head=64, attn_types=('full', 'axial_row'),
              shift_tokens=True).cuda()
    opt = torch.optim.Adam((p for p in d.parameters() if p.requires_grad), lr=1e-4)
    text = torch.randint(1, 1000, (2, 256), device='cuda')
    imgs = torch.rand(2, 3, 256, 256, device='cuda')
    with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
        loss = d(text, imgs, return_loss=True)
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


def test_reversible_gpu_matches_cpu_grads():
    """Reversible stack on GPU: finite grads, deterministic replay."""
    from dalle_pytorch_amd import DALLE, DiscreteVAE
    torch.manual_seed(0)
    vae = DiscreteVAE(image_size=64, num_layers=3, num_tokens=64,
                      codebook_dim=32, hidden_dim=8)
    d = DALLE(dim=128, vae=vae, num_text_tokens=100, text_seq_len=16, depth=2,
              heads=2, dim_head=64, reversible=True, attn_dropout=0.1,
              ff_dropout=0.1).cuda()
    text = torch.randint(1, 100, (2, 16), device='cuda')
    imgs = torch.rand(2, 3, 64, 64, device='cuda')
    with torch.autocast(device_type='cuda', dtype=torch.bfloat16):
        loss = d(text, imgs, return_loss=True)
    loss.backward()
    torch.cuda.synchronize()
    assert all(torch.isfinite(p.grad).all() for p in d.parameters()
               if p.requires_grad and p.grad is not None)


def test_generation_gpu_cached():
    from dalle_pytorch_amd import DALLE, DiscreteVAE
    torch.manual_seed(0)
    vae = DiscreteVAE(image_size=64, num_layers=3, num_tokens=64,
                      codebook_dim=32, hidden_dim=8)
    d = DALLE(dim=128, vae=vae, num_text_tokens=100, text_seq_len=16, depth=2,
              heads=2, dim_head=64, shift_tokens=True).cuda().eval()
    text = torch.randint(1, 100, (2, 16), device='cuda')
    imgs = d.generate_images(text, use_cache=True)
    assert imgs.shape == (2, 3, 64, 64)
    assert torch.isfinite(imgs).all()


def test_dvae_gemm_encoder_matches_conv():
    """The im2col+GEMM frozen-encoder path == the nn.Conv2d stack."""
    from dalle_pytorch_amd import DiscreteVAE
    torch.manual_seed(10)
    vae = DiscreteVAE(image_size=64, num_layers=3, num_tokens=128,
                      codebook_dim=64, hidden_dim=16,
                      num_resnet_blocks=1).cuda().eval()
    img = torch.rand(2, 3, 64, 64, device='cuda')
    with torch.no_grad():
        ref = vae.encoder(vae.norm(img))
        got = vae._encode_as_gemms(vae.norm(img))
    assert (got - ref).abs().max().item() < 1e-3
    codes = vae.get_codebook_indices(img)
    ref_codes = ref.argmax(dim=1).flatten(1)
    assert (codes == ref_codes).float().mean().item() > 0.99


def test_layer_norm_vs_oracle(ext):
    torch.manual_seed(11)
    from dalle_pytorch_amd.ops.fused import layer_norm
    x0 = torch.randn(300, 1024, device='cuda') * 2 + 0.5
    w0 = torch.randn(1024, device='cuda')
    b0 = torch.randn(1024, device='cuda')
    dy = torch.randn(300, 1024, device='cuda')

    x = x0.bfloat16().requires_grad_()
    w = w0.clone().requires_grad_()
    b = b0.clone().requires_grad_()
    y = layer_norm(x, w, b)
    y.backward(dy.bfloat16())

    xr = x0.clone().requires_grad_()
    wr = w0.clone().requires_grad_()
    br = b0.clone().requires_grad_()
    yr = torch.nn.functional.layer_norm(xr, (1024,), wr, br)
    yr.backward(dy)

    assert (y.float() - yr).abs().max() / yr.abs().max() < 2e-2
    for got, want, name in ((x.grad, xr.grad, 'dx'), (w.grad, wr.grad, 'dw'),
                            (b.grad, br.grad, 'db')):
        rel = (got.float() - want).abs().max() / want.abs().max().clamp(min=1e-6)
        assert rel < 2e-2, f'{name} {rel.item()}'


@pytest.mark.gpu
def test_sk2_decode_gemm_matches_linear():
    """sk2 (packed weights-streaming decode GEMM) vs fp32 F.linear across
    row counts and all three epilogue modes (bias / geglu / fp32 head)."""
    import torch.nn.functional as F
    import dalle_pytorch_amd._hip as ext
    from dalle_pytorch_amd.engine.decode import FastDecoder
    torch.manual_seed(0)
    dev = 'cuda'
    for rows, K, N in [(64, 1024, 3072), (64, 1024, 1024), (128, 1024, 1024),
                       (32, 4096, 1024), (16, 1024, 512), (64, 2048, 2048)]:
        x = (torch.randn(rows, K, device=dev) * 0.3).bfloat16()
        w = (torch.randn(N, K, device=dev) * 0.05).bfloat16()
        b = torch.randn(N, device=dev)
        pk = FastDecoder._sk2_pack(w)
        assert pk is not None
        out = ext.sk2(x, pk, b, N, K, 0)
        ref = F.linear(x.float(), w.float(), b)
        torch.testing.assert_close(out.float(), ref, rtol=3e-2, atol=3e-2)
        out_nb = ext.sk2(x, pk, None, N, K, 0)
        torch.testing.assert_close(out_nb.float(), ref - b, rtol=3e-2,
                                   atol=3e-2)
    # mode 1: fused geglu epilogue
    rows, K, N = 64, 1024, 8192
    x = (torch.randn(rows, K, device='cuda') * 0.3).bfloat16()
    w = (torch.randn(N, K, device='cuda') * 0.05).bfloat16()
    b = torch.randn(N, device='cuda')
    pk = FastDecoder._sk2_pack(w)
    y = F.linear(x.float(), w.float(), b)
    v, g = y.chunk(2, dim=-1)
    ref = v * F.gelu(g)
    out = ext.sk2(x, pk, b, N, K, 1)
    torch.testing.assert_close(out.float(), ref, rtol=3e-2, atol=3e-2)
    # mode 2: fp32 output (image-vocab head)
    out32 = ext.sk2(x, pk, b, N, K, 2)
    assert out32.dtype == torch.float32
    torch.testing.assert_close(out32, y, rtol=3e-2, atol=3e-2)


@pytest.mark.gpu
def test_sk2_fp8_weights_close_to_bf16():
    """Opt-in fp8-weight decode GEMM (sk2 e4m3 pack): per-tensor-scaled
    quantization error only — compare against the fp32 reference computed
    from the DEQUANTIZED weights (exact-path check) and loosely against the
    unquantized weights (end-to-end error budget)."""
    import torch.nn.functional as F
    import dalle_pytorch_amd._hip as ext
    from dalle_pytorch_amd.engine.decode import FastDecoder
    torch.manual_seed(1)
    for rows, K, N, mode in [(64, 1024, 3072, 0), (64, 1024, 8192, 1),
                             (64, 4096, 1024, 0), (32, 1024, 1024, 0)]:
        x = (torch.randn(rows, K, device='cuda') * 0.3).bfloat16()
        w = (torch.randn(N, K, device='cuda') * 0.05).bfloat16()
        b = torch.randn(N, device='cuda')
        pk8, ws = FastDecoder._sk2_pack_fp8(w)
        out = ext.sk2(x, pk8, b, N, K, mode, ws).float()
        # exact reference on dequantized weights (x also rides through e4m3)
        wdq = (w.float() / ws).clamp(-448, 448).to(torch.float8_e4m3fn) \
            .float() * ws
        xdq = x.float().clamp(-448, 448).to(torch.float8_e4m3fn).float()
        y = F.linear(xdq, wdq, b)
        if mode == 1:
            v, g = y.chunk(2, dim=-1)
            y = v * F.gelu(g)
        torch.testing.assert_close(out, y, rtol=3e-2, atol=3e-2)
        # loose end-to-end budget vs the unquantized math
        yref = F.linear(x.float(), w.float(), b)
        if mode == 1:
            v, g = yref.chunk(2, dim=-1)
            yref = v * F.gelu(g)
        rel = (out - yref).norm() / yref.norm()
        assert rel < 0.05, rel.item()


@pytest.mark.gpu
def test_fp8_decode_engine_close_to_bf16(monkeypatch):
    """DALLE_AMD_FP8_DECODE=1: the engine runs fp8-weight sk2 GEMMs; step
    logits stay within the fp8 error budget of the bf16 engine."""
    monkeypatch.setenv('DALLE_AMD_FP8_DECODE', '1')
    from dalle_pytorch_amd.engine.decode import FastDecoder
    from tests.test_decode_engine import tiny_dalle
    torch.manual_seed(11)
    d = tiny_dalle(attn_types=('axial_row', 'axial_col'), depth=2, dim=1024,
                   heads=16, dim_head=64).cuda().eval()
    text = torch.randint(1, 50, (16, 8), device='cuda')
    token = torch.randint(0, 64, (16,), device='cuda')
    dec8 = FastDecoder(d, batch_size=16, dtype=torch.bfloat16)
    assert dec8.states[0].w.get('qkv_pk8') is not None
    monkeypatch.setenv('DALLE_AMD_FP8_DECODE', '0')
    dec = FastDecoder(d, batch_size=16, dtype=torch.bfloat16)
    assert dec.states[0].w.get('qkv_pk8') is None
    with torch.no_grad():
        a = dec8.prefill(text)
        b = dec.prefill(text)
        for _ in range(3):
            a = dec8.step(token)
            b = dec.step(token)
    valid = b > -1e30
    rel = (a[valid].float() - b[valid].float()).norm() / \
        b[valid].float().norm()
    assert rel < 0.1, rel.item()
