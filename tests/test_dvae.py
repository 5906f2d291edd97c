"""DiscreteVAE: shapes, loss quirks, straight-through/reinmax."""

import pytest
import torch
import torch.nn.functional as F

from dalle_pytorch_amd import DiscreteVAE

torch.manual_seed(0)


def test_shapes_roundtrip():
    vae = DiscreteVAE(image_size=64, num_layers=3, num_tokens=128,
                      codebook_dim=32, hidden_dim=8)
    img = torch.rand(2, 3, 64, 64)
    logits = vae(img, return_logits=True)
    assert logits.shape == (2, 128, 8, 8)
    codes = vae.get_codebook_indices(img)
    assert codes.shape == (2, 64)
    recon = vae.decode(codes)
    assert recon.shape == (2, 3, 64, 64)
    out = vae(img)
    assert out.shape == img.shape


def test_resblocks_and_1x1_decoder_head():
    vae = DiscreteVAE(image_size=32, num_layers=2, num_tokens=32,
                      codebook_dim=16, hidden_dim=8, num_resnet_blocks=2)
    img = torch.rand(1, 3, 32, 32)
    assert vae(img).shape == img.shape
    # resblock path inserts the codebook->hidden 1x1 conv at decoder[0]
    assert isinstance(vae.decoder[0], torch.nn.Conv2d)
    assert vae.decoder[0].kernel_size == (1, 1)


def test_kl_term_reversed_args_quirk():
    """Reference computes F.kl_div(log_uniform, log_qy, log_target=True)
    with uniform as *input* (dalle_pytorch.py:256-261). Verify our loss
    reproduces that exact expression."""
    vae = DiscreteVAE(image_size=32, num_layers=2, num_tokens=16,
                      codebook_dim=8, hidden_dim=4, kl_div_loss_weight=1.0)
    img = torch.rand(2, 3, 32, 32)
    torch.manual_seed(11)
    loss = vae(img, return_loss=True)

    torch.manual_seed(11)
    normed = vae.norm(img)
    logits = vae.encoder(normed)
    one_hot = F.gumbel_softmax(logits, tau=vae.temperature, dim=1, hard=False)
    sampled = torch.einsum('bnhw,nd->bdhw', one_hot, vae.codebook.weight)
    out = vae.decoder(sampled)
    recon = F.mse_loss(normed, out)
    lg = logits.permute(0, 2, 3, 1).reshape(2, -1, 16)
    log_qy = F.log_softmax(lg, dim=-1)
    log_u = torch.log(torch.tensor([1. / 16]))
    kl = F.kl_div(log_u, log_qy, None, None, 'batchmean', log_target=True)
    assert torch.allclose(loss, recon + kl, atol=1e-5)


def test_normalization_applied():
    vae = DiscreteVAE(image_size=32, num_layers=2, num_tokens=16,
                      codebook_dim=8, hidden_dim=4)
    img = torch.rand(1, 3, 32, 32)
    n = vae.norm(img)
    assert torch.allclose(n, (img - 0.5) / 0.5)


@pytest.mark.parametrize('st,rm', [(True, False), (True, True)])
def test_straight_through_and_reinmax_backward(st, rm):
    vae = DiscreteVAE(image_size=32, num_layers=2, num_tokens=16,
                      codebook_dim=8, hidden_dim=4, straight_through=st,
                      reinmax=rm)
    img = torch.rand(2, 3, 32, 32)
    loss = vae(img, return_loss=True)
    loss.backward()
    assert torch.isfinite(loss)
    assert vae.codebook.weight.grad is not None


def test_smooth_l1_option():
    vae = DiscreteVAE(image_size=32, num_layers=1, num_tokens=8,
                      codebook_dim=4, hidden_dim=4, smooth_l1_loss=True)
    assert vae.loss_fn is F.smooth_l1_loss


def test_temperature_argument():
    vae = DiscreteVAE(image_size=32, num_layers=1, num_tokens=8,
                      codebook_dim=4, hidden_dim=4)
    img = torch.rand(1, 3, 32, 32)
    torch.manual_seed(0)
    a = vae(img, temp=0.1)
    torch.manual_seed(0)
    b = vae(img, temp=5.0)
    assert not torch.allclose(a, b)


def test_image_size_power_of_two_assert():
    with pytest.raises(AssertionError):
        DiscreteVAE(image_size=48)


def test_run_as_gemms_matches_modules():
    """unfold/fold+GEMM evaluation of the conv stacks == the nn modules
    (covers Conv2d 4x4/1x1, ConvTranspose2d 4x4 stride 2, ResBlock)."""
    from dalle_pytorch_amd.models.dvae import _run_as_gemms
    torch.manual_seed(5)
    vae = DiscreteVAE(image_size=64, num_layers=3, num_tokens=32,
                      codebook_dim=16, hidden_dim=8, num_resnet_blocks=1)
    vae.eval()
    img = torch.rand(2, 3, 64, 64)
    with torch.no_grad():
        enc_ref = vae.encoder(img)
        enc_gem = _run_as_gemms(vae.encoder, img)
        assert torch.allclose(enc_gem, enc_ref, atol=1e-4), \
            (enc_gem - enc_ref).abs().max()
        seq = torch.randint(0, 32, (2, 64))
        dec_ref = vae.decode(seq)           # CPU path: module stack
        emb = vae.codebook(seq).reshape(2, 8, 8, 16).permute(0, 3, 1, 2)
        dec_gem = _run_as_gemms(vae.decoder, emb)
        assert torch.allclose(dec_gem, dec_ref, atol=1e-4), \
            (dec_gem - dec_ref).abs().max()


def test_run_as_gemms_gradient_parity():
    """The GEMM conv walker is differentiable: grads through it match the
    module stack (enables --conv_gemm trainable-path MIOpen independence)."""
    from dalle_pytorch_amd.models.dvae import _run_as_gemms
    torch.manual_seed(9)
    vae = DiscreteVAE(image_size=32, num_layers=2, num_tokens=16,
                      codebook_dim=8, hidden_dim=6, num_resnet_blocks=1)
    img = torch.rand(2, 3, 32, 32)

    out_ref = vae.encoder(img)
    out_ref.square().sum().backward()
    g_ref = [p.grad.clone() for p in vae.encoder.parameters()]
    vae.encoder.zero_grad()

    out = _run_as_gemms(vae.encoder, img)
    out.square().sum().backward()
    assert torch.allclose(out, out_ref, atol=1e-5)
    for g, p in zip(g_ref, vae.encoder.parameters()):
        assert torch.allclose(g, p.grad, atol=1e-4), (g - p.grad).abs().max()

    x = torch.randn(2, 8, 8, 8, requires_grad=True)
    dec_ref = vae.decoder(x)
    dec_ref.square().sum().backward()
    gx_ref, x.grad = x.grad.clone(), None
    g_ref = [p.grad.clone() for p in vae.decoder.parameters()]
    vae.decoder.zero_grad()

    dec = _run_as_gemms(vae.decoder, x)
    dec.square().sum().backward()
    assert torch.allclose(dec, dec_ref, atol=1e-5)
    assert torch.allclose(x.grad, gx_ref, atol=1e-4)
    for g, p in zip(g_ref, vae.decoder.parameters()):
        assert torch.allclose(g, p.grad, atol=1e-4), (g - p.grad).abs().max()
