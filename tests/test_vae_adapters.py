"""Native OpenAI-dVAE and VQGAN adapters: interface + shape contracts."""

import torch

from dalle_pytorch_amd import DALLE, OpenAIDiscreteVAE, VQGanVAE

torch.manual_seed(0)


def test_openai_dvae_interface():
    vae = OpenAIDiscreteVAE(n_hid=32, vocab_size=256)  # slim for CPU test
    assert vae.image_size == 256 and vae.num_layers == 3
    img = torch.rand(1, 3, 256, 256)
    codes = vae.get_codebook_indices(img)
    assert codes.shape == (1, 32 * 32)
    assert codes.max() < 256
    out = vae.decode(codes)
    assert out.shape == (1, 3, 256, 256)
    assert (out >= 0).all() and (out <= 1).all()


def test_vqgan_interface_f16():
    vae = VQGanVAE(image_size=64, num_tokens=512, embed_dim=32, ch=16,
                   ch_mult=(1, 1, 2), num_res_blocks=1)
    assert vae.num_layers == 2  # f = 4
    img = torch.rand(1, 3, 64, 64)
    codes = vae.get_codebook_indices(img)
    assert codes.shape == (1, (64 // 4) ** 2)
    out = vae.decode(codes)
    assert out.shape == (1, 3, 64, 64)
    assert (out >= 0).all() and (out <= 1).all()


def test_vqgan_gumbel_variant():
    vae = VQGanVAE(image_size=32, num_tokens=128, embed_dim=16, ch=16,
                   ch_mult=(1, 2), num_res_blocks=1, gumbel=True)
    img = torch.rand(1, 3, 32, 32)
    codes = vae.get_codebook_indices(img)
    assert codes.max() < 128
    assert vae.decode(codes).shape == (1, 3, 32, 32)


def test_dalle_with_vqgan():
    vae = VQGanVAE(image_size=32, num_tokens=64, embed_dim=16, ch=16,
                   ch_mult=(1, 2), num_res_blocks=1)
    d = DALLE(dim=32, vae=vae, num_text_tokens=50, text_seq_len=4, depth=1,
              heads=2, dim_head=16)
    text = torch.randint(1, 50, (1, 4))
    imgs = torch.rand(1, 3, 32, 32)
    loss = d(text, imgs, return_loss=True)
    loss.backward()
    assert torch.isfinite(loss)
