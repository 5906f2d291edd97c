"""CLIP: shapes, masked pooling, symmetric InfoNCE."""

import torch

from dalle_pytorch_amd import CLIP
from dalle_pytorch_amd.models.clip import masked_mean

torch.manual_seed(0)


def tiny_clip():
    return CLIP(dim_text=32, dim_image=32, dim_latent=16, num_text_tokens=100,
                text_enc_depth=1, text_seq_len=12, text_heads=2,
                visual_enc_depth=1, visual_heads=2, visual_image_size=32,
                visual_patch_size=8)


def test_similarity_and_loss():
    clip = tiny_clip()
    text = torch.randint(0, 100, (4, 12))
    imgs = torch.rand(4, 3, 32, 32)
    mask = torch.ones(4, 12, dtype=torch.bool)
    sim = clip(text, imgs, text_mask=mask)
    assert sim.shape == (4,)
    loss = clip(text, imgs, text_mask=mask, return_loss=True)
    loss.backward()
    assert torch.isfinite(loss)


def test_masked_mean():
    t = torch.tensor([[[1.0], [3.0], [100.0]]])
    m = torch.tensor([[True, True, False]])
    assert torch.allclose(masked_mean(t, m), torch.tensor([[2.0]]))


def test_patch_embedding_layout():
    """Patch rearrange must match 'b c (h p1) (w p2) -> b (h w) (p1 p2 c)'."""
    clip = tiny_clip()
    # identity-ish probe: craft an image where each patch is constant, check
    # the patch embedding input ordering via the linear layer identity
    img = torch.zeros(1, 3, 32, 32)
    img[0, :, :8, :8] = 1.0   # patch (0, 0)
    p = clip.visual_patch_size
    hp = 32 // p
    patches = img.reshape(1, 3, hp, p, hp, p).permute(0, 2, 4, 3, 5, 1) \
                 .reshape(1, hp * hp, p * p * 3)
    assert patches[0, 0].sum() == 8 * 8 * 3
    assert patches[0, 1].sum() == 0
