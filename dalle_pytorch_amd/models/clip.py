"""CLIP: contrastive text/image ranker used for generation re-ranking.

Parity target: reference dalle_pytorch.py:272-348. Two non-causal
Transformers (which on GPU run the same fused CDNA4 attention core,
non-causal mode), linear patch embedding, masked-mean text pooling, and a
symmetric InfoNCE loss with a learned temperature.
"""

import torch
import torch.nn.functional as F
from torch import nn

from dalle_pytorch_amd.models.transformer import Transformer


def masked_mean(t, mask, dim=1):
    t = t.masked_fill(~mask[:, :, None], 0.)
    return t.sum(dim=dim) / mask.sum(dim=dim)[..., None]


class CLIP(nn.Module):
    def __init__(
        self,
        *,
        dim_text=512,
        dim_image=512,
        dim_latent=512,
        num_text_tokens=10000,
        text_enc_depth=6,
        text_seq_len=256,
        text_heads=8,
        num_visual_tokens=512,
        visual_enc_depth=6,
        visual_heads=8,
        visual_image_size=256,
        visual_patch_size=32,
        channels=3,
    ):
        super().__init__()
        self.text_emb = nn.Embedding(num_text_tokens, dim_text)
        self.text_pos_emb = nn.Embedding(text_seq_len, dim_text)
        self.text_transformer = Transformer(
            causal=False, seq_len=text_seq_len, dim=dim_text,
            depth=text_enc_depth, heads=text_heads, rotary_emb=False)
        self.to_text_latent = nn.Linear(dim_text, dim_latent, bias=False)

        assert visual_image_size % visual_patch_size == 0, \
            'image dimensions must be divisible by the patch size'
        num_patches = (visual_image_size // visual_patch_size) ** 2
        patch_dim = channels * visual_patch_size ** 2

        self.visual_patch_size = visual_patch_size
        self.to_visual_embedding = nn.Linear(patch_dim, dim_image)
        self.visual_pos_emb = nn.Embedding(num_patches, dim_image)
        self.visual_transformer = Transformer(
            causal=False, seq_len=num_patches, dim=dim_image,
            depth=visual_enc_depth, heads=visual_heads, rotary_emb=False)
        self.to_visual_latent = nn.Linear(dim_image, dim_latent, bias=False)

        self.temperature = nn.Parameter(torch.tensor(1.))

    def forward(self, text, image, text_mask=None, return_loss=False):
        b, device, p = text.shape[0], text.device, self.visual_patch_size

        text_emb = self.text_emb(text)
        text_emb = text_emb + self.text_pos_emb(
            torch.arange(text.shape[1], device=device))

        # b c (hp p)(wp p) -> b (hp wp) (p p c)
        bimg, c, H, W = image.shape
        hp, wp = H // p, W // p
        patches = image.reshape(bimg, c, hp, p, wp, p)
        patches = patches.permute(0, 2, 4, 3, 5, 1).reshape(bimg, hp * wp, p * p * c)
        image_emb = self.to_visual_embedding(patches)
        image_emb = image_emb + self.visual_pos_emb(
            torch.arange(image_emb.shape[1], device=device))

        enc_text = self.text_transformer(text_emb, mask=text_mask)
        enc_image = self.visual_transformer(image_emb)

        if text_mask is not None:
            text_latents = masked_mean(enc_text, text_mask, dim=1)
        else:
            text_latents = enc_text.mean(dim=1)
        image_latents = enc_image.mean(dim=1)

        text_latents = F.normalize(self.to_text_latent(text_latents), p=2, dim=-1)
        image_latents = F.normalize(self.to_visual_latent(image_latents), p=2, dim=-1)

        temp = self.temperature.exp()
        if not return_loss:
            return (text_latents * image_latents).sum(dim=-1) * temp

        sim = text_latents @ image_latents.t() * temp
        labels = torch.arange(b, device=device)
        return (F.cross_entropy(sim, labels) + F.cross_entropy(sim.t(), labels)) / 2
