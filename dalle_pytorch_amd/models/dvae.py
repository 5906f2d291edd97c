"""Trainable Gumbel-softmax discrete VAE.

Parity target: reference dalle_pytorch.py:87-268 (DiscreteVAE, ResBlock).
Checkpoint-compatible: ``codebook.weight``, ``encoder.{i}...``,
``decoder.{i}...`` with identical Sequential indexing. The trainable path
runs conv/deconv through MIOpen; the frozen encode (DALLE training) and
decode (generation) paths run as unfold/fold + hipBLASLt GEMMs instead
(``_run_as_gemms``) because MIOpen's kernel choice is box-dependent on
gfx950.

Known reference quirks kept on purpose:
* the KL term calls ``F.kl_div(log_uniform, log_qy, ..., log_target=True)``
  with the *uniform* distribution as input and the posterior as target —
  i.e. KL(q || u) with q detached-free gradients flowing through the target
  argument (reference dalle_pytorch.py:256-261),
* normalization defaults to mean 0.5 / std 0.5 per channel and is applied
  inside forward (reference :185-193,225).
"""

import os
from math import log2, sqrt

import torch
import torch.nn.functional as F
from torch import nn


def _log(t, eps=1e-20):
    return torch.log(t.clamp(min=eps))


class ResBlock(nn.Module):
    def __init__(self, chan):
        super().__init__()
        self.net = nn.Sequential(
            nn.Conv2d(chan, chan, 3, padding=1),
            nn.ReLU(),
            nn.Conv2d(chan, chan, 3, padding=1),
            nn.ReLU(),
            nn.Conv2d(chan, chan, 1),
        )

    def forward(self, x):
        return self.net(x) + x


def _run_as_gemms(mod, x):
    """Run a conv stack as unfold/fold + hipBLASLt GEMMs.

    MIOpen's kernel choice for these shapes is box-dependent on gfx950 — on
    untuned machines it falls back to ``naive_conv`` (measured at 60% of a
    training step for the encoder and ~10% of a generation for the
    decoder's transposed convs). Stride-2 4x4 convs become unfold+matmul,
    transposed convs matmul+fold, and 1x1 convs plain matmuls — MFMA GEMMs
    everywhere, no MIOpen. All ops are differentiable, but the intended use
    is the frozen encode/decode paths.
    """
    if isinstance(mod, (nn.Conv2d, nn.ConvTranspose2d)) and (
            mod.groups != 1 or mod.dilation != (1, 1)
            or getattr(mod, 'output_padding', (0, 0)) != (0, 0)):
        return mod(x)   # config outside the GEMM mapping: use the module
    if isinstance(mod, nn.Conv2d):
        b, c, h, w = x.shape
        kh, kw = mod.kernel_size
        if kh == 1 and kw == 1:
            out = torch.matmul(
                x.reshape(b, c, h * w).transpose(1, 2),
                mod.weight.reshape(mod.out_channels, c).t())
            out = out + mod.bias
            return out.transpose(1, 2).reshape(b, -1, h, w)
        sh, sw = mod.stride
        oh = (h + 2 * mod.padding[0] - kh) // sh + 1
        ow = (w + 2 * mod.padding[1] - kw) // sw + 1
        cols = F.unfold(x, (kh, kw), stride=(sh, sw),
                        padding=mod.padding)        # [b, c*kh*kw, L]
        out = torch.matmul(cols.transpose(1, 2),
                           mod.weight.reshape(mod.out_channels, -1).t())
        out = out + mod.bias
        return out.transpose(1, 2).reshape(b, -1, oh, ow)
    if isinstance(mod, nn.ConvTranspose2d):
        b, c, h, w = x.shape
        kh, kw = mod.kernel_size
        sh, sw = mod.stride
        oh = (h - 1) * sh - 2 * mod.padding[0] + kh
        ow = (w - 1) * sw - 2 * mod.padding[1] + kw
        # deconv forward == conv backward-data: GEMM then col2im (fold)
        cols = torch.matmul(
            mod.weight.reshape(c, -1).t().unsqueeze(0),   # [1, out*kh*kw, in]
            x.reshape(b, c, h * w))                        # -> [b, out*kh*kw, L]
        out = F.fold(cols, (oh, ow), (kh, kw), stride=(sh, sw),
                     padding=mod.padding)
        return out + mod.bias.reshape(1, -1, 1, 1)
    if isinstance(mod, nn.ReLU):
        return torch.relu(x)
    if isinstance(mod, ResBlock):
        y = x
        for sub in mod.net:
            y = _run_as_gemms(sub, y)
        return y + x
    if isinstance(mod, nn.Sequential):
        for sub in mod:
            x = _run_as_gemms(sub, x)
        return x
    return mod(x)


class DiscreteVAE(nn.Module):
    def __init__(
        self,
        image_size=256,
        num_tokens=512,
        codebook_dim=512,
        num_layers=3,
        num_resnet_blocks=0,
        hidden_dim=64,
        channels=3,
        smooth_l1_loss=False,
        temperature=0.9,
        straight_through=False,
        reinmax=False,
        kl_div_loss_weight=0.,
        normalization=((0.5, 0.5, 0.5, 0), (0.5, 0.5, 0.5, 1)),
    ):
        super().__init__()
        assert log2(image_size).is_integer(), 'image size must be a power of 2'
        assert num_layers >= 1, 'number of layers must be at least 1'
        has_resblocks = num_resnet_blocks > 0

        self.channels = channels
        self.image_size = image_size
        self.num_tokens = num_tokens
        self.num_layers = num_layers
        self.temperature = temperature
        self.straight_through = straight_through
        self.reinmax = reinmax

        self.codebook = nn.Embedding(num_tokens, codebook_dim)

        enc_chans = [hidden_dim] * num_layers
        dec_chans = list(reversed(enc_chans))
        enc_chans = [channels, *enc_chans]
        dec_init_chan = codebook_dim if not has_resblocks else dec_chans[0]
        dec_chans = [dec_init_chan, *dec_chans]

        enc_layers = []
        dec_layers = []
        for (ei, eo), (di, do) in zip(zip(enc_chans[:-1], enc_chans[1:]),
                                      zip(dec_chans[:-1], dec_chans[1:])):
            enc_layers.append(nn.Sequential(nn.Conv2d(ei, eo, 4, stride=2, padding=1),
                                            nn.ReLU()))
            dec_layers.append(nn.Sequential(nn.ConvTranspose2d(di, do, 4, stride=2, padding=1),
                                            nn.ReLU()))
        for _ in range(num_resnet_blocks):
            dec_layers.insert(0, ResBlock(dec_chans[1]))
            enc_layers.append(ResBlock(enc_chans[-1]))
        if has_resblocks:
            dec_layers.insert(0, nn.Conv2d(codebook_dim, dec_chans[1], 1))

        enc_layers.append(nn.Conv2d(enc_chans[-1], num_tokens, 1))
        dec_layers.append(nn.Conv2d(dec_chans[-1], channels, 1))

        self.encoder = nn.Sequential(*enc_layers)
        self.decoder = nn.Sequential(*dec_layers)

        self.loss_fn = F.smooth_l1_loss if smooth_l1_loss else F.mse_loss
        self.kl_div_loss_weight = kl_div_loss_weight
        self.normalization = tuple(t[:channels] for t in normalization) \
            if normalization is not None else None

    def norm(self, images):
        if self.normalization is None:
            return images
        means = torch.as_tensor(self.normalization[0]).to(images).reshape(1, -1, 1, 1)
        stds = torch.as_tensor(self.normalization[1]).to(images).reshape(1, -1, 1, 1)
        return (images - means) / stds

    def _encode_as_gemms(self, x):
        """Frozen-encoder forward as im2col + hipBLASLt GEMMs.

        MIOpen's kernel choice for these bf16 conv shapes is box-dependent
        on gfx950 — on untuned machines it falls back to ``naive_conv``
        (measured at 60% of the training step at batch 64). Expressing the
        stride-2 4x4 convs as unfold+matmul and the 1x1 convs as plain
        matmuls pins the work to hipBLASLt MFMA GEMMs everywhere.
        """
        return _run_as_gemms(self.encoder, x)

    @torch.no_grad()
    def get_codebook_indices(self, images):
        was_training = self.training
        self.eval()
        if images.is_cuda:
            logits = self._encode_as_gemms(self.norm(images))
        else:
            logits = self(images, return_logits=True)
        self.train(was_training)
        return logits.argmax(dim=1).flatten(1)

    def decode(self, img_seq):
        emb = self.codebook(img_seq)
        b, n, d = emb.shape
        hw = int(sqrt(n))
        emb = emb.reshape(b, hw, hw, d).permute(0, 3, 1, 2)
        if emb.is_cuda and not torch.is_grad_enabled():
            return _run_as_gemms(self.decoder, emb)
        return self.decoder(emb)

    def forward(self, img, return_loss=False, return_recons=False,
                return_logits=False, temp=None):
        num_tokens, image_size = self.num_tokens, self.image_size
        assert img.shape[-1] == image_size and img.shape[-2] == image_size, \
            f'input must have the correct image size {image_size}'

        img = self.norm(img)
        # DALLE_AMD_CONV_GEMM=1 routes the TRAINABLE conv stacks through the
        # differentiable unfold/fold+GEMM walker too — opt-in MIOpen
        # independence for train_vae.py on boxes where MIOpen falls back to
        # naive_conv (the frozen encode/decode paths always use it)
        gemm = img.is_cuda and os.environ.get('DALLE_AMD_CONV_GEMM') == '1'
        logits = _run_as_gemms(self.encoder, img) if gemm else self.encoder(img)
        if return_logits:
            return logits

        temp = temp if temp is not None else self.temperature
        one_hot = F.gumbel_softmax(logits, tau=temp, dim=1, hard=self.straight_through)

        if self.straight_through and self.reinmax:
            # ReinMax second-order straight-through (arXiv:2304.08612 alg. 2;
            # reference dalle_pytorch.py:236-244)
            one_hot = one_hot.detach()
            p0 = logits.softmax(dim=1)
            p1 = (one_hot + (logits / temp).softmax(dim=1)) / 2
            p1 = ((_log(p1) - logits).detach() + logits).softmax(dim=1)
            p2 = 2 * p1 - 0.5 * p0
            one_hot = p2 - p2.detach() + one_hot

        sampled = torch.einsum('bnhw,nd->bdhw', one_hot, self.codebook.weight)
        out = _run_as_gemms(self.decoder, sampled) if gemm else self.decoder(sampled)

        if not return_loss:
            return out

        recon_loss = self.loss_fn(img, out)

        logits = logits.permute(0, 2, 3, 1).reshape(img.shape[0], -1, num_tokens)
        log_qy = F.log_softmax(logits, dim=-1)
        log_uniform = torch.log(torch.tensor([1. / num_tokens], device=img.device))
        kl_div = F.kl_div(log_uniform, log_qy, None, None, 'batchmean', log_target=True)

        loss = recon_loss + kl_div * self.kl_div_loss_weight
        if not return_recons:
            return loss
        return loss, out
