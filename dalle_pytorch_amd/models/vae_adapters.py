"""Native adapters replacing the reference's downloaded pretrained VAEs.

The reference (vae.py:111-232) unpickles OpenAI's released dVAE modules
(pinned to torch<1.11, incompatible with ROCm torch 2.10) and wraps
taming-transformers' VQGAN through OmegaConf. Here both are first-party
modules with the same uniform interface the DALLE class consumes
(``image_size/num_tokens/num_layers/channels``, ``get_codebook_indices``,
``decode``) and state-dict loaders for offline-supplied weights:

* :class:`OpenAIDiscreteVAE` re-implements the dall_e encoder/decoder
  architecture (bottleneck res-blocks, maxpool/upsample pyramid) so the
  published ``encoder.pkl``/``decoder.pkl`` *state dicts* convert key-for-key
  — no unpickling of module objects.
* :class:`VQGanVAE` re-implements taming's VQModel/GumbelVQ encoder,
  decoder and quantizer with taming's attribute naming, so a taming
  ``state_dict`` loads directly; random init otherwise (BASELINE config D
  uses the 16384-codebook f=16 shape with random weights).
"""

import math
from collections import OrderedDict

import torch
import torch.nn.functional as F
from torch import nn

# --------------------------------------------------------------------------
# OpenAI dVAE (dall_e package architecture)
# --------------------------------------------------------------------------

LOGIT_LAPLACE_EPS = 0.1


def map_pixels(x):
    """[0,1] -> logit-laplace domain (reference vae.py:49-50)."""
    return (1 - 2 * LOGIT_LAPLACE_EPS) * x + LOGIT_LAPLACE_EPS


def unmap_pixels(x):
    return torch.clamp((x - LOGIT_LAPLACE_EPS) / (1 - 2 * LOGIT_LAPLACE_EPS), 0, 1)


class _OAIEncBlock(nn.Module):
    """dall_e bottleneck residual block: 3 relu-conv3 + relu-conv1, id path
    gains a 1x1 conv when channel counts differ."""

    def __init__(self, n_in, n_out):
        super().__init__()
        n_hid = n_out // 4
        self.id_path = nn.Conv2d(n_in, n_out, 1) if n_in != n_out else nn.Identity()
        self.res_path = nn.Sequential(
            nn.ReLU(), nn.Conv2d(n_in, n_hid, 3, padding=1),
            nn.ReLU(), nn.Conv2d(n_hid, n_hid, 3, padding=1),
            nn.ReLU(), nn.Conv2d(n_hid, n_hid, 3, padding=1),
            nn.ReLU(), nn.Conv2d(n_hid, n_out, 1))

    def forward(self, x):
        return self.id_path(x) + self.res_path(x)


class _OAIDecBlock(nn.Module):
    """dall_e decoder block: 1 relu-conv1 + 3 relu-conv3."""

    def __init__(self, n_in, n_out):
        super().__init__()
        n_hid = n_out // 4
        self.id_path = nn.Conv2d(n_in, n_out, 1) if n_in != n_out else nn.Identity()
        self.res_path = nn.Sequential(
            nn.ReLU(), nn.Conv2d(n_in, n_hid, 1),
            nn.ReLU(), nn.Conv2d(n_hid, n_hid, 3, padding=1),
            nn.ReLU(), nn.Conv2d(n_hid, n_hid, 3, padding=1),
            nn.ReLU(), nn.Conv2d(n_hid, n_out, 3, padding=1))

    def forward(self, x):
        return self.id_path(x) + self.res_path(x)


class OpenAIDiscreteVAE(nn.Module):
    """256x256 -> 32x32 tokens, 8192-way codebook (reference vae.py:111-143)."""

    def __init__(self, n_hid=256, vocab_size=8192, blocks_per_group=2):
        super().__init__()
        self.image_size = 256
        self.num_layers = 3
        self.num_tokens = vocab_size
        self.channels = 3

        nh = n_hid
        def group(n_in, n_out, blocks, enc=True):
            cls = _OAIEncBlock if enc else _OAIDecBlock
            mods = [cls(n_in if i == 0 else n_out, n_out) for i in range(blocks)]
            return nn.Sequential(*mods)

        n_init = nh // 2  # dall_e n_init = 128 at n_hid = 256
        self.encoder = nn.Sequential(OrderedDict([
            ('input', nn.Conv2d(3, nh, 7, padding=3)),
            ('group_1', group(nh, nh, blocks_per_group)),
            ('pool_1', nn.MaxPool2d(2)),
            ('group_2', group(nh, 2 * nh, blocks_per_group)),
            ('pool_2', nn.MaxPool2d(2)),
            ('group_3', group(2 * nh, 4 * nh, blocks_per_group)),
            ('pool_3', nn.MaxPool2d(2)),
            ('group_4', group(4 * nh, 8 * nh, blocks_per_group)),
            ('output', nn.Sequential(nn.ReLU(), nn.Conv2d(8 * nh, vocab_size, 1))),
        ]))
        # dall_e's decoder "input" is Conv2d(vocab, n_init, 1) applied to the
        # one-hot codes — numerically an embedding table, stored here as one
        # so decode() is a gather instead of an 8192-channel conv
        self.codebook = nn.Embedding(vocab_size, n_init)
        self.decoder = nn.Sequential(OrderedDict([
            ('group_1', group(n_init, 8 * nh, blocks_per_group, enc=False)),
            ('up_1', nn.Upsample(scale_factor=2, mode='nearest')),
            ('group_2', group(8 * nh, 4 * nh, blocks_per_group, enc=False)),
            ('up_2', nn.Upsample(scale_factor=2, mode='nearest')),
            ('group_3', group(4 * nh, 2 * nh, blocks_per_group, enc=False)),
            ('up_3', nn.Upsample(scale_factor=2, mode='nearest')),
            ('group_4', group(2 * nh, nh, blocks_per_group, enc=False)),
            ('output', nn.Sequential(nn.ReLU(), nn.Conv2d(nh, 6, 1))),
        ]))

    @torch.no_grad()
    def get_codebook_indices(self, img):
        img = map_pixels(img)
        if img.is_cuda:
            if not getattr(self, '_encoder_channels_last', False):
                self.encoder.to(memory_format=torch.channels_last)
                self._encoder_channels_last = True
            img = img.contiguous(memory_format=torch.channels_last)
        logits = self.encoder(img)
        return logits.argmax(dim=1).flatten(1)

    def decode(self, img_seq):
        b, n = img_seq.shape
        hw = int(math.sqrt(n))
        z = self.codebook(img_seq)                       # [b, n, d]
        z = z.reshape(b, hw, hw, -1).permute(0, 3, 1, 2)
        out = self.decoder(z)
        return unmap_pixels(torch.sigmoid(out[:, :3]))

    def forward(self, img):
        raise NotImplementedError('OpenAIDiscreteVAE is inference-only '
                                  '(reference vae.py:140-143)')


# --------------------------------------------------------------------------
# VQGAN (taming-transformers VQModel / GumbelVQ architecture)
# --------------------------------------------------------------------------

def _gn(c):
    # taming uses 32 groups (channels are multiples of 32 in real configs);
    # degrade gracefully for slim test shapes
    g = 32
    while c % g:
        g //= 2
    return nn.GroupNorm(g, c, eps=1e-6, affine=True)


class _VqResBlock(nn.Module):
    def __init__(self, c_in, c_out):
        super().__init__()
        self.norm1 = _gn(c_in)
        self.conv1 = nn.Conv2d(c_in, c_out, 3, padding=1)
        self.norm2 = _gn(c_out)
        self.conv2 = nn.Conv2d(c_out, c_out, 3, padding=1)
        self.nin_shortcut = nn.Conv2d(c_in, c_out, 1) if c_in != c_out else nn.Identity()

    def forward(self, x):
        h = self.conv1(F.silu(self.norm1(x)))
        h = self.conv2(F.silu(self.norm2(h)))
        return self.nin_shortcut(x) + h


class _VqAttnBlock(nn.Module):
    """Single-head spatial self-attention (taming's AttnBlock); on GPU this
    runs through rocBLAS batched GEMMs."""

    def __init__(self, c):
        super().__init__()
        self.norm = _gn(c)
        self.q = nn.Conv2d(c, c, 1)
        self.k = nn.Conv2d(c, c, 1)
        self.v = nn.Conv2d(c, c, 1)
        self.proj_out = nn.Conv2d(c, c, 1)

    def forward(self, x):
        h = self.norm(x)
        q, k, v = self.q(h), self.k(h), self.v(h)
        b, c, hh, ww = q.shape
        q = q.reshape(b, c, hh * ww).permute(0, 2, 1)
        k = k.reshape(b, c, hh * ww)
        attn = torch.softmax(torch.bmm(q, k) * c ** -0.5, dim=-1)
        v = v.reshape(b, c, hh * ww)
        out = torch.bmm(v, attn.transpose(1, 2)).reshape(b, c, hh, ww)
        return x + self.proj_out(out)


class _VqDownsample(nn.Module):
    """Strided conv with taming's ASYMMETRIC (0,1,0,1) zero pad — a plain
    padding=1 conv gives the same shape but spatially shifted activations,
    so real taming checkpoints would decode off-by-half-a-pixel. Nested
    ``conv`` attribute matches taming's ``down.{i}.downsample.conv.*`` keys."""

    def __init__(self, c):
        super().__init__()
        self.conv = nn.Conv2d(c, c, 3, stride=2, padding=0)

    def forward(self, x):
        return self.conv(F.pad(x, (0, 1, 0, 1)))


class _VqUpsample(nn.Module):
    """Nearest 2x upsample + conv (taming key layout ``up.{i}.upsample.conv.*``)."""

    def __init__(self, c):
        super().__init__()
        self.conv = nn.Conv2d(c, c, 3, padding=1)

    def forward(self, x):
        return self.conv(F.interpolate(x, scale_factor=2.0, mode='nearest'))


class _VqEncoder(nn.Module):
    def __init__(self, ch, ch_mult, num_res_blocks, z_channels, in_ch=3,
                 resolution=256, attn_resolutions=()):
        super().__init__()
        self.conv_in = nn.Conv2d(in_ch, ch, 3, padding=1)
        self.down = nn.ModuleList()
        c = ch
        res = resolution
        for i, m in enumerate(ch_mult):
            stage = nn.Module()
            stage.block = nn.ModuleList(
                [_VqResBlock(c if j == 0 else ch * m, ch * m) for j in range(num_res_blocks)])
            c = ch * m
            stage.attn = nn.ModuleList(
                [_VqAttnBlock(c) for _ in range(num_res_blocks)]
                if res in attn_resolutions else [])
            if i != len(ch_mult) - 1:
                stage.downsample = _VqDownsample(c)
                res //= 2
            else:
                stage.downsample = None
            self.down.append(stage)
        self.mid = nn.Module()
        self.mid.block_1 = _VqResBlock(c, c)
        self.mid.attn_1 = _VqAttnBlock(c)
        self.mid.block_2 = _VqResBlock(c, c)
        self.norm_out = _gn(c)
        self.conv_out = nn.Conv2d(c, z_channels, 3, padding=1)

    def forward(self, x):
        h = self.conv_in(x)
        for stage in self.down:
            for j, blk in enumerate(stage.block):
                h = blk(h)
                if len(stage.attn):
                    h = stage.attn[j](h)
            if stage.downsample is not None:
                h = stage.downsample(h)
        h = self.mid.block_2(self.mid.attn_1(self.mid.block_1(h)))
        return self.conv_out(F.silu(self.norm_out(h)))


class _VqDecoder(nn.Module):
    def __init__(self, ch, ch_mult, num_res_blocks, z_channels, out_ch=3,
                 resolution=256, attn_resolutions=()):
        super().__init__()
        c = ch * ch_mult[-1]
        self.conv_in = nn.Conv2d(z_channels, c, 3, padding=1)
        self.mid = nn.Module()
        self.mid.block_1 = _VqResBlock(c, c)
        self.mid.attn_1 = _VqAttnBlock(c)
        self.mid.block_2 = _VqResBlock(c, c)
        self.up = nn.ModuleList()
        res = resolution // 2 ** (len(ch_mult) - 1)
        for i, m in reversed(list(enumerate(ch_mult))):
            stage = nn.Module()
            stage.block = nn.ModuleList(
                [_VqResBlock(c if j == 0 else ch * m, ch * m) for j in range(num_res_blocks + 1)])
            c = ch * m
            stage.attn = nn.ModuleList(
                [_VqAttnBlock(c) for _ in range(num_res_blocks + 1)]
                if res in attn_resolutions else [])
            if i != 0:
                stage.upsample = _VqUpsample(c)
                res *= 2
            else:
                stage.upsample = None
            self.up.insert(0, stage)
        self.norm_out = _gn(c)
        self.conv_out = nn.Conv2d(c, out_ch, 3, padding=1)

    def forward(self, z):
        h = self.conv_in(z)
        h = self.mid.block_2(self.mid.attn_1(self.mid.block_1(h)))
        for stage in reversed(self.up):
            for j, blk in enumerate(stage.block):
                h = blk(h)
                if len(stage.attn):
                    h = stage.attn[j](h)
            if stage.upsample is not None:
                h = stage.upsample(h)
        return self.conv_out(F.silu(self.norm_out(h)))


class VQGanVAE(nn.Module):
    """Native VQGAN with taming-style attribute naming.

    Default shape matches BASELINE config D: f=16 (num_layers=4), 16384-way
    codebook, 256-dim embeddings. ``gumbel=True`` mirrors taming's GumbelVQ
    (reference vae.py:193,210-229). ``vqgan_model_path`` may point at a
    taming checkpoint whose ``state_dict`` loads into this module.
    """

    def __init__(self, vqgan_model_path=None, vqgan_config_path=None, *,
                 image_size=256, num_tokens=16384, embed_dim=256, ch=128,
                 ch_mult=(1, 1, 2, 2, 4), num_res_blocks=2, gumbel=False,
                 attn_resolutions=(16,)):
        super().__init__()
        if vqgan_config_path is not None:
            # taming-transformers yaml config (reference vae.py:148-158 uses
            # OmegaConf; plain yaml suffices for the fields we need)
            import yaml
            with open(vqgan_config_path) as f:
                cfg = yaml.safe_load(f)
            params = cfg['model']['params']
            gumbel = 'Gumbel' in cfg['model'].get('target', '')
            dd = params['ddconfig']
            ch = dd.get('ch', ch)
            ch_mult = tuple(dd.get('ch_mult', ch_mult))
            num_res_blocks = dd.get('num_res_blocks', num_res_blocks)
            image_size = dd.get('resolution', image_size)
            attn_resolutions = tuple(dd.get('attn_resolutions', attn_resolutions))
            num_tokens = params.get('n_embed', num_tokens)
            embed_dim = params.get('embed_dim', embed_dim)
        f = 2 ** (len(ch_mult) - 1)
        self.image_size = image_size
        self.num_tokens = num_tokens
        self.num_layers = int(math.log2(f))
        self.channels = 3
        self.gumbel = gumbel

        z_ch = embed_dim
        self.encoder = _VqEncoder(ch, ch_mult, num_res_blocks, z_ch,
                                  resolution=image_size,
                                  attn_resolutions=attn_resolutions)
        self.decoder = _VqDecoder(ch, ch_mult, num_res_blocks, z_ch,
                                  resolution=image_size,
                                  attn_resolutions=attn_resolutions)
        self.quant_conv = nn.Conv2d(z_ch, embed_dim, 1)
        self.post_quant_conv = nn.Conv2d(embed_dim, z_ch, 1)
        self.quantize = nn.Module()
        if gumbel:
            # taming GumbelQuantize: learned 1x1 proj -> vocab logits, then
            # an embedding lookup; indices come from proj logits, NOT from a
            # similarity against the codebook
            self.quantize.proj = nn.Conv2d(embed_dim, num_tokens, 1)
            self.quantize.embed = nn.Embedding(num_tokens, embed_dim)
        else:
            self.quantize.embedding = nn.Embedding(num_tokens, embed_dim)

        if vqgan_model_path is not None:
            state = torch.load(vqgan_model_path, map_location='cpu')
            state = state.get('state_dict', state)
            missing, unexpected = self.load_state_dict(state, strict=False)
            if missing:
                raise RuntimeError(f'VQGAN checkpoint missing keys: {missing[:8]}...')
            # silently dropping quantizer weights would produce wrong codes
            bad = [k for k in unexpected if k.startswith('quantize.')]
            if bad:
                raise RuntimeError(
                    f'VQGAN checkpoint has quantizer keys this module did not '
                    f'consume (wrong gumbel= setting?): {bad[:8]}')

    def _codebook_weight(self):
        return self.quantize.embed.weight if self.gumbel else self.quantize.embedding.weight

    @torch.no_grad()
    def get_codebook_indices(self, img):
        img = 2 * img - 1  # reference vae.py:212
        if img.is_cuda:
            if not getattr(self, '_encoder_channels_last', False):
                self.encoder.to(memory_format=torch.channels_last)
                self.quant_conv.to(memory_format=torch.channels_last)
                self._encoder_channels_last = True
            img = img.contiguous(memory_format=torch.channels_last)
        z = self.quant_conv(self.encoder(img))
        b, c, h, w = z.shape
        if self.gumbel:
            # GumbelVQ: learned proj produces vocab logits per position
            logits = self.quantize.proj(z)               # [b, vocab, h, w]
            idx = logits.argmax(dim=1).reshape(b, h * w)
            return idx
        flat = z.permute(0, 2, 3, 1).reshape(-1, c)
        book = self._codebook_weight()
        d = (flat.pow(2).sum(1, keepdim=True)
             - 2 * flat @ book.t()
             + book.pow(2).sum(1))
        idx = d.argmin(dim=-1)
        return idx.reshape(b, h * w)

    def decode(self, img_seq):
        b, n = img_seq.shape
        hw = int(math.sqrt(n))
        z = F.embedding(img_seq, self._codebook_weight())
        z = z.reshape(b, hw, hw, -1).permute(0, 3, 1, 2)
        img = self.decoder(self.post_quant_conv(z))
        return (img.clamp(-1., 1.) + 1) * 0.5  # reference vae.py:227-229

    def forward(self, img):
        raise NotImplementedError('VQGanVAE is used frozen for DALLE training '
                                  '(reference vae.py:231-232)')
