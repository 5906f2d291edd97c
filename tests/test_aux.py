"""Auxiliary subsystems: tar streaming, conversion mapping, flops profiler,
rainbow convergence, bench distributed launch."""

import io
import json
import subprocess
import sys
import tarfile
from pathlib import Path

import pytest
import torch

REPO = Path(__file__).resolve().parents[1]
sys.path.insert(0, str(REPO))


def _make_shard(path, n, size=32):
    from dalle_pytorch_amd.utils.vision import tensor_to_pil
    with tarfile.open(path, 'w') as tf:
        for i in range(n):
            img = tensor_to_pil(torch.rand(3, size, size))
            buf = io.BytesIO()
            img.save(buf, format='PNG')
            data = buf.getvalue()
            info = tarfile.TarInfo(f'{i:06d}.png')
            info.size = len(data)
            tf.addfile(info, io.BytesIO(data))
            cap = f'sample number {i}'.encode()
            info = tarfile.TarInfo(f'{i:06d}.txt')
            info.size = len(cap)
            tf.addfile(info, io.BytesIO(cap))


def test_tar_streaming_dataset(tmp_path):
    from dalle_pytorch_amd.utils.wds import TarImageTextDataset, expand_shards
    from dalle_pytorch_amd.utils.tokenizer import SimpleTokenizer
    for s in range(2):
        _make_shard(tmp_path / f'shard-{s:03d}.tar', 5)

    assert len(expand_shards(str(tmp_path / 'shard-{000..001}.tar'))) == 2
    assert len(expand_shards(str(tmp_path / '*.tar'))) == 2

    ds = TarImageTextDataset(str(tmp_path / '*.tar'),
                             tokenizer=SimpleTokenizer(), text_len=16,
                             image_size=32)
    samples = list(ds)
    assert len(samples) == 10
    tokens, image = samples[0]
    assert tokens.shape == (16,) and image.shape == (3, 32, 32)

    from torch.utils.data import DataLoader
    dl = DataLoader(ds, batch_size=4, drop_last=True)
    batch = next(iter(dl))
    assert batch[0].shape == (4, 16) and batch[1].shape == (4, 3, 32, 32)


def test_tar_dataset_repeats_shards_when_fewer_than_ranks(tmp_path, monkeypatch):
    """1 shard, 2 ranks: rank 1 must still yield data (a zero-batch rank
    stalls the other ranks' gradient all-reduce forever)."""
    from dalle_pytorch_amd.utils import wds as wds_mod
    from dalle_pytorch_amd.utils.tokenizer import SimpleTokenizer
    _make_shard(tmp_path / 'shard-000.tar', 3)
    ds = wds_mod.TarImageTextDataset(str(tmp_path / '*.tar'),
                                     tokenizer=SimpleTokenizer(), text_len=16,
                                     image_size=32)
    import dalle_pytorch_amd.parallel as par
    monkeypatch.setattr(par, 'get_world_size', lambda: 2)
    for rank in (0, 1):
        monkeypatch.setattr(par, 'get_rank', lambda r=rank: r)
        with pytest.warns(RuntimeWarning, match='repeating shards'):
            assert len(ds._my_shards()) >= 1
        assert len(list(ds)) == 3


def test_vqgan_downsample_matches_taming_asymmetric_pad():
    """taming pads (0,1,0,1) with a padding=0 stride-2 conv; a symmetric
    padding=1 conv is spatially shifted and decodes real ckpts wrong."""
    import torch.nn.functional as F
    from dalle_pytorch_amd.models.vae_adapters import _VqDownsample
    ds = _VqDownsample(4)
    x = torch.randn(2, 4, 16, 16)
    want = F.conv2d(F.pad(x, (0, 1, 0, 1)), ds.conv.weight, ds.conv.bias, stride=2)
    assert torch.allclose(ds(x), want)
    # and it must differ from the symmetric-pad variant on generic input
    sym = F.conv2d(x, ds.conv.weight, ds.conv.bias, stride=2, padding=1)
    assert ds(x).shape == sym.shape and not torch.allclose(ds(x), sym)


def test_vqgan_gumbel_proj_consumed_and_rejects_mismatch(tmp_path):
    from dalle_pytorch_amd import VQGanVAE
    g = VQGanVAE(image_size=32, num_tokens=128, embed_dim=16, ch=16,
                 ch_mult=(1, 2), num_res_blocks=1, gumbel=True)
    sd = g.state_dict()
    assert 'quantize.proj.weight' in sd  # taming GumbelQuantize layout
    ck = tmp_path / 'g.ckpt'
    torch.save({'state_dict': sd}, ck)
    # round-trips into a fresh gumbel module
    g2 = VQGanVAE(vqgan_model_path=str(ck), image_size=32, num_tokens=128,
                  embed_dim=16, ch=16, ch_mult=(1, 2), num_res_blocks=1,
                  gumbel=True)
    img = torch.rand(1, 3, 32, 32)
    assert torch.equal(g.get_codebook_indices(img), g2.get_codebook_indices(img))
    # a gumbel checkpoint must NOT silently load into a non-gumbel module
    with pytest.raises(RuntimeError):
        VQGanVAE(vqgan_model_path=str(ck), image_size=32, num_tokens=128,
                 embed_dim=16, ch=16, ch_mult=(1, 2), num_res_blocks=1,
                 gumbel=False)


def test_openai_dvae_conversion_mapping():
    """Synthesize a dall_e-layout state dict with matching shapes; the
    converter must cover every parameter of our module."""
    sys.path.insert(0, str(REPO / 'scripts'))
    import convert_openai_dvae as conv
    from dalle_pytorch_amd import OpenAIDiscreteVAE

    vae = OpenAIDiscreteVAE(n_hid=32, vocab_size=64)

    def fake_block_state(prefix, block, out):
        import torch.nn as nn
        if isinstance(block.id_path, nn.Conv2d):
            out[f'{prefix}id_path.w'] = torch.randn_like(block.id_path.weight)
            out[f'{prefix}id_path.b'] = torch.randn_like(block.id_path.bias)
        convs = [m for m in block.res_path if isinstance(m, nn.Conv2d)]
        for i, c in enumerate(convs, start=1):
            out[f'{prefix}res_path.conv_{i}.w'] = torch.randn_like(c.weight)
            out[f'{prefix}res_path.conv_{i}.b'] = torch.randn(1, c.out_channels, 1, 1)

    enc, dec = {}, {}
    enc['blocks.input.w'] = torch.randn_like(vae.encoder.input.weight)
    enc['blocks.input.b'] = torch.randn(1, 32, 1, 1)
    for g in range(1, 5):
        for j in range(1, 3):
            fake_block_state(f'blocks.group_{g}.block_{j}.',
                             getattr(vae.encoder, f'group_{g}')[j - 1], enc)
    enc['blocks.output.conv.w'] = torch.randn_like(vae.encoder.output[1].weight)
    enc['blocks.output.conv.b'] = torch.randn(1, 64, 1, 1)

    dec['blocks.input.w'] = torch.randn(16, 64, 1, 1)  # [n_init, vocab, 1, 1]
    for g in range(1, 5):
        for j in range(1, 3):
            fake_block_state(f'blocks.group_{g}.block_{j}.',
                             getattr(vae.decoder, f'group_{g}')[j - 1], dec)
    dec['blocks.output.conv.w'] = torch.randn_like(vae.decoder.output[1].weight)
    dec['blocks.output.conv.b'] = torch.randn(1, 3 * 2, 1, 1)

    state = conv.convert(enc, dec)
    missing, unexpected = vae.load_state_dict(state, strict=False)
    assert not unexpected, unexpected[:5]
    assert not missing, missing[:5]
    # codebook is the transposed decoder input conv
    assert torch.equal(state['codebook.weight'],
                       dec['blocks.input.w'][:, :, 0, 0].t())


def test_flops_profiler_table(capsys):
    from dalle_pytorch_amd import DALLE, DiscreteVAE
    from dalle_pytorch_amd.utils.flops import profile_step
    vae = DiscreteVAE(image_size=64, num_layers=3, num_tokens=64,
                      codebook_dim=32, hidden_dim=8)
    d = DALLE(dim=64, vae=vae, num_text_tokens=100, text_seq_len=8, depth=2,
              heads=2, dim_head=32, attn_types=('full', 'axial_row'),
              shift_tokens=True)
    fl = profile_step(d, batch_size=2, step_time_s=0.1)
    out = capsys.readouterr().out
    assert 'achieved' in out and 'layer0.attn' in out
    assert fl > 0


def test_vqgan_yaml_config(tmp_path):
    import yaml
    from dalle_pytorch_amd import VQGanVAE
    cfg = {'model': {'target': 'taming.models.vqgan.VQModel',
                     'params': {'n_embed': 128, 'embed_dim': 32,
                                'ddconfig': {'ch': 16, 'ch_mult': [1, 2],
                                             'num_res_blocks': 1,
                                             'resolution': 32}}}}
    p = tmp_path / 'cfg.yaml'
    p.write_text(yaml.safe_dump(cfg))
    vae = VQGanVAE(vqgan_config_path=str(p))
    assert vae.num_tokens == 128 and vae.num_layers == 1
    codes = vae.get_codebook_indices(torch.rand(1, 3, 32, 32))
    assert codes.max() < 128


@pytest.mark.slow
def test_rainbow_overfits_cpu():
    """Tiny end-to-end: loss must clearly fall on the compositional data
    (the reference notebook's correctness signal, bounded for CI time)."""
    sys.path.insert(0, str(REPO / 'examples'))
    import train_rainbow
    first, last, acc = train_rainbow.main([
        '--samples', '32', '--vae_steps', '30', '--dalle_steps', '60',
        '--dim', '64', '--depth', '1', '--batch_size', '8'])
    assert last < first * 0.7, (first, last)


def test_bench_distributed_launch_cpu():
    """bench.py tiny config under torch.distributed.run, 2 ranks on CPU
    (gloo): the exact launch shape the driver uses for multi-GPU."""
    cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
           '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
           '--master-port', '29701', str(REPO / 'bench.py'),
           '--config', 'tiny', '--steps', '2', '--warmup', '1']
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                         cwd=str(REPO))
    assert res.returncode == 0, res.stderr[-2000:]
    line = [l for l in res.stdout.splitlines() if l.startswith('{')][-1]
    rec = json.loads(line)
    assert rec['n_gpus'] == 2
    assert rec['config']['parallelism'] == 'dp2'
    assert rec['value'] > 0


def test_clip_train_and_rerank(tmp_path):
    """CLIP example trains (loss falls) and the saved checkpoint loads for
    re-ranking."""
    sys.path.insert(0, str(REPO / 'examples'))
    import train_clip
    out = tmp_path / 'clip.pt'
    first, last = train_clip.main(['--steps', '40', '--dim', '64',
                                   '--depth', '1', '--batch_size', '16',
                                   '--out', str(out)])
    assert last < first
    from dalle_pytorch_amd.utils.checkpoint import load_clip_checkpoint
    clip, _ = load_clip_checkpoint(out)
    text = torch.randint(0, 100, (4, 16))
    imgs = torch.rand(4, 3, 32, 32)
    scores = clip(text, imgs)
    assert scores.shape == (4,)
