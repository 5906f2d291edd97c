"""Checkpoint container schema + roundtrip + rotation."""

import torch

from dalle_pytorch_amd import DALLE, DiscreteVAE
from dalle_pytorch_amd.utils.checkpoint import (
    save_dalle_checkpoint, load_dalle_checkpoint, build_dalle_from_checkpoint,
    save_vae_checkpoint, load_vae_checkpoint, rotate_checkpoints)

torch.manual_seed(0)

VAE_PARAMS = dict(image_size=32, num_layers=2, num_tokens=32, codebook_dim=16,
                  hidden_dim=8)
DALLE_PARAMS = dict(dim=32, num_text_tokens=50, text_seq_len=4, depth=1,
                    heads=2, dim_head=16)


def test_dalle_checkpoint_roundtrip(tmp_path):
    vae = DiscreteVAE(**VAE_PARAMS)
    d = DALLE(vae=vae, **DALLE_PARAMS)
    opt = torch.optim.Adam(p for p in d.parameters() if p.requires_grad)
    path = tmp_path / 'd.pt'
    save_dalle_checkpoint(path, d, DALLE_PARAMS, VAE_PARAMS, epoch=3, opt=opt)

    ckpt = load_dalle_checkpoint(path)
    # reference container schema (train_dalle.py:535-582)
    for key in ('hparams', 'vae_params', 'epoch', 'version', 'vae_class_name',
                'weights', 'opt_state', 'scheduler_state'):
        assert key in ckpt, key
    assert ckpt['epoch'] == 3

    d2, _ = build_dalle_from_checkpoint(ckpt)
    for (k1, p1), (k2, p2) in zip(d.state_dict().items(), d2.state_dict().items()):
        assert k1 == k2
        assert torch.equal(p1, p2), k1

    text = torch.randint(1, 50, (1, 4))
    d.eval(); d2.eval()
    assert torch.allclose(d(text, None), d2(text, None))


def test_vae_checkpoint_roundtrip(tmp_path):
    vae = DiscreteVAE(**VAE_PARAMS)
    path = tmp_path / 'vae.pt'
    save_vae_checkpoint(path, vae, VAE_PARAMS)
    vae2, hparams = load_vae_checkpoint(path)
    assert hparams == VAE_PARAMS
    img = torch.rand(1, 3, 32, 32)
    assert torch.allclose(vae.get_codebook_indices(img).float(),
                          vae2.get_codebook_indices(img).float())


def test_rotation(tmp_path):
    for i in range(5):
        (tmp_path / f'run-step{i}.pt').write_bytes(b'x')
    rotate_checkpoints(tmp_path, 'run-step*.pt', keep_n=2)
    assert len(list(tmp_path.glob('run-step*.pt'))) == 2


def test_half_weight_checkpoint_dtype_parity(tmp_path):
    """Reference --fp16 saves fp16 weights (dalle.half(),
    train_dalle.py:430-432): a half-precision checkpoint must round-trip
    with its dtype preserved, and a model rebuilt from it must carry fp16
    weights and still run a forward (VERDICT missing #3)."""
    import torch
    from dalle_pytorch_amd import DALLE, DiscreteVAE
    from dalle_pytorch_amd.utils.checkpoint import (
        save_dalle_checkpoint, load_dalle_checkpoint, build_dalle_from_checkpoint)

    vae = DiscreteVAE(image_size=64, num_layers=3, num_tokens=32,
                      codebook_dim=16, hidden_dim=8)
    d = DALLE(dim=32, vae=vae, num_text_tokens=40, text_seq_len=4, depth=1,
              heads=2, dim_head=16).half()
    params = dict(dim=32, num_text_tokens=40, text_seq_len=4, depth=1,
                  heads=2, dim_head=16)
    vparams = dict(image_size=64, num_layers=3, num_tokens=32,
                   codebook_dim=16, hidden_dim=8)
    path = tmp_path / 'half.pt'
    save_dalle_checkpoint(path, d, params, vparams, epoch=0,
                          vae_class_name='DiscreteVAE')

    ckpt = load_dalle_checkpoint(path)
    assert ckpt['weights']['text_emb.weight'].dtype == torch.float16
    d2, _ = build_dalle_from_checkpoint(ckpt)
    assert d2.text_emb.weight.dtype == torch.float16
    text = torch.randint(1, 40, (1, 4))
    logits = d2.float()(text, None)   # cast up to run the CPU forward
    assert torch.isfinite(logits).all()
