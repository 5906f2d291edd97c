"""Checkpoint save/load with the reference container schema.

DALLE checkpoints (reference train_dalle.py:535-582):
``{'hparams', 'vae_params', 'epoch', 'version', 'vae_class_name',
'weights', 'opt_state', 'scheduler_state'}``.
VAE checkpoints (reference train_vae.py:203-223): ``{'hparams', 'weights'}``.
Model hyperparameters live inside the checkpoint and are rehydrated on
resume/generate — same convention as the reference.
"""

from pathlib import Path

import torch

from dalle_pytorch_amd.version import __version__


def save_dalle_checkpoint(path, dalle, dalle_params, vae_params, *, epoch=0,
                          vae_class_name='DiscreteVAE', opt=None, scheduler=None):
    state = {
        'hparams': dalle_params,
        'vae_params': vae_params,
        'epoch': epoch,
        'version': __version__,
        'vae_class_name': vae_class_name,
        'weights': dalle.state_dict(),
        'opt_state': opt.state_dict() if opt is not None else None,
        'scheduler_state': scheduler.state_dict() if scheduler is not None else None,
    }
    Path(path).parent.mkdir(parents=True, exist_ok=True)
    torch.save(state, path)


def load_dalle_checkpoint(path, map_location='cpu'):
    path = Path(path)
    assert path.exists(), f'DALL-E checkpoint {path} does not exist'
    ckpt = torch.load(str(path), map_location=map_location, weights_only=False)
    assert 'hparams' in ckpt and 'weights' in ckpt, \
        f'{path} is not a DALL-E checkpoint (missing hparams/weights)'
    return ckpt


def build_dalle_from_checkpoint(ckpt, vae=None, strict=True, extra_hparams=None):
    """Rebuild VAE + DALLE from a checkpoint dict (reference generate.py:82-107)."""
    from dalle_pytorch_amd import DALLE, DiscreteVAE, OpenAIDiscreteVAE, VQGanVAE

    vae_params = ckpt.get('vae_params')
    cls_name = ckpt.get('vae_class_name', 'DiscreteVAE')
    if vae is None:
        if vae_params is not None:
            vae = DiscreteVAE(**vae_params)
        elif cls_name == 'OpenAIDiscreteVAE':
            vae = OpenAIDiscreteVAE()
        elif cls_name == 'VQGanVAE':
            vae = VQGanVAE()
        else:
            raise ValueError(f'cannot reconstruct VAE of class {cls_name}')
    else:
        assert type(vae).__name__ == cls_name, \
            f'checkpoint was trained with {cls_name}, got {type(vae).__name__}'

    hparams = dict(ckpt['hparams'])
    hparams.update(extra_hparams or {})
    dalle = DALLE(vae=vae, **hparams)
    # adopt the checkpoint's storage dtype (a reference --fp16 run saves
    # fp16 weights via dalle.half(), train_dalle.py:430-432) — plain
    # load_state_dict would silently cast them up to the fresh model's fp32
    float_dtypes = {t.dtype for t in ckpt['weights'].values()
                    if torch.is_tensor(t) and t.is_floating_point()}
    if len(float_dtypes) == 1:
        dalle = dalle.to(next(iter(float_dtypes)))
    dalle.load_state_dict(ckpt['weights'], strict=strict)
    return dalle, vae


def save_vae_checkpoint(path, vae, vae_params):
    torch.save({'hparams': vae_params, 'weights': vae.state_dict()}, path)


def load_vae_checkpoint(path, map_location='cpu'):
    path = Path(path)
    assert path.exists(), f'VAE checkpoint {path} does not exist'
    ckpt = torch.load(str(path), map_location=map_location, weights_only=False)
    from dalle_pytorch_amd import DiscreteVAE
    vae = DiscreteVAE(**ckpt['hparams'])
    vae.load_state_dict(ckpt['weights'])
    return vae, ckpt['hparams']


def rotate_checkpoints(directory, pattern, keep_n):
    """Delete oldest matching checkpoints beyond ``keep_n`` (reference
    train_dalle.py:547-550)."""
    if keep_n is None or keep_n <= 0:
        return
    files = sorted(Path(directory).glob(pattern), key=lambda p: p.stat().st_mtime)
    for old in files[:-keep_n]:
        try:
            old.unlink()
        except OSError:
            pass


def save_clip_checkpoint(path, clip, clip_params):
    torch.save({'hparams': clip_params, 'weights': clip.state_dict()}, path)


def load_clip_checkpoint(path, map_location='cpu'):
    path = Path(path)
    assert path.exists(), f'CLIP checkpoint {path} does not exist'
    ckpt = torch.load(str(path), map_location=map_location, weights_only=False)
    from dalle_pytorch_amd import CLIP
    clip = CLIP(**ckpt['hparams'])
    clip.load_state_dict(ckpt['weights'])
    return clip, ckpt['hparams']
