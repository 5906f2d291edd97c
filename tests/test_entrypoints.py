"""End-to-end entry points on CPU with tiny configs (the reference's
runnable-scripts bar)."""

import json
import sys
from pathlib import Path

import torch

REPO = Path(__file__).resolve().parents[1]
sys.path.insert(0, str(REPO))


def test_train_vae_synthetic(tmp_path):
    import train_vae
    train_vae.main([
        '--image_size', '32', '--num_tokens', '32', '--num_layers', '2',
        '--emb_dim', '16', '--hidden_dim', '8', '--num_resnet_blocks', '0',
        '--batch_size', '2', '--epochs', '1', '--stop_after_steps', '3',
        '--output_dir', str(tmp_path)])
    assert (tmp_path / 'vae-final.pt').exists()
    ckpt = torch.load(tmp_path / 'vae-final.pt', weights_only=False)
    assert 'hparams' in ckpt and 'weights' in ckpt


def test_train_dalle_then_generate(tmp_path):
    import train_dalle
    import generate

    # first train a tiny VAE to feed in
    import train_vae
    train_vae.main([
        '--image_size', '32', '--num_tokens', '32', '--num_layers', '2',
        '--emb_dim', '16', '--hidden_dim', '8', '--batch_size', '2',
        '--epochs', '1', '--stop_after_steps', '1',
        '--output_dir', str(tmp_path)])

    train_dalle.main([
        '--vae_path', str(tmp_path / 'vae-final.pt'), '--synthetic',
        '--dim', '32', '--depth', '1', '--heads', '2', '--dim_head', '16',
        '--text_seq_len', '8', '--batch_size', '2', '--epochs', '1',
        '--stop_after_steps', '2', '--save_every_n_steps', '2',
        '--output_dir', str(tmp_path)])
    ckpt_path = tmp_path / 'dalle.pt'
    assert ckpt_path.exists()
    ckpt = torch.load(ckpt_path, weights_only=False)
    for key in ('hparams', 'vae_params', 'weights', 'opt_state'):
        assert key in ckpt

    # resume path
    train_dalle.main([
        '--dalle_path', str(ckpt_path), '--synthetic',
        '--batch_size', '2', '--epochs', '2', '--stop_after_steps', '1',
        '--output_dir', str(tmp_path)])

    generate.main([
        '--dalle_path', str(ckpt_path), '--text', 'a tiny test',
        '--num_images', '1', '--batch_size', '1',
        '--outputs_dir', str(tmp_path / 'gen')])
    outs = list((tmp_path / 'gen').glob('**/*'))
    assert any(p.suffix in ('.png', '.pt') for p in outs)

    # classifier-free guidance on the fast path (doubled-batch decoder)
    generate.main([
        '--dalle_path', str(ckpt_path), '--text', 'a guided test',
        '--num_images', '1', '--batch_size', '1', '--cond_scale', '2.0',
        '--outputs_dir', str(tmp_path / 'gen_guided')])
    assert any(p.suffix == '.png'
               for p in (tmp_path / 'gen_guided').glob('**/*'))

    # text completion first, then fast-path image decode
    generate.main([
        '--dalle_path', str(ckpt_path), '--text', 'a', '--gentxt',
        '--num_images', '1', '--batch_size', '1',
        '--outputs_dir', str(tmp_path / 'gen_txt')])
    assert any(p.suffix == '.png'
               for p in (tmp_path / 'gen_txt').glob('**/*'))


def test_train_dalle_grad_accum(tmp_path):
    import train_vae
    import train_dalle
    train_vae.main([
        '--image_size', '32', '--num_tokens', '32', '--num_layers', '2',
        '--emb_dim', '16', '--hidden_dim', '8', '--batch_size', '2',
        '--epochs', '1', '--stop_after_steps', '1',
        '--output_dir', str(tmp_path)])
    train_dalle.main([
        '--vae_path', str(tmp_path / 'vae-final.pt'), '--synthetic',
        '--dim', '32', '--depth', '1', '--heads', '2',
        '--dim_head', '16', '--text_seq_len', '8', '--batch_size', '2',
        '--ga_steps', '2', '--epochs', '1', '--stop_after_steps', '4',
        '--output_dir', str(tmp_path)])
    assert (tmp_path / 'log.jsonl').exists()
    records = [json.loads(l) for l in (tmp_path / 'log.jsonl').read_text().splitlines()]
    assert any('loss' in r for r in records)


def test_serve_endpoint(tmp_path):
    """HTTP serving front end (examples/serve.py): health + generate round
    trip on a tiny checkpoint through Starlette's in-process TestClient."""
    import base64
    import io
    import train_vae
    import train_dalle
    train_vae.main([
        '--image_size', '32', '--num_tokens', '32', '--num_layers', '2',
        '--emb_dim', '16', '--hidden_dim', '8', '--batch_size', '2',
        '--epochs', '1', '--stop_after_steps', '1',
        '--output_dir', str(tmp_path)])
    train_dalle.main([
        '--vae_path', str(tmp_path / 'vae-final.pt'), '--synthetic',
        '--dim', '32', '--depth', '1', '--heads', '2', '--dim_head', '16',
        '--text_seq_len', '8', '--batch_size', '2', '--epochs', '1',
        '--stop_after_steps', '1', '--output_dir', str(tmp_path)])

    sys.path.insert(0, str(REPO / 'examples'))
    import serve
    from dalle_pytorch_amd.utils.checkpoint import (
        build_dalle_from_checkpoint, load_dalle_checkpoint)
    from dalle_pytorch_amd.utils.tokenizer import tokenizer as tok
    ckpt = load_dalle_checkpoint(tmp_path / 'dalle.pt')
    dalle, _ = build_dalle_from_checkpoint(ckpt)
    app = serve.build_app(dalle.eval(), torch.device('cpu'), tok)

    from starlette.testclient import TestClient
    client = TestClient(app)
    r = client.get('/health')
    assert r.status_code == 200 and r.json()['status'] == 'ok'
    r = client.post('/generate', json={
        'text': 'a tiny test', 'num_images': 1, 'seed': 0})
    assert r.status_code == 200, r.text
    body = r.json()
    assert body['format'] == 'png' and len(body['images']) == 1
    from PIL import Image
    img = Image.open(io.BytesIO(base64.b64decode(body['images'][0])))
    assert img.size == (32, 32)
    r = client.post('/generate', json={'text': 'x', 'num_images': 999})
    assert r.status_code == 400
