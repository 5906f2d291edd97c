"""Minimal production-style serving front end for a trained DALLE checkpoint.

The reference ships only batch scripts (generate.py); this adds an HTTP
endpoint on the same rebuild-from-checkpoint flow, backed by the
static-shape FastDecoder so every request reuses one captured HIP graph
per (batch, guidance) shape instead of re-tracing the model:

    python examples/serve.py --dalle_path dalle.pt --port 8000
    curl -X POST localhost:8000/generate \
         -H 'content-type: application/json' \
         -d '{"text": "a red square", "num_images": 1}'

Returns base64 PNGs (or raw tensors with "format": "pt"). Requests are
serialized through a lock — decode throughput comes from batching inside a
request (num_images), which is how the decode engine is fastest anyway.
"""

import argparse
import base64
import io
import sys
import threading
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch


def build_app(dalle, device, tokenizer, max_batch=64):
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    class GenerateRequest(BaseModel):
        text: str
        num_images: int = 1
        cond_scale: float = 1.0
        temperature: float = 1.0
        filter_thres: float = 0.9
        format: str = 'png'      # 'png' | 'pt'
        seed: int | None = None

    app = FastAPI(title='dalle-pytorch-amd')
    lock = threading.Lock()
    decoders = {}

    def get_decoder(batch, guided):
        from dalle_pytorch_amd.engine import FastDecoder
        key = (batch, guided)
        if key not in decoders:
            decoders[key] = FastDecoder(
                dalle, batch_size=batch * (2 if guided else 1),
                use_graph=device.type == 'cuda')
        return decoders[key]

    @app.get('/health')
    def health():
        return {'status': 'ok', 'device': str(device),
                'image_size': dalle.vae.image_size,
                'text_seq_len': dalle.text_seq_len}

    @app.post('/generate')
    def generate(req: GenerateRequest):
        if not (1 <= req.num_images <= max_batch):
            raise HTTPException(400, f'num_images must be 1..{max_batch}')
        tokens = tokenizer.tokenize([req.text], dalle.text_seq_len,
                                    truncate_text=True).to(device)
        tokens = tokens.repeat(req.num_images, 1)
        guided = req.cond_scale != 1.0
        with lock:
            if req.seed is not None:
                torch.manual_seed(req.seed)
            try:
                dec = get_decoder(req.num_images, guided)
                images = dec.generate(tokens, temperature=req.temperature,
                                      filter_thres=req.filter_thres,
                                      cond_scale=req.cond_scale)
            except (ValueError, AssertionError):
                images = dalle.generate_images(
                    tokens, use_cache=True, temperature=req.temperature,
                    filter_thres=req.filter_thres, cond_scale=req.cond_scale)
        images = images.clamp(0, 1).cpu()
        if req.format == 'pt':
            buf = io.BytesIO()
            torch.save(images, buf)
            return {'format': 'pt',
                    'data': base64.b64encode(buf.getvalue()).decode()}
        out = []
        from dalle_pytorch_amd.utils.vision import tensor_to_pil
        for img in images:
            pil = tensor_to_pil(img)
            buf = io.BytesIO()
            pil.save(buf, format='PNG')
            out.append(base64.b64encode(buf.getvalue()).decode())
        return {'format': 'png', 'images': out}

    return app


def main(argv=None):
    p = argparse.ArgumentParser(description='serve a DALLE checkpoint')
    p.add_argument('--dalle_path', type=str, required=True)
    p.add_argument('--host', type=str, default='127.0.0.1')
    p.add_argument('--port', type=int, default=8000)
    p.add_argument('--max_batch', type=int, default=64)
    args = p.parse_args(argv)

    from dalle_pytorch_amd.utils.checkpoint import (
        build_dalle_from_checkpoint, load_dalle_checkpoint)
    from dalle_pytorch_amd.utils.tokenizer import tokenizer as tok

    device = torch.device('cuda:0') if torch.cuda.is_available() \
        else torch.device('cpu')
    ckpt = load_dalle_checkpoint(args.dalle_path)
    dalle, _ = build_dalle_from_checkpoint(ckpt)
    dalle = dalle.to(device).eval()

    app = build_app(dalle, device, tok, max_batch=args.max_batch)
    import uvicorn
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == '__main__':
    main()
