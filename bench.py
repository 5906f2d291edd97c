#!/usr/bin/env python3
"""Benchmark: flagship DALL-E training step on 1..8 MI355X GPUs.

Metric (BASELINE.json): img-tokens/sec training DALL-E dim=1024 / depth=12 /
heads=16, seq 1280 (256 text + 1024 image tokens), bf16, synthetic data,
random-init weights. ``value`` is the WHOLE-JOB aggregate across all ranks.

Run directly (single GPU) or under torchrun for N>1:
    python bench.py --gpus 1 --steps 10 --warmup 3
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 bench.py --gpus 8 --steps 10 --warmup 3

Each timed step is a full training iteration: frozen-VAE image encode,
transformer forward, loss, backward, bucketed RCCL gradient all-reduce,
grad clip, Adam step — nothing cached, nothing skipped.
"""

import argparse
import json
import os
import sys
import time

import torch

CONFIGS = {
    # BASELINE.json configs B-D (training). Default: the headline C shape.
    'b': dict(dim=512, depth=12, heads=8, attn_types=('full',),
              reversible=False, batch_size=8, vae='dvae'),
    'c': dict(dim=1024, depth=12, heads=16, attn_types=('axial_row', 'axial_col'),
              reversible=True, batch_size=64, vae='dvae'),
    'c_full': dict(dim=1024, depth=12, heads=16, attn_types=('full',),
                   reversible=True, batch_size=64, vae='dvae'),
    'd': dict(dim=1024, depth=64, heads=16, attn_types=('axial_row', 'axial_col'),
              reversible=True, batch_size=4, vae='vqgan16k'),
    # the other two sparse attention families at the flagship shape
    'c_conv': dict(dim=1024, depth=12, heads=16, attn_types=('conv_like',),
                   reversible=True, batch_size=64, vae='dvae'),
    'c_sparse': dict(dim=1024, depth=12, heads=16, attn_types=('sparse',),
                     reversible=True, batch_size=64, vae='dvae'),
    # plumbing smoke (CPU-runnable, used by the distributed-launch test)
    'tiny': dict(dim=64, depth=1, heads=1, attn_types=('full',),
                 reversible=False, batch_size=2, vae='dvae',
                 image_size=64, text_seq_len=16, vocab=100),
}


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=10)
    p.add_argument('--warmup', type=int, default=3)
    p.add_argument('--config', type=str, default='c', choices=sorted(CONFIGS))
    p.add_argument('--batch_size', type=int, default=None, help='per-GPU batch')
    p.add_argument('--mode', type=str, default='train', choices=['train', 'generate'])
    p.add_argument('--gen_batch', type=int, default=64)
    p.add_argument('--gen_cond_scale', type=float, default=1.0,
                   help='classifier-free guidance scale for --mode generate '
                        '(the decoder runs cond+null as one doubled batch)')
    p.add_argument('--no_graph', action='store_true',
                   help='disable HIP-graph decode replay (for kernel profiling)')
    p.add_argument('--fp8', action='store_true',
                   help='e4m3 forward path for the projection GEMMs')
    p.add_argument('--eager', action='store_true',
                   help='force eager ops on GPU (baseline comparison)')
    return p.parse_args()


def maybe_self_launch(args):
    """Convenience: `python bench.py --gpus N` outside torchrun re-execs
    itself under torch.distributed.run."""
    if args.gpus > 1 and 'RANK' not in os.environ:
        import subprocess
        cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
               f'--nproc-per-node={args.gpus}', '--master-addr', '127.0.0.1',
               '--master-port', '29517', sys.argv[0], *sys.argv[1:]]
        raise SystemExit(subprocess.call(cmd))


def build_model(cfg, device):
    from dalle_pytorch_amd import DALLE, DiscreteVAE, VQGanVAE
    torch.manual_seed(1234)
    image_size = cfg.get('image_size', 256)
    if cfg['vae'] == 'dvae':
        hidden = 8 if image_size < 256 else 64
        tokens = 512 if image_size < 256 else 8192
        vae = DiscreteVAE(image_size=image_size, num_layers=3, num_tokens=tokens,
                          codebook_dim=64 if image_size < 256 else 512,
                          hidden_dim=hidden)
    else:
        vae = VQGanVAE(num_tokens=16384)  # f=16 -> 256 image tokens (config D)
    dalle = DALLE(
        dim=cfg['dim'], vae=vae, num_text_tokens=cfg.get('vocab', 10000),
        text_seq_len=cfg.get('text_seq_len', 256),
        depth=cfg['depth'], heads=cfg['heads'], dim_head=64,
        attn_types=cfg['attn_types'], reversible=cfg['reversible'],
        shift_tokens=True, rotary_emb=True)
    return dalle.to(device)


def main():
    args = parse_args()
    maybe_self_launch(args)
    if args.eager:
        os.environ['DALLE_AMD_ALLOW_EAGER'] = '1'
        import dalle_pytorch_amd.ops.dispatch as _d
        _d._HIP, _d._TRIED = None, True

    if args.fp8:
        from dalle_pytorch_amd.ops.fp8 import set_fp8_enabled
        set_fp8_enabled(True)

    from dalle_pytorch_amd.parallel import init_distributed, barrier
    from dalle_pytorch_amd.parallel import DataParallelEngine
    from dalle_pytorch_amd.utils.tunable import maybe_enable_tunableop
    maybe_enable_tunableop()

    rank, world, local_rank = init_distributed()
    use_cuda = torch.cuda.is_available()
    device = torch.device(f'cuda:{local_rank}') if use_cuda else torch.device('cpu')

    cfg = dict(CONFIGS[args.config])
    if args.batch_size:
        cfg['batch_size'] = args.batch_size
    bsz = cfg['batch_size']

    dalle = build_model(cfg, device)
    image_seq_len = dalle.image_seq_len
    seq_len = dalle.total_seq_len

    engine = DataParallelEngine(dalle)
    params = [p for p in dalle.parameters() if p.requires_grad]
    try:
        # single fused HIP kernel per step vs foreach's several passes
        opt = torch.optim.Adam(params, lr=3e-4, fused=use_cuda)
    except (RuntimeError, ValueError):
        opt = torch.optim.Adam(params, lr=3e-4)

    # synthetic data pool: distinct per rank & step, generated once on device
    torch.manual_seed(4321 + rank)
    n_pool = 4
    vocab = cfg.get('vocab', 10000)
    tlen = dalle.text_seq_len
    isz = dalle.vae.image_size
    pool = [(torch.randint(1, vocab, (bsz, tlen), device=device),
             torch.rand(bsz, 3, isz, isz, device=device)) for _ in range(n_pool)]

    autocast = torch.autocast(device_type='cuda', dtype=torch.bfloat16,
                              enabled=use_cuda)

    def train_step(i):
        text, images = pool[i % n_pool]
        with autocast:
            loss = dalle(text, images, return_loss=True)
        loss.backward()
        engine.finish_gradient_sync()
        engine.clip_grad_norm_(0.5)
        opt.step()
        engine.zero_grad()
        if args.fp8:
            from dalle_pytorch_amd.ops.fp8 import fp8_mark_step
            fp8_mark_step()
        return loss

    decoder = None
    if args.mode == 'generate':
        # fp8-weight decode (DALLE_AMD_FP8_DECODE=1) measured SLOWER than
        # bf16 sk2 at these shapes (8.6 vs 6.9 us qkv: the kernels are
        # latency-bound, not weight-byte-bound, and the in-register
        # bf16->e4m3 conversion costs VALU), so --fp8 does NOT enable it
        dalle.eval()
        if not args.eager:
            from dalle_pytorch_amd.engine import FastDecoder
            guided = args.gen_cond_scale != 1.0
            decoder = FastDecoder(
                dalle, batch_size=args.gen_batch * (2 if guided else 1),
                use_graph=use_cuda and not args.no_graph)

    def gen_step(_):
        text = pool[0][0][:1].repeat(args.gen_batch, 1)
        if decoder is not None:
            decoder.generate(text, filter_thres=0.9,
                             cond_scale=args.gen_cond_scale)
        else:
            with autocast:
                dalle.generate_images(text, use_cache=True, filter_thres=0.9)

    step = train_step if args.mode == 'train' else gen_step

    for i in range(args.warmup):
        step(i)

    barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(args.warmup + i)
    if use_cuda:
        torch.cuda.synchronize()
    barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    t = torch.tensor([elapsed], device=device if use_cuda else 'cpu')
    if world > 1:
        import torch.distributed as dist
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    max_mem_gb = round(torch.cuda.max_memory_allocated() / 2**30, 2) \
        if use_cuda else None
    if rank == 0:
        ms_per_step = elapsed / args.steps * 1000
        if args.mode == 'train':
            global_batch = bsz * world
            value = global_batch * image_seq_len * args.steps / elapsed
            metric, unit = 'img-tokens/sec', 'img_tokens_per_sec'
            dtype = 'fp8-fwd/bf16' if args.fp8 else \
                'bf16' if use_cuda else 'fp32'
        else:
            value = args.gen_batch * world * args.steps / elapsed
            metric, unit = 'gen imgs/sec', 'images_per_sec'
            dtype = 'bf16/fp8-weights' \
                if os.environ.get('DALLE_AMD_FP8_DECODE') == '1' \
                else 'bf16' if use_cuda else 'fp32'
            global_batch = args.gen_batch * world
        print(json.dumps({
            'metric': metric, 'value': round(value, 2), 'unit': unit,
            'n_gpus': world, 'steps': args.steps, 'warmup': args.warmup,
            'ms_per_step': round(ms_per_step, 2),
            'higher_is_better': True, 'scaling': 'weak',
            'vs_baseline': None, 'dtype': dtype,
            'data': 'synthetic',
            'config': {
                'model': f"dalle-dim{cfg['dim']}-depth{cfg['depth']}-heads{cfg['heads']}",
                'attn_types': list(cfg['attn_types']),
                'reversible': cfg['reversible'],
                'global_batch': global_batch, 'seq_len': seq_len,
                'image_seq_len': image_seq_len,
                'parallelism': f'dp{world}', 'mode': args.mode,
                'max_mem_gb': max_mem_gb,
                'eager': bool(args.eager)},
        }))


if __name__ == '__main__':
    main()
    import torch.distributed as dist
    if dist.is_available() and dist.is_initialized():
        dist.destroy_process_group()
