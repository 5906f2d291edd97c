#!/usr/bin/env python3
"""Train DALL-E on an MI355X node.

Same CLI surface as the reference ``train_dalle.py`` (flags :31-141), with
the distributed backend zoo replaced by the pure-RCCL engine: launch with

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train_dalle.py --image_text_folder data/ ...

Single-process runs need no launcher. DeepSpeed/Horovod/apex flags from the
reference are accepted-and-ignored or mapped (--fp16 -> bf16 autocast, the
MI355X-native precision; --deepspeed/--horovod -> the RCCL engine).
"""

import argparse
import time
from pathlib import Path

import torch
from torch.optim import Adam
from torch.optim.lr_scheduler import ReduceLROnPlateau
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

from dalle_pytorch_amd import DALLE, DiscreteVAE, OpenAIDiscreteVAE, VQGanVAE
from dalle_pytorch_amd.parallel import (
    DataParallelEngine, init_distributed, barrier, average_scalar)
from dalle_pytorch_amd.utils.checkpoint import (
    save_dalle_checkpoint, load_dalle_checkpoint, rotate_checkpoints)
from dalle_pytorch_amd.utils.loader import TextImageDataset, SyntheticTextImageDataset
from dalle_pytorch_amd.utils.logging import RunLogger
from dalle_pytorch_amd.utils import tokenizer as tokenizer_mod


def parse_args(argv=None):
    p = argparse.ArgumentParser(description='DALL-E training (MI355X-native)')
    grp = p.add_mutually_exclusive_group(required=False)
    grp.add_argument('--vae_path', type=str, help='path to trained DiscreteVAE checkpoint')
    grp.add_argument('--dalle_path', type=str, help='path to partially trained DALL-E to resume')
    p.add_argument('--vqgan_model_path', type=str, default=None)
    p.add_argument('--vqgan_config_path', type=str, default=None)
    p.add_argument('--image_text_folder', type=str, default=None,
                   help='folder of image/text pairs (omit for synthetic data)')
    p.add_argument('--synthetic', action='store_true',
                   help='train on synthetic random data (no dataset needed)')
    p.add_argument('--wds', type=str, default='',
                   help='tar shard spec for streaming input, e.g. '
                        '"shards/train-{000..099}.tar" or a glob')
    p.add_argument('--truncate_captions', action='store_true')
    p.add_argument('--random_resize_crop_lower_ratio', dest='resize_ratio',
                   type=float, default=0.75)
    p.add_argument('--chinese', action='store_true')
    p.add_argument('--taming', action='store_true')
    p.add_argument('--hug', action='store_true')
    p.add_argument('--bpe_path', type=str, default=None)
    p.add_argument('--fp16', action='store_true',
                   help='mixed precision (bf16 autocast on MI355X)')
    p.add_argument('--amp', action='store_true', help='alias of --fp16 here')
    p.add_argument('--half_weights', choices=['fp16', 'bf16'], default=None,
                   help='store the model weights in half precision, as the '
                        'reference --fp16 does via dalle.half() '
                        '(train_dalle.py:430-432); bf16 keeps the HIP kernel '
                        'path, fp16 matches reference checkpoint dtype '
                        'exactly (runs on torch fallbacks)')
    p.add_argument('--wandb_name', default='dalle_train_transformer')
    p.add_argument('--wandb_entity', default=None)
    p.add_argument('--name_suffix', default='')
    p.add_argument('--output_dir', default='./outputs')
    # accepted for reference parity; the RCCL engine is always used
    p.add_argument('--deepspeed', action='store_true', help=argparse.SUPPRESS)
    p.add_argument('--horovod', action='store_true', help=argparse.SUPPRESS)
    p.add_argument('--local_rank', type=int, default=0, help=argparse.SUPPRESS)
    p.add_argument('--torch_profile', action='store_true',
                   help='capture a torch.profiler chrome trace of steps '
                        '10-13 into <output_dir>/trace.json')
    p.add_argument('--flops_profiler', action='store_true',
                   help='profile FLOPs at --profile_step then stop '
                        '(reference behavior)')
    p.add_argument('--profile_step', type=int, default=200)

    t = p.add_argument_group('Training settings')
    t.add_argument('--epochs', default=20, type=int)
    t.add_argument('--save_every_n_steps', default=1000, type=int)
    t.add_argument('--keep_n_checkpoints', default=None, type=int)
    t.add_argument('--batch_size', default=4, type=int,
                   help='per-GPU batch size')
    t.add_argument('--ga_steps', default=1, type=int,
                   help='gradient accumulation steps')
    t.add_argument('--learning_rate', default=3e-4, type=float)
    t.add_argument('--clip_grad_norm', default=0.5, type=float)
    t.add_argument('--lr_decay', dest='lr_decay', action='store_true')
    t.add_argument('--stop_after_steps', default=None, type=int)
    t.add_argument('--dalle_output_file_name', default='dalle')

    m = p.add_argument_group('Model settings')
    m.add_argument('--dim', default=512, type=int)
    m.add_argument('--text_seq_len', default=256, type=int)
    m.add_argument('--depth', default=2, type=int)
    m.add_argument('--heads', default=8, type=int)
    m.add_argument('--dim_head', default=64, type=int)
    m.add_argument('--reversible', dest='reversible', action='store_true')
    m.add_argument('--attn_dropout', default=0.0, type=float)
    m.add_argument('--ff_dropout', default=0.0, type=float)
    m.add_argument('--loss_img_weight', default=7, type=int)
    m.add_argument('--attn_types', default='full', type=str,
                   help='comma separated: full, sparse, axial_row, axial_col, conv_like')
    m.add_argument('--shift_tokens', help='token shift feature', action='store_true')
    m.add_argument('--rotary_emb', help='rotary positional embedding', action='store_true')
    m.add_argument('--shared_attn_ids', default=None, type=str)
    m.add_argument('--shared_ff_ids', default=None, type=str)
    m.add_argument('--share_input_output_emb', action='store_true')
    m.add_argument('--stable_softmax', dest='stable_softmax', action='store_true')

    return p.parse_args(argv)


def exists(v):
    return v is not None


def get_tokenizer(args):
    if args.chinese:
        return tokenizer_mod.ChineseTokenizer()
    if args.hug:
        assert exists(args.bpe_path), '--hug requires --bpe_path (json)'
        return tokenizer_mod.HugTokenizer(args.bpe_path)
    if exists(args.bpe_path):
        suffix = Path(args.bpe_path).suffix
        if suffix == '.json':
            return tokenizer_mod.HugTokenizer(args.bpe_path)
        if suffix == '.model':
            return tokenizer_mod.YttmTokenizer(args.bpe_path)
        return tokenizer_mod.SimpleTokenizer(args.bpe_path)
    return tokenizer_mod.tokenizer


def main(argv=None):
    args = parse_args(argv)
    from dalle_pytorch_amd.utils.tunable import maybe_enable_tunableop
    from dalle_pytorch_amd.ops.fp8 import fp8_mark_step as _fp8_mark_step
    maybe_enable_tunableop()
    rank, world, local_rank = init_distributed()
    is_root = rank == 0
    device = torch.device(f'cuda:{local_rank}') if torch.cuda.is_available() \
        else torch.device('cpu')

    tok = get_tokenizer(args)

    # ------------------------------------------------------------- VAE
    resume_ckpt = None
    dalle_params = None
    vae_params = None
    opt_state = sched_state = None
    start_epoch = 0

    if exists(args.dalle_path):
        resume_ckpt = load_dalle_checkpoint(args.dalle_path)
        dalle_params = dict(resume_ckpt['hparams'])
        vae_params = resume_ckpt.get('vae_params')
        opt_state = resume_ckpt.get('opt_state')
        sched_state = resume_ckpt.get('scheduler_state')
        start_epoch = resume_ckpt.get('epoch', 0)

    if vae_params is not None:
        vae = DiscreteVAE(**vae_params)
        vae_class_name = 'DiscreteVAE'
    elif exists(args.vae_path):
        from dalle_pytorch_amd.utils.checkpoint import load_vae_checkpoint
        vae, vae_params = load_vae_checkpoint(args.vae_path)
        vae_class_name = 'DiscreteVAE'
    elif args.taming:
        vae = VQGanVAE(args.vqgan_model_path, args.vqgan_config_path)
        vae_class_name = 'VQGanVAE'
    else:
        if is_root:
            print('using OpenAIDiscreteVAE-architecture dVAE (random init '
                  'unless weights are provided offline)')
        vae = OpenAIDiscreteVAE()
        vae_class_name = 'OpenAIDiscreteVAE'

    # ----------------------------------------------------------- DALLE
    if dalle_params is None:
        dalle_params = dict(
            num_text_tokens=tok.vocab_size,
            text_seq_len=args.text_seq_len,
            dim=args.dim,
            depth=args.depth,
            heads=args.heads,
            dim_head=args.dim_head,
            reversible=args.reversible,
            attn_dropout=args.attn_dropout,
            ff_dropout=args.ff_dropout,
            loss_img_weight=args.loss_img_weight,
            attn_types=tuple(args.attn_types.split(',')),
            shift_tokens=args.shift_tokens,
            rotary_emb=args.rotary_emb,
            shared_attn_ids=tuple(int(i) for i in args.shared_attn_ids.split(','))
                if exists(args.shared_attn_ids) else None,
            shared_ff_ids=tuple(int(i) for i in args.shared_ff_ids.split(','))
                if exists(args.shared_ff_ids) else None,
            share_input_output_emb=args.share_input_output_emb,
            stable=args.stable_softmax,
        )

    dalle = DALLE(vae=vae, **dalle_params)
    if resume_ckpt is not None:
        dalle.load_state_dict(resume_ckpt['weights'])
    dalle = dalle.to(device)
    if args.half_weights:
        # reference --fp16 semantics: the weights themselves are halved
        # (reference train_dalle.py:430-432 does dalle.half())
        dalle = dalle.to(torch.float16 if args.half_weights == 'fp16'
                         else torch.bfloat16)

    # --------------------------------------------------------- dataset
    text_seq_len = dalle_params['text_seq_len']  # checkpoint-authoritative on resume
    streaming = bool(args.wds)
    if streaming:
        from dalle_pytorch_amd.utils.wds import TarImageTextDataset
        ds = TarImageTextDataset(args.wds, tokenizer=tok, text_len=text_seq_len,
                                 image_size=vae.image_size,
                                 truncate_captions=args.truncate_captions,
                                 resize_ratio=args.resize_ratio)
        if is_root:
            print(f'streaming from {len(ds.shards)} tar shard(s)')
    elif args.synthetic or not exists(args.image_text_folder):
        ds = SyntheticTextImageDataset(
            length=max(args.batch_size * world * 64, 512),
            text_len=text_seq_len, image_size=vae.image_size,
            vocab_size=min(tok.vocab_size, dalle_params['num_text_tokens']))
    else:
        ds = TextImageDataset(
            args.image_text_folder, text_len=text_seq_len,
            image_size=vae.image_size, resize_ratio=args.resize_ratio,
            truncate_captions=args.truncate_captions, tokenizer=tok, shuffle=True)
        assert len(ds) > 0, 'dataset is empty'
    if is_root and not streaming:
        print(f'{len(ds)} image-text pairs found for training')

    sampler = DistributedSampler(ds, num_replicas=world, rank=rank, shuffle=True) \
        if (world > 1 and not streaming) else None
    dl = DataLoader(ds, batch_size=args.batch_size,
                    shuffle=(sampler is None and not streaming),
                    drop_last=True, sampler=sampler, num_workers=2,
                    pin_memory=device.type == 'cuda')

    # ------------------------------------------------------ optimizer
    engine = DataParallelEngine(dalle)
    # fused=True runs the whole Adam update as one multi-tensor HIP kernel
    # instead of foreach's several full passes over 320M params
    opt = Adam((p for p in dalle.parameters() if p.requires_grad),
               lr=args.learning_rate, fused=device.type == 'cuda')
    if opt_state:
        opt.load_state_dict(opt_state)
    scheduler = None
    if args.lr_decay:
        scheduler = ReduceLROnPlateau(opt, mode='min', factor=0.5, patience=10,
                                      cooldown=10, min_lr=1e-6)
        if sched_state:
            scheduler.load_state_dict(sched_state)

    run_name = args.dalle_output_file_name + args.name_suffix
    out_dir = Path(args.output_dir)
    ckpt_path = out_dir / f'{run_name}.pt'
    logger = RunLogger('dalle_train_transformer', config={
        **{k: v for k, v in dalle_params.items() if k != 'shared_attn_ids'},
        'batch_size': args.batch_size, 'world_size': world,
        'learning_rate': args.learning_rate},
        enabled=is_root, output_dir=args.output_dir, run_name=args.wandb_name,
        entity=args.wandb_entity)

    profiler = None
    if args.torch_profile and is_root:
        from torch.profiler import profile, schedule, ProfilerActivity
        acts = [ProfilerActivity.CPU]
        if device.type == 'cuda':
            acts.append(ProfilerActivity.CUDA)
        profiler = profile(
            activities=acts, schedule=schedule(wait=10, warmup=1, active=3),
            on_trace_ready=lambda p: p.export_chrome_trace(
                str(out_dir / 'trace.json')))
        profiler.__enter__()

    autocast_enabled = ((args.fp16 or args.amp) and device.type == 'cuda'
                        and not args.half_weights)

    def save(epoch):
        if not is_root:
            return
        save_dalle_checkpoint(ckpt_path, dalle, dalle_params, vae_params,
                              epoch=epoch, vae_class_name=vae_class_name,
                              opt=opt, scheduler=scheduler)
        if args.keep_n_checkpoints:
            step_path = out_dir / f'{run_name}-step{global_step}.pt'
            save_dalle_checkpoint(step_path, dalle, dalle_params, vae_params,
                                  epoch=epoch, vae_class_name=vae_class_name)
            rotate_checkpoints(out_dir, f'{run_name}-step*.pt', args.keep_n_checkpoints)
        logger.log_artifact(ckpt_path)

    # fail-early checkpoint (reference train_dalle.py:591-594)
    global_step = 0
    save(start_epoch)

    dalle.train()
    t_window = time.time()
    for epoch in range(start_epoch, args.epochs):
        if sampler is not None:
            sampler.set_epoch(epoch)
        for i, (text, images) in enumerate(dl):
            text = text.to(device, non_blocking=True)
            images = images.to(device, non_blocking=True)

            accum_boundary = (global_step + 1) % args.ga_steps == 0
            ctx = engine.no_sync() if not accum_boundary else _nullcontext()
            with ctx:
                with torch.autocast(device_type='cuda', dtype=torch.bfloat16,
                                    enabled=autocast_enabled):
                    loss = dalle(text, images, return_loss=True)
                (loss / args.ga_steps).backward()

            if accum_boundary:
                engine.finish_gradient_sync()
                if args.clip_grad_norm:
                    engine.clip_grad_norm_(args.clip_grad_norm)
                opt.step()
                _fp8_mark_step()
                engine.zero_grad()

            avg_loss = average_scalar(loss)
            global_step += 1

            if is_root and i % 10 == 0:
                dt = time.time() - t_window
                sample_per_sec = args.batch_size * world * 10 / max(dt, 1e-9)
                t_window = time.time()
                print(f'epoch {epoch} iter {i} loss {avg_loss.item():.4f} '
                      f'sample/s {sample_per_sec:.2f}')
                logger.log({'loss': avg_loss.item(), 'epoch': epoch, 'iter': i,
                            'lr': opt.param_groups[0]['lr'],
                            'sample_per_sec': sample_per_sec}, step=global_step)

            if args.save_every_n_steps and global_step % args.save_every_n_steps == 0:
                save(epoch)

            if is_root and i and i % 100 == 0:
                with torch.no_grad():
                    sample_img = dalle.generate_images(
                        text[:1], filter_thres=0.9, use_cache=True)
                logger.log_image('generated image', sample_img[0].float().cpu(),
                                 step=global_step)

            if args.flops_profiler and global_step == args.profile_step:
                # profile the next step's wall time, print, then abort
                # (reference train_dalle.py:492-499,656-657)
                import time as _t
                if device.type == 'cuda':
                    torch.cuda.synchronize()
                t0 = _t.perf_counter()
                with torch.autocast(device_type='cuda', dtype=torch.bfloat16,
                                    enabled=autocast_enabled):
                    ploss = dalle(text, images, return_loss=True)
                ploss.backward()
                engine.finish_gradient_sync()
                opt.step()
                _fp8_mark_step()
                engine.zero_grad()
                if device.type == 'cuda':
                    torch.cuda.synchronize()
                if is_root:
                    from dalle_pytorch_amd.utils.flops import profile_step
                    profile_step(dalle, text.shape[0], _t.perf_counter() - t0,
                                 reversible=dalle_params.get('reversible', False))
                args.stop_after_steps = global_step  # abort after profiling

            if profiler is not None:
                profiler.step()
            if args.stop_after_steps and global_step >= args.stop_after_steps:
                break
        if scheduler is not None:
            scheduler.step(avg_loss)
        save(epoch + 1)
        if args.stop_after_steps and global_step >= args.stop_after_steps:
            break

    if profiler is not None:
        profiler.__exit__(None, None, None)
    save(args.epochs)
    logger.finish()
    barrier()


class _nullcontext:
    def __enter__(self):
        return None

    def __exit__(self, *a):
        return False


if __name__ == '__main__':
    main()
