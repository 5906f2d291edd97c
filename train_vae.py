#!/usr/bin/env python3
"""Train the discrete VAE (reference train_vae.py parity, RCCL-native).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train_vae.py --image_folder data/ ...
"""

import argparse
import math
import os
from pathlib import Path

import torch
from torch.optim import Adam
from torch.optim.lr_scheduler import ExponentialLR
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

from dalle_pytorch_amd import DiscreteVAE
from dalle_pytorch_amd.parallel import (
    DataParallelEngine, init_distributed, average_scalar, barrier)
from dalle_pytorch_amd.utils.checkpoint import save_vae_checkpoint
from dalle_pytorch_amd.utils.logging import RunLogger


def parse_args(argv=None):
    p = argparse.ArgumentParser(description='DiscreteVAE training (MI355X-native)')
    p.add_argument('--image_folder', type=str, default=None,
                   help='folder of images (ImageFolder layout); omit for synthetic')
    p.add_argument('--image_size', type=int, default=128)
    p.add_argument('--epochs', type=int, default=20)
    p.add_argument('--batch_size', type=int, default=8)
    p.add_argument('--learning_rate', type=float, default=1e-3)
    p.add_argument('--lr_decay_rate', type=float, default=0.98)
    p.add_argument('--starting_temp', type=float, default=1.)
    p.add_argument('--temp_min', type=float, default=0.5)
    p.add_argument('--anneal_rate', type=float, default=1e-6)
    p.add_argument('--num_images_save', type=int, default=4)
    p.add_argument('--num_tokens', type=int, default=8192)
    p.add_argument('--num_layers', type=int, default=3)
    p.add_argument('--num_resnet_blocks', type=int, default=2)
    p.add_argument('--smooth_l1_loss', action='store_true')
    p.add_argument('--emb_dim', type=int, default=512)
    p.add_argument('--hidden_dim', type=int, default=256)
    p.add_argument('--kl_loss_weight', type=float, default=0.)
    p.add_argument('--transparent', action='store_true')
    p.add_argument('--straight_through', action='store_true')
    p.add_argument('--reinmax', action='store_true')
    p.add_argument('--conv_gemm', action='store_true',
                   help='run the trainable conv stacks as unfold/fold+GEMMs '
                        '(MIOpen independence on untuned boxes)')
    p.add_argument('--fp16', action='store_true', help='bf16 autocast on MI355X')
    p.add_argument('--stop_after_steps', type=int, default=None)
    p.add_argument('--output_dir', default='.')
    p.add_argument('--local_rank', type=int, default=0, help=argparse.SUPPRESS)
    return p.parse_args(argv)


class _SyntheticImages(torch.utils.data.Dataset):
    def __init__(self, n, size, channels):
        self.n, self.size, self.channels = n, size, channels

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(i)
        return torch.rand(self.channels, self.size, self.size, generator=g), 0


def main(argv=None):
    args = parse_args(argv)
    if args.conv_gemm:
        os.environ['DALLE_AMD_CONV_GEMM'] = '1'
    from dalle_pytorch_amd.utils.tunable import maybe_enable_tunableop
    maybe_enable_tunableop()
    rank, world, local_rank = init_distributed()
    is_root = rank == 0
    device = torch.device(f'cuda:{local_rank}') if torch.cuda.is_available() \
        else torch.device('cpu')

    channels = 4 if args.transparent else 3
    if args.image_folder:
        from dalle_pytorch_amd.utils.vision import ImageFolderDataset
        ds = ImageFolderDataset(args.image_folder, args.image_size,
                                transparent=args.transparent)
        assert len(ds) > 0, 'folder does not contain any images'
    else:
        ds = _SyntheticImages(max(args.batch_size * world * 64, 256),
                              args.image_size, channels)
    if is_root:
        print(f'{len(ds)} images found for training')

    sampler = DistributedSampler(ds, num_replicas=world, rank=rank) if world > 1 else None
    dl = DataLoader(ds, args.batch_size, shuffle=sampler is None, sampler=sampler,
                    drop_last=True, num_workers=2,
                    pin_memory=device.type == 'cuda')

    vae_params = dict(
        image_size=args.image_size,
        num_layers=args.num_layers,
        num_tokens=args.num_tokens,
        codebook_dim=args.emb_dim,
        hidden_dim=args.hidden_dim,
        num_resnet_blocks=args.num_resnet_blocks,
        channels=channels,
    )
    vae = DiscreteVAE(**vae_params, smooth_l1_loss=args.smooth_l1_loss,
                      kl_div_loss_weight=args.kl_loss_weight,
                      straight_through=args.straight_through,
                      reinmax=args.reinmax).to(device)

    engine = DataParallelEngine(vae)
    opt = Adam(vae.parameters(), lr=args.learning_rate)
    sched = ExponentialLR(optimizer=opt, gamma=args.lr_decay_rate)

    logger = RunLogger('dalle_train_vae', config={**vae_params,
                       'batch_size': args.batch_size, 'world_size': world},
                       enabled=is_root, output_dir=args.output_dir)

    out_dir = Path(args.output_dir)
    autocast = args.fp16 and device.type == 'cuda'
    global_step = 0
    temp = args.starting_temp

    vae.train()
    for epoch in range(args.epochs):
        if sampler is not None:
            sampler.set_epoch(epoch)
        for i, (images, _) in enumerate(dl):
            images = images.to(device, non_blocking=True)
            with torch.autocast(device_type='cuda', dtype=torch.bfloat16,
                                enabled=autocast):
                loss, recons = vae(images, return_loss=True, return_recons=True,
                                   temp=temp)
            loss.backward()
            engine.finish_gradient_sync()
            opt.step()
            engine.zero_grad()

            if i % 100 == 0:
                # temperature anneal + lr decay cadence (reference :278-284)
                temp = max(temp * math.exp(-args.anneal_rate * global_step),
                           args.temp_min)
                sched.step()
                if is_root:
                    with torch.no_grad():
                        k = args.num_images_save
                        codes = vae.get_codebook_indices(images[:k])
                        hard_recons = vae.decode(codes)
                    logger.log_image('original images', images[:k].float().cpu(),
                                     step=global_step)
                    logger.log_image('hard reconstructions',
                                     hard_recons.float().cpu(), step=global_step)
                    logger.log({'temperature': temp,
                                'codebook_usage': len(codes.unique())},
                               step=global_step)
                    save_vae_checkpoint(out_dir / 'vae.pt', vae, vae_params)

            avg_loss = average_scalar(loss)
            global_step += 1
            if is_root and i % 10 == 0:
                print(f'epoch {epoch} iter {i} loss {avg_loss.item():.4f} '
                      f'lr {sched.get_last_lr()[0]:.2e} temp {temp:.3f}')
                logger.log({'loss': avg_loss.item(), 'epoch': epoch,
                            'lr': sched.get_last_lr()[0]}, step=global_step)
            if args.stop_after_steps and global_step >= args.stop_after_steps:
                break
        if args.stop_after_steps and global_step >= args.stop_after_steps:
            break

    if is_root:
        save_vae_checkpoint(out_dir / 'vae-final.pt', vae, vae_params)
        logger.save(out_dir / 'vae-final.pt')
    logger.finish()
    barrier()


if __name__ == '__main__':
    main()
