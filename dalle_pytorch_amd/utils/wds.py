"""Tar-shard streaming dataset (WebDataset-style, no external package).

Replaces the reference's WebDataset wiring (train_dalle.py:364-423): shards
are plain tar files whose members share a basename per sample
(``000123.jpg`` + ``000123.txt``). Samples stream sequentially per shard;
shards are partitioned across ranks and dataloader workers; unreadable
members are skipped with a warning (the reference's ``warn_and_continue``
behavior).
"""

import io
import random
import tarfile
import warnings
from pathlib import Path

import torch
from torch.utils.data import IterableDataset

IMAGE_KEYS = ('img', 'image', 'jpg', 'jpeg', 'png')
TEXT_KEYS = ('cap', 'txt', 'text', 'caption')


def expand_shards(spec):
    """'a.tar', 'dir/*.tar', 'shard-{000..009}.tar', or list of those."""
    if isinstance(spec, (list, tuple)):
        out = []
        for s in spec:
            out.extend(expand_shards(s))
        return out
    spec = str(spec)
    if '{' in spec and '..' in spec:
        pre, rest = spec.split('{', 1)
        rng, post = rest.split('}', 1)
        lo, hi = rng.split('..')
        width = len(lo)
        return [f'{pre}{i:0{width}d}{post}' for i in range(int(lo), int(hi) + 1)]
    if any(c in spec for c in '*?['):
        p = Path(spec)
        return sorted(str(f) for f in p.parent.glob(p.name))
    return [spec]


class TarImageTextDataset(IterableDataset):
    """Yields (caption_tokens, image_tensor) from tar shards."""

    def __init__(self, shards, tokenizer, text_len=256, image_size=256,
                 truncate_captions=True, image_key=None, text_key=None,
                 shuffle_shards=True, seed=0, resize_ratio=0.75):
        super().__init__()
        self.shards = expand_shards(shards)
        assert self.shards, f'no shards matched {shards!r}'
        self.tokenizer = tokenizer
        self.text_len = text_len
        self.image_size = image_size
        self.truncate = truncate_captions
        self.image_key = image_key
        self.text_key = text_key
        self.shuffle_shards = shuffle_shards
        self.seed = seed
        self.resize_ratio = resize_ratio

    def _my_shards(self):
        from dalle_pytorch_amd.parallel import get_rank, get_world_size
        shards = list(self.shards)
        if self.shuffle_shards:
            random.Random(self.seed).shuffle(shards)
        world = max(get_world_size(), 1)
        info = torch.utils.data.get_worker_info()
        workers = info.num_workers if info is not None else 1
        if len(shards) < world * workers:
            # a rank/worker with zero shards yields zero batches, which stalls
            # the other ranks' gradient all-reduce forever: repeat the shard
            # list so every (rank, worker) slot sees at least one shard
            warnings.warn(
                f'{len(shards)} shards < world_size*workers = {world}*{workers};'
                ' repeating shards so every rank yields data (samples will '
                'repeat across ranks — add shards for proper sharding)',
                RuntimeWarning)
            reps = -(-(world * workers) // len(shards))
            shards = shards * reps
        shards = shards[get_rank()::world]
        if info is not None:
            shards = shards[info.id::info.num_workers]
        return shards

    def _decode_image(self, data):
        from PIL import Image
        from dalle_pytorch_amd.utils.vision import random_resized_crop, to_tensor
        img = Image.open(io.BytesIO(data))
        if img.mode != 'RGB':
            img = img.convert('RGB')
        return to_tensor(random_resized_crop(
            img, self.image_size, scale=(self.resize_ratio, 1.)))

    def __iter__(self):
        for shard in self._my_shards():
            try:
                tf = tarfile.open(shard)
            except (OSError, tarfile.TarError) as e:
                print(f'[wds] skipping unreadable shard {shard}: {e}')
                continue
            with tf:
                current_key, parts = None, {}
                for member in tf:
                    if not member.isfile():
                        continue
                    stem, _, ext = member.name.rpartition('.')
                    if current_key is not None and stem != current_key and parts:
                        sample = self._emit(parts)
                        if sample is not None:
                            yield sample
                        parts = {}
                    current_key = stem
                    try:
                        parts[ext.lower()] = tf.extractfile(member).read()
                    except (OSError, tarfile.TarError) as e:
                        print(f'[wds] skipping member {member.name}: {e}')
                if parts:
                    sample = self._emit(parts)
                    if sample is not None:
                        yield sample

    def _emit(self, parts):
        img_ext = self.image_key or next(
            (k for k in parts if k in IMAGE_KEYS), None)
        txt_ext = self.text_key or next(
            (k for k in parts if k in TEXT_KEYS), None)
        if img_ext is None or txt_ext is None:
            return None
        try:
            caption = parts[txt_ext].decode('utf-8', errors='replace')
            tokens = self.tokenizer.tokenize(
                caption, self.text_len, truncate_text=self.truncate).squeeze(0)
            image = self._decode_image(parts[img_ext])
        except Exception as e:  # corrupt member -> warn and continue
            print(f'[wds] skipping corrupt sample: {e}')
            return None
        return tokens, image
