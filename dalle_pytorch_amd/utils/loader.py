"""Datasets: folder-of-(txt,image) pairs, synthetic benchmark data, and the
rainbow-style compositional dataset used as the end-to-end correctness
signal (reference loader.py:10-103 and examples/rainbow_dalle.ipynb).
"""

import random
from pathlib import Path

import torch
from torch.utils.data import Dataset


IMAGE_EXTS = ('.png', '.jpg', '.jpeg', '.bmp')


class TextImageDataset(Dataset):
    """Folder of stem-matched ``*.txt`` / image files -> (tokens, image).

    Mirrors reference loader.py: a random caption line is chosen per item,
    corrupt files are skipped by resampling a neighbor, and the crop is a
    ratio-locked RandomResizedCrop.
    """

    def __init__(self, folder, text_len=256, image_size=128, truncate_captions=False,
                 resize_ratio=0.75, tokenizer=None, shuffle=False):
        super().__init__()
        from dalle_pytorch_amd.utils.vision import random_resized_crop, to_tensor

        self.shuffle = shuffle
        path = Path(folder)
        text_files = {p.stem: p for p in path.glob('**/*.txt')}
        image_files = {p.stem: p for ext in IMAGE_EXTS for p in path.glob(f'**/*{ext}')}
        keys = image_files.keys() & text_files.keys()
        self.keys = sorted(keys)
        self.text_files = {k: text_files[k] for k in self.keys}
        self.image_files = {k: image_files[k] for k in self.keys}
        self.text_len = text_len
        self.truncate_captions = truncate_captions
        self.tokenizer = tokenizer

        def transform(img):
            if img.mode != 'RGB':
                img = img.convert('RGB')
            img = random_resized_crop(img, image_size, scale=(resize_ratio, 1.))
            return to_tensor(img)

        self.image_transform = transform

    def __len__(self):
        return len(self.keys)

    def random_sample(self):
        return self[random.randint(0, len(self) - 1)]

    def sequential_sample(self, ind):
        return self[(ind + 1) % len(self)]

    def skip_sample(self, ind):
        if self.shuffle:
            return self.random_sample()
        return self.sequential_sample(ind)

    def __getitem__(self, ind):
        from PIL import UnidentifiedImageError, Image
        key = self.keys[ind]
        text_file = self.text_files[key]
        image_file = self.image_files[key]

        captions = [c for c in text_file.read_text().split('\n') if len(c) > 0]
        if not captions:
            return self.skip_sample(ind)
        description = random.choice(captions)

        try:
            tokens = self.tokenizer.tokenize(
                description, self.text_len,
                truncate_text=self.truncate_captions).squeeze(0)
        except RuntimeError:
            return self.skip_sample(ind)
        try:
            img = Image.open(image_file)
            image_tensor = self.image_transform(img)
        except (UnidentifiedImageError, OSError):
            print(f'skipping unreadable file: {image_file}')
            return self.skip_sample(ind)

        return tokens, image_tensor


class SyntheticTextImageDataset(Dataset):
    """Random-token captions + random images of the benchmark shape.

    Used by ``bench.py`` and smoke tests (BASELINE: synthetic 256x256 images
    + random token captions, random-init weights — no network for datasets).
    Deterministic per index so ranks agree without coordination.
    """

    def __init__(self, length=10000, text_len=256, image_size=256,
                 vocab_size=10000, channels=3, seed=0):
        self.length = length
        self.text_len = text_len
        self.image_size = image_size
        self.vocab_size = vocab_size
        self.channels = channels
        self.seed = seed

    def __len__(self):
        return self.length

    def __getitem__(self, ind):
        g = torch.Generator().manual_seed(self.seed * 1000003 + ind)
        text = torch.randint(1, self.vocab_size, (self.text_len,), generator=g)
        image = torch.rand(self.channels, self.image_size, self.image_size, generator=g)
        return text, image


class RainbowDataset(Dataset):
    """Compositional colored-shapes dataset (the reference notebook's
    integration-test workload): images of a colored square/cross on a
    colored background with the caption fully describing them, so a trained
    model's token accuracy is measurable.
    """

    COLORS = {
        'red': (1., 0., 0.), 'green': (0., 1., 0.), 'blue': (0., 0., 1.),
        'yellow': (1., 1., 0.), 'magenta': (1., 0., 1.), 'cyan': (0., 1., 1.),
        'white': (1., 1., 1.), 'gray': (0.5, 0.5, 0.5),
    }
    SHAPES = ('square', 'cross')

    def __init__(self, length=2048, image_size=32, text_len=16, tokenizer=None, seed=0):
        self.length = length
        self.image_size = image_size
        self.text_len = text_len
        self.tokenizer = tokenizer
        self.seed = seed

    def __len__(self):
        return self.length

    def caption_and_image(self, ind):
        g = random.Random(self.seed * 7919 + ind)
        fg_name, fg = g.choice(list(self.COLORS.items()))
        bg_name, bg = g.choice(list(self.COLORS.items()))
        while bg_name == fg_name:
            bg_name, bg = g.choice(list(self.COLORS.items()))
        shape = g.choice(self.SHAPES)
        s = self.image_size
        img = torch.tensor(bg).reshape(3, 1, 1).expand(3, s, s).clone()
        c, w = s // 2, s // 4
        if shape == 'square':
            img[:, c - w:c + w, c - w:c + w] = torch.tensor(fg).reshape(3, 1, 1)
        else:
            img[:, c - 2:c + 2, c - w:c + w] = torch.tensor(fg).reshape(3, 1, 1)
            img[:, c - w:c + w, c - 2:c + 2] = torch.tensor(fg).reshape(3, 1, 1)
        caption = f'a {fg_name} {shape} on a {bg_name} background'
        return caption, img

    def __getitem__(self, ind):
        caption, img = self.caption_and_image(ind)
        tokens = self.tokenizer.tokenize(caption, self.text_len,
                                         truncate_text=True).squeeze(0)
        return tokens, img
