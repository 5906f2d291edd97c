"""PIL-native image transforms (this environment has no torchvision).

Implements exactly what the data layer needs: ratio-locked
RandomResizedCrop (reference loader.py:50-57 semantics), resize+center-crop
(train_vae path), tensor conversion, and PNG saving.
"""

import math
import random
from pathlib import Path

import torch


def to_tensor(img):
    """PIL RGB image -> float tensor [C, H, W] in [0, 1]."""
    import numpy as np
    arr = np.asarray(img, dtype=np.uint8)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    t = torch.from_numpy(arr.copy()).permute(2, 0, 1).float() / 255.0
    return t


def tensor_to_pil(t):
    from PIL import Image
    import numpy as np
    t = t.detach().float().clamp(0, 1).mul(255).round().byte().cpu()
    arr = t.permute(1, 2, 0).numpy()
    if arr.shape[2] == 1:
        arr = arr[:, :, 0]
    return Image.fromarray(arr)


def save_image(tensor, path):
    tensor_to_pil(tensor).save(str(path))


def random_resized_crop(img, size, scale=(0.75, 1.0), ratio=(1.0, 1.0),
                        rng=random):
    """Square crop of random area in `scale`, resized to `size`."""
    from PIL import Image
    w, h = img.size
    area = w * h
    for _ in range(10):
        target_area = rng.uniform(*scale) * area
        aspect = math.exp(rng.uniform(math.log(ratio[0]), math.log(ratio[1])))
        cw = int(round(math.sqrt(target_area * aspect)))
        ch = int(round(math.sqrt(target_area / aspect)))
        if cw <= w and ch <= h:
            x = rng.randint(0, w - cw)
            y = rng.randint(0, h - ch)
            img = img.crop((x, y, x + cw, y + ch))
            return img.resize((size, size), Image.BICUBIC)
    # fallback: center crop of the short side
    s = min(w, h)
    x, y = (w - s) // 2, (h - s) // 2
    return img.crop((x, y, x + s, y + s)).resize((size, size), Image.BICUBIC)


def resize_center_crop(img, size):
    from PIL import Image
    w, h = img.size
    short = min(w, h)
    nw, nh = int(round(w * size / short)), int(round(h * size / short))
    img = img.resize((nw, nh), Image.BICUBIC)
    x, y = (nw - size) // 2, (nh - size) // 2
    return img.crop((x, y, x + size, y + size))


class ImageFolderDataset(torch.utils.data.Dataset):
    """Recursively collects images under a folder; returns (tensor, 0)."""

    EXTS = ('.png', '.jpg', '.jpeg', '.bmp', '.webp')

    def __init__(self, folder, image_size, transparent=False):
        self.paths = sorted(p for p in Path(folder).rglob('*')
                            if p.suffix.lower() in self.EXTS)
        self.image_size = image_size
        self.mode = 'RGBA' if transparent else 'RGB'

    def __len__(self):
        return len(self.paths)

    def __getitem__(self, i):
        from PIL import Image
        try:
            img = Image.open(self.paths[i])
            if img.mode != self.mode:
                img = img.convert(self.mode)
            img = resize_center_crop(img, self.image_size)
            return to_tensor(img), 0
        except OSError:
            return self[(i + 1) % len(self)]
