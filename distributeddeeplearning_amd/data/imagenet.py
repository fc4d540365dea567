"""Real ImageNet input pipeline (folder-per-class JPEGs).

The reference reads ImageNet either as raw folders (PyTorch/Keras scripts)
or TFRecords (TF script) — SURVEY.md §2.1.  Written against PIL + torch
DataLoader directly (torchvision is not a dependency): deterministic
per-rank sharding, random-resized-crop + horizontal-flip train transform,
normalisation folded into the tensor conversion.  Benchmarks use
``data/synthetic.py``; this path exists for real-training capability parity.
"""
from __future__ import annotations

import os
import random
from typing import Iterator, List, Tuple

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset

IMAGENET_MEAN = np.array([0.485, 0.456, 0.406], dtype=np.float32)
IMAGENET_STD = np.array([0.229, 0.224, 0.225], dtype=np.float32)
_EXTS = (".jpg", ".jpeg", ".png", ".bmp")


class ImageFolderDataset(Dataset):
    """<root>/<class_name>/<image> layout, classes sorted alphabetically."""

    def __init__(self, root: str, image_size: int = 224, train: bool = True):
        from PIL import Image  # noqa: F401 — fail here, loudly, if missing
        self.root = root
        self.image_size = image_size
        self.train = train
        classes = sorted(
            d for d in os.listdir(root) if os.path.isdir(os.path.join(root, d))
        )
        self.class_to_idx = {c: i for i, c in enumerate(classes)}
        self.samples: List[Tuple[str, int]] = []
        for c in classes:
            cdir = os.path.join(root, c)
            for fn in sorted(os.listdir(cdir)):
                if fn.lower().endswith(_EXTS):
                    self.samples.append((os.path.join(cdir, fn), self.class_to_idx[c]))
        if not self.samples:
            raise FileNotFoundError(f"no images found under {root}")

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, i):
        from PIL import Image
        path, label = self.samples[i]
        img = Image.open(path).convert("RGB")
        if self.train:
            img = _random_resized_crop(img, self.image_size)
            if random.random() < 0.5:
                img = img.transpose(Image.FLIP_LEFT_RIGHT)
        else:
            img = _center_crop(img, self.image_size)
        arr = np.asarray(img, dtype=np.float32) / 255.0  # HWC
        arr = (arr - IMAGENET_MEAN) / IMAGENET_STD
        return torch.from_numpy(arr.transpose(2, 0, 1).copy()), label


def _random_resized_crop(img, size, scale=(0.08, 1.0), ratio=(3 / 4, 4 / 3)):
    from PIL import Image
    w, h = img.size
    area = w * h
    for _ in range(10):
        target_area = random.uniform(*scale) * area
        aspect = random.uniform(*ratio)
        cw = int(round((target_area * aspect) ** 0.5))
        ch = int(round((target_area / aspect) ** 0.5))
        if 0 < cw <= w and 0 < ch <= h:
            x = random.randint(0, w - cw)
            y = random.randint(0, h - ch)
            return img.crop((x, y, x + cw, y + ch)).resize((size, size), Image.BILINEAR)
    return _center_crop(img, size)


def _center_crop(img, size):
    from PIL import Image
    w, h = img.size
    s = min(w, h)
    img = img.crop(((w - s) // 2, (h - s) // 2, (w + s) // 2, (h + s) // 2))
    return img.resize((size, size), Image.BILINEAR)


class ShardedSampler(torch.utils.data.Sampler):
    """Deterministic per-rank shard with per-epoch shuffle (the
    DistributedSampler role in the reference's ``_get_sampler``)."""

    def __init__(self, n: int, rank: int, world: int, shuffle=True, seed=0):
        self.n, self.rank, self.world = n, rank, world
        self.shuffle, self.seed, self.epoch = shuffle, seed, 0
        self.per_rank = n // world

    def set_epoch(self, e):
        self.epoch = e

    def __iter__(self):
        idx = list(range(self.n))
        if self.shuffle:
            g = random.Random(self.seed + self.epoch)
            g.shuffle(idx)
        return iter(idx[self.rank * self.per_rank:(self.rank + 1) * self.per_rank])

    def __len__(self):
        return self.per_rank


def imagenet_loader(root, batch_size, image_size=224, dtype=torch.bfloat16,
                    device=None, rank=0, world=1, train=True,
                    channels_last=True, num_workers=8) -> Iterator:
    split_dir = os.path.join(root, "train" if train else "val")
    if not os.path.isdir(split_dir):
        split_dir = root
    ds = ImageFolderDataset(split_dir, image_size, train)
    sampler = ShardedSampler(len(ds), rank, world, shuffle=train)
    dl = DataLoader(ds, batch_size=batch_size, sampler=sampler,
                    num_workers=num_workers, pin_memory=device is not None
                    and device.type == "cuda", drop_last=True,
                    persistent_workers=num_workers > 0)
    epoch = 0
    while True:
        sampler.set_epoch(epoch)
        for x, y in dl:
            if device is not None:
                x = x.to(device, non_blocking=True)
                y = y.to(device, non_blocking=True)
            x = x.to(dtype)
            if channels_last:
                x = x.contiguous(memory_format=torch.channels_last)
            yield x, y
        epoch += 1
