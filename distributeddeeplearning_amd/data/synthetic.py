"""Synthetic ImageNet data (the reference's fake-data benchmark mode).

The reference trainers build random tensors of ImageNet shape when no data
dir is given (``imagenet_pytorch_horovod.py`` synthetic path; TF
``input_fn`` from random tensors — SURVEY.md §4.1).  Headline metric is
measured on exactly this (BASELINE.json: "synthetic 224×224, random-init
weights").

``synthetic_loader`` yields pre-generated device-resident batches (NHWC,
compute dtype) so the input pipeline costs zero host↔device traffic in the
timed region — data is "already there", as HBM3E capacity (288 GB) makes
natural on MI355X.
"""
from __future__ import annotations

import itertools
from typing import Iterator, Optional, Tuple

import torch
from torch.utils.data import Dataset


class SyntheticImageNet(Dataset):
    """Seeded CPU dataset of random images/labels (tests, loader parity)."""

    def __init__(self, n: int = 1024, image_size: int = 224, num_classes: int = 1000,
                 channels: int = 3, seed: int = 0):
        self.n = n
        self.image_size = image_size
        self.num_classes = num_classes
        self.channels = channels
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, i) -> Tuple[torch.Tensor, int]:
        g = torch.Generator().manual_seed(self.seed * 1_000_003 + i)
        x = torch.randn(self.channels, self.image_size, self.image_size, generator=g)
        y = int(torch.randint(0, self.num_classes, (1,), generator=g))
        return x, y


def synthetic_loader(
    batch_size: int,
    image_size: int = 224,
    num_classes: int = 1000,
    dtype: torch.dtype = torch.bfloat16,
    device: Optional[torch.device] = None,
    n_batches: int = 4,
    seed: int = 1234,
    channels_last: bool = True,
    steps: Optional[int] = None,
) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
    """Cycle over ``n_batches`` pre-generated device batches."""
    device = device or (torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu"))
    g = torch.Generator(device="cpu").manual_seed(seed)
    batches = []
    for _ in range(n_batches):
        x = torch.randn(batch_size, 3, image_size, image_size, generator=g)
        y = torch.randint(0, num_classes, (batch_size,), generator=g)
        x = x.to(device=device, dtype=dtype)
        if channels_last:
            x = x.contiguous(memory_format=torch.channels_last)
        batches.append((x, y.to(device)))
    it = itertools.cycle(batches)
    if steps is None:
        return it
    return itertools.islice(it, steps)
