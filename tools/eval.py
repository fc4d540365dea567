#!/usr/bin/env python3
"""Standalone checkpoint evaluation (the reference scripts' validation
pass as its own entry point):

    python tools/eval.py --checkpoint ckpts/resnet50_epoch90.pt \
        --model resnet50 [--data /path/to/imagenet] [--batch-size 256]

With no --data, evaluates on the seeded synthetic validation set (sanity/
CI).  Multi-GPU: launch under tools/launch.py / torchrun — top-1 and
val_loss are averaged across ranks (C4 metric averaging).
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--checkpoint", required=True)
    p.add_argument("--model", default="resnet50")
    p.add_argument("--data", default=None)
    p.add_argument("--batch-size", type=int, default=256)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--dtype", default="bf16")
    p.add_argument("--max-batches", type=int, default=None)
    args = p.parse_args()

    import torch
    from distributeddeeplearning_amd.config import TrainConfig
    from distributeddeeplearning_amd.engine.trainer import Trainer
    from distributeddeeplearning_amd.parallel import api

    on_gpu = torch.cuda.is_available()
    cfg = TrainConfig(model=args.model, image_size=args.image_size,
                      num_classes=args.num_classes,
                      batch_size=args.batch_size,
                      dtype=args.dtype if on_gpu else "fp32",
                      device="auto", warmup_epochs=0,
                      resume=args.checkpoint)
    tr = Trainer(cfg, steps_per_epoch=10 ** 9)
    if args.data:
        from distributeddeeplearning_amd.data.imagenet import (
            ImageFolderDataset, ShardedSampler)
        from torch.utils.data import DataLoader
        ds = ImageFolderDataset(
            os.path.join(args.data, "val")
            if os.path.isdir(os.path.join(args.data, "val")) else args.data,
            args.image_size, train=False)
        dl = DataLoader(ds, batch_size=args.batch_size,
                        sampler=ShardedSampler(len(ds), tr.rank, tr.world,
                                               shuffle=False),
                        num_workers=8, drop_last=False)
        loader = iter(dl)
    else:
        from distributeddeeplearning_amd.data.synthetic import SyntheticImageNet
        from torch.utils.data import DataLoader
        ds = SyntheticImageNet(n=args.batch_size * 4,
                               image_size=args.image_size,
                               num_classes=args.num_classes, seed=7)
        loader = iter(DataLoader(ds, batch_size=args.batch_size))
    out = tr.evaluate(loader, max_batches=args.max_batches)
    if tr.rank == 0:
        import json
        print(json.dumps({"checkpoint": args.checkpoint, **out}))
    api.shutdown()


if __name__ == "__main__":
    main()
