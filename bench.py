#!/usr/bin/env python3
"""Driver benchmark contract: flagship training step on N GPUs of one node.

Measures BASELINE.json's headline metric — images/sec (whole node),
ResNet-50, synthetic ImageNet 224x224, random-init weights, bf16 compute —
with W untimed warmup steps and exactly K timed steps bracketed by
barrier + torch.cuda.synchronize on both sides; elapsed is MAX over ranks;
rank 0 prints ONE JSON line.

Launch (driver): N=1 plain, N>1 via torch.distributed.run with
--nproc-per-node N (one rank per GPU over RCCL).
"""
from __future__ import annotations

import argparse
import json
import os

import torch


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch-size", type=int, default=1024)
    p.add_argument("--model", type=str, default="resnet50")
    p.add_argument("--dtype", type=str, default="bf16")
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--bucket-mb", type=float, default=float(os.environ.get("DDL_BUCKET_MB", "25")))
    args = p.parse_args()

    from distributeddeeplearning_amd.config import TrainConfig
    from distributeddeeplearning_amd.engine.trainer import Trainer
    from distributeddeeplearning_amd.parallel import api

    on_gpu = torch.cuda.is_available()
    if on_gpu:
        cfg = TrainConfig(
            model=args.model, image_size=args.image_size, dtype=args.dtype,
            batch_size=args.batch_size, data="synthetic",
            bucket_mb=args.bucket_mb, warmup_epochs=0,
        )
    else:
        # no-GPU sanity fallback (dev container only); NOT the headline config
        cfg = TrainConfig(model="resnet18", image_size=32, num_classes=10,
                          batch_size=4, dtype="fp32", device="cpu",
                          warmup_epochs=0, comm="torchdist")
        args.steps = min(args.steps, 5)
        args.warmup = min(args.warmup, 2)

    trainer = Trainer(cfg, steps_per_epoch=10 ** 9)  # flat LR during bench
    n_ranks = trainer.world
    res = trainer.bench(args.steps, args.warmup)

    if trainer.rank == 0:
        # Provenance: record which comm path actually ran,
        # bucket layout and environment so every BENCH/SCALE record is
        # self-describing — a torchdist fallback can never be mistaken for
        # the native engine.
        try:
            import subprocess
            head = subprocess.run(
                ["git", "rev-parse", "--short=12", "HEAD"], cwd=os.path.dirname(
                    os.path.abspath(__file__)), capture_output=True, text=True,
                timeout=10).stdout.strip() or None
        except Exception:
            head = None  # gpurun snapshots exclude .git
        provenance = {
            "comm_backend": api.comm_backend_name(),
            "hipgraph": bool(getattr(trainer, "used_hipgraph", False)),
            "bucket_mb": cfg.bucket_mb,
            "n_buckets": len(trainer.opt.buckets.buckets),
            "git_head": head,
            "env": {k: v for k, v in sorted(os.environ.items())
                    if k.startswith(("DDL_", "NCCL_", "RCCL_"))},
        }
        metric = "images/sec (whole node) ResNet-50 synthetic ImageNet"
        if on_gpu and args.model != "resnet50":
            # side benches (R152/VGG/...) must not masquerade as the
            # BASELINE headline metric
            metric = f"images/sec (whole node) {args.model} synthetic ImageNet"
        out = {
            "metric": metric,
            "value": round(res["images_per_sec"], 2),
            "unit": "images/sec",
            "n_gpus": n_ranks,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(res["ms_per_step"], 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # BASELINE.json "published" is empty
            "dtype": cfg.dtype,
            "data": "synthetic",
            "config": {
                "model": cfg.model if on_gpu else f"{cfg.model}-cpu-fallback",
                "global_batch": cfg.batch_size * n_ranks,
                "image_size": cfg.image_size,
                "parallelism": f"dp{n_ranks}",
            },
            "provenance": provenance,
        }
        print(json.dumps(out), flush=True)
    api.shutdown()


if __name__ == "__main__":
    main()
