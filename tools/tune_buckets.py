#!/usr/bin/env python3
"""Gradient-bucket size sweep (the static replacement for Horovod's
HOROVOD_AUTOTUNE — SURVEY.md §2.2 Parameter autotuner row).

Run under the launcher on an N-GPU node:
    python tools/launch.py --nproc 8 tools/tune_buckets.py --model resnet50
Rank 0 prints images/sec per bucket size; export the winner via
DDL_BUCKET_MB.
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet50")
    p.add_argument("--batch-size", type=int, default=256)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--dtype", default="bf16")
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--sizes-mb", default="5,10,25,50,100")
    args = p.parse_args()

    from distributeddeeplearning_amd.config import TrainConfig
    from distributeddeeplearning_amd.engine.trainer import Trainer
    from distributeddeeplearning_amd.parallel import api

    results = {}
    for mb in [float(x) for x in args.sizes_mb.split(",")]:
        cfg = TrainConfig(model=args.model, batch_size=args.batch_size,
                          image_size=args.image_size,
                          num_classes=args.num_classes,
                          dtype=args.dtype, bucket_mb=mb, warmup_epochs=0)
        tr = Trainer(cfg, steps_per_epoch=10 ** 9)
        res = tr.bench(args.steps, args.warmup)
        results[mb] = res["images_per_sec"]
        if tr.rank == 0:
            print(f"bucket {mb:6.1f} MB -> {res['images_per_sec']:.1f} images/sec",
                  flush=True)
        # fresh model per size; free memory
        del tr
        import torch
        if torch.cuda.is_available():
            torch.cuda.empty_cache()
    if api.is_initialized() and api.rank() == 0:
        best = max(results, key=results.get)
        print(json.dumps({"results": results, "best_bucket_mb": best}))
    api.shutdown()


if __name__ == "__main__":
    main()
