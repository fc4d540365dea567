#!/usr/bin/env python3
"""Inference (serving) benchmark: eval-mode forward throughput/latency.

    python tools/infer_bench.py --model resnet50 --batch-size 256

Uses the same HIP kernel path as training (BN folds running stats via the
single-pass bn_infer kernel; dropout disabled).  Prints one JSON line with
images/sec and per-batch latency percentiles."""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet50")
    p.add_argument("--batch-size", type=int, default=256)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--dtype", default="bf16")
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--graph", action="store_true",
                   help="capture the forward in a hipGraph and replay it "
                        "(removes per-kernel launch overhead — the bound "
                        "at small serving batches)")
    args = p.parse_args()

    from distributeddeeplearning_amd import models
    from distributeddeeplearning_amd.ops import cast_model_
    from distributeddeeplearning_amd.data.synthetic import synthetic_loader

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    dtype = {"fp32": torch.float32, "bf16": torch.bfloat16,
             "fp16": torch.float16}[args.dtype if dev.type == "cuda" else "fp32"]
    model = models.build_model(args.model).to(dev)
    cast_model_(model, dtype)
    model.eval()
    it = synthetic_loader(args.batch_size, args.image_size, 1000, dtype, dev)

    graphed = None
    static_x = None
    if args.graph and dev.type == "cuda":
        # hipGraph capture (torch.cuda.CUDAGraph is hipGraph on ROCm):
        # warm up on a side stream, capture one eval forward on static
        # input, then replay with copy-in
        static_x, _ = next(it)
        static_x = static_x.clone()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side), torch.no_grad():
            for _ in range(3):
                model(static_x)
        torch.cuda.current_stream().wait_stream(side)
        graphed = torch.cuda.CUDAGraph()
        with torch.no_grad(), torch.cuda.graph(graphed):
            static_out = model(static_x)

    lat = []
    with torch.no_grad():
        for i in range(args.warmup + args.steps):
            x, _ = next(it)
            if dev.type == "cuda":
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            if graphed is not None:
                static_x.copy_(x)
                graphed.replay()
                out = static_out
            else:
                out = model(x)
            if dev.type == "cuda":
                torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            if i >= args.warmup:
                lat.append(dt)
    lat.sort()
    n = len(lat)
    total = sum(lat)
    print(json.dumps({
        "metric": "inference images/sec",
        "value": round(args.batch_size * n / total, 2),
        "model": args.model, "batch_size": args.batch_size,
        "dtype": args.dtype, "hipgraph": bool(graphed is not None),
        "latency_ms": {"p50": round(lat[n // 2] * 1e3, 3),
                       "p99": round(lat[min(n - 1, int(n * 0.99))] * 1e3, 3)},
    }))


if __name__ == "__main__":
    main()
