#!/usr/bin/env python3
"""Per-shape conv kernel microbenchmark (fwd/dgrad/wgrad TFLOP/s).

    python tools/kbench.py [--batch 256] [--dtype bf16]

Times each distinct ResNet-50 conv shape with hipEvent-accurate torch
events and prints achieved TFLOP/s per pass — the per-shape view behind
the aggregate images/sec number (guides round-2 kernel work).
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402

SHAPES = [  # (name, C, H, K, R, stride, pad)
    ("l1.c1 64->64 56^2 1x1", 64, 56, 64, 1, 1, 0),
    ("l1.c2 64->64 56^2 3x3", 64, 56, 64, 3, 1, 1),
    ("l1.c3 64->256 56^2 1x1", 64, 56, 256, 1, 1, 0),
    ("l2.c1 256->128 56^2 1x1", 256, 56, 128, 1, 1, 0),
    ("l2.c2 128->128 28^2 3x3s2", 128, 56, 128, 3, 2, 1),
    ("l2.c3 128->512 28^2 1x1", 128, 28, 512, 1, 1, 0),
    ("l3.c1 512->256 28^2 1x1", 512, 28, 256, 1, 1, 0),
    ("l3.c2 256->256 14^2 3x3", 256, 14, 256, 3, 1, 1),
    ("l3.c3 256->1024 14^2 1x1", 256, 14, 1024, 1, 1, 0),
    ("l4.c1 1024->512 14^2 1x1", 1024, 14, 512, 1, 1, 0),
    ("l4.c2 512->512 7^2 3x3", 512, 14, 512, 3, 2, 1),
    ("l4.c3 512->2048 7^2 1x1", 512, 7, 2048, 1, 1, 0),
]


def time_op(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e-3  # seconds


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--dtype", default="bf16")
    args = ap.parse_args()
    dt = {"bf16": torch.bfloat16, "fp16": torch.float16,
          "fp32": torch.float32}[args.dtype]
    from distributeddeeplearning_amd.ops._ext import load_kernels_ext
    ext = load_kernels_ext()
    cl = lambda t: t.contiguous(memory_format=torch.channels_last)
    n = args.batch
    print(f"{'shape':28s} {'fwd TF':>8s} {'dgrad TF':>9s} {'wgrad TF':>9s}")
    for name, c, h, k, r, s, p in SHAPES:
        x = cl(torch.randn(n, c, h, h, dtype=dt, device="cuda") * 0.1)
        w = cl(torch.randn(k, c, r, r, dtype=dt, device="cuda") * 0.1)
        po = (h + 2 * p - r) // s + 1
        dy = cl(torch.randn(n, k, po, po, dtype=dt, device="cuda") * 0.1)
        flops = 2.0 * n * po * po * k * c * r * r
        tf = flops / time_op(lambda: ext.conv_fwd(x, w, None, s, s, p, p, False)) / 1e12
        tg = flops / time_op(lambda: ext.conv_dgrad(dy, w, list(x.shape), s, s, p, p)) / 1e12
        tw = flops / time_op(lambda: ext.conv_wgrad(dy, x, list(w.shape), s, s, p, p)) / 1e12
        print(f"{name:28s} {tf:8.0f} {tg:9.0f} {tw:9.0f}")


if __name__ == "__main__":
    main()
