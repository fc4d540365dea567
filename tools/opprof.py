#!/usr/bin/env python3
"""torch.profiler view of the training step — op names + input shapes with
device time, where rocprofv3 only shows kernel symbols.  Used to attribute
the `at::native::vectorized_elementwise` population (autograd AccumulateGrad
adds vs fan-out gradient adds vs everything else).

    python tools/opprof.py [--batch-size 1024] [--steps 3]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch-size", type=int, default=1024)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--model", default="resnet50")
    args = ap.parse_args()

    from distributeddeeplearning_amd.config import TrainConfig
    from distributeddeeplearning_amd.engine.trainer import Trainer
    cfg = TrainConfig(model=args.model, image_size=224, num_classes=1000,
                      batch_size=args.batch_size, dtype="bf16",
                      warmup_epochs=0, comm="single")
    tr = Trainer(cfg, steps_per_epoch=10)
    it = tr.data_iter()
    for _ in range(3):  # warmup
        x, y = next(it)
        tr.train_step(x, y)
    torch.cuda.synchronize()
    with torch.profiler.profile(
            activities=[torch.profiler.ProfilerActivity.CPU,
                        torch.profiler.ProfilerActivity.CUDA],
            record_shapes=True) as prof:
        for _ in range(args.steps):
            x, y = next(it)
            tr.train_step(x, y)
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_input_shape=True).table(
        sort_by="cuda_time_total", row_limit=40, max_src_column_width=40))
    # targeted: the unfused elementwise population (autograd adds etc.)
    print("\n== elementwise attribution (self CUDA, by name+shape) ==")
    rows = []
    for ev in prof.key_averages(group_by_input_shape=True):
        if any(k in ev.key for k in ("add_", "add", "copy_", "zero_",
                                     "fill_", "mul_", "div_")):
            t = ev.self_device_time_total
            if t > 0:
                rows.append((t, ev.key, ev.count, str(ev.input_shapes)[:80]))
    for t, name, cnt, shp in sorted(rows, reverse=True)[:25]:
        print(f"{t/1e3:9.3f} ms  x{cnt:<5d} {name:<28.28s} {shp}")


if __name__ == "__main__":
    main()
