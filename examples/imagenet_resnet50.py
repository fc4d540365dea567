#!/usr/bin/env python3
"""ResNet-50 / ImageNet distributed data-parallel training — the
reference-script shape (``HorovodPytorch/src/imagenet_pytorch_horovod.py``)
ported call-for-call to distributeddeeplearning_amd.

Launch (one process per GPU, replacing the reference's ``mpirun -np N``):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/imagenet_resnet50.py \
        [--data /path/to/imagenet] [--epochs 90] [--batch-size 256]

Side-by-side with the Horovod original:

    hvd.init()                       -> ddl.init()
    torch.cuda.set_device(
        hvd.local_rank())            -> unchanged (ddl.local_rank())
    torchvision resnet50()           -> ddl.models.resnet50() (in-repo,
                                        same state-dict names)
    optimizer = optim.SGD(...)       -> ddl.ops.FusedSGD(...) (one fused
                                        HIP launch, fp32 masters)
    hvd.DistributedOptimizer(...)    -> ddl.DistributedOptimizer(...)
    hvd.broadcast_parameters(...)    -> ddl.broadcast_parameters(...)
    hvd.broadcast_optimizer_state()  -> ddl.broadcast_optimizer_state()
    LR = base * hvd.size(), warmup   -> WarmupPiecewiseLR (same recipe)
    rank-0 torch.save checkpoints    -> ddl.engine.checkpoint.save (same
                                        fp32/NCHW torchvision layout)
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402

import distributeddeeplearning_amd as ddl  # noqa: E402
from distributeddeeplearning_amd.data.synthetic import synthetic_loader  # noqa: E402
from distributeddeeplearning_amd.engine import checkpoint  # noqa: E402
from distributeddeeplearning_amd.engine.schedule import WarmupPiecewiseLR  # noqa: E402
from distributeddeeplearning_amd.ops import functional as DF  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--data", default=None, help="ImageNet dir (default: synthetic)")
    p.add_argument("--batch-size", type=int, default=256)
    p.add_argument("--epochs", type=int, default=90)
    p.add_argument("--steps-per-epoch", type=int, default=None)
    p.add_argument("--base-lr", type=float, default=0.1)
    p.add_argument("--dtype", default="bf16")
    p.add_argument("--checkpoint-dir", default=None)
    args = p.parse_args()

    # 1. init + device pinning (hvd.init / set_device(local_rank))
    ddl.init()
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda", ddl.local_rank()) if use_gpu else torch.device("cpu")
    if use_gpu:
        torch.cuda.set_device(device)
    dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
             "fp32": torch.float32}[args.dtype if use_gpu else "fp32"]

    # 2. model (+ NHWC/bf16 cast) and optimizer (lr scaled by world size via
    #    the warmup schedule, as in Goyal et al.)
    model = ddl.models.resnet50().to(device)
    ddl.ops.cast_model_(model, dtype)
    optimizer = ddl.DistributedOptimizer(
        ddl.ops.FusedSGD(model.parameters(), lr=args.base_lr, momentum=0.9,
                         weight_decay=1e-4),
        named_parameters=model.named_parameters())

    # 3. broadcast initial state from rank 0 (C2/C3)
    ddl.broadcast_parameters(model.state_dict(), root_rank=0)
    ddl.broadcast_optimizer_state(optimizer, root_rank=0)
    optimizer.optimizer.refresh_master()

    # 4. data
    global_batch = args.batch_size * ddl.size()
    steps_per_epoch = args.steps_per_epoch or max(1, 1_281_167 // global_batch)
    if args.data:
        from distributeddeeplearning_amd.data.imagenet import imagenet_loader
        batches = imagenet_loader(args.data, args.batch_size, 224, dtype,
                                  device, rank=ddl.rank(), world=ddl.size())
    else:
        batches = synthetic_loader(args.batch_size, 224, 1000, dtype, device,
                                   seed=1234 + ddl.rank())
    schedule = WarmupPiecewiseLR(optimizer, args.base_lr, global_batch,
                                 steps_per_epoch)

    # 5. the hot loop: forward / backward (hooks overlap
    #    the bucketed RCCL all-reduce) / fused step
    step = 0
    for epoch in range(args.epochs):
        model.train()
        t0, seen = time.time(), 0
        for _ in range(steps_per_epoch):
            x, y = next(batches)
            schedule.step(step)
            loss = DF.softmax_cross_entropy(model(x), y)
            loss.backward()
            optimizer.step()
            optimizer.zero_grad()
            step += 1
            seen += global_batch
            if ddl.rank() == 0 and step % 10 == 0:
                if use_gpu:
                    torch.cuda.synchronize()
                rate = seen / (time.time() - t0)
                print(f"epoch {epoch} step {step} loss "
                      f"{float(loss.detach().float()):.4f} {rate:.0f} images/sec",
                      flush=True)
                t0, seen = time.time(), 0
        if args.checkpoint_dir:  # rank-0-only, torchvision-compatible layout
            checkpoint.save(f"{args.checkpoint_dir}/resnet50_epoch{epoch+1}.pt",
                            model, optimizer, epoch=epoch + 1, step=step)
    ddl.shutdown()


if __name__ == "__main__":
    main()
