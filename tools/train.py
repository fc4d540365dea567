#!/usr/bin/env python3
"""Full training entry point (the reference's
``imagenet_pytorch_horovod.py`` role):

    # single GPU / CPU
    python tools/train.py --preset resnet50-bf16 --epochs 90 \
        --checkpoint-dir ckpts --data /path/to/imagenet   # or synthetic
    # 8 GPUs
    python tools/launch.py --nproc 8 tools/train.py --preset resnet50-bf16
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from distributeddeeplearning_amd.config import TrainConfig  # noqa: E402
from distributeddeeplearning_amd.engine.trainer import Trainer  # noqa: E402
from distributeddeeplearning_amd.parallel import api  # noqa: E402


def main():
    cfg = TrainConfig.from_args()
    trainer = Trainer(cfg)
    try:
        trainer.fit()
    finally:
        api.shutdown()


if __name__ == "__main__":
    main()
