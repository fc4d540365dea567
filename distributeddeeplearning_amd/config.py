"""Training configuration (dataclass + env/CLI overlay).

The reference spreads knobs over argparse flags, ``HOROVOD_*`` env vars,
Makefile vars and job-template JSON.  Here one dataclass
carries them all; ``PRESETS`` holds the five BASELINE.json acceptance
configs by name.
"""
from __future__ import annotations

import argparse
import dataclasses
import os
from dataclasses import dataclass
from typing import Optional

import torch

_DTYPES = {"fp32": torch.float32, "bf16": torch.bfloat16, "fp16": torch.float16}


@dataclass
class TrainConfig:
    model: str = "resnet50"
    image_size: int = 224
    num_classes: int = 1000
    batch_size: int = 256            # per GPU
    dtype: str = "bf16"              # fp32 | bf16 | fp16
    data: str = "synthetic"          # synthetic | path to ImageNet dir
    epochs: int = 90
    steps: Optional[int] = None      # cap steps (bench mode)
    warmup_steps: int = 0            # untimed bench warmup

    # optimizer (Goyal et al. recipe defaults; lr is PER-256-global-batch
    # base — engine scales linearly with world batch)
    base_lr: float = 0.1
    momentum: float = 0.9
    weight_decay: float = 1e-4
    warmup_epochs: int = 5
    lr_decay_epochs: tuple = (30, 60, 80)
    lr_decay_factor: float = 0.1
    label_smoothing: float = 0.0
    lars: bool = False
    lars_eta: float = 0.001
    loss_scale: float = 0.0          # 0 = dynamic (fp16 only)
    zero_init_residual: bool = False

    # comm
    comm: str = "auto"               # auto | native | torchdist | single
    bucket_mb: float = 25.0
    backward_passes_per_step: int = 1
    autotune_buckets: bool = False   # online bucket-size tuner (DDL_AUTOTUNE)

    # engine
    device: str = "auto"             # auto | cpu | cuda
    channels_last: bool = True
    log_interval: int = 10
    checkpoint_dir: Optional[str] = None
    checkpoint_interval_epochs: int = 1
    resume: Optional[str] = None
    timeline: Optional[str] = None   # chrome-trace output path
    metrics_jsonl: Optional[str] = None
    seed: int = 42

    def torch_dtype(self) -> torch.dtype:
        return _DTYPES[self.dtype]

    def resolved_device(self) -> torch.device:
        if self.device == "auto":
            return torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        return torch.device(self.device)

    @classmethod
    def from_yaml(cls, path: str) -> "TrainConfig":
        import yaml
        with open(path) as f:
            d = yaml.safe_load(f)
        d = {k: tuple(v) if isinstance(v, list) else v for k, v in d.items()}
        return cls(**d)

    @classmethod
    def from_args(cls, argv=None, **overrides):
        parser = argparse.ArgumentParser(description="distributeddeeplearning_amd trainer")
        defaults = cls(**overrides)
        for f in dataclasses.fields(cls):
            name = f.name.replace("_", "-")
            default = getattr(defaults, f.name)
            if f.type == "bool" or isinstance(default, bool):
                parser.add_argument(f"--{name}", type=lambda s: s.lower() in ("1", "true", "yes"),
                                    default=default)
            elif isinstance(default, tuple):
                parser.add_argument(f"--{name}", type=lambda s: tuple(int(x) for x in s.split(",")),
                                    default=default)
            else:
                if default is not None:
                    typ = type(default)
                else:
                    # Optional[...] fields: infer from the annotation so
                    # e.g. --steps parses as int, not str
                    ann = str(f.type)
                    typ = int if "int" in ann else (
                        float if "float" in ann else str)
                parser.add_argument(f"--{name}", type=typ, default=default)
        parser.add_argument("--preset", type=str, default=None,
                            help=f"named preset: {sorted(PRESETS)}")
        parser.add_argument("--config", type=str, default=None,
                            help="YAML config file (see configs/)")
        ns = parser.parse_args(argv)
        if ns.config:
            base = dataclasses.asdict(cls.from_yaml(ns.config))
            given = {a.dest for a in parser._actions
                     if getattr(ns, a.dest, None) != a.default
                     and a.dest not in ("preset", "config")}
            for k, v in base.items():
                if k not in given:
                    setattr(ns, k, v)
        if ns.preset:
            base = dataclasses.asdict(PRESETS[ns.preset])
            # CLI flags explicitly given override preset values
            given = {a.dest for a in parser._actions
                     if getattr(ns, a.dest, None) != a.default and a.dest != "preset"}
            for k, v in base.items():
                if k not in given:
                    setattr(ns, k, v if not isinstance(v, list) else tuple(v))
        kwargs = {f.name: getattr(ns, f.name) for f in dataclasses.fields(cls)}
        cfg = cls(**kwargs)
        # env overlay (DDL_*)
        if "DDL_BUCKET_MB" in os.environ:
            cfg.bucket_mb = float(os.environ["DDL_BUCKET_MB"])
        if "DDL_COMM" in os.environ:
            cfg.comm = os.environ["DDL_COMM"]
        if os.environ.get("DDL_AUTOTUNE") == "1":
            cfg.autotune_buckets = True
        return cfg


# The five acceptance configs (BASELINE.json:7-11).
PRESETS = {
    # 1. plumbing: runs without a GPU, CPU/gloo world_size=2
    "cpu-plumbing": TrainConfig(
        model="resnet18", image_size=32, num_classes=10, batch_size=32,
        dtype="fp32", device="cpu", comm="torchdist", bucket_mb=1.0,
        base_lr=0.01, warmup_epochs=0,
    ),
    # 2. headline: ResNet-50 bf16 synthetic ImageNet, DP=8
    "resnet50-bf16": TrainConfig(
        model="resnet50", image_size=224, batch_size=256, dtype="bf16",
    ),
    # 3. large-batch numerics: fp16 + LARS, global batch 8k on 8 GPUs
    "resnet50-fp16-lars-b8k": TrainConfig(
        model="resnet50", image_size=224, batch_size=1024, dtype="fp16",
        lars=True, base_lr=0.1, warmup_epochs=5, loss_scale=0.0,
        zero_init_residual=True,
    ),
    # 4. deeper net: stresses allreduce/backward overlap
    "resnet152-bf16": TrainConfig(
        model="resnet152", image_size=224, batch_size=128, dtype="bf16",
    ),
    # 5. bandwidth-bound: stresses gradient bucket fusion on xGMI
    "vgg16-bf16": TrainConfig(
        model="vgg16", image_size=224, batch_size=128, dtype="bf16",
        base_lr=0.01, weight_decay=5e-4,
    ),
}
