"""Engine-level CPU tests: gradient accumulation, evaluation, trainer
timeline, fp16-style scaler integration on the CPU path."""
import dataclasses
import json

import torch

from distributeddeeplearning_amd.config import PRESETS, TrainConfig
from distributeddeeplearning_amd.engine.trainer import Trainer


def _cfg(**kw):
    base = dict(model="resnet18", image_size=32, num_classes=10,
                batch_size=8, dtype="fp32", device="cpu", comm="single",
                warmup_epochs=0, base_lr=0.01)
    base.update(kw)
    return TrainConfig(**base)


def test_backward_passes_per_step_accumulates():
    torch.manual_seed(0)
    tr = Trainer(_cfg(backward_passes_per_step=2), steps_per_epoch=10)
    it = tr.data_iter()
    p = next(tr.model.parameters())
    before = p.detach().clone()
    x, y = next(it)
    tr.train_step(x, y)  # pass 1: accumulate only
    assert torch.equal(p.detach(), before), "param changed on accumulation pass"
    x, y = next(it)
    tr.train_step(x, y)  # pass 2: reduce + step
    assert not torch.equal(p.detach(), before), "no update after full step"


def test_evaluate_reports_top1():
    tr = Trainer(_cfg(), steps_per_epoch=10)
    loader = [(torch.randn(4, 3, 32, 32), torch.randint(0, 10, (4,)))
              for _ in range(2)]
    out = tr.evaluate(loader)
    assert 0.0 <= out["top1"] <= 1.0
    assert out["val_loss"] == out["val_loss"]


def test_trainer_timeline(tmp_path):
    cfg = _cfg(timeline=str(tmp_path / "tl"))
    tr = Trainer(cfg, steps_per_epoch=10)
    it = tr.data_iter()
    for _ in range(2):
        x, y = next(it)
        tr.train_step(x, y)
    tr.timeline.save()
    data = json.load(open(str(tmp_path / "tl") + ".rank0.json"))
    names = {e["name"] for e in data["traceEvents"]}
    assert {"forward", "backward+allreduce", "step"} <= names


def test_metrics_jsonl(tmp_path):
    cfg = _cfg(metrics_jsonl=str(tmp_path / "m.jsonl"), log_interval=1)
    tr = Trainer(cfg, steps_per_epoch=2)
    tr.train_epoch(steps=2)
    lines = [json.loads(l) for l in open(tmp_path / "m.jsonl")]
    assert lines and {"step", "loss", "lr", "images_per_sec"} <= set(lines[0])


def test_checkpoint_periodic_and_resume(tmp_path):
    cfg = _cfg(checkpoint_dir=str(tmp_path), epochs=1, steps=2)
    tr = Trainer(cfg, steps_per_epoch=2)
    tr.fit()
    ck = tmp_path / "resnet18_epoch1.pt"
    assert ck.exists()
    cfg2 = _cfg(resume=str(ck))
    tr2 = Trainer(cfg2, steps_per_epoch=2)
    assert tr2.epoch == 1 and tr2.global_step == 2


def test_eval_cli_roundtrip(tmp_path):
    """tools/eval.py end-to-end: train 2 steps, checkpoint, evaluate."""
    import json
    import os
    import subprocess
    import sys
    cfg = _cfg(checkpoint_dir=str(tmp_path), epochs=1, steps=2)
    tr = Trainer(cfg, steps_per_epoch=2)
    tr.fit()
    ck = tmp_path / "resnet18_epoch1.pt"
    root = os.path.join(os.path.dirname(__file__), "..")
    out = subprocess.run(
        [sys.executable, os.path.join(root, "tools", "eval.py"),
         "--checkpoint", str(ck), "--model", "resnet18",
         "--image-size", "32", "--num-classes", "10",
         "--batch-size", "8", "--max-batches", "2"],
        capture_output=True, text=True, timeout=300, cwd=root)
    assert out.returncode == 0, out.stderr[-1500:]
    rec = json.loads(out.stdout.strip().splitlines()[-1])
    assert 0.0 <= rec["top1"] <= 1.0 and rec["val_loss"] == rec["val_loss"]
