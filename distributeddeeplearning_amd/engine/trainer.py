"""Training engine: the reference's per-rank hot loop as a
class.  One process per GPU; backward overlaps bucketed all-reduce; the only
sync point is ``DistributedOptimizer.synchronize()`` inside ``step``."""
from __future__ import annotations

import os
import time
from typing import Iterator, Optional, Tuple

import torch

from .. import models as model_zoo
from ..config import TrainConfig
from ..data.synthetic import synthetic_loader
from ..ops import FusedSGD, cast_model_
from ..ops import functional as DF
from ..parallel import DistributedOptimizer, api
from . import checkpoint as ckpt
from .loss_scaler import DynamicLossScaler
from .metrics import MetricLogger, StepTimer, log, setup_logging
from .schedule import WarmupPiecewiseLR


class Trainer:
    def __init__(self, cfg: TrainConfig, steps_per_epoch: Optional[int] = None):
        self.cfg = cfg
        api.init(cfg.comm)
        self.rank, self.world = api.rank(), api.size()
        setup_logging(self.rank)
        torch.manual_seed(cfg.seed + self.rank)

        self.device = cfg.resolved_device()
        if self.device.type == "cuda":
            torch.cuda.set_device(api.local_rank())
            self.device = torch.device("cuda", api.local_rank())

        dtype = cfg.torch_dtype()
        model = model_zoo.build_model(
            cfg.model, num_classes=cfg.num_classes,
            zero_init_residual=cfg.zero_init_residual,
        ) if cfg.model.startswith("resnet") else model_zoo.build_model(
            cfg.model, num_classes=cfg.num_classes)
        model = model.to(self.device)
        cast_model_(model, dtype, channels_last=cfg.channels_last)
        self.model = model
        self.dtype = dtype

        # steps/epoch: synthetic pretends a 1.28 M-image epoch like ImageNet
        global_batch = cfg.batch_size * self.world
        self.steps_per_epoch = steps_per_epoch or max(1, 1_281_167 // global_batch)
        self.global_batch = global_batch

        fp16 = dtype == torch.float16
        self.scaler = DynamicLossScaler(
            init_scale=cfg.loss_scale if cfg.loss_scale > 0 else 2.0 ** 16,
            enabled=fp16,
        )
        if cfg.loss_scale > 0:
            self.scaler.growth_interval = 10 ** 9  # static scale

        inner = FusedSGD(
            model.parameters(), lr=cfg.base_lr, momentum=cfg.momentum,
            weight_decay=cfg.weight_decay, lars=cfg.lars,
            lars_eta=cfg.lars_eta, grad_scale=self.scaler.scale,
        )
        self.opt = DistributedOptimizer(
            inner, named_parameters=model.named_parameters(),
            backward_passes_per_step=cfg.backward_passes_per_step,
            bucket_bytes=int(cfg.bucket_mb * (1 << 20)),
        )
        self.lr_sched = WarmupPiecewiseLR(
            self.opt, cfg.base_lr, global_batch, self.steps_per_epoch,
            cfg.warmup_epochs, cfg.lr_decay_epochs, cfg.lr_decay_factor,
        )
        self.metrics = MetricLogger(cfg.metrics_jsonl, self.rank)
        self.global_step = 0
        self.epoch = 0

        # Chrome-trace timeline (HOROVOD_TIMELINE equivalent): engine-level
        # spans from Python, per-collective spans from the native comm engine.
        tl_path = cfg.timeline or os.environ.get("DDL_TIMELINE")
        self.timeline = None
        if tl_path:
            from ..utils.timeline import Timeline
            self.timeline = Timeline(f"{tl_path}.rank{self.rank}.json", self.rank)
            self.opt.set_timeline(self.timeline)
            backend = api._backend
            if hasattr(backend, "engine"):
                backend.engine.enable_timeline(
                    f"{tl_path}.rank{self.rank}.comm.json")

        if cfg.resume:
            payload = ckpt.load(cfg.resume, self.model, self.opt, self.scaler,
                                device=self.device)
            self.epoch = payload.get("epoch", 0)
            self.global_step = payload.get("step", 0)
            log.info("resumed from %s at epoch %d step %d", cfg.resume,
                     self.epoch, self.global_step)
        elif self.world > 1:
            # C2/C3: one fused broadcast each of params and optimizer state
            api.broadcast_parameters(self.model.state_dict(), root_rank=0)
            self.opt.optimizer.refresh_master()

        self._loader: Optional[Iterator] = None
        self._micro = 0  # backward passes since the last optimizer step
        # online bucket autotune (Horovod parameter_manager role): walks a
        # candidate ladder during early training, every rank picks the
        # allreduced winner, buckets rebuild once
        self.bucket_tuner = None
        if (cfg.autotune_buckets
                or os.environ.get("DDL_AUTOTUNE") == "1"):
            from ..parallel.autotune import OnlineBucketTuner
            self.bucket_tuner = OnlineBucketTuner(self.opt)

    # ------------------------------------------------------------------

    def data_iter(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        if self._loader is None:
            if self.cfg.data == "synthetic":
                self._loader = synthetic_loader(
                    self.cfg.batch_size, self.cfg.image_size,
                    self.cfg.num_classes, self.dtype, self.device,
                    seed=self.cfg.seed + self.rank,
                    channels_last=self.cfg.channels_last,
                )
            else:
                from ..data.imagenet import imagenet_loader
                self._loader = imagenet_loader(
                    self.cfg.data, self.cfg.batch_size, self.cfg.image_size,
                    self.dtype, self.device, rank=self.rank, world=self.world,
                    channels_last=self.cfg.channels_last,
                )
        return self._loader

    def train_step(self, x, y) -> torch.Tensor:
        """One micro-batch: forward+backward; every
        ``backward_passes_per_step``-th call also reduces and applies the
        update (the Horovod accumulation contract: N backwards, one step)."""
        tl = self.timeline
        # the schedule counts OPTIMIZER steps: with gradient accumulation a
        # window of N micro-batches advances the LR once (Goyal warmup is
        # defined over global-batch steps)
        lr = self.lr_sched.step(
            self.global_step // self.cfg.backward_passes_per_step)
        if tl:
            tl.begin("forward")
            tl.begin_device("forward")
        logits = self.model(x)
        loss = DF.softmax_cross_entropy(logits, y, self.cfg.label_smoothing)
        if tl:
            tl.end("forward")
            tl.end_device("forward")
            tl.begin("backward+allreduce")
            tl.begin_device("backward+allreduce")
        self.scaler.scale_loss(loss).backward()
        if tl:
            tl.end("backward+allreduce")
            tl.end_device("backward+allreduce")
        self._micro += 1
        if self._micro < self.cfg.backward_passes_per_step:
            # mid-accumulation: grads keep accumulating in the flat buckets
            self.global_step += 1
            return loss
        self._micro = 0
        if tl:
            tl.begin("step")
            tl.begin_device("step")
        if self.scaler.enabled:
            self.opt.synchronize()
            flats = [b.flat for b in self.opt.buckets.buckets]
            self.opt.optimizer.grad_scale = self.scaler.scale
            if self.scaler.check_and_update(flats):
                self.opt.step()
            # on overflow: skip the update; synchronize() already re-armed
            # the accumulation counters and zero_grad below discards the
            # non-finite gradients, so the next window starts clean
        else:
            self.opt.step()
        self.opt.zero_grad()
        if self.bucket_tuner is not None and not self.bucket_tuner.done:
            self.bucket_tuner.step()
        if tl:
            tl.end("step")
            tl.end_device("step")
            if self.global_step % 50 == 0:
                tl.save()
        self.global_step += 1
        return loss

    def train_epoch(self, steps: Optional[int] = None) -> float:
        self.model.train()
        steps = steps or self.steps_per_epoch
        it = self.data_iter()
        timer = StepTimer(self.device)
        last_loss = float("nan")
        for i in range(steps):
            x, y = next(it)
            loss = self.train_step(x, y)
            timer.add(self.global_batch)
            if (i + 1) % self.cfg.log_interval == 0:
                last_loss = float(loss.detach().float())
                rate = timer.rate()
                log.info("epoch %d step %d/%d loss %.4f lr %.4f %.1f images/sec",
                         self.epoch, i + 1, steps, last_loss,
                         self.lr_sched.last_lr, rate)
                self.metrics.write(step=self.global_step, epoch=self.epoch,
                                   loss=last_loss, lr=self.lr_sched.last_lr,
                                   images_per_sec=rate,
                                   loss_scale=self.scaler.scale)
                timer.reset()
        self.epoch += 1
        return last_loss

    def fit(self) -> None:
        cfg = self.cfg
        while self.epoch < cfg.epochs:
            self.train_epoch(cfg.steps)
            if cfg.checkpoint_dir and (self.epoch % cfg.checkpoint_interval_epochs == 0):
                path = f"{cfg.checkpoint_dir}/{cfg.model}_epoch{self.epoch}.pt"
                ckpt.save(path, self.model, self.opt, self.epoch,
                          self.global_step, self.scaler)
                log.info("checkpoint: %s", path)
            if cfg.steps:  # bench/CI mode: bounded total work
                break

    # ------------------------------------------------------------------

    @torch.no_grad()
    def evaluate(self, loader, max_batches: Optional[int] = None):
        from .metrics import average_metric
        self.model.eval()
        correct = total = 0
        loss_sum = 0.0
        for bi, (x, y) in enumerate(loader):
            if max_batches and bi >= max_batches:
                break
            x = x.to(self.device, self.dtype)
            y = y.to(self.device)
            if self.cfg.channels_last and x.dim() == 4:
                x = x.contiguous(memory_format=torch.channels_last)
            logits = self.model(x)
            # per-batch mean loss, weighted by batch size so the epoch mean
            # is per-sample regardless of ragged final batches
            loss_sum += float(DF.softmax_cross_entropy(logits, y).float()) * y.numel()
            correct += int((logits.float().argmax(1) == y).sum())
            total += y.numel()
        top1 = correct / max(1, total)
        val_loss = loss_sum / max(1, total)
        # C4: metric averaging across ranks
        top1 = average_metric(top1, device="cpu")
        val_loss = average_metric(val_loss, device="cpu")
        self.model.train()
        return {"top1": top1, "val_loss": val_loss}

    # ------------------------------------------------------------------

    def _can_graph(self) -> bool:
        """hipGraph-capture eligibility for the bench loop: single GPU (RCCL
        collectives are not captured), no dynamic loss scaling (host-side
        control flow) and no dropout (philox seed would freeze per graph).

        Measured (gpurun, b64/128/256/1024): replay is NEUTRAL for this
        training step — the big MFMA/BN kernels average tens to hundreds
        of us, so eager launches are fully hidden; opt-in via
        DDL_HIPGRAPH=1 (it pays in launch-bound regimes like small-batch
        inference — tools/infer_bench.py)."""
        return (os.environ.get("DDL_HIPGRAPH", "0") == "1"
                and self.device.type == "cuda" and self.world == 1
                and not self.scaler.enabled
                and self.cfg.data == "synthetic"  # loader must CYCLE
                and self.cfg.model.startswith("resnet"))

    def _capture_graphs(self, it) -> list:
        """Capture one hipGraph per distinct cycled device batch (the
        synthetic loader cycles n_batches resident tensors, so replays see
        the same data stream as the eager loop — no input copies, no work
        skipped; each replay is a full fwd+bwd+optimizer step).  Launch
        overhead (~1.5-2 us per kernel boundary across ~1k launches/step)
        collapses to one graph replay."""
        graphs = []
        seen = set()
        torch.cuda.synchronize(self.device)
        while len(graphs) < 16:  # hard cap: a non-cycling loader must not
            x, y = next(it)      # capture unboundedly
            if x.data_ptr() in seen:
                break
            seen.add(x.data_ptr())
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self.train_step(x, y)
            graphs.append(g)
        torch.cuda.synchronize(self.device)
        return graphs

    def bench(self, steps: int, warmup: int) -> dict:
        """Timed benchmark: W untimed warmups, barrier+sync bracket, K timed
        steps, MAX elapsed over ranks → whole-job images/sec."""
        self.model.train()
        it = self.data_iter()
        for _ in range(warmup):
            x, y = next(it)
            self.train_step(x, y)
        graphs = None
        if self._can_graph():
            try:
                graphs = self._capture_graphs(it)
            except Exception as e:  # noqa: BLE001 — eager fallback
                log.warning("hipGraph capture failed (%r); running eager", e)
                graphs = None
        self.used_hipgraph = bool(graphs)
        if self.world > 1:
            api.barrier()
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        t0 = time.perf_counter()
        if graphs:
            for i in range(steps):
                graphs[i % len(graphs)].replay()
        else:
            for _ in range(steps):
                x, y = next(it)
                self.train_step(x, y)
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        elapsed = time.perf_counter() - t0
        if self.world > 1:
            t = torch.tensor([elapsed], dtype=torch.float64)
            # MAX over ranks: a job is as fast as its slowest rank
            import torch.distributed as torch_dist
            if torch_dist.is_initialized():
                torch_dist.all_reduce(t, op=torch_dist.ReduceOp.MAX)
            elapsed = float(t.item())
        if self.timeline:
            self.timeline.save()  # flush all spans for overlap analysis
        ms_per_step = elapsed / steps * 1000.0
        images_per_sec = self.global_batch * steps / elapsed
        return {"images_per_sec": images_per_sec, "ms_per_step": ms_per_step,
                "elapsed_s": elapsed}
