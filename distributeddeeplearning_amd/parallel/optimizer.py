"""DistributedOptimizer — hvd.DistributedOptimizer-shaped wrapper.

Reference mechanism (Horovod ``horovod/torch/optimizer.py``): a hook on each
parameter's grad accumulator launches an async allreduce during backward;
``step()`` synchronises every handle then runs the inner optimizer.  Here the
hook marks the param ready in its static bucket; a full bucket launches ONE
allreduce of its persistent flat buffer (no negotiation round, no fusion
memcpys — SURVEY.md §2.2 "Controller" row: static bucket order).

Overlap: on GPU the allreduce runs on the comm backend's side HIP stream;
``synchronize()`` makes the compute stream wait on comm events device-side,
so the host never blocks between backward and step.
"""
from __future__ import annotations

import os
import time
import warnings
from typing import Optional

import torch

from . import api
from .bucket import BucketManager
from .compression import Compression

STALL_WARN_SECONDS = float(os.environ.get("DDL_STALL_WARN_SECONDS", "60"))
_DIRECT_WGRAD = os.environ.get("DDL_DIRECT_WGRAD", "1") == "1"


class _DirectGradRegistry:
    """Routes conv weight gradients straight into their flat-bucket views
    (ops/functional.py consults this from the conv backward): `claim`
    hands out the view only on the FIRST backward pass of an accumulation
    window (later passes must take the autograd-accumulate path so the
    running sum is preserved); `commit` replays exactly what the
    post-accumulate-grad hook would have done — stall timestamp, delay
    decrement, bucket-ready bookkeeping, collective launch.

    Caveat: a conv weight REUSED at several graph sites would commit at
    its first use, before later contributions land — set
    DDL_DIRECT_WGRAD=0 for weight-sharing models (none of the in-repo
    model families share conv weights)."""

    def __init__(self, opt: "DistributedOptimizer"):
        self.opt = opt
        self._members = {id(p) for b in opt.buckets.buckets for p in b.params}

    def claim(self, p):
        opt = self.opt
        d = opt._delay.get(id(p))
        if (id(p) not in self._members or d is None
                or d != opt.backward_passes_per_step):
            return None
        g = p.grad
        if g is None or g.data_ptr() != opt.buckets.ptr_of(p):
            return None
        return g

    def commit(self, p):
        opt = self.opt
        opt._last_ready_time = time.monotonic()
        opt._delay[id(p)] -= 1
        if opt._delay[id(p)] == 0:
            bucket = opt.buckets.mark_ready(p)
            if bucket is not None:
                opt._launch(bucket)


class DistributedOptimizer:
    """Wraps a torch-style optimizer with bucketed gradient all-reduce.

    Parameters mirror hvd.DistributedOptimizer: the wrapped ``optimizer``,
    ``named_parameters`` (defining bucket membership and order), and
    ``backward_passes_per_step`` for gradient accumulation.
    """

    def __init__(
        self,
        optimizer: torch.optim.Optimizer,
        named_parameters=None,
        backward_passes_per_step: int = 1,
        bucket_bytes: Optional[int] = None,
        average: bool = True,
        compression=Compression.none,
    ):
        self.optimizer = optimizer
        self.backward_passes_per_step = int(backward_passes_per_step)
        self.average = average
        self.compression = compression
        self._timeline = None
        # debug collective-order check: hash the bucket
        # launch sequence and compare across ranks every step
        self._order_check = os.environ.get("DDL_DEBUG_ORDER", "0") == "1"
        self._launch_order: list[int] = []

        if named_parameters is None:
            named_parameters = [
                (f"param.{gi}.{pi}", p)
                for gi, g in enumerate(optimizer.param_groups)
                for pi, p in enumerate(g["params"])
            ]
        else:
            named_parameters = list(named_parameters)

        self.buckets = BucketManager(named_parameters, bucket_bytes=bucket_bytes)
        self._hooks = []
        self._last_ready_time: Optional[float] = None
        self._last_stall_warn = 0.0
        # Horovod-style per-param accumulation counters
        # (horovod/torch/optimizer.py `_allreduce_delay`): each backward pass
        # decrements; the bucket launches on the pass that reaches zero.
        # Supports BOTH usage patterns: N backwards then one step() (the
        # upstream contract) and step() after every backward (step() no-ops
        # mid-accumulation).
        self._delay: dict[int, int] = {}
        self._synchronized = False
        for b in self.buckets.buckets:
            for p in b.params:
                self._delay[id(p)] = self.backward_passes_per_step
                h = p.register_post_accumulate_grad_hook(self._make_hook())
                self._hooks.append(h)
        if _DIRECT_WGRAD:
            from ..ops import functional as DF
            DF.set_direct_wgrad(_DirectGradRegistry(self))

    # -- hook & launch ------------------------------------------------------

    def _make_hook(self):
        def hook(p: torch.nn.Parameter):
            self._last_ready_time = time.monotonic()
            g = p.grad
            if g is None or g.data_ptr() != self.buckets.ptr_of(p):
                # e.g. zero_grad(set_to_none=True) detached the flat view:
                # reducing the stale bucket would be silently wrong.
                raise RuntimeError(
                    "param.grad no longer aliases its flat bucket view "
                    "(was zero_grad(set_to_none=True) called on the model?). "
                    "Use DistributedOptimizer.zero_grad() instead.")
            d = self._delay[id(p)]
            if d <= 0:
                if api.size() == 1:
                    # no collective to mis-schedule: extra backwards just
                    # accumulate locally (plain-PyTorch semantics)
                    return
                raise RuntimeError(
                    f"more than backward_passes_per_step="
                    f"{self.backward_passes_per_step} backward passes since "
                    "the last step() — call step() (and zero_grad()) after "
                    "every backward_passes_per_step backwards")
            self._delay[id(p)] = d - 1
            if d == 1:  # this was the Nth (final) pass for this param
                bucket = self.buckets.mark_ready(p)
                if bucket is not None:
                    self._launch(bucket)

        return hook

    def _launch(self, bucket) -> None:
        if api.size() > 1:
            if self._timeline is not None:
                self._timeline.mark(f"bucket{bucket.index}/launch")
            if self._order_check:
                self._launch_order.append(bucket.index)
            wire, ctx = self.compression.compress(bucket.flat)
            h = api.allreduce_async_(wire, average=self.average)
            if ctx is not None:
                inner = h

                class _DecompressHandle:
                    def wait(self2):
                        t = inner.wait()
                        return self.compression.decompress(t, ctx)

                bucket.handle = _DecompressHandle()
            else:
                bucket.handle = h

    # -- hvd surface --------------------------------------------------------

    def _mid_accumulation(self) -> bool:
        """True while some backward passes of the accumulation window ran but
        the window is not complete (no param has reached its final pass)."""
        mind = min(self._delay.values())
        return 0 < mind < self.backward_passes_per_step

    def synchronize(self) -> None:
        """Wait every in-flight bucket (device-side on GPU backends), then
        re-arm the per-param accumulation counters for the next window.

        Also launches any bucket that never filled (a param that got no grad
        this pass would otherwise deadlock other ranks — the stall-inspector
        case).
        """
        for b in self.buckets.buckets:
            if b.handle is None and api.size() > 1:
                if 0 < b.ready < len(b.params):
                    warnings.warn(
                        f"bucket {b.index}: only {b.ready}/{len(b.params)} grads "
                        "produced this step; reducing it anyway (did part of the "
                        "model not participate in the loss?)"
                    )
                    self._launch(b)
            if b.handle is not None:
                b.handle.wait()
                b.handle = None
        self._inspect_stall()
        if self._order_check and api.size() > 1 and self._launch_order:
            import hashlib
            dig = hashlib.sha1(str(self._launch_order).encode()).digest()[:8]
            t = torch.tensor(list(dig), dtype=torch.float64)
            mine = t.clone()
            api.allreduce_(t, average=True)
            if not torch.equal(t, mine):
                raise RuntimeError(
                    f"collective order diverged across ranks: {self._launch_order}")
            self._launch_order.clear()
        for k in self._delay:
            self._delay[k] = self.backward_passes_per_step
        self.buckets.reset()
        self._synchronized = True

    def _inspect_stall(self) -> None:
        """Stall inspector (Horovod stall_inspector.cc role): warn when the
        native engine reports collectives outstanding for too long.
        Rate-limited to one warning per STALL_WARN_SECONDS."""
        backend = api._backend
        eng = getattr(backend, "engine", None)
        now = time.monotonic()
        if eng is not None and now - self._last_stall_warn > STALL_WARN_SECONDS:
            age = eng.oldest_age_seconds()
            if age > STALL_WARN_SECONDS:
                self._last_stall_warn = now
                warnings.warn(
                    f"collective outstanding for {age:.0f}s "
                    f"({eng.outstanding()} in flight) — possible rank "
                    "desync/deadlock")

    def step(self, closure=None):
        if not self._synchronized:
            if self._mid_accumulation():
                # legacy pattern: step() called after every backward while
                # accumulating — no reduce, no inner step until the window
                # completes
                return None
            self.synchronize()
        self._synchronized = False
        return self.optimizer.step(closure)

    def zero_grad(self, set_to_none: bool = False):
        # Grads are views into flat buckets; zero the buckets and keep views.
        # Mid-accumulation (backward_passes_per_step > 1, before the real
        # step) the call is a no-op so accumulated gradients survive.
        if self._mid_accumulation():
            return
        self.buckets.zero_grads()
        self.buckets.reset()
        if not self.buckets.grad_views_valid():
            self.buckets.rebind_grads()

    # Pass-through of the torch optimizer surface ---------------------------
    @property
    def param_groups(self):
        return self.optimizer.param_groups

    @property
    def state(self):
        return self.optimizer.state

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, sd):
        return self.optimizer.load_state_dict(sd)

    def add_param_group(self, g):
        raise NotImplementedError(
            "add_param_group after DistributedOptimizer construction is not "
            "supported: bucket membership is static"
        )

    def rebucket(self, bucket_bytes: int) -> None:
        """Rebuild the buckets at a new size (online autotune): new flat
        buffers + grad views, hooks re-registered, delay counters re-armed,
        direct-wgrad registry refreshed.  Must run at an optimizer-step
        boundary, at the SAME step on every rank (the collective order
        changes with the bucket layout)."""
        for h in self._hooks:
            h.remove()
        named = list(zip(self.buckets.names[::-1],
                         [p for b in self.buckets.buckets for p in b.params]))
        # names list is reverse-registration order matching the flattened
        # bucket params; restore registration order for the new manager
        named = named[::-1]
        self.buckets = BucketManager(named, bucket_bytes=bucket_bytes)
        self._hooks = []
        self._delay = {}
        for b in self.buckets.buckets:
            for p in b.params:
                self._delay[id(p)] = self.backward_passes_per_step
                h = p.register_post_accumulate_grad_hook(self._make_hook())
                self._hooks.append(h)
        self._synchronized = False
        if _DIRECT_WGRAD:
            from ..ops import functional as DF
            DF.set_direct_wgrad(_DirectGradRegistry(self))
        if hasattr(self.optimizer, "invalidate_table"):
            self.optimizer.invalidate_table()  # grad pointers changed

    def set_timeline(self, timeline):
        self._timeline = timeline

    def __repr__(self):
        nb = len(self.buckets.buckets)
        mb = self.buckets.total_bytes() / (1 << 20)
        return f"DistributedOptimizer({self.optimizer.__class__.__name__}, buckets={nb}, {mb:.1f} MB)"
