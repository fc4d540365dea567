"""Gradient bucketing with persistent flat buffers.

Horovod fuses gradients by memcpy-ing each tensor into a 64 MB fusion buffer
per cycle (``horovod/common/fusion_buffer_manager.cc``).  The MI355X-native
design removes those copies entirely: each bucket owns ONE flat buffer for
its lifetime and every ``param.grad`` is a *view* into it, so backward
accumulates directly into collective-ready memory (SURVEY.md §2.3 C6:
"eliminated for steady state").

Bucket order is static: parameters in reverse registration order (the order
backward produces gradients in a sequential CNN), packed greedily up to
``bucket_bytes``.  The default bucket size targets the 8×MI355X xGMI mesh:
a ring all-reduce is per-link bound at ≈153 GB/s, so 25 MB
buckets give ≈0.4 ms per-bucket on-link time — large enough to amortise
launch overhead, small enough that the final bucket (first conv layers)
doesn't serialise after backward ends.  Tune with tools/tune_buckets.py.
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import List, Optional

import torch

DEFAULT_BUCKET_MB = float(os.environ.get("DDL_BUCKET_MB", "25"))


def _grad_view(flat: torch.Tensor, off: int, p: torch.nn.Parameter) -> torch.Tensor:
    """A view of flat[off:off+numel] with p's PHYSICAL layout, so autograd
    accumulates coalescedly and the flat buffer holds the same element order
    as the parameter storage (KRSC for channels_last weights)."""
    sl = flat[off : off + p.numel()]
    if p.dim() == 4 and p.is_contiguous(memory_format=torch.channels_last):
        n, c, h, w = p.shape
        return sl.view(n, h, w, c).permute(0, 3, 1, 2)
    return sl.view_as(p)


@dataclass
class Bucket:
    index: int
    params: List[torch.nn.Parameter]
    flat: torch.Tensor  # persistent flat gradient buffer
    offsets: List[int]
    ready: int = 0
    handle: object = None

    def numel(self) -> int:
        return self.flat.numel()


class BucketManager:
    """Assigns params to buckets, owns flat buffers, tracks readiness."""

    def __init__(
        self,
        named_params,
        bucket_bytes: Optional[int] = None,
        grad_dtype: Optional[torch.dtype] = None,
    ):
        named = [(n, p) for n, p in named_params if p.requires_grad]
        if not named:
            raise ValueError("no trainable parameters")
        self.names = [n for n, _ in named]
        params = [p for _, p in named]
        # Reverse registration order ≈ backward completion order.
        params_rev = list(reversed(params))
        names_rev = list(reversed(self.names))

        bucket_bytes = bucket_bytes or int(DEFAULT_BUCKET_MB * (1 << 20))
        self.buckets: List[Bucket] = []
        self._param_bucket: dict[int, tuple[Bucket, int]] = {}

        cur: list[torch.nn.Parameter] = []
        cur_bytes = 0
        groups: list[list[torch.nn.Parameter]] = []
        for p in params_rev:
            nbytes = p.numel() * p.element_size()
            if cur and cur_bytes + nbytes > bucket_bytes:
                groups.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += nbytes
        if cur:
            groups.append(cur)

        for bi, group in enumerate(groups):
            dt = grad_dtype or group[0].dtype
            total = sum(p.numel() for p in group)
            flat = torch.zeros(total, dtype=dt, device=group[0].device)
            offsets = []
            off = 0
            for p in group:
                offsets.append(off)
                # p.grad becomes a view into the flat buffer: autograd
                # accumulates in place, the collective reads the flat buffer
                # with zero staging copies.
                p.grad = _grad_view(flat, off, p)
                off += p.numel()
            b = Bucket(index=bi, params=list(group), flat=flat, offsets=offsets)
            self.buckets.append(b)
            for pi, p in enumerate(group):
                self._param_bucket[id(p)] = (b, pi)

        self._names_rev = names_rev
        # precomputed grad-view base pointers for the cheap aliasing guard
        self._expected_ptr = {}
        for b in self.buckets:
            for p, off in zip(b.params, b.offsets):
                self._expected_ptr[id(p)] = b.flat[off].data_ptr()

    # -- readiness protocol -------------------------------------------------

    def ptr_of(self, p: torch.nn.Parameter) -> int:
        """Expected storage pointer of p's grad view (aliasing guard)."""
        return self._expected_ptr[id(p)]

    def mark_ready(self, p: torch.nn.Parameter) -> Optional[Bucket]:
        """Mark one param's grad complete; return the bucket if it is full."""
        b, _ = self._param_bucket[id(p)]
        b.ready += 1
        if b.ready == len(b.params):
            return b
        return None

    def reset(self) -> None:
        for b in self.buckets:
            b.ready = 0
            b.handle = None

    def zero_grads(self) -> None:
        for b in self.buckets:
            b.flat.zero_()

    def grad_views_valid(self) -> bool:
        """True if every param.grad still aliases its flat-buffer slot."""
        for b in self.buckets:
            for p, off in zip(b.params, b.offsets):
                g = p.grad
                if g is None or g.data_ptr() != b.flat[off].data_ptr():
                    return False
        return True

    def rebind_grads(self) -> None:
        """Re-point param.grad at the flat views (after set_to_none etc.)."""
        for b in self.buckets:
            for p, off in zip(b.params, b.offsets):
                p.grad = _grad_view(b.flat, off, p)

    def total_bytes(self) -> int:
        return sum(b.flat.numel() * b.flat.element_size() for b in self.buckets)
