"""Communication backends.

The reference's comm plane is Horovod's C++ core dispatching to NCCL
(``horovod/common/ops/nccl_operations.cc``) / Gloo / MPI.  Here the
MI355X-native data plane is RCCL over xGMI.  Two backends implement one
interface:

* :class:`NativeRcclBackend` — our own C++ engine (``csrc/comm``): one
  ncclComm built from a unique id exchanged over the torch.distributed TCP
  store, collectives issued on a dedicated side HIP stream with hipEvent
  handshakes against the compute stream (the overlap machinery of
  BASELINE.json's north star).  GPU only.
* :class:`TorchDistBackend` — torch.distributed (gloo on CPU; nccl==RCCL on
  GPU).  This is the bootstrap/control plane everywhere and the data plane
  for CPU plumbing tests (BASELINE config 1) or as an explicit opt-out
  (``DDL_COMM=torchdist``).

Handles returned by ``*_async`` are opaque; pass them to ``synchronize``.
"""
from __future__ import annotations

import datetime
import os
from typing import Any, Optional

import torch
import torch.distributed as torch_dist


_STALL_WARN_S = float(os.environ.get("DDL_STALL_WARN_SECONDS", "60"))


class Handle:
    """An async-op handle: backend-specific payload plus the target tensor."""

    __slots__ = ("work", "tensor", "post", "_done", "stall_guard")

    def __init__(self, work: Any, tensor: torch.Tensor, post=None,
                 stall_guard: bool = False):
        self.work = work
        self.tensor = tensor
        self.post = post  # optional callable run once at synchronize()
        self._done = False
        # poll-with-warning instead of a silent indefinite block (gloo
        # control/data plane only — NCCL work keeps its efficient
        # stream-semantics wait)
        self.stall_guard = stall_guard

    def _guarded_wait(self) -> None:
        import time as _time
        import warnings as _warnings
        t0 = _time.monotonic()
        warned = False
        while not self.work.is_completed():
            if not warned and _time.monotonic() - t0 > _STALL_WARN_S:
                _warnings.warn(
                    f"collective outstanding for {_STALL_WARN_S:.0f}s on the "
                    "torch.distributed path — possible rank desync/deadlock "
                    "(stall inspector)")
                warned = True
            _time.sleep(0.002)

    def wait(self) -> torch.Tensor:
        if not self._done:
            if self.work is not None:
                if self.stall_guard and hasattr(self.work, "is_completed"):
                    self._guarded_wait()
                else:
                    self.work.wait()
            if self.post is not None:
                self.post()
            self._done = True
        return self.tensor


class CommBackend:
    """Interface shared by all backends."""

    name = "abstract"  # reported in bench provenance (VERDICT weak #1)

    def rank(self) -> int:
        raise NotImplementedError

    def size(self) -> int:
        raise NotImplementedError

    def allreduce_async_(self, tensor: torch.Tensor, average: bool = True) -> Handle:
        raise NotImplementedError

    def broadcast_async_(self, tensor: torch.Tensor, root_rank: int) -> Handle:
        raise NotImplementedError

    def allgather_async(self, tensor: torch.Tensor) -> Handle:
        raise NotImplementedError

    def barrier(self) -> None:
        raise NotImplementedError

    def shutdown(self) -> None:
        pass

    # Mark a point where subsequent compute depends on all issued comm
    # (no-op for synchronous backends; the native engine makes the compute
    # stream wait on the comm stream's events without blocking the host).
    def join_stream(self) -> None:
        pass


class TorchDistBackend(CommBackend):
    """torch.distributed-backed comm (gloo CPU / nccl==RCCL GPU)."""

    name = "torchdist"

    def __init__(self, process_group=None):
        self.pg = process_group
        self._is_nccl = torch_dist.get_backend(self.pg) == "nccl"

    def rank(self) -> int:
        return torch_dist.get_rank(self.pg)

    def size(self) -> int:
        return torch_dist.get_world_size(self.pg)

    def allreduce_async_(self, tensor: torch.Tensor, average: bool = True) -> Handle:
        # gloo aborts on 0-dim tensors; hvd supports them — reduce through
        # a storage-sharing 1-element view
        wire = tensor.unsqueeze(0) if tensor.dim() == 0 else tensor
        work = torch_dist.all_reduce(wire, op=torch_dist.ReduceOp.SUM, group=self.pg, async_op=True)
        post = None
        if average and self.size() > 1:
            n = self.size()

            def post(t=tensor, n=n):
                t.div_(n)

        # per-tensor: CUDA collectives keep NCCL's efficient stream-
        # semantics wait; host-side (gloo) waits get the stall warning
        return Handle(work, tensor, post, stall_guard=not tensor.is_cuda)

    def broadcast_async_(self, tensor: torch.Tensor, root_rank: int) -> Handle:
        wire = tensor.unsqueeze(0) if tensor.dim() == 0 else tensor
        work = torch_dist.broadcast(wire, src=root_rank, group=self.pg, async_op=True)
        return Handle(work, tensor, stall_guard=not tensor.is_cuda)

    def allgather_async(self, tensor: torch.Tensor) -> Handle:
        ws = self.size()
        shards = [torch.empty_like(tensor) for _ in range(ws)]
        work = torch_dist.all_gather(shards, tensor.contiguous(), group=self.pg, async_op=True)
        out = {"shards": shards}
        h = Handle(work, tensor)

        def post(h=h, out=out):
            h.tensor = torch.cat(out["shards"], dim=0)

        h.post = post
        return h

    def barrier(self) -> None:
        torch_dist.barrier(group=self.pg)

    def shutdown(self) -> None:
        pass


class SingleProcessBackend(CommBackend):
    """world_size==1 loopback: every collective is the identity.

    Mirrors Horovod's behaviour when run un-launched; also the test fake
    (SURVEY.md §4.3 "LoopbackComm")."""

    name = "single"

    def rank(self) -> int:
        return 0

    def size(self) -> int:
        return 1

    def allreduce_async_(self, tensor, average=True):
        return Handle(None, tensor)

    def broadcast_async_(self, tensor, root_rank):
        return Handle(None, tensor)

    def allgather_async(self, tensor):
        return Handle(None, tensor)

    def barrier(self):
        pass


class NativeRcclBackend(CommBackend):
    """Our C++ RCCL engine (csrc/comm/rccl_engine.*) behind the same interface.

    torch.distributed (gloo) stays alive as the control plane: rank/size and
    the ncclUniqueId exchange go through it; every data-plane collective is
    issued by the native engine on its own HIP stream.
    """

    name = "native"

    def __init__(self, device_index: int, control_pg=None):
        from ..ops._ext import load_comm_ext

        self._ctrl = TorchDistBackend(control_pg)
        # Pre-init consensus: every rank must have the extension loaded and
        # (on rank 0) a unique id BEFORE anyone enters ncclCommInitRank —
        # a rank that fails after its peers enter the blocking init would
        # hang the job instead of falling back.
        err: Optional[Exception] = None
        uid = None
        try:
            self._ext = load_comm_ext()  # raises if the .so is missing
            if self._ctrl.rank() == 0:
                uid = self._ext.get_unique_id()
        except Exception as e:  # noqa: BLE001
            err = e
        ok = torch.tensor([0 if err is not None else 1], dtype=torch.int64)
        torch_dist.all_reduce(ok, op=torch_dist.ReduceOp.MIN, group=control_pg)
        if int(ok.item()) != 1:
            raise RuntimeError(
                f"native RCCL engine unavailable on at least one rank "
                f"(local error: {err!r})")
        obj = [uid]
        torch_dist.broadcast_object_list(obj, src=0, group=control_pg)
        uid = obj[0]
        self.engine = self._ext.CommEngine(
            self._ctrl.rank(), self._ctrl.size(), device_index, uid
        )

    def rank(self) -> int:
        return self.engine.rank()

    def size(self) -> int:
        return self.engine.size()

    def _data_plane(self, tensor: torch.Tensor) -> bool:
        """GPU tensors of RCCL-supported dtypes go to the native engine;
        host-side scalars/metrics (CPU tensors, float64) ride the gloo
        control plane instead (C4 call sites: metric averaging, debug-order
        checksums — never perf-critical)."""
        return tensor.is_cuda and tensor.dtype != torch.float64

    def allreduce_async_(self, tensor: torch.Tensor, average: bool = True) -> Handle:
        if not self._data_plane(tensor):
            return self._ctrl.allreduce_async_(tensor, average=average)
        hid = self.engine.allreduce_async(tensor, bool(average))
        return Handle(_NativeWork(self.engine, hid), tensor)

    def broadcast_async_(self, tensor: torch.Tensor, root_rank: int) -> Handle:
        if not self._data_plane(tensor):
            return self._ctrl.broadcast_async_(tensor, root_rank)
        hid = self.engine.broadcast_async(tensor, int(root_rank))
        return Handle(_NativeWork(self.engine, hid), tensor)

    def allgather_async(self, tensor: torch.Tensor) -> Handle:
        if not self._data_plane(tensor):
            return self._ctrl.allgather_async(tensor)
        t = tensor.contiguous()
        out_shape = list(t.shape)
        out_shape[0] *= self.size()
        out = torch.empty(out_shape, dtype=t.dtype, device=t.device)
        hid = self.engine.allgather_async(t, out)
        return Handle(_NativeWork(self.engine, hid), out)

    def barrier(self) -> None:
        self._ctrl.barrier()

    def join_stream(self) -> None:
        # Make the CURRENT compute stream wait on every outstanding comm
        # event — device-side, no host sync.
        self.engine.join_compute_stream()

    def shutdown(self) -> None:
        self.engine.shutdown()


class _NativeWork:
    __slots__ = ("engine", "hid")

    def __init__(self, engine, hid):
        self.engine = engine
        self.hid = hid

    def wait(self):
        # Device-side: compute stream waits on the comm event.  No host block.
        self.engine.wait_handle(self.hid)


def _default_device_index() -> int:
    lr = int(os.environ.get("LOCAL_RANK", os.environ.get("DDL_LOCAL_RANK", "0")))
    return lr


def create_backend(comm: Optional[str] = None) -> CommBackend:
    """Build the comm backend from the environment.

    Rendezvous follows the torchrun convention (RANK/WORLD_SIZE/MASTER_ADDR/
    MASTER_PORT; the reference used mpirun env — Horovod
    ``horovod/runner``), with gloo as the bootstrap process group.
    """
    # DDL_COMM env wins over an explicit "auto" so the launcher/driver can
    # force the backend (e.g. DDL_COMM=native makes a native-init failure
    # fatal instead of falling back)
    if comm in (None, "auto"):
        comm = os.environ.get("DDL_COMM", "auto")
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1 and "RANK" not in os.environ:
        return SingleProcessBackend()

    if not torch_dist.is_initialized():
        use_gpu = torch.cuda.is_available()
        backend = "cpu:gloo,cuda:nccl" if use_gpu else "gloo"
        torch_dist.init_process_group(
            backend=backend, timeout=datetime.timedelta(seconds=300)
        )

    if comm == "auto":
        comm = "native-or-torchdist" if torch.cuda.is_available() else "torchdist"

    if comm == "native":
        dev = _default_device_index()
        torch.cuda.set_device(dev)
        return NativeRcclBackend(dev)
    elif comm == "native-or-torchdist":
        dev = _default_device_index()
        torch.cuda.set_device(dev)
        native = None
        err = None
        try:
            native = NativeRcclBackend(dev)
        except Exception as e:  # noqa: BLE001 — any init failure
            err = e
        # Consensus fallback: the backend choice must be IDENTICAL on every
        # rank — a per-rank fallback (some ranks native, some torchdist)
        # would deadlock the first collective.  NativeRcclBackend's own
        # pre-init consensus covers the common failure points; this second
        # vote covers a rank failing inside engine construction itself.
        flag = torch.tensor([1 if native is not None else 0], dtype=torch.int64)
        torch_dist.all_reduce(flag, op=torch_dist.ReduceOp.MIN)
        if int(flag.item()) == 1:
            return native
        import warnings
        warnings.warn(
            f"native RCCL engine init failed on at least one rank "
            f"(local error: {err!r}); ALL ranks falling back to "
            "torch.distributed nccl (still RCCL, without the native "
            "side-stream engine). Set DDL_COMM=native to make this fatal.")
        if native is not None:
            native.shutdown()
        return TorchDistBackend()
    elif comm == "torchdist":
        return TorchDistBackend()
    elif comm == "single":
        return SingleProcessBackend()
    raise ValueError(f"unknown comm backend {comm!r}")
