"""Horovod-shaped public API (reference surface: ``horovod/torch/__init__.py``
as called by ``HorovodPytorch/src/imagenet_pytorch_horovod.py``).

``init()`` builds the process-wide comm backend; the free functions mirror
hvd semantics: ``allreduce`` averages by default, ``broadcast_parameters`` /
``broadcast_optimizer_state`` fuse every tensor into a single flat broadcast
from the root (the reference does one broadcast per tensor through Horovod's
fusion buffer; here the fusion is explicit and single-shot — SURVEY.md §2.3
C2/C3).
"""
from __future__ import annotations

import os
from typing import Iterable, Optional

import torch

from .backend import CommBackend, Handle, create_backend

_backend: Optional[CommBackend] = None


def init(comm: Optional[str] = None) -> None:
    """Initialise the distributed context (hvd.init equivalent).

    Reads the torchrun env convention (RANK / WORLD_SIZE / LOCAL_RANK /
    MASTER_ADDR / MASTER_PORT).  Safe to call when un-launched: world size 1.
    """
    global _backend
    if _backend is not None:
        return
    _backend = create_backend(comm)


def shutdown() -> None:
    global _backend
    if _backend is not None:
        _backend.shutdown()
        _backend = None


def is_initialized() -> bool:
    return _backend is not None


def _b() -> CommBackend:
    if _backend is None:
        raise RuntimeError("distributeddeeplearning_amd: call init() first")
    return _backend


def comm_backend_name() -> str:
    """Which backend actually carries the gradient collectives
    ("native" | "torchdist" | "single") — reported in bench provenance so a
    silent fallback can never masquerade as the native engine."""
    return _b().name


def rank() -> int:
    return _b().rank()


def size() -> int:
    return _b().size()


def local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", "0"))


def local_size() -> int:
    return int(os.environ.get("LOCAL_WORLD_SIZE", os.environ.get("DDL_LOCAL_SIZE", str(size()))))


# ---- collectives ---------------------------------------------------------

def allreduce_async_(tensor: torch.Tensor, average: bool = True, name: str | None = None) -> Handle:
    return _b().allreduce_async_(tensor, average=average)


def allreduce_(tensor: torch.Tensor, average: bool = True, name: str | None = None) -> torch.Tensor:
    return allreduce_async_(tensor, average, name).wait()


def allreduce_async(tensor: torch.Tensor, average: bool = True, name: str | None = None) -> Handle:
    out = tensor.clone()
    return _b().allreduce_async_(out, average=average)


def allreduce(tensor: torch.Tensor, average: bool = True, name: str | None = None) -> torch.Tensor:
    return allreduce_async(tensor, average, name).wait()


def broadcast_async_(tensor: torch.Tensor, root_rank: int = 0, name: str | None = None) -> Handle:
    return _b().broadcast_async_(tensor, root_rank)


def broadcast_(tensor: torch.Tensor, root_rank: int = 0, name: str | None = None) -> torch.Tensor:
    return broadcast_async_(tensor, root_rank, name).wait()


def broadcast_async(tensor: torch.Tensor, root_rank: int = 0, name: str | None = None) -> Handle:
    out = tensor.clone()
    return _b().broadcast_async_(out, root_rank)


def broadcast(tensor: torch.Tensor, root_rank: int = 0, name: str | None = None) -> torch.Tensor:
    return broadcast_async(tensor, root_rank, name).wait()


def allgather(tensor: torch.Tensor, name: str | None = None) -> torch.Tensor:
    return _b().allgather_async(tensor).wait()


def synchronize(handle: Handle) -> torch.Tensor:
    return handle.wait()


def barrier() -> None:
    _b().barrier()


# ---- fused state broadcast (C2/C3 call sites) ----------------------------

_FUSED_BCAST_CHUNK = 64 << 20  # bytes; VGG-16 fp32 would otherwise stage 553 MB


def _fused_broadcast(tensors: Iterable[torch.Tensor], root_rank: int) -> None:
    """Broadcast many tensors as flat buffers per (device, dtype) group.

    Chunked to ~64 MB staging buffers so startup never materialises a full
    model copy; each chunk is one collective, still
    orders of magnitude fewer than the reference's per-tensor broadcasts."""
    groups: dict[tuple, list[torch.Tensor]] = {}
    for t in tensors:
        if not torch.is_tensor(t) or t.numel() == 0:
            continue
        groups.setdefault((t.device, t.dtype), []).append(t)
    for (_dev, dtype), ts in groups.items():
        max_el = max(1, _FUSED_BCAST_CHUNK // max(1, ts[0].element_size()))
        chunk: list[torch.Tensor] = []
        n_el = 0

        def flush():
            nonlocal chunk, n_el
            if not chunk:
                return
            if len(chunk) == 1 and chunk[0].detach().reshape(-1).data_ptr() \
                    == chunk[0].data_ptr() and chunk[0].is_contiguous():
                # single contiguous tensor: broadcast in place, no staging
                broadcast_(chunk[0].detach().reshape(-1), root_rank)
            else:
                flat = torch.cat([t.detach().reshape(-1) for t in chunk])
                broadcast_(flat, root_rank)
                off = 0
                for t in chunk:
                    n = t.numel()
                    # NB: copy_ INTO the tensor (reshape(-1) on a
                    # channels_last tensor returns a copy, so writing
                    # through it would be lost)
                    t.detach().copy_(flat[off : off + n].view(t.shape))
                    off += n
            chunk, n_el = [], 0

        for t in ts:
            if n_el and n_el + t.numel() > max_el:
                flush()
            chunk.append(t)
            n_el += t.numel()
            if t.numel() >= max_el:
                flush()
        flush()


def broadcast_parameters(params, root_rank: int = 0) -> None:
    """hvd.broadcast_parameters: accepts a state_dict or named_parameters."""
    if _b().size() <= 1:
        return
    if isinstance(params, dict):
        tensors = [v for v in params.values() if torch.is_tensor(v) and v.is_floating_point()]
        # integer buffers (num_batches_tracked) ride along too
        tensors += [v for v in params.values() if torch.is_tensor(v) and not v.is_floating_point()]
    else:
        tensors = [p for _name, p in params]
    _fused_broadcast(tensors, root_rank)


def broadcast_optimizer_state(optimizer, root_rank: int = 0) -> None:
    """hvd.broadcast_optimizer_state: sync optimizer state tensors from root."""
    if _b().size() <= 1:
        return
    tensors = []
    for group_state in optimizer.state_dict().get("state", {}).values():
        for v in group_state.values():
            if torch.is_tensor(v):
                tensors.append(v)
    # State may be lazily created; materialise missing momentum buffers first
    # is the caller's job (the engine calls this right after a dummy step or
    # at t=0 when state is empty, matching Horovod's behaviour of only
    # broadcasting existing entries).
    if tensors:
        _fused_broadcast(tensors, root_rank)
    # also re-sync hyperparameters scalars? Horovod broadcasts only tensors.
