"""Checkpoint save/resume, layout-compatible with the reference.

The reference PyTorch script ``torch.save``-s ``{model.state_dict(),
optimizer.state_dict(), epoch}`` on rank 0 per epoch and resumes by load +
broadcast.  Compatibility contract here: the saved model
``state_dict`` is plain fp32 NCHW-contiguous tensors under torchvision-style
names, regardless of the in-memory compute dtype / NHWC layout (the K12
boundary conversion is the ``.contiguous()`` + cast at save/load)."""
from __future__ import annotations

import os
from typing import Optional

import torch

from ..parallel import api


def _externalize_state_dict(model: torch.nn.Module) -> dict:
    sd = {}
    for k, v in model.state_dict().items():
        if torch.is_tensor(v) and v.is_floating_point():
            v = v.detach().to(torch.float32)
        if torch.is_tensor(v) and v.dim() == 4:
            v = v.contiguous(memory_format=torch.contiguous_format)
        sd[k] = v.cpu() if torch.is_tensor(v) else v
    return sd


def save(path: str, model, optimizer=None, epoch: int = 0, step: int = 0,
         scaler=None, extra: Optional[dict] = None) -> None:
    """Rank-0-only checkpoint write (all ranks may call; others no-op)."""
    if api.is_initialized() and api.rank() != 0:
        return
    payload = {
        "model": _externalize_state_dict(model),
        "epoch": epoch,
        "step": step,
        "format": "ddl-v1/nchw-fp32",
    }
    if optimizer is not None:
        payload["optimizer"] = optimizer.state_dict()
    if scaler is not None:
        payload["loss_scaler"] = scaler.state_dict()
    if extra:
        payload["extra"] = extra
    tmp = path + ".tmp"
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    torch.save(payload, tmp)
    os.replace(tmp, path)


def load(path: str, model, optimizer=None, scaler=None,
         device=None, broadcast: bool = True) -> dict:
    """Load on every rank (or rank-0 + broadcast), restoring dtype/layout.

    Returns the payload dict (epoch/step for the resume loop)."""
    payload = torch.load(path, map_location="cpu", weights_only=False)
    ref_sd = model.state_dict()
    cast_sd = {}
    for k, v in payload["model"].items():
        if k in ref_sd and torch.is_tensor(v):
            tgt = ref_sd[k]
            v = v.to(dtype=tgt.dtype, device=tgt.device)
            if v.dim() == 4 and tgt.is_contiguous(memory_format=torch.channels_last):
                v = v.contiguous(memory_format=torch.channels_last)
        cast_sd[k] = v
    model.load_state_dict(cast_sd)
    if optimizer is not None and "optimizer" in payload:
        optimizer.load_state_dict(payload["optimizer"])
        if hasattr(optimizer, "optimizer"):  # DistributedOptimizer wrapper
            inner = optimizer.optimizer
        else:
            inner = optimizer
        if hasattr(inner, "refresh_master"):
            # Only re-derive fp32 masters from the (possibly bf16/fp16)
            # params when the checkpoint did NOT carry them — a restored
            # master is exact and re-deriving would truncate it to the
            # low-precision params (breaking bit-exact resume).
            has_master = any(
                torch.is_tensor(st.get("master"))
                for st in getattr(inner, "state", {}).values()
            )
            if not has_master:
                inner.refresh_master()
        if hasattr(inner, "invalidate_table"):
            inner.invalidate_table()
    if scaler is not None and "loss_scaler" in payload:
        scaler.load_state_dict(payload["loss_scaler"])
    if broadcast and api.is_initialized() and api.size() > 1:
        # broadcast-after-restore ordering, as the reference does (§3.2)
        api.broadcast_parameters(model.state_dict(), root_rank=0)
    return payload
