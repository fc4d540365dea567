"""Native extension loading.

Two in-tree extensions (built by ``python setup.py build_ext --inplace`` or
``__graft_entry__.build()``; the .so files live inside the package so they
travel to the GPU box with the repo snapshot):

* ``_C_kernels`` — hand-written HIP/CDNA4 kernels (csrc/kernels/*.hip).
* ``_C_comm``    — native RCCL comm engine (csrc/comm, csrc/core).

Policy (task rule: "make your ops fail loudly if their extension is missing
on a GPU box"): on a machine WITH a GPU these loaders raise ImportError with
build instructions; only CPU-only processes may run without the native code
(they use the fp32 torch reference path, which is also what kernels are
tested against).
"""
from __future__ import annotations

import importlib
import os

import torch

_kernels = None
_comm = None
_kernels_err: Exception | None = None


def _try_import(name: str):
    return importlib.import_module(f"distributeddeeplearning_amd.{name}")


def load_kernels_ext(required: bool | None = None):
    """Return the _C_kernels module; raise if required and missing.

    required defaults to torch.cuda.is_available(): a GPU process must never
    silently fall back to eager PyTorch.
    """
    global _kernels, _kernels_err
    if _kernels is not None:
        return _kernels
    if required is None:
        required = torch.cuda.is_available()
    try:
        _kernels = _try_import("_C_kernels")
        return _kernels
    except ImportError as e:
        _kernels_err = e
        if required and os.environ.get("DDL_ALLOW_EAGER_FALLBACK", "0") != "1":
            raise ImportError(
                "distributeddeeplearning_amd._C_kernels is not built but this "
                "process has a GPU. Build it in-tree first:\n"
                "    python setup.py build_ext --inplace\n"
                "(or run __graft_entry__.build()). Refusing to fall back to "
                "eager PyTorch on GPU — the HIP path must be the one that runs. "
                f"Original error: {e}"
            ) from e
        return None


def load_comm_ext():
    global _comm
    if _comm is not None:
        return _comm
    try:
        _comm = _try_import("_C_comm")
    except ImportError as e:
        raise ImportError(
            "distributeddeeplearning_amd._C_comm (native RCCL engine) is not "
            "built. Build with: python setup.py build_ext --inplace. "
            f"Original error: {e}"
        ) from e
    return _comm


def kernels_available() -> bool:
    try:
        return load_kernels_ext(required=False) is not None
    except ImportError:
        return False
