"""Gradient compression (hvd.Compression parity — Horovod
``horovod/torch/compression.py``).

``Compression.fp16`` halves on-wire bytes for fp32 gradients by casting the
flat bucket before the all-reduce and back after.  With the recommended
bf16/fp16 compute dtypes the gradients are already 2 bytes so ``none`` is
the default; fp32 training over a bandwidth-bound fabric is where this
pays."""
from __future__ import annotations

import torch


class _NoneCompressor:
    name = "none"

    @staticmethod
    def compress(tensor):
        return tensor, None

    @staticmethod
    def decompress(tensor, ctx):
        return tensor


class _FP16Compressor:
    name = "fp16"

    @staticmethod
    def compress(tensor):
        if tensor.dtype == torch.float32:
            return tensor.to(torch.float16), tensor
        return tensor, None

    @staticmethod
    def decompress(tensor, ctx):
        if ctx is not None:
            ctx.copy_(tensor.to(ctx.dtype))
            return ctx
        return tensor


class Compression:
    none = _NoneCompressor
    fp16 = _FP16Compressor
