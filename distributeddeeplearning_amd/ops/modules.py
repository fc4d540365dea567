"""nn.Module wrappers over the functional ops.

Module/parameter naming matches torchvision's ResNet/VGG so checkpoints stay
layout-compatible with the reference scripts' ``torch.save(model.state_dict())``
(standard NCHW ``state_dict`` with torchvision-style names).
Physical layout on GPU is NHWC (channels_last); logical shapes stay NCHW.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

import os

from . import functional as DF

# dgrad->BN-backward-stats fusion: architecturally sound (saves the
# bn_bwd_stats (dy, x) pass) and numerically verified, but measured -2%
# end-to-end on ResNet-50 b1024 once the mode-3 mask landed — the extra
# epilogue reads sit on the serial dgrad path while the stats kernels
# they replace are cheap and overlap well.  Opt-in for >LLC-working-set
# regimes (DDL_FUSE_BNBWD=1).
_FUSE_BNBWD = os.environ.get("DDL_FUSE_BNBWD", "0") == "1"


class Conv2d(nn.Module):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, bias=False, emit_bn_stats=False):
        super().__init__()
        # emit_bn_stats: this conv feeds a BatchNorm — the MFMA epilogue
        # also emits the BN {sum,sumsq} partials so bn skips its stats pass
        self.emit_bn_stats = emit_bn_stats
        k = (kernel_size, kernel_size) if isinstance(kernel_size, int) else kernel_size
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = k
        self.stride = (stride, stride) if isinstance(stride, int) else stride
        self.padding = (padding, padding) if isinstance(padding, int) else padding
        self.weight = nn.Parameter(torch.empty(out_channels, in_channels, *k))
        if bias:
            self.bias = nn.Parameter(torch.zeros(out_channels))
        else:
            self.register_parameter("bias", None)
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.kaiming_normal_(self.weight, mode="fan_out", nonlinearity="relu")

    def forward(self, x):
        fuse = getattr(x, "_ddl_bn_bwd_fuse", None)
        if fuse is not None:
            del x._ddl_bn_bwd_fuse  # single consumer takes it
        if self.emit_bn_stats and self.training and x.is_cuda:
            y, partials = DF.conv2d_stats(x, self.weight, self.bias,
                                          self.stride, self.padding,
                                          bn_fuse=fuse)
            if partials is not None:
                y._ddl_bn_partials = partials
            return y
        return DF.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                         bn_fuse=fuse)

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, "
                f"kernel_size={self.kernel_size}, stride={self.stride}, "
                f"padding={self.padding}, bias={self.bias is not None}")


class BatchNorm2d(nn.Module):
    """BatchNorm with optional fused residual-add + ReLU at call time.

    ``forward(x)`` is plain BN; ``forward(x, relu=True, residual=r)`` fuses
    the bottleneck exit ``relu(bn(x) + r)`` into one kernel (K4/K14).
    Weight/bias follow the model compute dtype; running stats stay fp32
    (fp32 statistics under bf16/fp16 activations, BASELINE.json north star).
    """

    def __init__(self, num_features, eps=1e-5, momentum=0.1,
                 fuse_bwd_stats=False):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        # fuse_bwd_stats: this BN's output has exactly ONE consumer — a
        # Conv2d — so that conv's dgrad epilogue can emit this BN's
        # backward stats partials (set by the model builder for the
        # bn+relu layers inside a conv chain; NEVER set when the output
        # fans out, the partials would miss the other consumers' grads)
        self.fuse_bwd_stats = fuse_bwd_stats
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked", torch.tensor(0, dtype=torch.long))
        # python-side shadow: `buf += 1` on a GPU scalar is one kernel
        # launch per BN per step (53/step on ResNet-50); count here and
        # flush into the buffer whenever a state_dict is taken
        self._nbt_pending = 0

    def _flush_nbt(self):
        if self._nbt_pending:
            self.num_batches_tracked += self._nbt_pending
            self._nbt_pending = 0

    def _save_to_state_dict(self, destination, prefix, keep_vars):
        self._flush_nbt()
        super()._save_to_state_dict(destination, prefix, keep_vars)

    def _load_from_state_dict(self, *args, **kwargs):
        self._nbt_pending = 0
        super()._load_from_state_dict(*args, **kwargs)

    def forward(self, x, relu: bool = False, residual: Optional[torch.Tensor] = None):
        if self.training:
            self._nbt_pending += 1
        partials = getattr(x, "_ddl_bn_partials", None)
        if partials is not None:
            del x._ddl_bn_partials
        holder = None
        # no-residual BNs: the consumer conv recomputes the ReLU mask;
        # residual-join BNs: the forward's bit-packed mask rides along
        # (consumed by the next block's fork-conv dgrad epilogue)
        if (self.fuse_bwd_stats and _FUSE_BNBWD and self.training
                and x.is_cuda and relu
                and x.dtype != torch.float32
                and torch.is_grad_enabled()):
            holder = DF.BnBwdFusion()
        y = DF.bn_add_relu(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.training, self.momentum, self.eps, relu, residual,
            conv_partials=partials, bwd_holder=holder)
        if holder is not None:
            # picked up by the consuming Conv2d (read-only refs)
            y._ddl_bn_bwd_fuse = (holder, x, self.weight, self.bias)
        return y

    def extra_repr(self):
        return f"{self.num_features}, eps={self.eps}, momentum={self.momentum}"


class MaxPool2d(nn.Module):
    def __init__(self, kernel_size, stride=None, padding=0):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride if stride is not None else kernel_size
        self.padding = padding

    def forward(self, x):
        return DF.max_pool2d(x, self.kernel_size, self.stride, self.padding)


class GlobalAvgPool2d(nn.Module):
    """AdaptiveAvgPool2d((1,1)) + flatten, as one kernel (K7)."""

    def forward(self, x):
        return DF.global_avg_pool(x)


class Linear(nn.Module):
    def __init__(self, in_features, out_features, bias=True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        if bias:
            self.bias = nn.Parameter(torch.zeros(out_features))
        else:
            self.register_parameter("bias", None)
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.normal_(self.weight, 0, 0.01)
        if self.bias is not None:
            nn.init.zeros_(self.bias)

    def forward(self, x):
        return DF.linear(x, self.weight, self.bias)

    def extra_repr(self):
        return f"in_features={self.in_features}, out_features={self.out_features}"


class Dropout(nn.Module):
    def __init__(self, p=0.5):
        super().__init__()
        self.p = p

    def forward(self, x):
        return DF.dropout(x, self.p, self.training)


class ReLU(nn.Module):
    def forward(self, x):
        return torch.relu(x)


def cast_model_(model: nn.Module, dtype: torch.dtype,
                channels_last: bool = True) -> nn.Module:
    """Cast parameters to the compute dtype; keep buffers (BN running stats)
    fp32; apply channels_last memory format to 4-D parameters."""
    for p in model.parameters():
        with torch.no_grad():
            data = p.data.to(dtype)
            if channels_last and data.dim() == 4:
                data = data.contiguous(memory_format=torch.channels_last)
            p.data = data
    return model
