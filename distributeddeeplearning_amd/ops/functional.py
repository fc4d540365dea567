"""Functional op surface with two implementations per op:

* GPU (``tensor.is_cuda``): hand-written HIP/CDNA4 kernels via ``_C_kernels``
  wrapped in autograd Functions.  NHWC physical layout (torch
  ``channels_last``), bf16/fp16/fp32 compute.  Missing extension on a GPU
  machine raises (no silent eager fallback).
* CPU: plain fp32 PyTorch composition.  This is both the tier-0 test path
  (BASELINE config 1) and the numerical reference the HIP kernels are
  validated against (SURVEY.md §4.3 Tier 1).

Reference call sites these ops replace: the cuDNN/cuBLAS launches issued by
torchvision resnet50 inside ``HorovodPytorch/src/imagenet_pytorch_horovod.py``
(SURVEY.md §2.3 table K1–K15).
"""
from __future__ import annotations

from typing import Optional

import os

import torch
import torch.nn.functional as F

from ._ext import load_kernels_ext


def _pair(v):
    return (v, v) if isinstance(v, int) else tuple(v)


def to_channels_last(t: torch.Tensor) -> torch.Tensor:
    if t.dim() == 4:
        return t.contiguous(memory_format=torch.channels_last)
    return t.contiguous()


# --------------------------------------------------------------------------
# Conv2d (K1/K2/K3: implicit-GEMM NHWC, MFMA)
# --------------------------------------------------------------------------

# Direct-wgrad registry (set by DistributedOptimizer when active): conv
# weight gradients are written straight into their flat-bucket views and
# readiness is marked explicitly, eliminating the per-param AccumulateGrad
# `grad += dw` kernels (the last at::native residue on the hot path —
# measured 0.75 ms/step at ResNet-50 b1024).  claim(p) returns the view
# only on the FIRST backward pass of an accumulation window; later passes
# return None and take the normal autograd-accumulate path.
_grad_direct = None


def set_direct_wgrad(registry) -> None:
    global _grad_direct
    _grad_direct = registry


class BnBwdFusion:
    """Side-channel between a BatchNorm and the conv that consumes its
    output: the conv's dgrad epilogue emits the BN's backward stats
    partials (the bwd analog of conv_fwd_stats), so the BN skips its
    bn_bwd_stats full (dy, x) pass.  The BN fills mean/rstd at forward;
    the conv's backward fills partials; the BN's backward (which autograd
    runs after the conv's) consumes them."""

    __slots__ = ("mean", "rstd", "partials", "mask")

    def __init__(self):
        self.mean = None
        self.rstd = None
        self.partials = None
        self.mask = None  # bit-packed ReLU mask (residual-join BNs)


class _Conv2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding, relu, emit_stats=False,
                bn_fuse=None):
        ext = load_kernels_ext()
        x = to_channels_last(x)
        weight = to_channels_last(weight)
        epu = 4 if x.dtype == torch.float32 else 8
        ctx.small_c = x.shape[1] % epu != 0
        partials = None
        if ctx.small_c:
            # stem path: keep the im2col buffer for wgrad reuse
            y, col, partials = ext.conv_fwd_smallc(
                x, weight, bias if bias is not None else None,
                stride[0], stride[1], padding[0], padding[1], relu)
            if not emit_stats:
                partials = None
            saved = [x, weight, col]
        else:
            if emit_stats:
                # K1+K4 fusion: the MFMA epilogue also emits the following
                # BatchNorm's {sum, sumsq} partials, skipping bn_stats_k's
                # full re-read of y (empty partials = unsupported path)
                y, partials = ext.conv_fwd_stats(
                    x, weight, bias if bias is not None else None,
                    stride[0], stride[1], padding[0], padding[1], relu)
            else:
                y = ext.conv_fwd(x, weight,
                                 bias if bias is not None else None,
                                 stride[0], stride[1], padding[0],
                                 padding[1], relu)
            saved = [x, weight]
        if relu:
            saved.append(y)
        ctx.save_for_backward(*saved)
        ctx.stride = stride
        ctx.padding = padding
        ctx.has_bias = bias is not None
        ctx.relu = relu
        ctx.emit_stats = emit_stats
        # (holder, bnx, bn_w, bn_b) — read-only refs, no grads through here
        ctx.bn_fuse = bn_fuse if not ctx.small_c else None
        if emit_stats:
            if partials is None:
                partials = torch.empty(0, dtype=torch.float32,
                                       device=x.device)
            ctx.mark_non_differentiable(partials)
            return y, partials
        return y

    @staticmethod
    def backward(ctx, dy, *unused_dpartials):
        ext = load_kernels_ext()
        saved = list(ctx.saved_tensors)
        if ctx.relu:
            y = saved.pop()
        if ctx.small_c:
            x, weight, col = saved
        else:
            x, weight = saved
            col = None
        if ctx.relu:
            dy = ext.relu_bwd(to_channels_last(dy), y)
        else:
            dy = to_channels_last(dy)
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            if ctx.small_c:
                # stem path: dcol = dy @ w_pad (library GEMM) + col2im
                # scatter-add — completes K2 for C % 8 != 0 inputs
                dx = ext.conv_dgrad_smallc(dy, weight, list(x.shape),
                                           ctx.stride[0], ctx.stride[1],
                                           ctx.padding[0], ctx.padding[1])
                fuse = None
            else:
                fuse = ctx.bn_fuse
                if (fuse is not None and fuse[0].mean is not None
                        and x.dtype != torch.float32):
                    holder, bnx, bn_w, bn_b = fuse
                    dx, bparts = ext.conv_dgrad_bnstats(
                        dy, weight, list(x.shape), ctx.stride[0],
                        ctx.stride[1], ctx.padding[0], ctx.padding[1],
                        to_channels_last(bnx), holder.mean, holder.rstd,
                        bn_w, bn_b, None, holder.mask)
                    holder.partials = bparts
                else:
                    dx = ext.conv_dgrad(dy, weight, list(x.shape),
                                        ctx.stride[0], ctx.stride[1],
                                        ctx.padding[0], ctx.padding[1])
        if ctx.needs_input_grad[1]:
            if ctx.small_c:
                dw = ext.conv_wgrad_smallc(dy, col, list(weight.shape),
                                           list(x.shape))
            else:
                dst = _grad_direct.claim(weight) if _grad_direct else None
                dw = ext.conv_wgrad(dy, x, list(weight.shape),
                                    ctx.stride[0], ctx.stride[1],
                                    ctx.padding[0], ctx.padding[1], dst)
                if dst is not None:
                    _grad_direct.commit(weight)
                    dw = None  # grad landed in the bucket view directly
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = dy.sum(dim=(0, 2, 3))
        return dx, dw, db, None, None, None, None, None


def conv2d(x, weight, bias=None, stride=1, padding=0, relu=False,
           bn_fuse=None):
    stride, padding = _pair(stride), _pair(padding)
    if x.is_cuda:
        return _Conv2dFn.apply(x, weight, bias, stride, padding, relu, False,
                               bn_fuse)
    y = F.conv2d(x, weight, bias, stride, padding)
    return F.relu(y) if relu else y


def conv2d_stats(x, weight, bias=None, stride=1, padding=0, bn_fuse=None):
    """conv2d that also returns the following BatchNorm's {sum, sumsq}
    partials computed in the conv epilogue (None when unsupported: CPU,
    fp32, or the C%8!=0 stem)."""
    stride, padding = _pair(stride), _pair(padding)
    if x.is_cuda:
        y, partials = _Conv2dFn.apply(x, weight, bias, stride, padding,
                                      False, True, bn_fuse)
        return y, (partials if partials.numel() > 0 else None)
    return F.conv2d(x, weight, bias, stride, padding), None


class _ForkConvFn(torch.autograd.Function):
    """(identity, y, partials) = (x, conv(x), bn-partials).

    Routing the residual fork THROUGH the conv's autograd Function gives
    the block input a single consumer, so backward receives both the
    conv-path gradient (dy) and the identity-path gradient (dident) and
    fuses ``dx = conv_dgrad(dy, w) + dident`` in the dgrad epilogue —
    removing the separate full-activation autograd add at every residual
    join (~5.5 ms/step at ResNet-50 b1024, the largest single
    at::native elementwise cost)."""

    @staticmethod
    def forward(ctx, x, weight, stride, padding, emit_stats, bn_fuse=None):
        ext = load_kernels_ext()
        x = to_channels_last(x)
        weight = to_channels_last(weight)
        if emit_stats:
            y, partials = ext.conv_fwd_stats(x, weight, None, stride[0],
                                             stride[1], padding[0],
                                             padding[1], False)
        else:
            y = ext.conv_fwd(x, weight, None, stride[0], stride[1],
                             padding[0], padding[1], False)
            partials = torch.empty(0, dtype=torch.float32, device=x.device)
        ctx.save_for_backward(x, weight)
        ctx.stride = stride
        ctx.padding = padding
        ctx.bn_fuse = bn_fuse
        ident = x.view_as(x)
        ctx.mark_non_differentiable(partials)
        return ident, y, partials

    @staticmethod
    def backward(ctx, dident, dy, _dpartials):
        ext = load_kernels_ext()
        x, weight = ctx.saved_tensors
        dy = to_channels_last(dy)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            acc = to_channels_last(dident) if dident is not None else None
            fuse = ctx.bn_fuse
            if (fuse is not None and fuse[0].mean is not None
                    and fuse[0].mask is not None
                    and x.dtype != torch.float32):
                # residual-join BN backward-stats fusion: the accumulated
                # dgrad output IS the bn3 dy; the saved bit-mask supplies
                # the ReLU activity (bn.hip mode 3)
                holder, bnx, bn_w, bn_b = fuse
                dx, bparts = ext.conv_dgrad_bnstats(
                    dy, weight, list(x.shape), ctx.stride[0], ctx.stride[1],
                    ctx.padding[0], ctx.padding[1], to_channels_last(bnx),
                    holder.mean, holder.rstd, bn_w, bn_b, acc, holder.mask)
                holder.partials = bparts
            else:
                dx = ext.conv_dgrad(dy, weight, list(x.shape), ctx.stride[0],
                                    ctx.stride[1], ctx.padding[0],
                                    ctx.padding[1], acc)
        if ctx.needs_input_grad[1]:
            dst = _grad_direct.claim(weight) if _grad_direct else None
            dw = ext.conv_wgrad(dy, x, list(weight.shape), ctx.stride[0],
                                ctx.stride[1], ctx.padding[0],
                                ctx.padding[1], dst)
            if dst is not None:
                _grad_direct.commit(weight)
                dw = None
        return dx, dw, None, None, None, None


_FUSE_RESJOIN = os.environ.get("DDL_FUSE_RESJOIN", "1") == "1"


def fork_conv2d(x, conv_module):
    """Residual-block entry: returns (identity_source, conv_output) with
    the join-gradient add fused into the conv's dgrad (GPU training path;
    everything else falls back to the plain fork)."""
    run_fused = (_FUSE_RESJOIN and x.is_cuda and conv_module.training
                 and torch.is_grad_enabled()
                 and x.shape[1] % (4 if x.dtype == torch.float32 else 8) == 0)
    if not run_fused:
        return x, conv_module(x)
    fuse = getattr(x, "_ddl_bn_bwd_fuse", None)
    if fuse is not None:
        del x._ddl_bn_bwd_fuse  # this fork is the single autograd consumer
    emit = (conv_module.emit_bn_stats and conv_module.training
            and x.dtype != torch.float32)
    ident, y, partials = _ForkConvFn.apply(
        x, conv_module.weight, conv_module.stride, conv_module.padding,
        emit, fuse)
    if partials.numel() > 0:
        y._ddl_bn_partials = partials
    return ident, y


# --------------------------------------------------------------------------
# Fused BatchNorm (+ residual add) (+ ReLU)   (K4/K5/K14)
# --------------------------------------------------------------------------

class _BnAddReluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var,
                momentum, eps, relu, residual, conv_partials=None,
                bwd_holder=None):
        ext = load_kernels_ext()
        x = to_channels_last(x)
        if residual is not None:
            residual = to_channels_last(residual)
        y, mean, rstd, relu_mask = ext.bn_relu_fwd(
            x, weight, bias, running_mean, running_var,
            float(momentum), float(eps), relu,
            residual if residual is not None else None,
            conv_partials,
        )
        ctx.save_for_backward(x, weight, bias, mean, rstd, y, relu_mask)
        ctx.relu = relu
        ctx.has_residual = residual is not None
        ctx.bwd_holder = bwd_holder
        if bwd_holder is not None:
            bwd_holder.mean = mean
            bwd_holder.rstd = rstd
            bwd_holder.mask = relu_mask if relu_mask.numel() > 0 else None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_kernels_ext()
        x, weight, bias, mean, rstd, y, relu_mask = ctx.saved_tensors
        dy = to_channels_last(dy)
        holder = ctx.bwd_holder
        pre = holder.partials if holder is not None else None
        if pre is not None:
            holder.partials = None  # consume once
            dx, dw, db, dres = ext.bn_relu_bwd(
                dy, x, weight, bias, mean, rstd, y, ctx.relu,
                ctx.has_residual, pre, relu_mask)
        else:
            dx, dw, db, dres = ext.bn_relu_bwd(
                dy, x, weight, bias, mean, rstd, y, ctx.relu,
                ctx.has_residual, None, relu_mask)
        return (dx, dw, db, None, None, None, None, None,
                dres if ctx.has_residual else None, None, None)


def bn_add_relu(x, weight, bias, running_mean, running_var, training,
                momentum=0.1, eps=1e-5, relu=True, residual=None,
                conv_partials=None, bwd_holder=None):
    """y = relu?(bn(x) + residual?) — single fused kernel each direction.

    The reference issues separate cuDNN BN and ReLU (and a separate add for
    the bottleneck exit); on MI355X the op is HBM-bound so fusing saves two
    full activation round-trips."""
    if x.is_cuda:
        if training:
            return _BnAddReluFn.apply(x, weight, bias, running_mean,
                                      running_var, momentum, eps, relu,
                                      residual, conv_partials, bwd_holder)
        ext = load_kernels_ext()
        return ext.bn_relu_infer(to_channels_last(x), weight, bias,
                                 running_mean, running_var, float(eps), relu,
                                 to_channels_last(residual) if residual is not None else None)
    # CPU reference
    y = F.batch_norm(x, running_mean, running_var, weight, bias,
                     training, momentum, eps)
    if residual is not None:
        y = y + residual
    if relu:
        y = F.relu(y)
    return y


# --------------------------------------------------------------------------
# Pooling (K6/K7)
# --------------------------------------------------------------------------

class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kernel, stride, padding):
        ext = load_kernels_ext()
        x = to_channels_last(x)
        y, idx = ext.maxpool_fwd(x, kernel[0], kernel[1], stride[0], stride[1],
                                 padding[0], padding[1])
        ctx.save_for_backward(idx)
        ctx.x_shape = list(x.shape)
        ctx.params = (kernel, stride, padding)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_kernels_ext()
        (idx,) = ctx.saved_tensors
        kernel, stride, padding = ctx.params
        dx = ext.maxpool_bwd(to_channels_last(dy), idx, ctx.x_shape,
                             kernel[0], kernel[1], stride[0], stride[1],
                             padding[0], padding[1])
        return dx, None, None, None


def max_pool2d(x, kernel_size, stride=None, padding=0):
    kernel = _pair(kernel_size)
    stride = _pair(stride) if stride is not None else kernel
    padding = _pair(padding)
    if x.is_cuda:
        return _MaxPoolFn.apply(x, kernel, stride, padding)
    return F.max_pool2d(x, kernel, stride, padding)


class _GlobalAvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = load_kernels_ext()
        x = to_channels_last(x)
        ctx.x_shape = list(x.shape)
        return ext.gap_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        ext = load_kernels_ext()
        return ext.gap_bwd(dy.contiguous(), ctx.x_shape)


def global_avg_pool(x):
    """[N,C,H,W] -> [N,C] mean over H,W."""
    if x.is_cuda:
        return _GlobalAvgPoolFn.apply(x)
    return x.mean(dim=(2, 3))


# --------------------------------------------------------------------------
# Linear (K8: plain library GEMM — rocBLAS/hipBLASLt via torch.matmul)
# --------------------------------------------------------------------------

def linear(x, weight, bias=None):
    return F.linear(x, weight, bias)


# --------------------------------------------------------------------------
# Fused softmax + cross-entropy (K9)
# --------------------------------------------------------------------------

class _SoftmaxXentFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target, label_smoothing):
        ext = load_kernels_ext()
        losses, probs = ext.softmax_xent_fwd(logits.contiguous(), target,
                                             float(label_smoothing))
        ctx.save_for_backward(probs, target)
        ctx.label_smoothing = label_smoothing
        ctx.logits_dtype = logits.dtype
        return losses.mean()

    @staticmethod
    def backward(ctx, dloss):
        ext = load_kernels_ext()
        probs, target = ctx.saved_tensors
        code = {torch.float32: 0, torch.bfloat16: 1, torch.float16: 2}[ctx.logits_dtype]
        dlogits = ext.softmax_xent_bwd(probs, target,
                                       float(ctx.label_smoothing),
                                       dloss.to(torch.float32), code)
        return dlogits, None, None


def softmax_cross_entropy(logits, target, label_smoothing: float = 0.0):
    if logits.is_cuda:
        return _SoftmaxXentFn.apply(logits, target, label_smoothing)
    return F.cross_entropy(logits.float(), target, label_smoothing=label_smoothing)


# --------------------------------------------------------------------------
# Dropout (K15, VGG path)
# --------------------------------------------------------------------------

class _DropoutFn(torch.autograd.Function):
    _seed_counter = [0]

    @staticmethod
    def forward(ctx, x, p, seed):
        ext = load_kernels_ext()
        y, mask = ext.dropout_fwd(x.contiguous(), float(p), int(seed))
        ctx.save_for_backward(mask)
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_kernels_ext()
        (mask,) = ctx.saved_tensors
        return ext.dropout_bwd(dy.contiguous(), mask, float(ctx.p)), None, None


def dropout(x, p=0.5, training=True, seed: Optional[int] = None):
    if not training or p == 0.0:
        return x
    if x.is_cuda:
        if seed is None:
            _DropoutFn._seed_counter[0] += 1
            seed = torch.initial_seed() % (2**31) + _DropoutFn._seed_counter[0]
        return _DropoutFn.apply(x, p, seed)
    return F.dropout(x, p, training)


# --------------------------------------------------------------------------
# Loss scaling utilities (K13, fp16 path)
# --------------------------------------------------------------------------

def unscale_and_check_finite(flats, inv_scale: float) -> bool:
    """In-place multiply each flat grad buffer by inv_scale; returns True if
    all values finite (fp16 dynamic loss scaling)."""
    if flats and flats[0].is_cuda:
        ext = load_kernels_ext()
        found_inf = torch.zeros(1, dtype=torch.int32, device=flats[0].device)
        for f in flats:
            ext.unscale_check(f, float(inv_scale), found_inf)
        return int(found_inf.item()) == 0
    ok = True
    for f in flats:
        f.mul_(inv_scale)
        ok = ok and bool(torch.isfinite(f).all())
    return ok
