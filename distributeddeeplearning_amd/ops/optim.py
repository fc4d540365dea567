"""Fused optimizers (K10 SGD+momentum, K11 LARS).

The reference updates 161 ResNet-50 tensors with framework foreach kernels;
here the whole model updates in ONE multi-tensor HIP kernel launch: a chunk
table (pointers + sizes, built once — gradient pointers are stable views
into the persistent flat buckets) drives a single grid over every parameter.
fp32 master weights + momentum state under bf16/fp16 model parameters; the
1/loss_scale unscale folds into the same pass (K13).

Math matches torch.optim.SGD (weight_decay as L2, dampening=0) so the CPU
path is bit-comparable with the stock optimizer in tier-0 tests; LARS
implements You et al. 2017 (layer-wise trust ratio, excluded for BN params
and biases) as required by the batch-8k config (BASELINE.json:9).
"""
from __future__ import annotations

import torch

from ._ext import load_kernels_ext

CHUNK = 1 << 16  # elements per kernel block


class FusedSGD(torch.optim.Optimizer):
    """SGD + momentum with fp32 master weights, fused on GPU.

    ``lars=True`` turns on layer-wise adaptive rate scaling for tensors with
    ``ndim > 1`` (conv/linear weights); BN parameters and biases always take
    the plain momentum update (standard LARS exclusion list).
    """

    def __init__(self, params, lr, momentum=0.9, weight_decay=0.0,
                 nesterov=False, lars=False, lars_eta=0.001, lars_eps=1e-9,
                 grad_scale=1.0):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        nesterov=nesterov, lars=lars, lars_eta=lars_eta,
                        lars_eps=lars_eps)
        super().__init__(params, defaults)
        self.grad_scale = grad_scale  # multiply grads by 1/scale during update
        self._tables = {}  # per-param-group cached GPU chunk tables

    # ---- state ------------------------------------------------------------

    def _init_state(self, p):
        st = self.state[p]
        if "master" not in st:
            st["master"] = p.detach().to(torch.float32).clone()
            st["momentum_buffer"] = torch.zeros_like(st["master"])
        return st

    def refresh_master(self):
        """Re-copy params into masters (after load_state_dict of the model
        or broadcast_parameters)."""
        for group in self.param_groups:
            for p in group["params"]:
                st = self._init_state(p)
                st["master"].copy_(p.detach().to(torch.float32))

    # ---- GPU fused path ----------------------------------------------------

    def _build_table(self, group):
        """Chunk table: one row per block = [param_ptr, grad_ptr, master_ptr,
        mom_ptr, numel, use_lars, tensor_idx]."""
        ext = load_kernels_ext()
        rows = []
        norms_len = 0
        tensors = []
        for p in group["params"]:
            if p.grad is None:
                continue
            st = self._init_state(p)
            tensors.append((p, st))
        for ti, (p, st) in enumerate(tensors):
            n = p.numel()
            use_lars = 1 if (group["lars"] and p.dim() > 1) else 0
            esz = p.element_size()
            for off in range(0, n, CHUNK):
                cnt = min(CHUNK, n - off)
                rows.append([
                    p.data_ptr() + off * esz,
                    p.grad.data_ptr() + off * esz,
                    st["master"].data_ptr() + off * 4,
                    st["momentum_buffer"].data_ptr() + off * 4,
                    cnt, use_lars, ti,
                ])
        table = torch.tensor(rows, dtype=torch.int64, device=tensors[0][0].device)
        return table, len(tensors)

    def _sig(self, group):
        return tuple(
            (p.data_ptr(), p.grad.data_ptr() if p.grad is not None else 0)
            for p in group["params"]
        )

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for gi, group in enumerate(self.param_groups):
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            if params[0].is_cuda:
                self._step_gpu(gi, group)
            else:
                self._step_cpu(group)
        return loss

    def _step_gpu(self, gi, group):
        ext = load_kernels_ext()
        sig = self._sig(group)
        cached = self._tables.get(gi)
        if cached is None or cached[0] != sig:
            table, ntensors = self._build_table(group)
            norms = torch.zeros(2, ntensors, dtype=torch.float32,
                                device=table.device)
            cached = (sig, table, norms)
            self._tables[gi] = cached
        _, table, norms = cached
        dt = group["params"][0].dtype
        if group["lars"]:
            norms.zero_()
            ext.lars_norms(table, norms, _dtype_code(dt),
                           float(group["weight_decay"]),
                           float(1.0 / self.grad_scale))
        ext.fused_sgd(
            table, norms if group["lars"] else None,
            _dtype_code(dt),
            float(group["lr"]), float(group["momentum"]),
            float(group["weight_decay"]), bool(group["nesterov"]),
            float(group["lars_eta"]), float(group["lars_eps"]),
            float(1.0 / self.grad_scale),
        )

    def _step_cpu(self, group):
        inv_scale = 1.0 / self.grad_scale
        for p in group["params"]:
            if p.grad is None:
                continue
            st = self._init_state(p)
            master, buf = st["master"], st["momentum_buffer"]
            g = p.grad.detach().to(torch.float32)
            if inv_scale != 1.0:
                g = g * inv_scale
            if group["weight_decay"]:
                g = g.add(master, alpha=group["weight_decay"])
            if group["lars"] and p.dim() > 1:
                wn = master.norm()
                gn = g.norm()
                ratio = torch.where(
                    (wn > 0) & (gn > 0),
                    group["lars_eta"] * wn / (gn + group["lars_eps"]),
                    torch.ones_like(wn),
                )
                g = g * ratio
            if group["momentum"]:
                buf.mul_(group["momentum"]).add_(g)
                g = g.add(buf, alpha=group["momentum"]) if group["nesterov"] else buf
            master.add_(g, alpha=-group["lr"])
            p.data.copy_(master.to(p.dtype))

    def state_dict(self):
        sd = super().state_dict()
        return sd

    def load_state_dict(self, sd):
        """Exact-dtype state restore.

        torch's default ``Optimizer.load_state_dict`` casts floating state
        tensors to the PARAM dtype — under bf16/fp16 params that would
        truncate the fp32 masters and momentum (and break the fused kernel,
        which reads them as 4-byte words).  Restore manually, moving device
        but never changing dtype."""
        groups = self.param_groups
        saved_groups = sd["param_groups"]
        if len(groups) != len(saved_groups):
            raise ValueError("loaded state dict has a different number of "
                             "parameter groups")
        id_map = {}
        for g, sg in zip(groups, saved_groups):
            if len(g["params"]) != len(sg["params"]):
                raise ValueError("parameter group size mismatch in "
                                 "load_state_dict")
            for p, pid in zip(g["params"], sg["params"]):
                id_map[pid] = p
            for k, v in sg.items():
                if k != "params":
                    g[k] = v
        self.state.clear()
        for pid, st in sd["state"].items():
            p = id_map.get(pid, pid)
            dev = p.device if torch.is_tensor(p) else None
            self.state[p] = {
                k: (v.to(device=dev) if torch.is_tensor(v) and dev is not None
                    else v)
                for k, v in st.items()
            }
        self._tables = {}

    def invalidate_table(self):
        self._tables = {}


def _dtype_code(dt: torch.dtype) -> int:
    return {torch.float32: 0, torch.bfloat16: 1, torch.float16: 2}[dt]
