"""Tier-0 distributed CPU tests (BASELINE config 1: gloo, world_size=2).

Mirrors Horovod's upstream test themes (``test/parallel/test_torch.py``):
allreduce correctness, broadcast state equality, DistributedOptimizer ==
large-batch single-process SGD, checkpoint resume.
"""
import os

import pytest
import torch

from tests.dist_helpers import run_distributed


# ---- worker functions (module-level for spawn pickling) -------------------

def _w_allreduce(rank, world):
    from distributeddeeplearning_amd.parallel import api
    api.init("torchdist")
    t = torch.arange(8, dtype=torch.float32) * (rank + 1)
    out = api.allreduce(t, average=False)
    avg = api.allreduce(torch.full((4,), float(rank)), average=True)
    bc = torch.full((3,), float(rank))
    api.broadcast_(bc, root_rank=1)
    ag = api.allgather(torch.tensor([float(rank)]))
    return {
        "sum": out.tolist(), "avg": avg.tolist(), "bcast": bc.tolist(),
        "gathered": ag.tolist(), "rank": api.rank(), "size": api.size(),
    }


def _make_mlp(seed=7):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.Tanh(), torch.nn.Linear(32, 4))


def _w_dist_opt(rank, world):
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    from distributeddeeplearning_amd.parallel import DistributedOptimizer, api
    api.init("torchdist")
    model = _make_mlp(seed=100 + rank)  # intentionally different init
    api.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = DistributedOptimizer(
        FusedSGD(model.parameters(), lr=0.05, momentum=0.9),
        named_parameters=model.named_parameters(), bucket_bytes=4096)
    opt.optimizer.refresh_master()
    losses = []
    for step in range(4):
        g = torch.Generator().manual_seed(step)
        x = torch.randn(8, 16, generator=g)  # identical across ranks
        y = torch.randn(8, 4, generator=g)
        xr = x[rank * 4:(rank + 1) * 4]
        yr = y[rank * 4:(rank + 1) * 4]
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(xr), yr)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    params = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    return {"params": params, "losses": losses}


def _w_ckpt_resume(rank, world, tmpdir):
    from distributeddeeplearning_amd.engine import checkpoint as ckpt
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    from distributeddeeplearning_amd.parallel import DistributedOptimizer, api
    api.init("torchdist")
    torch.manual_seed(3)

    def build():
        model = _make_mlp(seed=3)
        opt = DistributedOptimizer(
            FusedSGD(model.parameters(), lr=0.05, momentum=0.9),
            named_parameters=model.named_parameters(), bucket_bytes=4096)
        return model, opt

    def one_step(model, opt, step):
        g = torch.Generator().manual_seed(1000 + step)
        x = torch.randn(4, 16, generator=g)
        y = torch.randn(4, 4, generator=g)
        opt.zero_grad()
        torch.nn.functional.mse_loss(model(x), y).backward()
        opt.step()

    model, opt = build()
    for s in range(3):
        one_step(model, opt, s)
    path = os.path.join(tmpdir, "ck.pt")
    ckpt.save(path, model, opt, epoch=1, step=3)
    api.barrier()
    one_step(model, opt, 3)  # the "step N+1" reference continuation
    ref = torch.cat([p.detach().reshape(-1) for p in model.parameters()])

    # fresh replica resumes and repeats step N+1
    model2, opt2 = build()
    ckpt.load(path, model2, opt2)
    one_step(model2, opt2, 3)
    got = torch.cat([p.detach().reshape(-1) for p in model2.parameters()])
    return {"max_diff": float((ref - got).abs().max())}


def _single_process_reference():
    """batch-8 single-process trajectory to compare with 2x4 DP."""
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    model = _make_mlp(seed=100)
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
    for step in range(4):
        g = torch.Generator().manual_seed(step)
        x = torch.randn(8, 16, generator=g)
        y = torch.randn(8, 4, generator=g)
        opt.zero_grad()
        torch.nn.functional.mse_loss(model(x), y).backward()
        opt.step()
    return torch.cat([p.detach().reshape(-1) for p in model.parameters()])


# ---- tests ----------------------------------------------------------------

def test_collectives_cpu_gloo():
    res = run_distributed(_w_allreduce, world=2)
    expect_sum = (torch.arange(8.0) * 1 + torch.arange(8.0) * 2).tolist()
    for r in res:
        assert r["size"] == 2
        assert r["sum"] == expect_sum
        assert r["avg"] == [0.5] * 4
        assert r["bcast"] == [1.0] * 3  # root_rank=1's value
        assert r["gathered"] == [0.0, 1.0]


def test_distributed_optimizer_rank_equality_and_dp_equivalence():
    res = run_distributed(_w_dist_opt, world=2)
    p0, p1 = res[0]["params"], res[1]["params"]
    assert torch.equal(p0, p1), "ranks diverged after DP steps"
    ref = _single_process_reference()
    # DP of two half-batches with grad averaging == full-batch single process
    assert torch.allclose(p0, ref, atol=1e-5), float((p0 - ref).abs().max())


def test_checkpoint_resume_reproduces(tmp_path):
    res = run_distributed(_w_ckpt_resume, world=2, args=(str(tmp_path),))
    for r in res:
        assert r["max_diff"] < 1e-6


def _w_trainer_smoke(rank, world):
    from distributeddeeplearning_amd.config import PRESETS
    from distributeddeeplearning_amd.engine.trainer import Trainer
    import dataclasses
    cfg = dataclasses.replace(PRESETS["cpu-plumbing"], batch_size=8,
                              log_interval=2)
    tr = Trainer(cfg, steps_per_epoch=4)
    loss = tr.train_epoch(steps=4)
    params = torch.cat([p.detach().reshape(-1) for p in tr.model.parameters()])
    return {"loss": loss, "psum": float(params.sum()), "pall": params}


def test_trainer_cpu_plumbing_world2():
    """BASELINE config 1 end-to-end: ResNet-18 synthetic 32x32, gloo ws=2."""
    res = run_distributed(_w_trainer_smoke, world=2, timeout=300)
    assert torch.equal(res[0]["pall"], res[1]["pall"]), "ranks diverged"
    assert res[0]["loss"] == res[0]["loss"]  # finite


def _w_compression(rank, world):
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    from distributeddeeplearning_amd.parallel import (Compression,
                                                      DistributedOptimizer,
                                                      api)
    api.init("torchdist")
    os.environ["DDL_DEBUG_ORDER"] = "1"  # collective-order checksum on
    model = _make_mlp(seed=5)
    api.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = DistributedOptimizer(
        FusedSGD(model.parameters(), lr=0.05, momentum=0.9),
        named_parameters=model.named_parameters(), bucket_bytes=4096,
        compression=Compression.fp16)
    for step in range(2):
        g = torch.Generator().manual_seed(step)
        x = torch.randn(4, 16, generator=g)
        y = torch.randn(4, 4, generator=g)
        opt.zero_grad()
        torch.nn.functional.mse_loss(model(x), y).backward()
        opt.step()
    params = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    return {"params": params}


def test_fp16_compression_keeps_ranks_synced():
    res = run_distributed(_w_compression, world=2)
    assert torch.equal(res[0]["params"], res[1]["params"])


def _w_world4(rank, world):
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    from distributeddeeplearning_amd.parallel import DistributedOptimizer, api
    api.init("torchdist")
    model = _make_mlp(seed=9)
    api.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = DistributedOptimizer(
        FusedSGD(model.parameters(), lr=0.05, momentum=0.9),
        named_parameters=model.named_parameters(), bucket_bytes=2048,
        backward_passes_per_step=2)
    for step in range(4):  # 2 optimizer steps with 2-pass accumulation
        g = torch.Generator().manual_seed(step)
        x = torch.randn(8, 16, generator=g)
        y = torch.randn(8, 4, generator=g)
        xr = x[rank * 2:(rank + 1) * 2]
        yr = y[rank * 2:(rank + 1) * 2]
        torch.nn.functional.mse_loss(model(xr), yr).backward()
        opt.step()
        opt.zero_grad()
    params = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    return {"params": params}


def test_world4_with_grad_accumulation():
    """4 ranks x 2-pass accumulation over gloo: all ranks stay identical."""
    res = run_distributed(_w_world4, world=4, timeout=300)
    for r in res[1:]:
        assert torch.equal(res[0]["params"], r["params"])


def _w_dtype_matrix(rank, world):
    """Horovod-suite theme (test_torch.py): allreduce/broadcast across
    dtypes and shapes, including 0-dim and large tensors."""
    import torch
    from distributeddeeplearning_amd.parallel import api
    api.init("torchdist")
    out = {}
    for dt in (torch.float32, torch.float64, torch.int64):
        for shape in ((), (1,), (7, 3), (100_000,)):
            t = torch.ones(shape, dtype=dt) * (rank + 1)
            r = api.allreduce(t, average=False)
            out[f"ar/{dt}/{shape}"] = bool((r == world * (world + 1) // 2).all())
            b = torch.full(shape, float(rank)).to(dt)
            api.broadcast_(b, root_rank=0)
            out[f"bc/{dt}/{shape}"] = bool((b == 0).all())
    # average on float dtypes
    t = torch.full((5,), float(rank + 1))
    r = api.allreduce(t, average=True)
    out["avg"] = bool((r == (world + 1) / 2).all())
    return out


def test_collective_dtype_shape_matrix():
    res = run_distributed(_w_dtype_matrix, world=2)
    for r in res:
        bad = [k for k, v in r.items() if not v]
        assert not bad, bad


def _w_autotune(rank, world):
    """Online bucket autotune: every rank walks the same candidate ladder,
    picks the allreduced winner, and the ranks stay bit-identical."""
    import dataclasses
    from distributeddeeplearning_amd.config import PRESETS
    from distributeddeeplearning_amd.engine.trainer import Trainer
    cfg = dataclasses.replace(PRESETS["cpu-plumbing"], batch_size=8,
                              autotune_buckets=True)
    tr = Trainer(cfg, steps_per_epoch=100)
    tr.bucket_tuner.sizes = [0.05, 0.2]          # tiny ladder for CI
    tr.bucket_tuner.steps_per = 3
    tr.bucket_tuner.warmup = 1
    it = tr.data_iter()
    for _ in range(10):
        x, y = next(it)
        tr.train_step(x, y)
    params = torch.cat([p.detach().reshape(-1)
                        for p in tr.model.parameters()])
    return {"params": params, "done": tr.bucket_tuner.done,
            "chosen": tr.bucket_tuner.chosen_mb,
            "n_buckets": len(tr.opt.buckets.buckets)}


def test_online_bucket_autotune_world2():
    res = run_distributed(_w_autotune, world=2, timeout=300)
    assert all(r["done"] for r in res)
    assert res[0]["chosen"] == res[1]["chosen"], "ranks chose different sizes"
    assert res[0]["n_buckets"] == res[1]["n_buckets"]
    assert torch.equal(res[0]["params"], res[1]["params"]), "ranks diverged"
