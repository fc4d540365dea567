"""Multi-rank exercise of NativeRcclBackend's control flow with a mocked
RCCL engine (VERDICT r1 next-round #1b).

The real engine needs one GPU per rank; here a fake ``_C_comm`` whose
CommEngine delegates to gloo runs the EXACT Python-side sequence the 8-GPU
job will run — uid generation on rank 0, broadcast_object_list exchange,
pre-init consensus, engine construction, per-bucket allreduce_async /
wait_handle through DistributedOptimizer — multi-process on CPU.  Numerics
flow through the fake engine, so rank-divergence bugs in the sequence are
caught, not just API shape.
"""
import torch

from tests.dist_helpers import run_distributed


class _FakeEngine:
    """Stands in for csrc/comm CommEngine: same call surface, gloo data
    plane, and strict checks on the init contract."""

    def __init__(self, rank, size, device, uid):
        import torch.distributed as torch_dist
        assert isinstance(uid, bytes) and len(uid) == 128, \
            "uid must be the raw ncclUniqueId payload"
        # every rank must hold rank 0's uid bit-for-bit
        mine = torch.tensor(list(uid), dtype=torch.uint8)
        root = mine.clone()
        torch_dist.broadcast(root, src=0)
        assert torch.equal(mine, root), "uid differs from rank 0's"
        self._rank, self._size = rank, size
        self._next = 0
        self._inflight = {}
        self.enqueue_log = []  # (op, numel) in issue order

    def rank(self):
        return self._rank

    def size(self):
        return self._size

    def allreduce_async(self, t, average):
        import torch.distributed as torch_dist
        assert t.is_contiguous()
        self.enqueue_log.append(("allreduce", t.numel()))
        work = torch_dist.all_reduce(t, async_op=True)
        h = self._next
        self._next += 1
        self._inflight[h] = (work, t, average)
        return h

    def broadcast_async(self, t, root):
        import torch.distributed as torch_dist
        self.enqueue_log.append(("broadcast", t.numel()))
        work = torch_dist.broadcast(t, src=root, async_op=True)
        h = self._next
        self._next += 1
        self._inflight[h] = (work, t, False)
        return h

    def allgather_async(self, src, dst):
        import torch.distributed as torch_dist
        self.enqueue_log.append(("allgather", src.numel()))
        shards = list(dst.chunk(self._size, dim=0))
        work = torch_dist.all_gather(shards, src, async_op=True)
        h = self._next
        self._next += 1
        self._inflight[h] = (work, None, False)
        return h

    def wait_handle(self, h):
        if h not in self._inflight:
            return
        work, t, average = self._inflight.pop(h)
        work.wait()
        if average and t is not None:
            t.div_(self._size)

    def join_compute_stream(self):
        for h in list(self._inflight):
            self.wait_handle(h)

    def outstanding(self):
        return len(self._inflight)

    def oldest_age_seconds(self):
        return 0.0

    def enable_timeline(self, path):
        pass

    def shutdown(self):
        self.join_compute_stream()


class _FakeExt:
    CommEngine = _FakeEngine

    @staticmethod
    def get_unique_id():
        # shape/size-faithful stand-in for ncclGetUniqueId (128-byte id)
        import os
        return os.urandom(128)


def _w_native_sequence(rank, world):
    import distributeddeeplearning_amd.parallel.backend as backend_mod
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    from distributeddeeplearning_amd.parallel import (DistributedOptimizer,
                                                      api)
    import torch.distributed as torch_dist

    # bootstrap gloo exactly as create_backend would
    if not torch_dist.is_initialized():
        torch_dist.init_process_group("gloo")

    # patch the extension loader + the CUDA-only data-plane predicate
    import distributeddeeplearning_amd.ops._ext as ext_mod
    ext_mod.load_comm_ext = lambda: _FakeExt
    backend_mod.NativeRcclBackend._data_plane = lambda self, t: True

    nb = backend_mod.NativeRcclBackend(device_index=0)
    api._backend = nb

    assert nb.rank() == rank and nb.size() == world

    torch.manual_seed(21)
    model = torch.nn.Sequential(torch.nn.Linear(16, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4))
    api.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = DistributedOptimizer(
        FusedSGD(model.parameters(), lr=0.05, momentum=0.9),
        named_parameters=model.named_parameters(), bucket_bytes=1024)
    for step in range(3):
        g = torch.Generator().manual_seed(step)
        x = torch.randn(2 * world, 16, generator=g)
        y = torch.randn(2 * world, 4, generator=g)
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(
            model(x[rank * 2:(rank + 1) * 2]), y[rank * 2:(rank + 1) * 2])
        loss.backward()
        opt.step()
    params = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    # uid exchange + engine ctor + per-bucket enqueue all ran through the
    # fake engine; return the log so the test can assert the launch ORDER
    # is identical across ranks (the static-order invariant RCCL needs)
    api._backend = None
    return {"params": params, "log": nb.engine.enqueue_log,
            "outstanding": nb.engine.outstanding()}


def test_native_backend_multirank_sequence():
    res = run_distributed(_w_native_sequence, world=2)
    assert torch.equal(res[0]["params"], res[1]["params"]), "ranks diverged"
    assert res[0]["log"] == res[1]["log"], \
        "collective enqueue order differs across ranks"
    # every bucket allreduce was drained by synchronize()
    assert all(r["outstanding"] == 0 for r in res)
    # buckets (not per-param tensors) were the allreduce granularity
    ops = [o for o, _ in res[0]["log"]]
    assert ops.count("allreduce") >= 3  # >= one bucket per step


def _w_native_init_failure_consensus(rank, world):
    """One rank's extension load fails -> EVERY rank must raise from
    NativeRcclBackend (no half-native/half-fallback split)."""
    import distributeddeeplearning_amd.parallel.backend as backend_mod
    import distributeddeeplearning_amd.ops._ext as ext_mod
    import torch.distributed as torch_dist
    if not torch_dist.is_initialized():
        torch_dist.init_process_group("gloo")

    if rank == 1:
        def _boom():
            raise ImportError("simulated missing _C_comm.so")
        ext_mod.load_comm_ext = _boom
    else:
        ext_mod.load_comm_ext = lambda: _FakeExt
    try:
        backend_mod.NativeRcclBackend(device_index=0)
    except RuntimeError as e:
        return {"raised": "unavailable on at least one rank" in str(e)}
    return {"raised": False}


def test_native_init_failure_is_collective():
    res = run_distributed(_w_native_init_failure_consensus, world=2)
    assert all(r["raised"] for r in res), res
