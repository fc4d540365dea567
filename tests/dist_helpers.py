"""Spawn N CPU processes over gloo for tier-0 distributed tests
(BASELINE config 1: CPU/gloo world_size=2)."""
from __future__ import annotations

import os
import pickle
import tempfile

import torch.multiprocessing as mp


def _worker(rank: int, world: int, port: int, fn, args, result_dir: str):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    out = fn(rank, world, *args)
    with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
        pickle.dump(out, f)
    from distributeddeeplearning_amd.parallel import api
    api.shutdown()


def run_distributed(fn, world: int = 2, args: tuple = (), timeout: int = 120):
    """Run fn(rank, world, *args) in `world` processes; return list of results."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    with tempfile.TemporaryDirectory() as d:
        ctx = mp.start_processes(
            _worker, args=(world, port, fn, args, d), nprocs=world,
            join=True, start_method="spawn",
        )
        results = []
        for r in range(world):
            with open(os.path.join(d, f"rank{r}.pkl"), "rb") as f:
                results.append(pickle.load(f))
        return results
