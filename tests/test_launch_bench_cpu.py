"""End-to-end launcher test: the exact shape of the driver's multi-GPU
invocation (torchrun-style env rendezvous -> bench.py -> one JSON line),
run with 2 CPU ranks over gloo.  Catches rendezvous/provenance/JSON
regressions before they can cost the real 8-GPU run."""
import json
import os
import subprocess
import sys

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def test_launch_bench_world2_cpu():
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "launch.py"),
         "--nproc", "2", os.path.join(REPO, "bench.py"),
         "--steps", "3", "--warmup", "1"],
        capture_output=True, text=True, timeout=420, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = None
    for line in out.stdout.splitlines():
        line = line.strip()
        if line.startswith("{") and '"metric"' in line:
            rec = json.loads(line)
    assert rec is not None, out.stdout[-2000:]
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["value"] > 0
    prov = rec["provenance"]
    assert prov["comm_backend"] == "torchdist"  # CPU: gloo data plane
    assert prov["n_buckets"] >= 1
