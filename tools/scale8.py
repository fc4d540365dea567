#!/usr/bin/env python3
"""One-shot 8-GPU tuning + evidence run (VERDICT r1 next-round #1d).

On an N-GPU MI355X node this single command produces the distributed
tuning artifacts the 1-GPU rounds could not measure:

    python tools/scale8.py --gpus 8 --out profiles

It runs, in order, each as a child `tools/launch.py --nproc N` job:

1. scaling curve   — bench.py at 1, 2, 4, N GPUs (native backend);
2. bucket sweep    — tools/tune_buckets.py at N GPUs;
3. channel sweep   — bench.py at N GPUs under NCCL_MIN/MAX_NCHANNELS in
                     {1, 2, 4, 7, 14, 28} (the 7-link xGMI mesh: a ring
                     uses one link/GPU, concurrent channels spread over
                     the 7 point-to-point links — SURVEY.md §5.8);
4. overlap check   — a DDL_TIMELINE run at N GPUs; the report states how
                     much of the RCCL span time lies inside the
                     backward+allreduce device span (dev/ spans are
                     hipEvent-timed).

Everything is appended to <out>/scale8_results.md (tracked in git) plus
raw JSON at <out>/scale8_results.json.
"""
from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def run_bench(nproc: int, extra_env: dict | None = None, steps: int = 20,
              warmup: int = 5, batch: int = 1024, timeout: int = 1200,
              bench_args: list[str] | None = None) -> dict | None:
    env = dict(os.environ)
    env.setdefault("DDL_COMM", "native" if nproc > 1 else "auto")
    if extra_env:
        env.update({k: str(v) for k, v in extra_env.items()})
    cmd = [sys.executable, os.path.join(REPO, "tools", "launch.py"),
           "--nproc", str(nproc), os.path.join(REPO, "bench.py"),
           "--gpus", str(nproc), "--steps", str(steps),
           "--warmup", str(warmup), "--batch-size", str(batch)]
    if bench_args:
        cmd += bench_args
    if nproc == 1:
        cmd = [sys.executable, os.path.join(REPO, "bench.py"),
               "--gpus", "1", "--steps", str(steps), "--warmup", str(warmup),
               "--batch-size", str(batch)] + (bench_args or [])
    t0 = time.time()
    try:
        out = subprocess.run(cmd, env=env, capture_output=True, text=True,
                             timeout=timeout, cwd=REPO)
    except subprocess.TimeoutExpired:
        return None
    for line in reversed(out.stdout.splitlines()):
        line = line.strip()
        if line.startswith("{") and '"images/sec' in line or \
           line.startswith("{") and '"metric"' in line:
            try:
                rec = json.loads(line)
                rec["_wall_s"] = round(time.time() - t0, 1)
                return rec
            except json.JSONDecodeError:
                continue
    sys.stderr.write(f"-- bench nproc={nproc} produced no JSON; rc="
                     f"{out.returncode}\nstderr tail: {out.stderr[-2000:]}\n")
    return None


def overlap_from_timelines(tl_prefix: str, nproc: int) -> dict:
    """Fraction of native-engine RCCL span time that lies inside the
    hipEvent-timed backward window (rank files written by DDL_TIMELINE)."""
    res = {}
    for r in range(nproc):
        try:
            eng = json.load(open(f"{tl_prefix}.rank{r}.comm.json"))
            host = json.load(open(f"{tl_prefix}.rank{r}.json"))
        except (FileNotFoundError, json.JSONDecodeError):
            continue
        bwd = [(e["ts"], e["ts"] + e["dur"])
               for e in host.get("traceEvents", [])
               if e.get("name") == "dev/backward+allreduce"]
        spans = [(e["ts"], e["ts"] + e["dur"])
                 for e in eng.get("traceEvents", [])
                 if e.get("name", "").startswith("span/allreduce")]
        total = sum(b - a for a, b in spans)
        inside = 0.0
        for a, b in spans:
            for c, d in bwd:
                lo, hi = max(a, c), min(b, d)
                if hi > lo:
                    inside += hi - lo
        res[f"rank{r}"] = {
            "rccl_span_ms": round(total / 1000, 2),
            "inside_backward_ms": round(inside / 1000, 2),
            "overlap_frac": round(inside / total, 3) if total else None,
        }
    return res


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=8)
    p.add_argument("--out", default=os.path.join(REPO, "profiles"))
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--batch", type=int, default=1024)
    p.add_argument("--skip", default="", help="comma list: scaling,buckets,channels,overlap")
    args = p.parse_args()
    skip = set(args.skip.split(",")) if args.skip else set()
    os.makedirs(args.out, exist_ok=True)
    results: dict = {"gpus": args.gpus}

    if "scaling" not in skip:
        curve = {}
        for n in [1, 2, 4, args.gpus]:
            if n > args.gpus or n in curve:
                continue
            rec = run_bench(n, steps=args.steps, batch=args.batch)
            if rec:
                curve[n] = {"images_per_sec": rec["value"],
                            "ms_per_step": rec["ms_per_step"],
                            "comm_backend": rec.get("provenance", {}).get("comm_backend")}
                print(f"scaling n={n}: {rec['value']:.0f} img/s "
                      f"({curve[n]['comm_backend']})", flush=True)
        results["scaling"] = curve

    if "buckets" not in skip:
        sweep = {}
        for mb in [10, 25, 50, 100]:
            rec = run_bench(args.gpus, extra_env={"DDL_BUCKET_MB": mb},
                            steps=args.steps, batch=args.batch)
            if rec:
                sweep[mb] = rec["value"]
                print(f"bucket {mb} MB: {rec['value']:.0f} img/s", flush=True)
        results["bucket_sweep_mb"] = sweep

    if "channels" not in skip:
        ch = {}
        for n_ch in [1, 2, 4, 7, 14, 28]:
            rec = run_bench(args.gpus,
                            extra_env={"NCCL_MIN_NCHANNELS": n_ch,
                                       "NCCL_MAX_NCHANNELS": n_ch},
                            steps=args.steps, batch=args.batch)
            if rec:
                ch[n_ch] = rec["value"]
                print(f"channels {n_ch}: {rec['value']:.0f} img/s", flush=True)
        results["channel_sweep"] = ch

    if "overlap" not in skip:
        tl = os.path.join(args.out, "scale8_tl")
        rec = run_bench(args.gpus, extra_env={"DDL_TIMELINE": tl},
                        steps=args.steps, batch=args.batch)
        results["overlap"] = overlap_from_timelines(tl, args.gpus)
        print("overlap:", json.dumps(results["overlap"]), flush=True)

    with open(os.path.join(args.out, "scale8_results.json"), "w") as f:
        json.dump(results, f, indent=1)
    with open(os.path.join(args.out, "scale8_results.md"), "w") as f:
        f.write("# 8-GPU tuning run (tools/scale8.py)\n\n")
        if "scaling" in results:
            f.write("## Scaling curve (native backend)\n\n|GPUs|img/s|ms/step|backend|\n|--|--|--|--|\n")
            for n, r in sorted(results["scaling"].items()):
                f.write(f"|{n}|{r['images_per_sec']:.0f}|{r['ms_per_step']:.1f}|{r['comm_backend']}|\n")
        if results.get("bucket_sweep_mb"):
            f.write("\n## Bucket-size sweep (DDL_BUCKET_MB)\n\n|MB|img/s|\n|--|--|\n")
            for mb, v in sorted(results["bucket_sweep_mb"].items()):
                f.write(f"|{mb}|{v:.0f}|\n")
        if results.get("channel_sweep"):
            f.write("\n## RCCL channel sweep (NCCL_MIN/MAX_NCHANNELS)\n\n|channels|img/s|\n|--|--|\n")
            for c, v in sorted(results["channel_sweep"].items()):
                f.write(f"|{c}|{v:.0f}|\n")
        if results.get("overlap"):
            f.write("\n## Backward/RCCL overlap (hipEvent device spans)\n\n"
                    "|rank|RCCL ms|inside backward ms|overlap|\n|--|--|--|--|\n")
            for r, o in sorted(results["overlap"].items()):
                f.write(f"|{r}|{o['rccl_span_ms']}|{o['inside_backward_ms']}|"
                        f"{o['overlap_frac']}|\n")
    print("wrote", os.path.join(args.out, "scale8_results.md"), flush=True)


if __name__ == "__main__":
    main()
