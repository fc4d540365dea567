#!/usr/bin/env python3
"""Reporting utilities:

* --profile results.db  -> markdown per-kernel time table (rocprofv3 SQLite)
* --scaling a.json b.json ... -> images/sec scaling table from bench.py
  output lines (the reference's README scaling-chart role).
"""
import argparse
import json
import re
import sqlite3
import sys


def profile_table(db_path, out=None):
    db = sqlite3.connect(db_path)
    sfx = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE name LIKE 'rocpd_kernel_dispatch%'"
    )][0].replace("rocpd_kernel_dispatch_", "")
    q = f"""SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
                   AVG(k.end-k.start)/1e3
            FROM rocpd_kernel_dispatch_{sfx} k
            JOIN rocpd_info_kernel_symbol_{sfx} ks ON k.kernel_id = ks.id
            GROUP BY ks.display_name ORDER BY 3 DESC"""
    lines = ["| kernel | calls | total ms | avg µs |", "|---|---|---|---|"]
    total = 0.0
    for name, n, ms, avg in db.execute(q):
        nm = re.sub(r"<[^>]*>", "<>", name)
        nm = re.sub(r"void \(anonymous namespace\)::", "", nm)[:70]
        total += ms
        lines.append(f"| `{nm}` | {n} | {ms:.2f} | {avg:.1f} |")
    lines.append(f"| **total** | | **{total:.1f}** | |")
    text = "\n".join(lines)
    if out:
        with open(out, "w") as f:
            f.write("# Kernel time breakdown (rocprofv3 --kernel-trace --stats)\n\n")
            f.write(text + "\n")
    else:
        print(text)


def scaling_table(paths):
    rows = []
    for p in paths:
        text = open(p).read().strip()
        if text.startswith("["):  # whole-file JSON array (driver SCALE file)
            rows.extend(r for r in json.loads(text)
                        if isinstance(r, dict) and "n_gpus" in r)
            continue
        for line in text.splitlines():
            line = line.strip()
            if line.startswith("{") and '"images/sec' in line:
                rows.append(json.loads(line))
    rows.sort(key=lambda r: r["n_gpus"])
    if not rows:
        print("no bench JSON lines found")
        return
    base = rows[0]["value"] / rows[0]["n_gpus"]
    print("| GPUs | images/sec | per-GPU | scaling eff. |")
    print("|---|---|---|---|")
    for r in rows:
        eff = r["value"] / (base * r["n_gpus"])
        print(f"| {r['n_gpus']} | {r['value']:.0f} | "
              f"{r['value']/r['n_gpus']:.0f} | {eff*100:.1f}% |")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--profile")
    ap.add_argument("--out")
    ap.add_argument("--scaling", nargs="*")
    a = ap.parse_args()
    if a.profile:
        profile_table(a.profile, a.out)
    if a.scaling:
        scaling_table(a.scaling)
