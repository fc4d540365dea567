#!/usr/bin/env python3
"""Aggregate a rocprofv3 --pmc counter_collection.csv into a per-kernel
markdown table (mean counter value per dispatch, top kernels by calls)."""
import csv
import sys
from collections import defaultdict

src, out = sys.argv[1], sys.argv[2]

# rows: per dispatch x counter
vals = defaultdict(lambda: defaultdict(list))  # kernel -> counter -> [v]
with open(src) as f:
    r = csv.DictReader(f)
    kcol = next(c for c in r.fieldnames if "Kernel_Name" in c)
    ccol = next(c for c in r.fieldnames if "Counter_Name" in c)
    vcol = next(c for c in r.fieldnames if "Counter_Value" in c)
    for row in r:
        k = row[kcol]
        k = k.replace("void (anonymous namespace)::", "")
        k = k.replace("(anonymous namespace)::", "")
        k = k.split("(")[0][:56]
        try:
            vals[k][row[ccol]].append(float(row[vcol]))
        except ValueError:
            pass

counters = sorted({c for k in vals for c in vals[k]})
rows = []
for k, cs in vals.items():
    n = max(len(v) for v in cs.values())
    rows.append((n, k, {c: sum(v) / len(v) for c, v in cs.items()}))
rows.sort(reverse=True)

with open(out, "w") as f:
    f.write("# PMC counters (rocprofv3 --pmc), per-kernel dispatch means\n\n")
    f.write("| kernel | calls | " + " | ".join(counters) + " |\n")
    f.write("|---|--|" + "--|" * len(counters) + "\n")
    for n, k, cs in rows[:25]:
        f.write(f"| `{k}` | {n} | " +
                " | ".join(f"{cs.get(c, float('nan')):.1f}" for c in counters)
                + " |\n")
print("wrote", out)
