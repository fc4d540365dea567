#!/bin/bash
# rocprofv3 wrapper (SURVEY.md §5.1 deliverable): per-kernel stats of the
# flagship bench step.  Run on a GPU box:
#   tools/profile.sh [bench args...]
# Writes the results DB + a text summary under profiles/.
# Counter runs (--pmc) must be separate from trace domains; see
# tools/pmc_counters.txt for the MFMA/LDS/HBM set.
set -e
cd "$(dirname "$0")/.."
export TMPDIR=/tmp
OUT=${PROFILE_OUT:-profiles}
mkdir -p "$OUT"
rocprofv3 --kernel-trace --stats -d /tmp/ddl_prof -o ddl \
    -- python bench.py --steps 4 --warmup 2 "$@"
cp /tmp/ddl_prof/*results.db "$OUT/" 2>/dev/null || true
python tools/report.py --profile "$OUT"/ddl_results.db --out "$OUT/kernel_stats.md" || true
echo "profile written to $OUT/"
