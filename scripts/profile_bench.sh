#!/bin/bash
# Profile a bench config with rocprofv3 and keep ONLY the small stats
# summaries (kernel traces are huge and blow the gpurun copy-back cap).
# Usage: scripts/profile_bench.sh <tag> <bench args...>
set -u
TAG=$1; shift
ROOT=${GRAFT_REPO_ROOT:-$(cd "$(dirname "$0")/.." && pwd)}
OUT=$ROOT/gpurun_out/prof_$TAG
mkdir -p "$OUT"
export TMPDIR=/tmp
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d "$OUT" -- \
    python "$ROOT/bench.py" "$@" > "$OUT/run.log" 2>&1
rc=$?
echo "rocprof rc=$rc"
# keep stats csv, drop raw traces / dbs
find "$OUT" -type f ! -name "*stats*" ! -name "run.log" -delete
find "$OUT" -type d -empty -delete 2>/dev/null
for f in $(find "$OUT" -name "*kernel_stats*"); do
  echo "== $f (top 25 by total duration) =="
  head -1 "$f"
  tail -n +2 "$f" | sort -t, -k4 -rn | head -25
done
