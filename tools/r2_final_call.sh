#!/usr/bin/env bash
# Round-2 final GPU call: full suite + headline bench + rocprof stats + aux.
# Run: /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash tools/r2_final_call.sh'
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
OUT="$R/gpurun_out"
mkdir -p "$OUT"
cd "$R"

echo "== 1. full GPU suite"
timeout -k 10 600 python -m pytest tests -m gpu -q 2>&1 | tail -3

echo "== 2. headline bench (driver contract, default flags)"
timeout -k 10 600 python bench.py --steps 8 --warmup 2 \
  2> "$OUT/r2_bench.err" | tee "$OUT/r2_bench.json"

echo "== 3. rocprof kernel stats over a short bench (cross-check)"
cd /tmp && export TMPDIR=/tmp
mkdir -p "$OUT/r2_bench_db"
timeout -k 10 420 rocprofv3 --kernel-trace --stats -d "$OUT/r2_bench_db" -- \
  python "$R/bench.py" --steps 4 --warmup 1 > "$OUT/r2_bench_prof.json" 2>&1
echo "prof rc=$?"
cd "$R"

echo "== 4. aux benches (warmed: scan e2e + verify batched/single)"
timeout -k 10 600 python bench_aux.py --scan-labels 24 --verify-proofs 10000 \
  > "$OUT/r2_aux.json" 2>&1
tail -6 "$OUT/r2_aux.json"
echo done
