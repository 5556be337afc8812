#!/usr/bin/env bash
# Round-2 GPU call: verify host-prep optimization A/B + fresh PMC traffic.
# Run: /usr/local/graft/bin/gpurun --timeout 900 -- 'bash tools/r2_verify_opt.sh'
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
OUT="$R/gpurun_out"
mkdir -p "$OUT"
cd "$R"

echo "== 1. verify-path parity gate"
timeout -k 10 300 python -m pytest tests/test_engine_gpu.py -x -q \
  -k "verify or verdicts or batch or malformed or seeds or mixed" 2>&1 | tail -2

echo "== 2. warmed verify numbers (was: 130K K3=1 / 39K full-K2 / 81.7 single)"
POST_VERIFY_DEBUG=1 timeout -k 10 300 python bench_aux.py --scan-labels 20 \
  --verify-proofs 10000 > "$OUT/r2_verify_opt.json" 2>"$OUT/r2_verify_opt.dbg"
grep verify "$OUT/r2_verify_opt.json"
echo "-- debug splits (last full-K2 batch):"
grep "\[verify\]" "$OUT/r2_verify_opt.dbg" | tail -8

echo "== 3. fresh PMC traffic on the romix kernel (refresh roofline_traffic.json)"
cd /tmp && export TMPDIR=/tmp
mkdir -p "$OUT/r2_pmc_w" "$OUT/r2_pmc_f"
timeout -k 10 300 rocprofv3 --pmc WRITE_SIZE -d "$OUT/r2_pmc_w" -- \
  python "$R/bench.py" --steps 2 --warmup 1 > "$OUT/r2_pmc_w.log" 2>&1
echo "write pass rc=$?"
timeout -k 10 300 rocprofv3 --pmc FETCH_SIZE -d "$OUT/r2_pmc_f" -- \
  python "$R/bench.py" --steps 2 --warmup 1 > "$OUT/r2_pmc_f.log" 2>&1
echo "fetch pass rc=$?"
echo done
