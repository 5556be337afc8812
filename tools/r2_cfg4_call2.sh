#!/usr/bin/env bash
# Round-2 GPU call: mixed-ops test traceback + cfg4 64 GiB soak (fixed import).
# Run: /usr/local/graft/bin/gpurun --timeout 1200 -- 'bash tools/r2_cfg4_call2.sh'
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
OUT="$R/gpurun_out"
mkdir -p "$OUT"
cd "$R"

echo "== 1. mixed-ops test with full traceback"
timeout -k 10 300 python -m pytest tests/test_engine_gpu.py::test_concurrent_mixed_ops \
  -q --tb=long 2>&1 | tail -40

echo "== 2. cfg4 soak: 64 GiB file-mode init (kill+resume) + disk prove + verify"
timeout -k 10 700 python tools/cfg4_soak.py --gib 64 --dir /tmp/cfg4data \
  > "$OUT/r2_cfg4_soak.log" 2>&1
rc=$?
grep -E '"phase"' "$OUT/r2_cfg4_soak.log" || tail -15 "$OUT/r2_cfg4_soak.log"
echo "soak rc=$rc"
echo done
