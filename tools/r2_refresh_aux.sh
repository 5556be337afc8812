#!/usr/bin/env bash
# Round-2: full suite re-validation after api/verify changes + aux refresh.
# Run: /usr/local/graft/bin/gpurun --timeout 900 -- 'bash tools/r2_refresh_aux.sh'
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
OUT="$R/gpurun_out"
mkdir -p "$OUT"
cd "$R"
echo "== 1. full GPU suite"
timeout -k 10 600 python -m pytest tests -m gpu -q 2>&1 | tail -2
echo "== 2. aux refresh (with ABI-boundary verify rates)"
timeout -k 10 600 python bench_aux.py --scan-labels 24 --verify-proofs 10000 \
  > "$OUT/r2_aux2.json" 2>&1
cat "$OUT/r2_aux2.json" | tail -5
echo done
