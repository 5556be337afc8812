#!/usr/bin/env bash
# Round-2 GPU call 5: full GPU suite + cfg4-scale soak (64 GiB).
# Run: /usr/local/graft/bin/gpurun --timeout 1200 -- 'bash tools/r2_cfg4_call.sh'
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
OUT="$R/gpurun_out"
mkdir -p "$OUT"
cd "$R"

echo "== 1. full GPU test suite (incl. nonce-resume, mixed-ops, variant-agreement)"
timeout -k 10 600 python -m pytest tests -m gpu -q 2>&1 | tail -3

echo "== 2. cfg4 soak: 64 GiB file-mode init (kill+resume) + disk prove + verify"
df -B1G /tmp | tail -1
timeout -k 10 540 python tools/cfg4_soak.py --gib 64 --dir /tmp/cfg4data \
  2>&1 | tee "$OUT/r2_cfg4_soak.log" | grep -E '"phase"' || tail -5 "$OUT/r2_cfg4_soak.log"
echo "soak rc=$?"
df -B1G /tmp | tail -1
echo done
