#!/usr/bin/env bash
# Round-2 GPU call 2: bank-replicated scan kernel — parity + ILP sweep.
# Build locally first:  make && make scan1 && make scan4
# Run:  /usr/local/graft/bin/gpurun --timeout 600 -- 'bash tools/r2_scan_bankrep.sh'
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
OUT="$R/gpurun_out"
mkdir -p "$OUT"
cd "$R"

echo "== 0. box facts (cfg4 soak sizing)"
df -B1G /tmp "$R" 2>/dev/null | tail -2
free -g | head -2

echo "== 1. prove-path parity with the bankrep kernel as default"
timeout -k 10 300 python -m pytest tests/test_engine_gpu.py -x -q \
  -k "proof or scan or prove or verify or frozen" 2>&1 | tail -3

echo "== 2. scan kernel sweep: bankrep ILP1/2/4 vs shared baseline (same box)"
cd /tmp && export TMPDIR=/tmp
run_one() { # name lib mode
  local name=$1 lib=$2 mode=$3
  [ -f "$R/go-spacemesh_amd/$lib" ] || { echo "-- $name: $lib missing"; return; }
  mkdir -p "$OUT/r2b_db_$name"
  POST_ENGINE_LIB="$R/go-spacemesh_amd/$lib" POST_SCAN_MODE=$mode \
    timeout -k 10 180 rocprofv3 --kernel-trace --stats -d "$OUT/r2b_db_$name" -- \
    python -c "import sys; sys.path.insert(0,'$R'); from bench_aux import bench_scan; bench_scan(24)" \
    > "$OUT/r2b_scan_$name.log" 2>&1
  echo "-- $name rc=$? nonce=$(grep -o '\"nonce\": [0-9]*' "$OUT/r2b_scan_$name.log" | head -1)"
  grep -h "post_scan" "$OUT/r2b_db_$name"/*/*_kernel_stats.csv 2>/dev/null | head -2 || true
}
run_one bankrep2 libpost_hip.so bankrep
run_one bankrep1 libpost_hip_scan1.so bankrep
run_one bankrep4 libpost_hip_scan4.so bankrep
run_one shared   libpost_hip.so shared
echo done
