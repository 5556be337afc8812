#!/usr/bin/env bash
# Round-2 GPU call 4: scan kernel — occupancy (512-thread WG) and
# batched-gather scheduling A/B, then pick the default.
# Build locally: make && make scan4
# Run: /usr/local/graft/bin/gpurun --timeout 600 -- 'bash tools/r2_scan_round3.sh'
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
OUT="$R/gpurun_out"
mkdir -p "$OUT"
cd /tmp && export TMPDIR=/tmp

run_one() { # name lib mode
  local name=$1 lib=$2 mode=$3
  [ -f "$R/go-spacemesh_amd/$lib" ] || { echo "-- $name: $lib missing"; return; }
  mkdir -p "$OUT/r2d_db_$name"
  POST_ENGINE_LIB="$R/go-spacemesh_amd/$lib" POST_SCAN_MODE=$mode \
    timeout -k 10 180 rocprofv3 --kernel-trace --stats -d "$OUT/r2d_db_$name" -- \
    python -c "import sys; sys.path.insert(0,'$R'); from bench_aux import bench_scan; bench_scan(24)" \
    > "$OUT/r2d_scan_$name.log" 2>&1
  echo "-- $name rc=$? $(grep -o '\"nonce\": [0-9]*' "$OUT/r2d_scan_$name.log" | head -1)"
}
run_one b2        libpost_hip.so       bankrep
run_one b2_512    libpost_hip.so       bankrep512
run_one b2_bg     libpost_hip.so       bankrepbg
run_one b2_512bg  libpost_hip.so       bankrep512bg
run_one b4_512    libpost_hip_scan4.so bankrep512
run_one b4_512bg  libpost_hip_scan4.so bankrep512bg
echo done
