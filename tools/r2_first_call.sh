#!/usr/bin/env bash
# Round-2 first gpurun call: answer every measurement question round 1
# left open, in one box visit (~8 min).  Run from the repo root ON the
# GPU box:   /usr/local/graft/bin/gpurun --timeout 720 -- 'bash tools/r2_first_call.sh'
# Build the variant .so files HERE first (they travel with the snapshot):
#   make && make scan4
#   hipcc --offload-arch=gfx950 -O3 -std=c++17 -fPIC -Wall -DPOSTE_SCAN_GLOBAL_TT=1 \
#     -shared go-spacemesh_amd/csrc/{kernels.hip,engine.cpp,crypto_host.cpp} \
#     -o go-spacemesh_amd/libpost_hip_gtt.so
#   hipcc --offload-arch=gfx950 -O3 -std=c++17 -fPIC -Wall -DPOSTE_SCAN_SPLIT_TT=1 \
#     -shared go-spacemesh_amd/csrc/{kernels.hip,engine.cpp,crypto_host.cpp} \
#     -o go-spacemesh_amd/libpost_hip_split.so
#   hipcc --offload-arch=gfx950 -O3 -std=c++17 go-spacemesh_amd/csrc/probe_romix.hip -o probe_romix
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
OUT="$R/gpurun_out"
mkdir -p "$OUT"
cd "$R"

echo "== 1. parity gate on the current tree (must stay green)"
timeout -k 10 300 python -m pytest tests -m gpu -x -q 2>&1 | tail -3

echo "== 2. gather8: is a 128-B request pattern faster than the 64-B one?"
#   gather4 7.7 TB/s was round 1's design basis; if gather8 is >=15%
#   faster, the oct-cooperative ROMix redesign (ROUND2.md item 3) is live.
timeout -k 10 120 ./probe_romix 8192 768 2>&1 | tee "$OUT/r2_probe.log" | grep gather

echo "== 3. scan T-table variants: can the L1 bypass the LDS-conflict wall?"
#   Round 1's GLOBAL_TT attempt faulted ~4 s in (before the scan kernel
#   ran) — likely a box artifact; this re-test decides.  Each run prints
#   nonce=74 on success (parity witness at these fixed inputs).
for pair in default:libpost_hip.so gtt:libpost_hip_gtt.so split:libpost_hip_split.so; do
  name=${pair%%:*}; lib=${pair##*:}
  [ -f "go-spacemesh_amd/$lib" ] || { echo "$name: lib missing, skipped"; continue; }
  POST_ENGINE_LIB="$R/go-spacemesh_amd/$lib" \
    timeout -k 10 120 python -c "import sys; sys.path.insert(0,'$R'); from bench_aux import bench_scan; bench_scan(24)" \
    > "$OUT/r2_scan_$name.log" 2>&1
  echo "-- $name rc=$? $(grep -o '\"value\": [0-9.]*' "$OUT/r2_scan_$name.log" | head -1)"
done

echo "== 4. kernel-only rates for the same three variants (rocprofv3)"
cd /tmp && export TMPDIR=/tmp
for name in default gtt split; do
  lib=libpost_hip_${name}.so; [ "$name" = default ] && lib=libpost_hip.so
  [ -f "$R/go-spacemesh_amd/$lib" ] || continue
  mkdir -p "$OUT/r2_db_$name"
  POST_ENGINE_LIB="$R/go-spacemesh_amd/$lib" \
    timeout -k 10 120 rocprofv3 --kernel-trace --stats -d "$OUT/r2_db_$name" -- \
    python -c "import sys; sys.path.insert(0,'$R'); from bench_aux import bench_scan; bench_scan(24)" \
    >> "$OUT/r2_scan_$name.log" 2>&1
  echo "-- $name profiled rc=$?"
done
echo "done — query the r2_db_*/runc/*.db files locally (sqlite3 via python)"
