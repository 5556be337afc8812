#!/usr/bin/env bash
# Round-2 GPU call 3: scan kernel — rk-preload bankrep vs tt4, plus PMC.
# Build locally first: make && make scan4
# Run: /usr/local/graft/bin/gpurun --timeout 600 -- 'bash tools/r2_scan_round2.sh'
set -u
R=${GRAFT_REPO_ROOT:-/root/repo}
OUT="$R/gpurun_out"
mkdir -p "$OUT"
cd "$R"

echo "== 1. scan parity quick gate (bankrep default + tt4 mode)"
timeout -k 10 240 python -m pytest tests/test_engine_gpu.py -x -q \
  -k "proof or prove or frozen" 2>&1 | tail -2
POST_SCAN_MODE=tt4 timeout -k 10 240 python -m pytest tests/test_engine_gpu.py -x -q \
  -k "proof or prove or frozen" 2>&1 | tail -2

echo "== 2. kernel timing: bankrep(rk-preload) ilp2/ilp4, tt4 ilp2/ilp4"
cd /tmp && export TMPDIR=/tmp
run_one() { # name lib mode
  local name=$1 lib=$2 mode=$3
  [ -f "$R/go-spacemesh_amd/$lib" ] || { echo "-- $name: $lib missing"; return; }
  mkdir -p "$OUT/r2c_db_$name"
  POST_ENGINE_LIB="$R/go-spacemesh_amd/$lib" POST_SCAN_MODE=$mode \
    timeout -k 10 180 rocprofv3 --kernel-trace --stats -d "$OUT/r2c_db_$name" -- \
    python -c "import sys; sys.path.insert(0,'$R'); from bench_aux import bench_scan; bench_scan(24)" \
    > "$OUT/r2c_scan_$name.log" 2>&1
  echo "-- $name rc=$? $(grep -o '\"nonce\": [0-9]*' "$OUT/r2c_scan_$name.log" | head -1)"
}
run_one bankrep2 libpost_hip.so bankrep
run_one bankrep4 libpost_hip_scan4.so bankrep
run_one tt4i2    libpost_hip.so tt4
run_one tt4i4    libpost_hip_scan4.so tt4

echo "== 3. PMC pass on both (SQ slots: wave/wait/issue + LDS)"
pmc_one() { # name mode counters tag
  local name=$1 mode=$2 counters=$3 tag=$4
  mkdir -p "$OUT/r2c_pmc_${name}_$tag"
  POST_ENGINE_LIB="$R/go-spacemesh_amd/libpost_hip.so" POST_SCAN_MODE=$mode \
    timeout -k 10 240 rocprofv3 --pmc $counters -d "$OUT/r2c_pmc_${name}_$tag" -- \
    python -c "import sys; sys.path.insert(0,'$R'); from bench_aux import bench_scan; bench_scan(23)" \
    > "$OUT/r2c_pmc_${name}_$tag.log" 2>&1
  echo "-- pmc $name $tag rc=$?"
}
SQA="SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_WAIT_INST_LDS SQ_ACTIVE_INST_ANY SQ_INSTS_VALU SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE"
pmc_one bankrep bankrep "$SQA" sq
pmc_one tt4 tt4 "$SQA" sq
echo done
