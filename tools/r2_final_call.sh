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

echo "== 5. 2-rank bench dry run on one GPU (SCALE-path validation: rank"
echo "      sharding + nonce min-reduce + two engine sessions on one device)"
POST_BENCH_BACKEND=gloo POST_BENCH_SCRATCH=$((60 * 1024 * 1024 * 1024)) \
  POST_SKIP_CPU_BASELINE=1 \
  timeout -k 10 420 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29317 \
  bench.py --gpus 2 --steps 2 --warmup 1 > "$OUT/r2_bench2rank.json" 2>&1
echo "2rank rc=$?"
grep -o '"metric[^}]*' "$OUT/r2_bench2rank.json" | head -1 || tail -5 "$OUT/r2_bench2rank.json"

echo "== 6. 8-rank dry run on one GPU (the driver's N=8 torchrun shape;"
echo "      8 engine sessions share the device at bounded scratch)"
POST_BENCH_BACKEND=gloo POST_BENCH_SCRATCH=$((16 * 1024 * 1024 * 1024)) \
  POST_SKIP_CPU_BASELINE=1 \
  timeout -k 10 420 python -m torch.distributed.run --nnodes=1 \
  --nproc-per-node 8 --master-addr 127.0.0.1 --master-port 29319 \
  bench.py --gpus 8 --steps 1 --warmup 1 > "$OUT/r2_bench8rank.json" 2>&1
echo "8rank rc=$?"
grep -o '"metric[^}]*' "$OUT/r2_bench8rank.json" | head -1 || tail -5 "$OUT/r2_bench8rank.json"
echo done
