#!/usr/bin/env python3
"""bench.py — POST labeling throughput on MI355X (BASELINE.json metric).

A "step" is one pass of the init hot path over one batch of synthetic input:
STEP_LABELS labels of the mainnet configuration (scryptN=8192, 4 units of
2^32 labels per unit = 256 GiB total space) streamed to the in-memory sink —
BASELINE config 2, the largest single-GPU configuration (config 1 is the
reference's CPU-runnable case; configs 3-5 are covered by SCALE runs and the
parity/verify tests).

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W]
N>1 is launched by the driver via torch.distributed.run, one rank per GPU
over RCCL; ranks shard the label index space (weak scaling: per-GPU labels
fixed) and the only exchange is the final VRF-nonce min-reduce.

Rank 0 prints ONE JSON line: whole-job labels/s over all ranks, kernel-level
roofline of the dominant kernel (HIP events on the engine stream), and the
CPU-oracle baseline timed on this box's host cores.
"""
import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

STEP_LABELS = 1 << 21          # labels per step (~0.5-1 s/step on target)
SCRYPT_N = 8192                # mainnet (activation/post.go:155)
NUM_UNITS = 4                  # mainnet minimum (config/mainnet.go:184)
LABELS_PER_UNIT = 1 << 32      # config/mainnet.go:186
HBM_PEAK_GBS = 8000.0          # gfx950 spec peak (MI355X_MICROARCH.md)

# Algorithmic HBM bytes per label at lookup-gap 1:
# ROMix writes N blocks of 128 B once and reads N blocks of 128 B once
# (random), the working block stages through xbuf twice each way (512 B),
# plus the 16-B label store the metric names.
BYTES_PER_LABEL = 128 * SCRYPT_N * 2 + 512 + 16


def cpu_baseline():
    """Time the CPU oracle (the reference CPU provider's role) on this
    box's host cores over a bounded sample."""
    if os.environ.get("POST_SKIP_CPU_BASELINE"):
        return {"value": None, "unit": "labels/s", "cores": 0,
                "kind": "port", "sample": "skipped (A/B run)"}
    try:
        import multiprocessing
        cores = multiprocessing.cpu_count()
        oracle_dir = os.path.join(REPO, "oracle")
        bench_bin = os.path.join(oracle_dir, "oracle_bench")
        if not os.path.exists(bench_bin):
            subprocess.run(["make", "-C", oracle_dir], check=True,
                           capture_output=True, timeout=300)
        sample = max(2048, 1024 * cores)  # ~5-15 s wall on all cores
        out = subprocess.run(
            [bench_bin, "bench", "--labels", str(sample), "--scrypt-n",
             str(SCRYPT_N)],
            capture_output=True, text=True, timeout=600, check=True)
        r = json.loads(out.stdout.strip())
        return {"value": r["labels_per_sec"], "unit": "labels/s",
                "cores": r["threads"], "kind": "port",
                "sample": f"{sample} labels at scryptN={SCRYPT_N}, "
                          f"OpenMP {r['threads']} threads "
                          f"({r['seconds']:.1f}s)"}
    except Exception as e:  # noqa: BLE001
        return {"value": None, "unit": "labels/s", "cores": 0,
                "kind": "port", "sample": f"failed: {e}"}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=2)
    args = ap.parse_args()

    import torch
    if not torch.cuda.is_available():
        print(json.dumps({"error": "no GPU: bench.py measures the HIP "
                                   "engine and has no CPU fallback"}))
        sys.exit(1)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    # RCCL in production; POST_BENCH_BACKEND=gloo lets the world>1 rank
    # logic run with several ranks sharing one GPU (functional testing)
    backend = os.environ.get("POST_BENCH_BACKEND", "nccl")
    device_id = local_rank % torch.cuda.device_count()
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        torch.cuda.set_device(device_id)
        dist.init_process_group(backend)

    # 8 ranks/node each burn a host core on the per-batch reference-label
    # scrypt; stretch the self-check period so the checks don't contend
    # (coverage semantics unchanged, engine.cpp self-check)
    if world > 1:
        os.environ.setdefault("POST_SELFCHECK_PERIOD", "8")

    import gsm_amd
    from importlib import import_module
    sharding = import_module("go-spacemesh_amd.sharding")

    total_labels = NUM_UNITS * LABELS_PER_UNIT
    start, end = sharding.shard_range(total_labels, world, rank)

    cfg = gsm_amd.PostConfig()  # mainnet params
    opts = gsm_amd.PostSetupOpts(
        num_units=NUM_UNITS, scrypt_n=SCRYPT_N, provider_id=device_id,
        index_start=start, index_end=end, data_dir=None,
        scratch_bytes=int(os.environ.get("POST_BENCH_SCRATCH", "0")))
    mgr = gsm_amd.PostSetupManager(bytes([0xA5]) * 32, bytes([0x5A]) * 32,
                                   cfg, opts)
    mgr.prepare_initializer()

    for _ in range(args.warmup):
        done, _ = mgr.step(STEP_LABELS)
        assert done == STEP_LABELS

    if dist:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    kernel_ms_total = 0.0
    for _ in range(args.steps):
        done, kms = mgr.step(STEP_LABELS)
        assert done == STEP_LABELS, "ran out of shard range"
        kernel_ms_total += kms
    torch.cuda.synchronize()
    if dist:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # whole-job time = MAX over ranks
    if dist:
        dev = "cuda" if backend == "nccl" else "cpu"
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # the one cross-GPU exchange of this path: VRF-nonce min-reduce
    local_nonce = mgr.vrf_nonce()
    global_nonce = sharding.allreduce_nonce(
        (local_nonce[0], local_nonce[1]) if local_nonce else None) \
        if dist else local_nonce

    labels_total = args.steps * STEP_LABELS * world
    value = labels_total / elapsed

    if rank == 0:
        kern_s = kernel_ms_total / 1e3
        achieved_gbs = (args.steps * STEP_LABELS * BYTES_PER_LABEL /
                        kern_s / 1e9) if kern_s > 0 else None
        # traffic: HBM bytes per launch from PMC counters.  Counters cannot
        # be collected inside a normal run (separate rocprofv3 --pmc passes,
        # MI355X_MICROARCH.md §HBM), so the bench line carries the committed
        # counter measurement of this same kernel+workload
        # (profiles/roofline_traffic.json, provenance inside), overridable
        # with POST_ROOFLINE_TRAFFIC_BYTES_PER_LAUNCH for fresh A/B runs.
        traffic_env = os.environ.get("POST_ROOFLINE_TRAFFIC_BYTES_PER_LAUNCH")
        traffic = float(traffic_env) if traffic_env else None
        traffic_src = "env" if traffic_env else None
        if traffic is None:
            try:
                rec = json.load(open(os.path.join(
                    REPO, "profiles", "roofline_traffic.json")))
                traffic = float(rec["traffic_bytes_per_launch"])
                traffic_src = (f"pmc-counters ({rec['source']}, "
                               f"{rec['labels_per_launch']} labels/launch)")
            except Exception:  # noqa: BLE001
                pass
        roofline = {
            "bound": "hbm",
            "achieved": round(achieved_gbs, 1) if achieved_gbs else None,
            "peak": HBM_PEAK_GBS,
            "unit": "GB/s",
            "frac": round(achieved_gbs / HBM_PEAK_GBS, 4)
            if achieved_gbs else None,
            "traffic": traffic,
            "traffic_source": traffic_src,
        }
        result = {
            "metric": "post_labels_per_sec",
            "value": round(value, 1),
            "unit": "labels/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # no published reference number
            "dtype": "u32",
            "data": "synthetic",
            "config": {
                "workload": "init-mainnet-scryptN8192-4units-256GiB-sink",
                "step_labels": STEP_LABELS,
                "scrypt_n": SCRYPT_N,
                "num_units": NUM_UNITS,
                "labels_per_unit": LABELS_PER_UNIT,
                "parallelism": f"index-range shards x{world}, RCCL nonce "
                               f"min-reduce",
                "vrf_nonce_found": bool(global_nonce),
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline(),
            "hbm_write_fraction_16B": round(
                value / world * 16 / (HBM_PEAK_GBS * 1e9), 6),
        }
        print(json.dumps(result))

    mgr.reset()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
