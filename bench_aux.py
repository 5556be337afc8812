#!/usr/bin/env python3
"""bench_aux.py — auxiliary measurements: the proving scan (BASELINE
config 4 shape) and batched verification (config 5 shape) on one MI355X.

Prints one JSON line per measurement.  Run on a GPU box:
    python bench_aux.py [--scan-labels LOG2] [--verify-proofs N]

These are recorded alongside the main bench (profiles/aux_rNN.json); the
headline metric stays bench.py's labels/s.
"""
import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

import gsm_amd  # noqa: E402

NODE = bytes([0xA5]) * 32
ATX = bytes([0x5A]) * 32
MAINNET_POW_DIFF = bytes.fromhex(
    "000dfb23b0979b4b000000000000000000000000000000000000000000000000")


def bench_scan(log2_labels: int):
    """Proving index scan: 288 nonces / 144 AES-128 ciphers per 16-B label
    (config 4 shape; label count shrunk from 2^34, per-byte work identical).
    Includes k2pow (host) and H2D chunk uploads, as post_prove does."""
    total = 1 << log2_labels
    cfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=total,
                             k1=26, k2=37,
                             pow_difficulty=MAINNET_POW_DIFF)
    opts = gsm_amd.PostSetupOpts(num_units=1, scrypt_n=8192)
    mgr = gsm_amd.PostSetupManager(NODE, ATX, cfg, opts)
    mgr.prepare_initializer()
    t0 = time.perf_counter()
    mgr.start_session()
    init_s = time.perf_counter() - t0
    labels = mgr.copy_labels(0, total)
    mgr.reset()

    t0 = time.perf_counter()
    proof = gsm_amd.api.prove_buffer(labels, total, NODE, ATX, bytes(32),
                                     cfg, gsm_amd.ProveOpts(nonces=288))
    scan_s = time.perf_counter() - t0
    print(json.dumps({
        "metric": "prove_scan_labels_per_sec",
        "value": round(total / scan_s, 1),
        "unit": "labels/s",
        "label_bytes_per_sec": round(total * 16 / scan_s, 1),
        "seconds": round(scan_s, 3),
        "init_seconds": round(init_s, 3),
        "config": {"labels": total, "nonces": 288, "k1": 26, "k2": 37,
                   "pow": "blake3-mode, mainnet difficulty",
                   "includes": "k2pow host search + H2D chunk uploads"},
        "nonce": proof.nonce,
    }))
    return labels, cfg, proof


def bench_verify(n_proofs: int):
    """Batched verification (config 5 shape): label space 2^20 so proofs
    generate quickly, but label recompute at mainnet scryptN=8192 — the
    dominant verification cost."""
    total = 1 << 20
    cfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=total,
                             k1=26, k2=37, k3=1,
                             pow_difficulty=MAINNET_POW_DIFF)
    opts = gsm_amd.PostSetupOpts(num_units=1, scrypt_n=8192)
    mgr = gsm_amd.PostSetupManager(NODE, ATX, cfg, opts)
    mgr.prepare_initializer()
    mgr.start_session()
    labels = mgr.copy_labels(0, total)
    mgr.reset()

    base_proofs = []
    for c in range(4):
        challenge = bytes([c]) * 32
        base_proofs.append((challenge, gsm_amd.api.prove_buffer(
            labels, total, NODE, ATX, challenge, cfg,
            gsm_amd.ProveOpts(nonces=288))))

    proofs, metas = [], []
    for i in range(n_proofs):
        ch, pr = base_proofs[i % len(base_proofs)]
        proofs.append(pr)
        metas.append(gsm_amd.PostProofMetadata(NODE, ATX, ch, 1, total))

    for k3, seed in [(1, b"peer-seed"), (37, None)]:
        vcfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=total,
                                  k1=26, k2=37, k3=k3,
                                  pow_difficulty=MAINNET_POW_DIFF)
        ver = gsm_amd.PostVerifier(vcfg, scrypt_n=8192)
        vopts = gsm_amd.VerifyOpts(subset_seed=seed)
        # warmup at FULL batch size: the cached workspace reserves its
        # buffers here (a first large hipMalloc costs seconds on some
        # boxes and must not land in the timed region)
        ver.verify_batch(proofs, metas, vopts)
        t0 = time.perf_counter()
        res = ver.verify_batch(proofs, metas, vopts)
        dt = time.perf_counter() - t0
        ok = sum(1 for s, _ in res if s == gsm_amd.api.Status.OK)
        assert ok == n_proofs, f"{ok}/{n_proofs} verified"
        # ABI-boundary rate: marshal once untimed (a cgo shim holds these
        # layouts natively; the ctypes build is Python-mirror overhead)
        mar = ver.marshal_batch(proofs, metas)
        t0 = time.perf_counter()
        res2 = ver.verify_batch(proofs, metas, vopts, marshalled=mar)
        dt_abi = time.perf_counter() - t0
        assert sum(1 for s, _ in res2
                   if s == gsm_amd.api.Status.OK) == n_proofs
        print(json.dumps({
            "metric": "verify_proofs_per_sec",
            "value": round(n_proofs / dt, 1),
            "unit": "proofs/s",
            "seconds": round(dt, 3),
            "abi_boundary_proofs_per_sec": round(n_proofs / dt_abi, 1),
            "config": {"proofs": n_proofs, "k3": k3, "k2": 37,
                       "scrypt_n": 8192,
                       "mode": "full-K2" if k3 >= 37 else f"K3={k3} subset",
                       "batched": True},
        }))

    # worker-pool shape: one proof per call (the reference's inner-verifier
    # call pattern, post_verifier.go:150-160)
    ver = gsm_amd.PostVerifier(
        gsm_amd.PostConfig(min_num_units=1, labels_per_unit=total, k1=26,
                           k2=37, k3=1, pow_difficulty=MAINNET_POW_DIFF),
        scrypt_n=8192)
    vopts = gsm_amd.VerifyOpts(subset_seed=b"peer-seed")
    n_single = min(64, n_proofs)
    t0 = time.perf_counter()
    for i in range(n_single):
        ver.verify(proofs[i], metas[i], vopts)
    dt = time.perf_counter() - t0
    print(json.dumps({
        "metric": "verify_proofs_per_sec_single_call",
        "value": round(n_single / dt, 1),
        "unit": "proofs/s",
        "config": {"proofs": n_single, "k3": 1, "batched": False},
    }))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--scan-labels", type=int, default=23)  # 2^23 = 128 MiB
    ap.add_argument("--verify-proofs", type=int, default=1000)
    args = ap.parse_args()
    import torch
    assert torch.cuda.is_available(), "needs a GPU"
    bench_scan(args.scan_labels)
    bench_verify(args.verify_proofs)


if __name__ == "__main__":
    main()
