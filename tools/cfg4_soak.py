#!/usr/bin/env python3
"""cfg4 soak: disk-backed init (kill+resume) + prove over >=64 GiB of
postdata files on one MI355X (BASELINE configs 3-4 at real scale).

The init runs at --scrypt-n (default 32) so the soak exercises the
file-writer / resume / prove IO paths at full cfg3/4 data scale without
spending ~35 GPU-minutes of mainnet-N labeling — the mainnet-N kernel
rate is bench.py's separately-measured headline, and neither the file
split, the resume scan, the k2pow, the AES index scan nor the proof
assembly depends on N (the scan is over label bytes).
Pass --scrypt-n 8192 for the gold-plated version when GPU budget allows.

Prints one JSON line per phase.  Usage (on the GPU box):
    python tools/cfg4_soak.py --gib 64 --dir /tmp/cfg4data
"""
import argparse
import json
import os
import random
import shutil
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))  # oracle/oracle.py as module

import gsm_amd  # noqa: E402
from oracle import Oracle  # noqa: E402  (checker only: parity spot-checks)

NODE = bytes([0xA5]) * 32
ATX = bytes([0x5A]) * 32
CHALLENGE = bytes([0x07]) * 32
MAINNET_POW_DIFF = bytes.fromhex(
    "000dfb23b0979b4b000000000000000000000000000000000000000000000000")


def emit(**kw):
    print(json.dumps(kw), flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gib", type=int, default=64)
    ap.add_argument("--dir", default="/tmp/cfg4data")
    ap.add_argument("--scrypt-n", type=int, default=32)
    ap.add_argument("--max-file-gib", type=int, default=2)
    ap.add_argument("--keep", action="store_true")
    args = ap.parse_args()

    total_labels = args.gib << 26           # GiB * 2^30 / 16 B per label
    num_units = 4
    lpu = total_labels // num_units
    mfs = args.max_file_gib << 30
    cfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=lpu,
                             k1=26, k2=37,
                             pow_difficulty=MAINNET_POW_DIFF)

    def mgr():
        return gsm_amd.PostSetupManager(
            NODE, ATX, cfg,
            gsm_amd.PostSetupOpts(num_units=num_units,
                                  scrypt_n=args.scrypt_n,
                                  data_dir=args.dir, max_file_size=mfs))

    # ---- phase 1: init to ~50%, then kill (reset mid-session) ----
    m = mgr()
    m.prepare_initializer()
    half = total_labels // 2
    t0 = time.perf_counter()
    written = 0
    while written < half:
        done, _ = m.step(1 << 24)
        assert done > 0, "init stalled"
        written = m.status()["num_labels_written"]
    t1 = time.perf_counter()
    nonce_before = m.vrf_nonce()
    m.reset()
    emit(phase="init-half-then-kill", labels=written,
         gib=round(written * 16 / 2**30, 1), seconds=round(t1 - t0, 1),
         write_gib_per_sec=round(written * 16 / 2**30 / (t1 - t0), 2),
         labels_per_sec=round(written / (t1 - t0), 1),
         scrypt_n=args.scrypt_n, nonce_found=bool(nonce_before))

    # ---- phase 2: resume and complete ----
    m = mgr()
    m.prepare_initializer()
    st = m.status()
    resume_at = st["num_labels_written"]
    assert resume_at >= written - (1 << 24), (resume_at, written)
    nonce_resumed = m.vrf_nonce()
    assert nonce_resumed == nonce_before, "persisted nonce lost on resume"
    t0 = time.perf_counter()
    m.start_session()
    t1 = time.perf_counter()
    final = m.status()["num_labels_written"]
    nonce_final = m.vrf_nonce()
    m.reset()
    remaining = total_labels - resume_at
    emit(phase="resume-complete", resumed_at=resume_at, labels=final,
         gib=round(final * 16 / 2**30, 1), seconds=round(t1 - t0, 1),
         write_gib_per_sec=round(remaining * 16 / 2**30 / (t1 - t0), 2),
         nonce_preserved=True, nonce_found=bool(nonce_final))
    assert final == total_labels

    # ---- phase 3: spot parity of disk content vs oracle ----
    o = Oracle()
    commit = o.commitment(NODE, ATX)
    per_file = mfs // 16
    rng = random.Random(7)
    checked = 0
    for _ in range(6):
        idx = rng.randrange(total_labels - 256)
        want, _ = o.init_range(commit, idx, idx + 256, args.scrypt_n)
        fi, off = divmod(idx, per_file)
        with open(os.path.join(args.dir, f"postdata_{fi}.bin"), "rb") as f:
            f.seek(off * 16)
            got = f.read(256 * 16)
        if len(got) < 256 * 16:  # window straddles a file boundary
            with open(os.path.join(args.dir, f"postdata_{fi+1}.bin"),
                      "rb") as f2:
                got += f2.read(256 * 16 - len(got))
        assert got == want, f"disk content diverges at label {idx}"
        checked += 1
    emit(phase="disk-spot-parity", windows=checked, labels_each=256,
         result="bit-exact vs oracle")

    # ---- phase 4: disk-backed prove over the full set ----
    popts = gsm_amd.ProveOpts(nonces=288)
    t0 = time.perf_counter()
    proof = gsm_amd.api.prove_dir(args.dir, CHALLENGE, cfg, popts)
    t1 = time.perf_counter()
    emit(phase="disk-prove", gib=args.gib, seconds=round(t1 - t0, 1),
         read_gib_per_sec=round(args.gib / (t1 - t0), 2),
         labels_per_sec=round(total_labels / (t1 - t0), 1),
         nonce=proof.nonce, indices_len=len(proof.indices),
         k1=26, k2=37, nonces=288, pow="blake3-mode mainnet difficulty")

    # ---- phase 5: verify the proof (GPU, full K2 + subset) ----
    meta = gsm_amd.PostProofMetadata(NODE, ATX, CHALLENGE, num_units, lpu)
    for k3, seed, mode in [(37, None, "full-K2"), (1, b"peer", "K3=1")]:
        vcfg = gsm_amd.PostConfig(min_num_units=1, labels_per_unit=lpu,
                                  k1=26, k2=37, k3=k3,
                                  pow_difficulty=MAINNET_POW_DIFF)
        ver = gsm_amd.PostVerifier(vcfg, scrypt_n=args.scrypt_n)
        t0 = time.perf_counter()
        ver.verify(proof, meta, gsm_amd.VerifyOpts(subset_seed=seed))
        emit(phase="verify", mode=mode,
             seconds=round(time.perf_counter() - t0, 3), result="accept")

    if not args.keep:
        shutil.rmtree(args.dir, ignore_errors=True)
    emit(phase="done", gib=args.gib, scrypt_n=args.scrypt_n)


if __name__ == "__main__":
    main()
