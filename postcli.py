#!/usr/bin/env python3
"""postcli — operator CLI over the MI355X POST engine.

Plays the role of the post-rs tooling around the reference node: data
initialization into a postdata directory, proof generation over it (the
post-service GenProof role, api/grpcserver/post_client.go:69-143), proof
verification, provider listing and benchmarking
(activation/post_supervisor.go:105-127).

    python postcli.py providers
    python postcli.py benchmark [--scrypt-n 8192]
    python postcli.py init --datadir DIR --node-id HEX32 --atx-id HEX32 \
        --num-units U [--labels-per-unit L] [--scrypt-n N] \
        [--max-file-size B] [--shard R/W]
    python postcli.py prove --datadir DIR --challenge HEX32 [--nonces 288]
    python postcli.py verify --datadir DIR --proof proof.json \
        [--k3 K] [--seed HEX]

init resumes automatically from existing postdata_*.bin
(activation/post.go:267-271); --shard R/W initializes only rank R of W
contiguous index-range shards (the multi-GPU axis).
"""
import argparse
import base64
import json
import sys
import threading
import time

sys.path.insert(0, __file__.rsplit("/", 1)[0])

import gsm_amd  # noqa: E402
from gsm_amd import wire  # noqa: E402


def cmd_providers(_args):
    provs = list(gsm_amd.Engine().providers())
    for p in provs:
        print(json.dumps(p))
    if not provs:
        print("no MI355X providers found (no GPU visible)", file=sys.stderr)
        return 1
    return 0


def cmd_benchmark(args):
    lps = gsm_amd.Engine().benchmark(args.provider, args.scrypt_n)
    print(json.dumps({"provider": args.provider, "scrypt_n": args.scrypt_n,
                      "labels_per_sec": lps}))


def _mk_cfg(args):
    if getattr(args, "preset", None):
        import importlib
        presets = importlib.import_module("go-spacemesh_amd.presets")
        cfg, popts = presets.get(args.preset)
        args.labels_per_unit = cfg.labels_per_unit
        args.scrypt_n = popts.scrypt_n
        if args.num_units is None:
            args.num_units = popts.num_units
    else:
        cfg = gsm_amd.PostConfig(labels_per_unit=args.labels_per_unit,
                                 min_num_units=1)
    if getattr(args, "pow_difficulty", None):
        cfg.pow_difficulty = bytes.fromhex(args.pow_difficulty)
    return cfg


def cmd_init(args):
    node = bytes.fromhex(args.node_id)
    atx = bytes.fromhex(args.atx_id)
    cfg = _mk_cfg(args)
    if args.num_units is None:
        raise SystemExit("either --num-units or --preset is required")
    start = end = 0
    if args.shard:
        import importlib
        sharding = importlib.import_module("go-spacemesh_amd.sharding")
        r, w = (int(x) for x in args.shard.split("/"))
        total = args.num_units * args.labels_per_unit
        start, end = sharding.shard_range(total, w, r)
    opts = gsm_amd.PostSetupOpts(
        data_dir=args.datadir, num_units=args.num_units,
        max_file_size=args.max_file_size, provider_id=args.provider,
        scrypt_n=args.scrypt_n, index_start=start, index_end=end)
    mgr = gsm_amd.PostSetupManager(node, atx, cfg, opts)
    mgr.prepare_initializer()
    st = mgr.status()
    total = (end or args.num_units * args.labels_per_unit) - start
    print(f"resuming at {st['num_labels_written']}/{total} labels",
          file=sys.stderr)

    stop = threading.Event()

    def progress():
        while not stop.wait(5):
            s = mgr.status()
            print(f"  {s['num_labels_written']}/{total} labels",
                  file=sys.stderr)

    t = threading.Thread(target=progress, daemon=True)
    t.start()
    resumed_at = st["num_labels_written"]
    t0 = time.time()
    mgr.start_session()
    stop.set()
    dt = time.time() - t0
    done = total - resumed_at
    nonce = mgr.vrf_nonce()
    print(json.dumps({
        "labels": total, "labels_this_session": done,
        "seconds": round(dt, 1),
        "labels_per_sec": round(done / dt, 1) if dt > 0 and done else None,
        "vrf_nonce": nonce[0] if nonce else None,
    }))
    mgr.reset()


def cmd_prove(args):
    md = wire.PostMetadata.read(args.datadir)
    cfg = gsm_amd.PostConfig(labels_per_unit=md.labels_per_unit,
                             min_num_units=1)
    if args.pow_difficulty:
        cfg.pow_difficulty = bytes.fromhex(args.pow_difficulty)
    t0 = time.time()
    proof = gsm_amd.api.prove_dir(args.datadir,
                                  bytes.fromhex(args.challenge), cfg,
                                  gsm_amd.ProveOpts(nonces=args.nonces))
    out = {"nonce": proof.nonce,
           "indices": base64.b64encode(proof.indices).decode(),
           "pow": proof.pow,
           "challenge": args.challenge,
           "seconds": round(time.time() - t0, 2),
           "scale_encoded_postv1": wire.PostV1(
               proof.nonce, proof.indices, proof.pow).encode().hex()}
    print(json.dumps(out))


def cmd_verify(args):
    md = wire.PostMetadata.read(args.datadir)
    with open(args.proof) as f:
        pj = json.load(f)
    proof = gsm_amd.PostProof(pj["nonce"],
                              base64.b64decode(pj["indices"]), pj["pow"])
    cfg = gsm_amd.PostConfig(labels_per_unit=md.labels_per_unit,
                             min_num_units=1, k3=args.k3)
    if args.pow_difficulty:
        cfg.pow_difficulty = bytes.fromhex(args.pow_difficulty)
    ver = gsm_amd.PostVerifier(cfg, scrypt_n=md.scrypt_n)
    meta = gsm_amd.PostProofMetadata(
        md.node_id, md.commitment_atx_id, bytes.fromhex(pj["challenge"]),
        md.num_units, md.labels_per_unit)
    seed = bytes.fromhex(args.seed) if args.seed else None
    try:
        ver.verify(proof, meta, gsm_amd.VerifyOpts(subset_seed=seed))
        print(json.dumps({"valid": True}))
    except gsm_amd.EngineError as e:
        print(json.dumps({"valid": False, "error": str(e)}))
        sys.exit(1)


def main():
    ap = argparse.ArgumentParser()
    sub = ap.add_subparsers(dest="cmd", required=True)
    sub.add_parser("providers").set_defaults(fn=cmd_providers)
    b = sub.add_parser("benchmark")
    b.add_argument("--provider", type=int, default=0)
    b.add_argument("--scrypt-n", type=int, default=8192)
    b.set_defaults(fn=cmd_benchmark)
    i = sub.add_parser("init")
    i.add_argument("--datadir", required=True)
    i.add_argument("--node-id", required=True)
    i.add_argument("--atx-id", required=True)
    i.add_argument("--num-units", type=int, default=None)
    i.add_argument("--preset", default=None,
                   help="mainnet|testnet|fastnet parameter preset")
    i.add_argument("--labels-per-unit", type=int, default=4294967296)
    i.add_argument("--scrypt-n", type=int, default=8192)
    i.add_argument("--max-file-size", type=int, default=4294967296)
    i.add_argument("--provider", type=int, default=0)
    i.add_argument("--shard", default=None, help="R/W index-range shard")
    i.add_argument("--pow-difficulty", default=None)
    i.set_defaults(fn=cmd_init)
    p = sub.add_parser("prove")
    p.add_argument("--datadir", required=True)
    p.add_argument("--challenge", required=True)
    p.add_argument("--nonces", type=int, default=288)
    p.add_argument("--pow-difficulty", default=None)
    p.set_defaults(fn=cmd_prove)
    v = sub.add_parser("verify")
    v.add_argument("--datadir", required=True)
    v.add_argument("--proof", required=True)
    v.add_argument("--k3", type=int, default=37)
    v.add_argument("--seed", default=None)
    v.add_argument("--pow-difficulty", default=None)
    v.set_defaults(fn=cmd_verify)
    args = ap.parse_args()
    sys.exit(args.fn(args))


if __name__ == "__main__":
    main()
