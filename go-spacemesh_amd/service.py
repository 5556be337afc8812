"""post-service — the out-of-process prover child (the role of the post-rs
`post-service` binary, spawned by the node's supervisor with the argv
contract of activation/post_supervisor.go:228-261).

Dials the node's gRPC server and calls spacemesh.v1.PostService/Register
(the bidi stream of api/grpcserver/post_service.go:91-141); answers
NodeRequest{Metadata} with the data-dir's metadata and
NodeRequest{GenProof{challenge}} with in-progress polls until the proof is
done (post_client.go:104-109 polls every 2 s).  `--watch-pid` kills the
service when the watched process dies (post_supervisor.go:246);
`--max-retries` bounds proving retries (:248-251).

Run:  python -m go-spacemesh_amd.service is not importable (hyphen) — use
      python go-spacemesh_amd/service.py --address HOST:PORT --dir DATADIR
          [--watch-pid PID] [--max-retries N] [--threads T] [--nonces N]
          [--mock-prover]   (CPU tests only)
"""
from __future__ import annotations

import argparse
import os
import queue
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import grpc  # noqa: E402

import gsm_amd  # noqa: E402
from gsm_amd import service_proto as sp  # noqa: E402
from gsm_amd import wire  # noqa: E402


class Prover:
    """Engine-backed prover over a postdata dir.  K1/K2/pow-difficulty
    arrive via argv like the reference's post-service
    (post_supervisor.go:228-247)."""

    def __init__(self, datadir: str, nonces: int, threads: int,
                 k1: int = 26, k2: int = 37,
                 pow_difficulty: bytes = None) -> None:
        self.datadir = datadir
        self.md = wire.PostMetadata.read(datadir)
        self.nonces = nonces
        self.threads = threads
        self.k1 = k1
        self.k2 = k2
        self.pow_difficulty = pow_difficulty

    def metadata(self) -> sp.Metadata:
        return sp.Metadata(node_id=self.md.node_id,
                           commitment_atx_id=self.md.commitment_atx_id,
                           nonce=self.md.nonce,
                           num_units=self.md.num_units,
                           labels_per_unit=self.md.labels_per_unit)

    def prove(self, challenge: bytes) -> sp.Proof:
        cfg = gsm_amd.PostConfig(labels_per_unit=self.md.labels_per_unit,
                                 min_num_units=1, k1=self.k1, k2=self.k2)
        if self.pow_difficulty:
            cfg.pow_difficulty = self.pow_difficulty
        p = gsm_amd.api.prove_dir(
            self.datadir, challenge, cfg,
            gsm_amd.ProveOpts(nonces=self.nonces, threads=self.threads))
        return sp.Proof(nonce=p.nonce, indices=p.indices, pow=p.pow)


class MockProver:
    """CPU-test stand-in (no GPU): deterministic fake proof after a delay."""

    def __init__(self, datadir: str, delay: float = 0.5) -> None:
        self.md = wire.PostMetadata.read(datadir)
        self.delay = delay

    def metadata(self) -> sp.Metadata:
        return sp.Metadata(node_id=self.md.node_id,
                           commitment_atx_id=self.md.commitment_atx_id,
                           nonce=self.md.nonce,
                           num_units=self.md.num_units,
                           labels_per_unit=self.md.labels_per_unit)

    def prove(self, challenge: bytes) -> sp.Proof:
        time.sleep(self.delay)
        return sp.Proof(nonce=7, indices=challenge[:8], pow=42)


def watch_pid(pid: int) -> None:
    """Exit when the watched process dies (post_supervisor.go:246)."""
    def loop():
        while True:
            try:
                os.kill(pid, 0)
            except OSError:
                print("watched process gone; exiting", file=sys.stderr)
                os._exit(1)
            time.sleep(1.0)
    threading.Thread(target=loop, daemon=True).start()


def serve(address: str, prover, max_retries: int, stop_event=None) -> None:
    channel = grpc.insecure_channel(address)
    method = channel.stream_stream(
        sp.REGISTER_METHOD,
        request_serializer=lambda b: b,
        response_deserializer=lambda b: b)

    outq: "queue.Queue[bytes]" = queue.Queue()
    proving_state = {"thread": None, "result": None, "error": None,
                     "retries": 0}

    def requests():
        while True:
            item = outq.get()
            if item is None:
                return
            yield item

    call = method(requests())

    def start_proving(challenge: bytes):
        def run():
            try:
                proving_state["result"] = prover.prove(challenge)
            except Exception as e:  # noqa: BLE001
                proving_state["retries"] += 1
                print(f"proving failed ({e}); retry "
                      f"{proving_state['retries']}/{max_retries}",
                      file=sys.stderr)
                if proving_state["retries"] >= max_retries:
                    proving_state["error"] = str(e)
                else:
                    proving_state["thread"] = None  # retried on next poll
        t = threading.Thread(target=run, daemon=True)
        proving_state["thread"] = t
        proving_state["challenge"] = challenge
        t.start()

    try:
        for raw in call:
            if stop_event is not None and stop_event.is_set():
                break
            req = sp.NodeRequest.decode(raw)
            if req.gen_proof_challenge is None:
                outq.put(sp.ServiceResponse(
                    metadata=prover.metadata()).encode())
                continue
            # GenProof: kick off or poll (the node polls every 2 s)
            if proving_state["result"] is not None:
                outq.put(sp.ServiceResponse(
                    gen_proof_status=sp.GEN_PROOF_STATUS_OK,
                    gen_proof_proof=proving_state["result"]).encode())
                proving_state["result"] = None
                proving_state["thread"] = None
                proving_state["retries"] = 0
            elif proving_state["error"] is not None:
                outq.put(sp.ServiceResponse(
                    gen_proof_status=sp.GEN_PROOF_STATUS_ERROR).encode())
                proving_state["error"] = None
            else:
                if proving_state["thread"] is None:
                    start_proving(req.gen_proof_challenge)
                outq.put(sp.ServiceResponse(
                    gen_proof_status=sp.GEN_PROOF_STATUS_IN_PROGRESS
                ).encode())
    finally:
        outq.put(None)
        channel.close()


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--address", required=True)
    ap.add_argument("--dir", required=True)
    ap.add_argument("--watch-pid", type=int, default=0)
    ap.add_argument("--max-retries", type=int, default=3)
    ap.add_argument("--threads", type=int, default=0)
    ap.add_argument("--nonces", type=int, default=288)
    ap.add_argument("--k1", type=int, default=26)
    ap.add_argument("--k2", type=int, default=37)
    ap.add_argument("--pow-difficulty", default=None)
    ap.add_argument("--mock-prover", action="store_true")
    args = ap.parse_args()
    if args.watch_pid:
        watch_pid(args.watch_pid)
    pow_diff = bytes.fromhex(args.pow_difficulty) \
        if args.pow_difficulty else None
    prover = (MockProver(args.dir) if args.mock_prover
              else Prover(args.dir, args.nonces, args.threads,
                          args.k1, args.k2, pow_diff))
    print(f"post-service registering at {args.address}", file=sys.stderr)
    serve(args.address, prover, args.max_retries)


if __name__ == "__main__":
    main()
