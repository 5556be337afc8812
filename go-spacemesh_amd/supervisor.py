"""Node-side post-service plumbing: the gRPC server the services dial into,
per-identity clients, and the child-process supervisor.

Mirrors (file:line under /root/reference/):
  - api/grpcserver/post_service.go:24-31,91-141 — PostService.Register bidi
    stream; registry of connected services keyed by node id (multi-smesher).
  - api/grpcserver/post_client.go:36-143 — PostClient.{Info,Proof}: send
    NodeRequest over the stream, poll GenProof every 2 s until OK.
  - activation/post_supervisor.go:133-299 — spawn the service child with
    the argv contract (:228-261), re-log its stderr (:207-218), treat an
    unexpected exit as fatal (:288-298); the child watches our pid (:246).
"""
from __future__ import annotations

import os
import queue
import subprocess
import sys
import threading
import time
from concurrent import futures
from typing import Callable, Dict, Optional

import grpc

from . import service_proto as sp


class _Client:
    """One registered post-service (one bidi stream)."""

    def __init__(self, metadata: sp.Metadata) -> None:
        self.metadata = metadata
        self.cmd_q: "queue.Queue" = queue.Queue()
        self.closed = threading.Event()

    def roundtrip(self, request: sp.NodeRequest,
                  timeout: float = 30.0) -> sp.ServiceResponse:
        if self.closed.is_set():
            raise ConnectionError("post service disconnected")
        reply_q: "queue.Queue" = queue.Queue()
        self.cmd_q.put((request, reply_q))
        out = reply_q.get(timeout=timeout)
        if isinstance(out, Exception):
            raise out
        return out


class PostClient:
    """PostClient.{Info,Proof} (post_client.go:36-143)."""

    def __init__(self, client: _Client, poll_interval: float = 2.0) -> None:
        self._c = client
        self.poll_interval = poll_interval  # post_service.go:65

    def info(self) -> sp.Metadata:
        resp = self._c.roundtrip(sp.NodeRequest())
        assert resp.metadata is not None
        return resp.metadata

    def proof(self, challenge: bytes, timeout: float = 600.0) -> sp.Proof:
        deadline = time.monotonic() + timeout
        while True:
            resp = self._c.roundtrip(
                sp.NodeRequest(gen_proof_challenge=challenge))
            if resp.gen_proof_status == sp.GEN_PROOF_STATUS_OK:
                assert resp.gen_proof_proof is not None
                return resp.gen_proof_proof
            if resp.gen_proof_status == sp.GEN_PROOF_STATUS_ERROR:
                raise RuntimeError("post service failed to generate proof")
            if time.monotonic() > deadline:
                raise TimeoutError("proof generation timed out")
            time.sleep(self.poll_interval)


class PostServiceServer:
    """The node's spacemesh.v1.PostService endpoint (post_service.go:91)."""

    def __init__(self, address: str = "127.0.0.1:0") -> None:
        self._clients: Dict[bytes, _Client] = {}
        self._mu = threading.Lock()
        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=8))
        handler = grpc.method_handlers_generic_handler(
            "spacemesh.v1.PostService",
            {"Register": grpc.stream_stream_rpc_method_handler(
                self._register,
                request_deserializer=lambda b: b,
                response_serializer=lambda b: b)})
        self._server.add_generic_rpc_handlers((handler,))
        self.port = self._server.add_insecure_port(address)
        self._server.start()

    @property
    def address(self) -> str:
        return f"127.0.0.1:{self.port}"

    def _register(self, request_iterator, context):
        """Bidi stream handler: ask for metadata, register the service,
        then serve queued node requests in lockstep."""
        client: Optional[_Client] = None
        try:
            yield sp.NodeRequest().encode()  # metadata request
            raw = next(request_iterator)
            md = sp.ServiceResponse.decode(raw).metadata
            if md is None:
                return
            client = _Client(md)
            with self._mu:
                self._clients[md.node_id] = client  # post_service.go:24-31
            while True:
                try:
                    req, reply_q = client.cmd_q.get(timeout=0.5)
                except queue.Empty:
                    if not context.is_active():
                        break
                    continue
                try:
                    yield req.encode()
                    raw = next(request_iterator)
                    reply_q.put(sp.ServiceResponse.decode(raw))
                except Exception as e:  # noqa: BLE001
                    reply_q.put(e)
                    raise
        except (StopIteration, grpc.RpcError):
            pass
        finally:
            if client is not None:
                client.closed.set()
                with self._mu:
                    if self._clients.get(client.metadata.node_id) is client:
                        del self._clients[client.metadata.node_id]

    def client(self, node_id: bytes,
               poll_interval: float = 2.0) -> Optional[PostClient]:
        with self._mu:
            c = self._clients.get(node_id)
        return PostClient(c, poll_interval) if c else None

    def wait_for_client(self, timeout: float = 10.0,
                        poll_interval: float = 2.0) -> PostClient:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            with self._mu:
                if self._clients:
                    c = next(iter(self._clients.values()))
                    return PostClient(c, poll_interval)
            time.sleep(0.05)
        raise TimeoutError("no post service registered")

    def stop(self) -> None:
        self._server.stop(grace=1)


class PostSupervisor:
    """Child-process lifecycle (post_supervisor.go:133-299)."""

    SERVICE_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "service.py")

    def __init__(self, address: str, datadir: str, nonces: int = 288,
                 threads: int = 0, max_retries: int = 3,
                 k1: int = 26, k2: int = 37,
                 pow_difficulty: Optional[bytes] = None,
                 mock_prover: bool = False,
                 on_fatal: Optional[Callable[[int], None]] = None) -> None:
        self.address = address
        self.datadir = datadir
        self.argv = [sys.executable, self.SERVICE_PATH,
                     "--address", address, "--dir", datadir,
                     "--watch-pid", str(os.getpid()),       # :246
                     "--max-retries", str(max_retries),     # :248-251
                     "--nonces", str(nonces),
                     "--k1", str(k1), "--k2", str(k2),      # :228-247
                     "--threads", str(threads)]
        if pow_difficulty:
            self.argv += ["--pow-difficulty", pow_difficulty.hex()]
        if mock_prover:
            self.argv.append("--mock-prover")
        self._proc: Optional[subprocess.Popen] = None
        self._stopping = threading.Event()
        self._on_fatal = on_fatal
        self.stderr_lines: list[str] = []

    def start(self) -> None:
        if self._proc is not None:
            raise RuntimeError("already started")  # :138-141
        self._stopping.clear()
        self._proc = subprocess.Popen(self.argv, stderr=subprocess.PIPE,
                                      text=True)

        def relog():  # stderr re-logging (:207-218)
            assert self._proc and self._proc.stderr
            for line in self._proc.stderr:
                self.stderr_lines.append(line.rstrip())
                print(f"[post-service] {line.rstrip()}", file=sys.stderr)

        def monitor():  # unexpected exit is fatal (:288-298)
            assert self._proc
            rc = self._proc.wait()
            if not self._stopping.is_set():
                print(f"post service exited unexpectedly (rc={rc})",
                      file=sys.stderr)
                if self._on_fatal:
                    self._on_fatal(rc)

        threading.Thread(target=relog, daemon=True).start()
        threading.Thread(target=monitor, daemon=True).start()

    def stop(self) -> None:
        if self._proc is None:
            return
        self._stopping.set()
        self._proc.terminate()
        try:
            self._proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            self._proc.kill()
            self._proc.wait()
        self._proc = None

    @property
    def pid(self) -> Optional[int]:
        return self._proc.pid if self._proc else None
