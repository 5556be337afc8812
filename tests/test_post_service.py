"""node <-> post-service process boundary tests (CPU, mock prover):
register/metadata exchange, GenProof polling, multi-service registry,
unexpected-exit fatality, watch-pid — mirroring the reference's
api/grpcserver/post_service_test.go:118-163 (Test_GenerateProof), :300
(multiple services) and post_supervisor lifecycle, with the real
post-service child process over real localhost gRPC."""
import importlib
import os
import subprocess
import sys
import threading
import time

import pytest

sup_mod = importlib.import_module("go-spacemesh_amd.supervisor")
sp = importlib.import_module("go-spacemesh_amd.service_proto")

NODE = bytes([0xA5]) * 32
ATX = bytes([0x5A]) * 32


@pytest.fixture()
def datadir(tmp_path, oracle):
    subprocess.run(
        [os.path.join("oracle", "oracle_bench"), "init",
         "--out", str(tmp_path), "--node-id", NODE.hex(),
         "--atx-id", ATX.hex(), "--num-units", "1",
         "--labels-per-unit", "128", "--scrypt-n", "2",
         "--max-file-size", "4096"],
        check=True, capture_output=True)
    return str(tmp_path)


def test_proto_roundtrips():
    md = sp.Metadata(NODE, ATX, 17, 4, 1 << 20)
    assert sp.Metadata.decode(md.encode()) == md
    md2 = sp.Metadata(NODE, ATX, None, 1, 64)
    assert sp.Metadata.decode(md2.encode()) == md2
    for r in [sp.NodeRequest(), sp.NodeRequest(gen_proof_challenge=b"x" * 32)]:
        got = sp.NodeRequest.decode(r.encode())
        assert got == r
    pr = sp.Proof(3, b"\x01\x02", 99)
    for resp in [sp.ServiceResponse(metadata=md),
                 sp.ServiceResponse(gen_proof_status=sp.GEN_PROOF_STATUS_OK,
                                    gen_proof_proof=pr),
                 sp.ServiceResponse(
                     gen_proof_status=sp.GEN_PROOF_STATUS_IN_PROGRESS)]:
        assert sp.ServiceResponse.decode(resp.encode()) == resp


def test_register_metadata_and_proof(datadir):
    """Test_GenerateProof shape: launch the real service child against an
    in-process server, exchange metadata, generate a (mock) proof with
    in-progress polling."""
    server = sup_mod.PostServiceServer()
    sup = sup_mod.PostSupervisor(server.address, datadir, mock_prover=True)
    sup.start()
    try:
        client = server.wait_for_client(timeout=15, poll_interval=0.1)
        info = client.info()
        assert info.node_id == NODE
        assert info.commitment_atx_id == ATX
        assert info.num_units == 1
        assert info.labels_per_unit == 128
        assert info.nonce is not None  # from postdata_metadata.json
        proof = client.proof(b"\x07" * 32, timeout=30)
        assert proof.nonce == 7
        assert proof.indices == b"\x07" * 8
        assert proof.pow == 42
        # the service registered under its node id (post_service.go:24-31)
        assert server.client(NODE) is not None
        assert server.client(b"\x00" * 32) is None
    finally:
        sup.stop()
        server.stop()


def test_multiple_services(tmp_path, oracle):
    """Two identities, two services, one node server
    (Test_GenerateProof_MultipleServices, post_service_test.go:300)."""
    ids = [bytes([1]) * 32, bytes([2]) * 32]
    dirs = []
    for i, nid in enumerate(ids):
        d = tmp_path / f"s{i}"
        d.mkdir()
        subprocess.run(
            [os.path.join("oracle", "oracle_bench"), "init",
             "--out", str(d), "--node-id", nid.hex(), "--atx-id", ATX.hex(),
             "--num-units", "1", "--labels-per-unit", "64",
             "--scrypt-n", "2", "--max-file-size", "4096"],
            check=True, capture_output=True)
        dirs.append(str(d))
    server = sup_mod.PostServiceServer()
    sups = [sup_mod.PostSupervisor(server.address, d, mock_prover=True)
            for d in dirs]
    for s in sups:
        s.start()
    try:
        deadline = time.time() + 15
        while time.time() < deadline:
            if all(server.client(nid) for nid in ids):
                break
            time.sleep(0.05)
        for nid in ids:
            c = server.client(nid, poll_interval=0.1)
            assert c is not None
            assert c.info().node_id == nid
    finally:
        for s in sups:
            s.stop()
        server.stop()


def test_unexpected_exit_is_fatal(datadir):
    server = sup_mod.PostServiceServer()
    fatal = threading.Event()
    sup = sup_mod.PostSupervisor(server.address, datadir, mock_prover=True,
                                 on_fatal=lambda rc: fatal.set())
    sup.start()
    try:
        server.wait_for_client(timeout=15, poll_interval=0.1)
        os.kill(sup.pid, 9)  # crash the child
        assert fatal.wait(timeout=10)
    finally:
        sup.stop()
        server.stop()


def test_clean_stop_is_not_fatal(datadir):
    server = sup_mod.PostServiceServer()
    fatal = threading.Event()
    sup = sup_mod.PostSupervisor(server.address, datadir, mock_prover=True,
                                 on_fatal=lambda rc: fatal.set())
    sup.start()
    server.wait_for_client(timeout=15, poll_interval=0.1)
    sup.stop()
    time.sleep(0.5)
    assert not fatal.is_set()
    server.stop()


def test_watch_pid_kills_orphan(datadir):
    """--watch-pid: the child dies when the watched process does
    (post_supervisor.go:246).  Watch a short-lived process."""
    dummy = subprocess.Popen([sys.executable, "-c", "import time; "
                              "time.sleep(1.5)"])
    svc = subprocess.Popen(
        [sys.executable, sup_mod.PostSupervisor.SERVICE_PATH,
         "--address", "127.0.0.1:1",  # nothing listening: it will retry
         "--dir", datadir, "--watch-pid", str(dummy.pid), "--mock-prover"],
        stderr=subprocess.DEVNULL)
    try:
        rc = svc.wait(timeout=30)
        assert rc == 1  # exited because the watched pid vanished
    finally:
        if svc.poll() is None:
            svc.kill()
        dummy.wait()
