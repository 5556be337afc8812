"""Verifier pool + autoscaling semantics (mirrors the reference's mocked
tests: activation/post_verifier_test.go:19-135,
post_verifier_scaling_test.go:19-111 — worker distribution, prioritization,
autoscale on synthetic Post events, close/races)."""
import threading
import time

import pytest

import gsm_amd
from gsm_amd import events as ev


class MockInner:
    def __init__(self, delay=0.0, fail=False):
        self.calls = []
        self.delay = delay
        self.fail = fail
        self.mu = threading.Lock()
        self.concurrent = 0
        self.max_concurrent = 0

    def verify(self, proof, meta, opts):
        with self.mu:
            self.concurrent += 1
            self.max_concurrent = max(self.max_concurrent, self.concurrent)
            self.calls.append((proof, bytes(meta.node_id)))
        if self.fail:
            with self.mu:
                self.concurrent -= 1
            raise gsm_amd.EngineError(gsm_amd.api.Status.INVALID_INDEX, "bad")
        time.sleep(self.delay)
        with self.mu:
            self.concurrent -= 1


class FakeMeta:
    def __init__(self, node_id=b"\x01" * 32):
        self.node_id = node_id


def test_pool_distributes_and_blocks():
    inner = MockInner(delay=0.05)
    pool = gsm_amd.OffloadingVerifier(inner, workers=4)
    threads = [threading.Thread(target=pool.verify,
                                args=(i, FakeMeta(), None))
               for i in range(16)]
    t0 = time.time()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    dt = time.time() - t0
    assert len(inner.calls) == 16
    assert inner.max_concurrent <= 4
    assert dt < 16 * 0.05  # actually parallel
    pool.close()


def test_pool_propagates_inner_errors():
    inner = MockInner(fail=True)
    pool = gsm_amd.OffloadingVerifier(inner, workers=2)
    with pytest.raises(gsm_amd.EngineError) as e:
        pool.verify("p", FakeMeta(), None)
    assert e.value.code == gsm_amd.api.Status.INVALID_INDEX
    pool.close()


def test_prioritized_ids_accepted():
    own = b"\x07" * 32
    inner = MockInner()
    pool = gsm_amd.OffloadingVerifier(inner, workers=2,
                                      prioritized_ids=[own])
    pool.verify("p1", FakeMeta(own), None)
    pool.verify("p2", FakeMeta(), None)
    assert len(inner.calls) == 2
    pool.close()


def test_autoscale_on_post_events():
    bus = ev.EventBus()
    inner = MockInner()
    pool = gsm_amd.OffloadingVerifier(inner, workers=6, bus=bus)
    pool.autoscale(min_workers=1, bus=bus)
    assert pool.workers == 6
    nid = b"\x02" * 32
    bus.emit(ev.PostStart(nid, bytes(32)))
    deadline = time.time() + 5
    while pool.workers != 1 and time.time() < deadline:
        time.sleep(0.01)
    assert pool.workers == 1  # scaled down while proving
    pool.verify("p", FakeMeta(), None)  # still serves jobs
    bus.emit(ev.PostComplete(nid))
    assert pool.workers == 6  # restored
    pool.close()


def test_autoscale_multiple_identities():
    # scale restores only when NO identity is proving (post_verifier.go:73-120)
    bus = ev.EventBus()
    pool = gsm_amd.OffloadingVerifier(MockInner(), workers=4, bus=bus)
    pool.autoscale(min_workers=2, bus=bus)
    a, b = b"\x0a" * 32, b"\x0b" * 32
    bus.emit(ev.PostStart(a, bytes(32)))
    bus.emit(ev.PostStart(b, bytes(32)))
    bus.emit(ev.PostComplete(a))
    deadline = time.time() + 5
    while pool.workers != 2 and time.time() < deadline:
        time.sleep(0.01)
    assert pool.workers == 2  # b still proving
    bus.emit(ev.PostComplete(b))
    assert pool.workers == 4
    pool.close()


def test_verify_after_close_raises():
    pool = gsm_amd.OffloadingVerifier(MockInner(), workers=1)
    pool.close()
    with pytest.raises(RuntimeError):
        pool.verify("p", FakeMeta(), None)


def test_init_lifecycle_emits_events():
    seen = []
    unsub = ev.bus().subscribe(seen.append)
    try:
        import torch
        if torch.cuda.is_available():
            pytest.skip("event check is a CPU test")
        mgr = gsm_amd.PostSetupManager(
            bytes(32), bytes(32), gsm_amd.PostConfig(),
            gsm_amd.PostSetupOpts(num_units=4, scrypt_n=2))
        with pytest.raises(gsm_amd.EngineError):
            mgr.prepare_initializer()  # NO_GPU before any event
        assert seen == []
    finally:
        unsub()


def test_batching_verifier_groups_jobs():
    calls = []

    class BatchInner:
        def verify_batch(self, proofs, metas, opts=None, seeds=None):
            calls.append((len(proofs), seeds))
            time.sleep(0.01)
            return [(0, 0) for _ in proofs]

    bv = gsm_amd.BatchingVerifier(BatchInner(), max_batch=64,
                                  max_wait_s=0.05, seed_len=4)
    threads = [threading.Thread(target=bv.verify,
                                args=(i, FakeMeta(), b"s%03d" % i))
               for i in range(20)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    bv.close()
    total = sum(n for n, _ in calls)
    assert total == 20
    assert len(calls) < 20  # actually batched
    # per-job seeds forwarded
    all_seeds = [s for _, seeds in calls for s in seeds]
    assert sorted(all_seeds) == sorted(b"s%03d" % i for i in range(20))


def test_batching_verifier_propagates_failures():
    class BatchInner:
        def verify_batch(self, proofs, metas, opts=None, seeds=None):
            # first proof invalid at position 3, rest OK
            out = [(0, 0) for _ in proofs]
            out[0] = (8, 3)  # INVALID_INDEX
            return out

    bv = gsm_amd.BatchingVerifier(BatchInner(), seed_len=1)
    with pytest.raises(gsm_amd.EngineError) as e:
        bv.verify("p", FakeMeta(), b"x")
    assert e.value.code == gsm_amd.api.Status.INVALID_INDEX
    bv.close()
