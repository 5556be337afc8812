"""Offloading verifier pool with autoscaling — the host-side worker layer
the reference wraps around the native verifier (activation/post_verifier.go
:230-390), restated for this engine.

Semantics mirrored (file:line under /root/reference/):
  - newOffloadingPostVerifier(inner, workers, prioritized...) :230-268:
    jobs distribute over N workers; callers block while all are busy.
  - prioritized node IDs :303-320: verifications for own identities jump
    the queue (a dedicated channel in the reference; a priority queue here).
  - autoscaling :55-120, wired at :270-299: while ANY local identity is
    proving (PostStart..PostComplete events), scale down to min_workers to
    leave compute for the prover; restore afterwards.
  - Close() drains and joins workers; Verify after close raises.

The inner verifier must be safe for concurrent use (SAFETY note :227) —
engine.post_verify is."""
from __future__ import annotations

import itertools
import queue
import threading
from typing import Iterable, Optional

from . import events as ev
from . import metrics as _metrics


class _Job:
    __slots__ = ("proof", "meta", "opts", "done", "error")

    def __init__(self, proof, meta, opts):
        self.proof = proof
        self.meta = meta
        self.opts = opts
        self.done = threading.Event()
        self.error: Optional[BaseException] = None


class OffloadingVerifier:
    def __init__(self, inner, workers: int,
                 prioritized_ids: Iterable[bytes] = (),
                 bus: Optional[ev.EventBus] = None) -> None:
        if workers < 1:
            raise ValueError("workers must be >= 1")
        self._inner = inner
        self._target = workers
        self._max_workers = workers
        self._prioritized = {bytes(p) for p in prioritized_ids}
        self._q: "queue.PriorityQueue" = queue.PriorityQueue()
        self._seq = itertools.count()
        self._mu = threading.Lock()
        self._workers: list[threading.Thread] = []
        self._closed = False
        self._unsub = None
        self._proving: set[bytes] = set()
        self._min_workers = workers
        self._bus = bus
        self._scale(workers)

    # -- worker management (scale(), post_verifier.go:270-299) --
    def _scale(self, n: int) -> None:
        with self._mu:
            if self._closed:
                return
            self._target = n
            while len(self._workers) < n:
                t = threading.Thread(target=self._worker, daemon=True)
                t.start()
                self._workers.append(t)
            # excess workers exit on the next poison pill they pick up
            excess = len(self._workers) - n
            for _ in range(excess):
                self._q.put((0, next(self._seq), None))

    def _worker(self) -> None:
        while True:
            _prio, _seq, job = self._q.get()
            if job is None:
                with self._mu:
                    if len(self._workers) > self._target:
                        try:
                            self._workers.remove(threading.current_thread())
                        except ValueError:
                            pass
                        return
                # stale pill (pool scaled back up): drop it
                continue
            try:
                self._inner.verify(job.proof, job.meta, job.opts)
            except BaseException as e:  # noqa: BLE001 - propagate to caller
                job.error = e
            job.done.set()

    # -- autoscaling (post_verifier.go:55-120) --
    def autoscale(self, min_workers: int, bus: Optional[ev.EventBus] = None
                  ) -> None:
        """Subscribe to Post{Start,Complete}: scale to min_workers while any
        identity is proving, restore to the full pool when none is."""
        self._min_workers = min(min_workers, self._max_workers)
        b = bus or self._bus or ev.bus()

        def on_event(e: object) -> None:
            if isinstance(e, ev.PostStart):
                with self._mu:
                    self._proving.add(e.node_id)
                    proving = bool(self._proving)
                self._scale(self._min_workers if proving
                            else self._max_workers)
            elif isinstance(e, ev.PostComplete):
                with self._mu:
                    self._proving.discard(e.node_id)
                    proving = bool(self._proving)
                self._scale(self._min_workers if proving
                            else self._max_workers)

        self._unsub = b.subscribe(on_event)

    @property
    def workers(self) -> int:
        with self._mu:
            return len(self._workers)

    # -- verification entry (post_verifier.go:303-390) --
    def verify(self, proof, meta, opts=None, prioritized: bool = False
               ) -> None:
        """Blocking verify through the pool.  Raises whatever the inner
        verifier raised (e.g. EngineError(INVALID_INDEX))."""
        with self._mu:
            if self._closed:
                raise RuntimeError("verifier closed")
        own = prioritized or bytes(meta.node_id) in self._prioritized
        job = _Job(proof, meta, opts)
        _metrics.post_verification_queue.inc()  # post_verifier.go:319
        import time as _time
        t0 = _time.monotonic()
        try:
            self._q.put((0 if own else 1, next(self._seq), job))
            job.done.wait()
        finally:
            _metrics.post_verification_queue.dec()
            _metrics.post_verification_latency.observe(
                _time.monotonic() - t0)  # validation.go:216-220
        if job.error is not None:
            raise job.error

    def close(self) -> None:
        with self._mu:
            if self._closed:
                return
            self._closed = True
            n = len(self._workers)
            self._target = 0
        if self._unsub:
            self._unsub()
        for _ in range(n):
            self._q.put((2, next(self._seq), None))
        for t in list(self._workers):
            t.join(timeout=10)


class BatchingVerifier:
    """MI355X-first variant of the offloading verifier: callers still block
    per proof (the reference's worker-pool call shape), but a collector
    groups concurrent jobs into one engine verify_batch call — the engine's
    natural grain (one label-recompute launch + one predicate launch per
    batch).  Each job carries its own subset seed, matching
    the per-peer Subset(k3, seed) of gossip verification
    (validation.go:206-209).

    inner must expose verify_batch(proofs, metas, opts, seeds) — a
    PostVerifier does."""

    def __init__(self, inner, max_batch: int = 256,
                 max_wait_s: float = 0.002, seed_len: int = 32) -> None:
        self._inner = inner
        self._max_batch = max_batch
        self._max_wait = max_wait_s
        self._seed_len = seed_len
        self._q: "queue.Queue" = queue.Queue()
        self._closed = False
        self._t = threading.Thread(target=self._collector, daemon=True)
        self._t.start()

    def _collector(self) -> None:
        import time as _time
        while True:
            item = self._q.get()
            if item is None:
                return
            batch = [item]
            deadline = _time.monotonic() + self._max_wait
            while len(batch) < self._max_batch:
                timeout = deadline - _time.monotonic()
                if timeout <= 0:
                    break
                try:
                    nxt = self._q.get(timeout=timeout)
                except queue.Empty:
                    break
                if nxt is None:
                    self._run(batch)
                    return
                batch.append(nxt)
            self._run(batch)

    def _run(self, batch) -> None:
        _metrics.post_verification_queue.set(self._q.qsize())
        try:
            res = self._inner.verify_batch(
                [j[0] for j in batch], [j[1] for j in batch],
                seeds=[j[2] for j in batch])
            for (job, (status, inv)) in zip(batch, res):
                job[3]["status"] = (status, inv)
                job[4].set()
        except BaseException as e:  # noqa: BLE001
            for job in batch:
                job[3]["error"] = e
                job[4].set()

    def verify(self, proof, meta, seed: bytes) -> None:
        """Blocking; raises EngineError(INVALID_INDEX/POW/...) like the
        single-proof verifier."""
        if self._closed:
            raise RuntimeError("verifier closed")
        if len(seed) != self._seed_len:
            raise ValueError(f"seed must be {self._seed_len} bytes")
        import time as _time
        out: dict = {}
        done = threading.Event()
        t0 = _time.monotonic()
        self._q.put((proof, meta, seed, out, done))
        done.wait()
        _metrics.post_verification_latency.observe(_time.monotonic() - t0)
        if "error" in out:
            raise out["error"]
        status, inv = out["status"]
        if int(status) != 0:
            from .api import EngineError
            raise EngineError(int(status), f"position {inv}")

    def close(self) -> None:
        self._closed = True
        self._q.put(None)
        self._t.join(timeout=10)
