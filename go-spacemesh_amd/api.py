"""ctypes bindings for libpost_hip.so + host-side mirrors of the reference's
Go interfaces.

Interface parity map (file:line under /root/reference/):
  PostSetupManager.{PrepareInitializer,StartSession,Status,Reset}
      -> activation/post.go:245,271,341,438
  PostConfig / PostSetupOpts fields -> activation/post.go:27-61
  PostVerifier.Verify(proof, metadata, opts) -> activation/interface.go:26-29,
      activation/post_verifier.go:150-160
  verify options Subset(k3, seed) / SelectedIndex
      -> activation/validation.go:206-209, activation/malfeasance.go:165
  proof shape {nonce, indices, pow} -> api/grpcserver/post_client.go:124-141
"""
from __future__ import annotations

import ctypes
import dataclasses
import enum
import os
import threading
from ctypes import (POINTER, byref, c_char_p, c_int, c_int32, c_size_t,
                    c_uint8, c_uint16, c_uint32, c_uint64, c_void_p)
from typing import Optional

from . import events as _ev
from . import metrics as _metrics

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.environ.get("POST_ENGINE_LIB",
                           os.path.join(_DIR, "libpost_hip.so"))

LABEL_SIZE = 16
FULL_LABEL_SIZE = 32

POW_MODE_RANDOMX = 0
POW_MODE_BLAKE3 = 1


class Status(enum.IntEnum):
    OK = 0
    ERR = 1
    INVALID_ARGS = 2
    NO_GPU = 3
    OOM = 4
    IO = 5
    CANCELLED = 6
    POW = 7
    INVALID_INDEX = 8
    UNSUPPORTED = 9
    NO_NONCE = 10


class EngineError(RuntimeError):
    def __init__(self, code: int, detail: str = ""):
        self.code = Status(code)
        super().__init__(f"{self.code.name}: {detail}")


class _CProvider(ctypes.Structure):
    _fields_ = [("id", c_uint32), ("model", ctypes.c_char * 256),
                ("device_type", c_uint32), ("memory_bytes", c_uint64),
                ("performance", c_uint64)]


class _CInitConfig(ctypes.Structure):
    _fields_ = [("node_id", c_uint8 * 32), ("commitment_atx_id", c_uint8 * 32),
                ("num_units", c_uint32), ("labels_per_unit", c_uint64),
                ("max_file_size", c_uint64), ("scrypt_n", c_uint32),
                ("provider_id", c_uint32), ("index_start", c_uint64),
                ("index_end", c_uint64), ("data_dir", c_char_p),
                ("scratch_bytes", c_uint64)]


class CProof(ctypes.Structure):
    _fields_ = [("nonce", c_uint32), ("pow", c_uint64),
                ("indices", c_uint8 * 800), ("indices_len", c_uint32),
                ("num_indices", c_uint16)]


class CProofMetadata(ctypes.Structure):
    _fields_ = [("node_id", c_uint8 * 32), ("commitment_atx_id", c_uint8 * 32),
                ("challenge", c_uint8 * 32), ("num_units", c_uint32),
                ("labels_per_unit", c_uint64)]


class _CProveConfig(ctypes.Structure):
    _fields_ = [("challenge", c_uint8 * 32), ("k1", c_uint32),
                ("k2", c_uint32), ("nonces", c_uint32),
                ("pow_difficulty", c_uint8 * 32), ("pow_mode", c_uint32),
                ("pow_threads", c_uint32), ("provider_id", c_uint32)]


class _CVerifyConfig(ctypes.Structure):
    _fields_ = [("k1", c_uint32), ("k2", c_uint32), ("k3", c_uint32),
                ("subset_seed", c_void_p), ("subset_seed_len", c_size_t),
                ("selected_index", c_int32),
                ("pow_difficulty", c_uint8 * 32), ("pow_mode", c_uint32),
                ("scrypt_n", c_uint32), ("provider_id", c_uint32)]


_lib_lock = threading.Lock()
_lib: Optional[ctypes.CDLL] = None


def load_engine() -> ctypes.CDLL:
    """Load libpost_hip.so.  Raises loudly when the extension is missing —
    a GPU box must never silently run without the HIP engine."""
    global _lib
    with _lib_lock:
        if _lib is not None:
            return _lib
        if not os.path.exists(_LIB_PATH):
            raise EngineError(
                Status.ERR,
                f"HIP engine not built: {_LIB_PATH} missing. "
                "Run `make` (or __graft_entry__.build()).")
        lib = ctypes.CDLL(_LIB_PATH)
        sigs = {
            "post_last_error": (c_char_p, []),
            "post_engine_version": (c_char_p, []),
            "post_providers": (c_int, [POINTER(_CProvider), c_uint32,
                                       POINTER(c_uint32)]),
            "post_benchmark": (c_int, [c_uint32, c_uint32,
                                       POINTER(c_uint64)]),
            "post_init_new": (c_int, [POINTER(_CInitConfig),
                                      POINTER(c_void_p)]),
            "post_init_run": (c_int, [c_void_p]),
            "post_init_step": (c_int, [c_void_p, c_uint64,
                                       POINTER(c_uint64)]),
            "post_init_last_kernel_ms": (ctypes.c_double, [c_void_p]),
            "post_init_num_labels_written": (c_uint64, [c_void_p]),
            "post_init_cancel": (None, [c_void_p]),
            "post_init_nonce": (c_int, [c_void_p, POINTER(c_uint64),
                                        ctypes.c_char_p]),
            "post_init_copy_labels": (c_int, [c_void_p, c_uint64, c_uint64,
                                              ctypes.c_char_p]),
            "post_init_free": (None, [c_void_p]),
            "post_prove": (c_int, [c_char_p, POINTER(_CProveConfig),
                                   POINTER(CProof)]),
            "post_prove_buffer": (c_int, [ctypes.c_char_p, c_uint64,
                                          ctypes.c_char_p, ctypes.c_char_p,
                                          POINTER(_CProveConfig),
                                          POINTER(CProof)]),
            "post_verify": (c_int, [POINTER(CProof), POINTER(CProofMetadata),
                                    POINTER(_CVerifyConfig),
                                    POINTER(c_uint32)]),
            "post_verify_batch": (c_int, [POINTER(CProof),
                                          POINTER(CProofMetadata), c_uint32,
                                          POINTER(_CVerifyConfig),
                                          POINTER(c_int), POINTER(c_uint32)]),
            "post_verify_batch_seeded": (c_int, [
                POINTER(CProof), POINTER(CProofMetadata), c_uint32,
                POINTER(_CVerifyConfig), ctypes.c_char_p, c_size_t,
                POINTER(c_int), POINTER(c_uint32)]),
            "post_verify_vrf_nonce": (c_int, [POINTER(CProofMetadata),
                                              c_uint64, c_uint32, c_uint32]),
            "post_selftest_blake3": (None, [ctypes.c_char_p, c_size_t,
                                            ctypes.c_char_p]),
            "post_selftest_aes128": (None, [ctypes.c_char_p, ctypes.c_char_p,
                                            ctypes.c_char_p]),
            "post_selftest_label": (c_int, [ctypes.c_char_p, ctypes.c_char_p,
                                            c_uint64, c_uint32,
                                            ctypes.c_char_p]),
        }
        for name, (res, args) in sigs.items():
            fn = getattr(lib, name)
            fn.restype = res
            fn.argtypes = args
        _lib = lib
        return lib


# ------------------------- config dataclasses -------------------------

@dataclasses.dataclass
class PostConfig:
    """Mirror of activation/post.go:27-49 PostConfig (protocol params)."""
    min_num_units: int = 4
    max_num_units: int = 2**32 - 1
    labels_per_unit: int = 4294967296   # config/mainnet.go:186
    k1: int = 26                        # config/mainnet.go:187
    k2: int = 37
    k3: int = 1
    pow_difficulty: bytes = bytes.fromhex(
        "000dfb23b0979b4b000000000000000000000000000000000000000000000000")
    pow_mode: int = POW_MODE_BLAKE3


@dataclasses.dataclass
class PostSetupOpts:
    """Mirror of activation/post.go:52-61 PostSetupOpts."""
    data_dir: Optional[str] = None
    num_units: int = 4
    max_file_size: int = 4294967296
    provider_id: int = 0
    scrypt_n: int = 8192                # activation/post.go:155 dep default
    index_start: int = 0                # shard range
    index_end: int = 0
    scratch_bytes: int = 0


@dataclasses.dataclass
class ProveOpts:
    """Mirror of PostProvingOpts (activation/post.go:64-74)."""
    nonces: int = 288                   # config/mainnet.go:61
    threads: int = 0
    provider_id: int = 0


@dataclasses.dataclass
class VerifyOpts:
    """verifying options (validation.go:206-209, malfeasance.go:165)."""
    subset_seed: Optional[bytes] = None
    selected_index: int = -1
    provider_id: int = 0


@dataclasses.dataclass
class PostProof:
    nonce: int
    indices: bytes
    pow: int

    def to_c(self) -> CProof:
        c = CProof()
        c.nonce = self.nonce
        c.pow = self.pow
        c.indices_len = len(self.indices)
        c.num_indices = 0  # set by callers that know k2
        ctypes.memmove(c.indices, self.indices, len(self.indices))
        return c


@dataclasses.dataclass
class PostProofMetadata:
    """shared.ProofMetadata (validation.go:193-199)."""
    node_id: bytes
    commitment_atx_id: bytes
    challenge: bytes
    num_units: int
    labels_per_unit: int

    def to_c(self) -> CProofMetadata:
        return CProofMetadata(
            (c_uint8 * 32)(*self.node_id),
            (c_uint8 * 32)(*self.commitment_atx_id),
            (c_uint8 * 32)(*self.challenge),
            self.num_units, self.labels_per_unit)


class Engine:
    """Thin OO wrapper over the C-ABI."""

    def __init__(self) -> None:
        self.lib = load_engine()

    def _check(self, rc: int) -> None:
        if rc != 0:
            raise EngineError(rc, self.lib.post_last_error().decode())

    def version(self) -> str:
        return self.lib.post_engine_version().decode()

    def providers(self):
        arr = (_CProvider * 16)()
        count = c_uint32(0)
        self._check(self.lib.post_providers(arr, 16, byref(count)))
        return [{"id": arr[i].id, "model": arr[i].model.decode(),
                 "device_type": arr[i].device_type,
                 "memory_bytes": arr[i].memory_bytes}
                for i in range(min(count.value, 16))]

    def benchmark(self, provider_id: int = 0, scrypt_n: int = 8192) -> int:
        out = c_uint64(0)
        self._check(self.lib.post_benchmark(provider_id, scrypt_n,
                                            byref(out)))
        return out.value

    # self-test hooks (used by tests to cross-check vs the oracle)
    def selftest_blake3(self, msg: bytes) -> bytes:
        out = ctypes.create_string_buffer(32)
        self.lib.post_selftest_blake3(msg, len(msg), out)
        return out.raw

    def selftest_aes128(self, key: bytes, block: bytes) -> bytes:
        out = ctypes.create_string_buffer(16)
        self.lib.post_selftest_aes128(key, block, out)
        return out.raw

    def selftest_label(self, node_id: bytes, atx_id: bytes, index: int,
                       scrypt_n: int) -> bytes:
        out = ctypes.create_string_buffer(32)
        rc = self.lib.post_selftest_label(node_id, atx_id, index, scrypt_n,
                                          out)
        self._check(rc)
        return out.raw


class PostSetupManager:
    """Mirror of activation/post.go PostSetupManager: PrepareInitializer /
    StartSession (blocking; resumable) / Status / Reset."""

    NOT_STARTED, PREPARED, IN_PROGRESS, STOPPED, COMPLETE, ERROR = range(1, 7)

    def __init__(self, node_id: bytes, commitment_atx_id: bytes,
                 cfg: PostConfig, opts: PostSetupOpts) -> None:
        self.engine = Engine()
        self.node_id = node_id
        self.commitment_atx_id = commitment_atx_id
        self.cfg = cfg
        self.opts = opts
        self.state = self.NOT_STARTED
        self._session: Optional[c_void_p] = None

    def prepare_initializer(self) -> None:
        if self.state == self.PREPARED:
            raise EngineError(Status.ERR, "already prepared")
        if self.opts.num_units < self.cfg.min_num_units or \
                self.opts.num_units > self.cfg.max_num_units:
            raise EngineError(Status.INVALID_ARGS, "numUnits out of range")
        c = _CInitConfig()
        ctypes.memmove(c.node_id, self.node_id, 32)
        ctypes.memmove(c.commitment_atx_id, self.commitment_atx_id, 32)
        c.num_units = self.opts.num_units
        c.labels_per_unit = self.cfg.labels_per_unit
        c.max_file_size = self.opts.max_file_size
        c.scrypt_n = self.opts.scrypt_n
        c.provider_id = self.opts.provider_id
        c.index_start = self.opts.index_start
        c.index_end = self.opts.index_end
        c.data_dir = self.opts.data_dir.encode() if self.opts.data_dir \
            else None
        c.scratch_bytes = self.opts.scratch_bytes
        handle = c_void_p()
        self.engine._check(self.engine.lib.post_init_new(byref(c),
                                                         byref(handle)))
        self._session = handle
        self.state = self.PREPARED

    def start_session(self) -> None:
        if self.state != self.PREPARED:
            raise EngineError(Status.ERR, "post session not prepared")
        self.state = self.IN_PROGRESS
        _ev.bus().emit(_ev.InitStart(self.node_id, self.commitment_atx_id))
        _metrics.init_size.labels(step="start").set(self.opts.num_units)
        rc = self.engine.lib.post_init_run(self._session)
        if rc == Status.CANCELLED:
            self.state = self.STOPPED
            raise EngineError(rc, "stopped")
        if rc != 0:
            self.state = self.ERROR
            err = self.engine.lib.post_last_error().decode()
            _ev.bus().emit(_ev.InitFailure(self.node_id, err))
            raise EngineError(rc, err)
        self.state = self.COMPLETE
        _metrics.init_size.labels(step="complete").set(self.opts.num_units)
        _ev.bus().emit(_ev.InitComplete(self.node_id))

    def step(self, max_labels: int) -> tuple:
        """Process up to max_labels labels; returns (labels_done,
        kernel_ms).  One bench step of the init hot path."""
        done = c_uint64(0)
        rc = self.engine.lib.post_init_step(self._session, max_labels,
                                            byref(done))
        self.engine._check(rc)
        kms = self.engine.lib.post_init_last_kernel_ms(self._session)
        return done.value, kms

    def status(self):
        written = 0
        if self._session:
            written = self.engine.lib.post_init_num_labels_written(
                self._session)
        return {"state": self.state, "num_labels_written": written}

    def stop(self) -> None:
        if self._session:
            self.engine.lib.post_init_cancel(self._session)

    def vrf_nonce(self):
        idx = c_uint64(0)
        label = ctypes.create_string_buffer(32)
        rc = self.engine.lib.post_init_nonce(self._session, byref(idx), label)
        if rc != 0:
            return None
        return idx.value, label.raw

    def copy_labels(self, first: int, count: int) -> bytes:
        buf = ctypes.create_string_buffer(count * LABEL_SIZE)
        self.engine._check(self.engine.lib.post_init_copy_labels(
            self._session, first, count, buf))
        return buf.raw

    def reset(self) -> None:
        if self._session:
            self.engine.lib.post_init_free(self._session)
            self._session = None
        self.state = self.NOT_STARTED

    def __del__(self):
        try:
            self.reset()
        except Exception:
            pass


def prove_buffer(labels: bytes, num_labels: int, node_id: bytes,
                 atx_id: bytes, challenge: bytes, cfg: PostConfig,
                 opts: ProveOpts) -> PostProof:
    import time as _time
    _ev.bus().emit(_ev.PostStart(node_id, challenge))
    t0 = _time.monotonic()
    try:
        return _prove_buffer(labels, num_labels, node_id, atx_id, challenge,
                             cfg, opts)
    finally:
        dt = _time.monotonic() - t0
        _metrics.post_seconds.set(dt)          # smh_post_seconds
        _metrics.post_duration.set(dt * 1e9)   # activation_post_duration
        _ev.bus().emit(_ev.PostComplete(node_id))


def _prove_buffer(labels: bytes, num_labels: int, node_id: bytes,
                  atx_id: bytes, challenge: bytes, cfg: PostConfig,
                  opts: ProveOpts) -> PostProof:
    eng = Engine()
    pc = _CProveConfig()
    ctypes.memmove(pc.challenge, challenge, 32)
    pc.k1, pc.k2 = cfg.k1, cfg.k2
    pc.nonces = opts.nonces
    ctypes.memmove(pc.pow_difficulty, cfg.pow_difficulty, 32)
    pc.pow_mode = cfg.pow_mode
    pc.pow_threads = opts.threads
    pc.provider_id = opts.provider_id
    out = CProof()
    eng._check(eng.lib.post_prove_buffer(labels, num_labels, node_id, atx_id,
                                         byref(pc), byref(out)))
    return PostProof(nonce=out.nonce,
                     indices=bytes(out.indices[:out.indices_len]),
                     pow=out.pow)


def prove_dir(data_dir: str, challenge: bytes, cfg: PostConfig,
              opts: ProveOpts) -> PostProof:
    eng = Engine()
    pc = _CProveConfig()
    ctypes.memmove(pc.challenge, challenge, 32)
    pc.k1, pc.k2 = cfg.k1, cfg.k2
    pc.nonces = opts.nonces
    ctypes.memmove(pc.pow_difficulty, cfg.pow_difficulty, 32)
    pc.pow_mode = cfg.pow_mode
    pc.pow_threads = opts.threads
    pc.provider_id = opts.provider_id
    out = CProof()
    eng._check(eng.lib.post_prove(data_dir.encode(), byref(pc), byref(out)))
    return PostProof(nonce=out.nonce,
                     indices=bytes(out.indices[:out.indices_len]),
                     pow=out.pow)


class PostVerifier:
    """Mirror of the reference PostVerifier (interface.go:26-29): the inner
    verifier the Go worker pool wraps (post_verifier.go:150-160).  Safe for
    concurrent use."""

    def __init__(self, cfg: PostConfig, scrypt_n: int = 8192) -> None:
        self.engine = Engine()
        self.cfg = cfg
        self.scrypt_n = scrypt_n

    def _vc(self, opts: VerifyOpts) -> tuple:
        vc = _CVerifyConfig()
        vc.k1, vc.k2, vc.k3 = self.cfg.k1, self.cfg.k2, self.cfg.k3
        seed_buf = None
        if opts.subset_seed is not None:
            seed_buf = ctypes.create_string_buffer(opts.subset_seed,
                                                   len(opts.subset_seed))
            vc.subset_seed = ctypes.cast(seed_buf, c_void_p)
            vc.subset_seed_len = len(opts.subset_seed)
        vc.selected_index = opts.selected_index
        ctypes.memmove(vc.pow_difficulty, self.cfg.pow_difficulty, 32)
        vc.pow_mode = self.cfg.pow_mode
        vc.scrypt_n = self.scrypt_n
        vc.provider_id = opts.provider_id
        return vc, seed_buf

    def verify(self, proof: PostProof, meta: PostProofMetadata,
               opts: VerifyOpts = VerifyOpts()) -> None:
        """Raises EngineError(INVALID_INDEX/POW/...) on invalid proofs."""
        vc, _seed = self._vc(opts)
        cp = proof.to_c()
        cp.num_indices = self.cfg.k2
        inv = c_uint32(0)
        rc = self.engine.lib.post_verify(byref(cp), byref(meta.to_c()),
                                         byref(vc), byref(inv))
        if rc == Status.INVALID_INDEX:
            raise EngineError(rc, f"invalid POST index at position "
                                  f"{inv.value}")
        self.engine._check(rc)

    def marshal_batch(self, proofs, metas):
        """Pre-build the C argument arrays for verify_batch.  A cgo shim
        holds these layouts natively; the ctypes conversion is a
        Python-mirror cost only, so callers that re-verify (or benchmark
        the ABI boundary) can marshal once and pass the result to
        verify_batch as `marshalled=`."""
        n = len(proofs)
        cps = (CProof * n)()
        cms = (CProofMetadata * n)()
        for i, (p, m) in enumerate(zip(proofs, metas)):
            cps[i] = p.to_c()
            cps[i].num_indices = self.cfg.k2
            cms[i] = m.to_c()
        return cps, cms, n

    def verify_batch(self, proofs, metas, opts: VerifyOpts = VerifyOpts(),
                     seeds=None, marshalled=None):
        # seeds (optional): equal-length per-proof subset seeds — each
        # gossip verify samples with its own peer seed
        # (validation.go:206-209).
        vc, _seed = self._vc(opts)
        cps, cms, n = (marshalled if marshalled is not None
                       else self.marshal_batch(proofs, metas))
        statuses = (c_int * n)()
        invs = (c_uint32 * n)()
        if seeds is not None:
            assert len(seeds) == n and n > 0
            slen = len(seeds[0])
            assert all(len(x) == slen for x in seeds)
            blob = b"".join(seeds)
            self.engine._check(self.engine.lib.post_verify_batch_seeded(
                cps, cms, n, byref(vc), blob, slen, statuses, invs))
        else:
            self.engine._check(self.engine.lib.post_verify_batch(
                cps, cms, n, byref(vc), statuses, invs))
        return [(Status(statuses[i]), invs[i]) for i in range(n)]

    def verify_vrf_nonce(self, meta: PostProofMetadata, index: int,
                         provider_id: int = 0) -> None:
        rc = self.engine.lib.post_verify_vrf_nonce(byref(meta.to_c()), index,
                                                   self.scrypt_n, provider_id)
        self.engine._check(rc)
