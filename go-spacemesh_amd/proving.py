"""Sharded proving coordination (proving row): each GPU scans
its label index range for all nonces; the host merges the per-nonce passing
index lists and packs the winner — identical semantics to the single-GPU
prover in engine.cpp (winner = nonce whose K2-th smallest passing index is
smallest, ties to the lowest nonce; indices ascending, LSB-first packed).

k2pow is computed once per nonce group and shared by every shard (the AES
cipher keys derive from it)."""
from __future__ import annotations

import ctypes
from ctypes import POINTER, byref, c_uint32, c_uint64
from typing import Dict, List, Optional, Sequence, Tuple

from . import api as _api
from . import wire as _wire

NONCE_GROUP = 16


class _CHit(ctypes.Structure):
    _fields_ = [("index", c_uint64), ("nonce", c_uint32), ("pad", c_uint32)]


def _bind(lib):
    if getattr(lib, "_proving_bound", False):
        return
    lib.post_prove_scan.restype = ctypes.c_int
    lib.post_prove_scan.argtypes = [
        ctypes.c_char_p, c_uint64, c_uint64, c_uint64,
        POINTER(_api._CProveConfig), POINTER(c_uint64), POINTER(_CHit),
        c_uint32, POINTER(c_uint32)]
    lib.post_k2pow_search.restype = ctypes.c_int
    lib.post_k2pow_search.argtypes = [
        ctypes.c_char_p, c_uint32, ctypes.c_char_p, c_uint32, c_uint32,
        POINTER(c_uint64)]
    lib._proving_bound = True


def group_pows(challenge: bytes, nonces: int, pow_difficulty: bytes,
               pow_mode: int = _api.POW_MODE_BLAKE3,
               threads: int = 0) -> List[int]:
    eng = _api.Engine()
    _bind(eng.lib)
    out = []
    for g in range(nonces // NONCE_GROUP):
        v = c_uint64(0)
        rc = eng.lib.post_k2pow_search(challenge, g, pow_difficulty,
                                       pow_mode, threads, byref(v))
        eng._check(rc)
        out.append(v.value)
    return out


def scan_shard(labels: bytes, index_base: int, total_labels: int,
               challenge: bytes, cfg: _api.PostConfig, nonces: int,
               pows: Sequence[int], provider_id: int = 0,
               cap: int = 1 << 22) -> List[Tuple[int, int]]:
    """Scan one shard's labels; returns (label_index, nonce) hits."""
    eng = _api.Engine()
    _bind(eng.lib)
    pc = _api._CProveConfig()
    ctypes.memmove(pc.challenge, challenge, 32)
    pc.k1, pc.k2 = cfg.k1, cfg.k2
    pc.nonces = nonces
    ctypes.memmove(pc.pow_difficulty, cfg.pow_difficulty, 32)
    pc.pow_mode = cfg.pow_mode
    pc.provider_id = provider_id
    parr = (c_uint64 * len(pows))(*pows)
    hits = (_CHit * cap)()
    n = c_uint32(0)
    rc = eng.lib.post_prove_scan(labels, len(labels) // 16, index_base,
                                 total_labels, byref(pc), parr, hits, cap,
                                 byref(n))
    eng._check(rc)
    return [(hits[i].index, hits[i].nonce) for i in range(n.value)]


def merge_shards(hit_lists: Sequence[Sequence[Tuple[int, int]]],
                 nonces: int, k2: int, total_labels: int,
                 pows: Sequence[int]) -> Optional[_api.PostProof]:
    """Merge per-shard hits into the final proof (the host-side tail of
    prove_core, engine.cpp): per nonce, ascending indices; winner = nonce
    with the smallest K2-th index; ties -> lowest nonce."""
    per_nonce: Dict[int, List[int]] = {}
    for hits in hit_lists:
        for idx, nonce in hits:
            per_nonce.setdefault(nonce, []).append(idx)
    best_nonce, best_kth = None, None
    for nonce in range(nonces):
        lst = per_nonce.get(nonce)
        if lst is None or len(lst) < k2:
            continue
        lst.sort()
        kth = lst[k2 - 1]
        if best_kth is None or kth < best_kth:
            best_nonce, best_kth = nonce, kth
    if best_nonce is None:
        return None
    idx = per_nonce[best_nonce][:k2]
    bpi = max(1, (total_labels - 1).bit_length())
    packed = bytearray((k2 * bpi + 7) // 8)
    pos = 0
    for v in idx:
        for b in range(bpi):
            if (v >> b) & 1:
                packed[pos >> 3] |= 1 << (pos & 7)
            pos += 1
    if len(packed) > _wire.PostV1.MAX_INDICES:
        return None
    return _api.PostProof(nonce=best_nonce, indices=bytes(packed),
                          pow=pows[best_nonce // NONCE_GROUP])


def prove_sharded(shards: Sequence[Tuple[bytes, int]], total_labels: int,
                  challenge: bytes, cfg: _api.PostConfig, nonces: int,
                  provider_ids: Optional[Sequence[int]] = None,
                  threads: int = 0) -> Optional[_api.PostProof]:
    """Single-process multi-shard prove: shards = [(labels, index_base)].
    In the 8-GPU deployment each rank calls scan_shard on its own device
    and rank 0 merges (hits are a few hundred bytes — latency-only)."""
    pows = group_pows(challenge, nonces, cfg.pow_difficulty, cfg.pow_mode,
                      threads)
    hit_lists = []
    for i, (labels, base) in enumerate(shards):
        pid = provider_ids[i] if provider_ids else 0
        hit_lists.append(scan_shard(labels, base, total_labels, challenge,
                                    cfg, nonces, pows, pid))
    return merge_shards(hit_lists, nonces, cfg.k2, total_labels, pows)
