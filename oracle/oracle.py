"""ctypes wrapper around the CPU oracle (liboracle.so).

TEST INFRASTRUCTURE ONLY — see oracle/oracle.h for the restatement and
pinning notes.  Only tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg may import this module; the product path is
go-spacemesh_amd/ and must fail loudly rather than fall back here.
"""
from __future__ import annotations

import ctypes
import os
import subprocess
from ctypes import (POINTER, byref, c_int, c_int32, c_size_t,
                    c_uint8, c_uint16, c_uint32, c_uint64, create_string_buffer)

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_DIR, "liboracle.so")

LABEL_SIZE = 16
FULL_LABEL_SIZE = 32
NONCES_PER_AES = 2
NONCE_GROUP = 16


class Proof(ctypes.Structure):
    _fields_ = [
        ("nonce", c_uint32),
        ("pow", c_uint64),
        ("num_indices", c_uint16),
        ("indices", c_uint8 * 800),
        ("indices_len", c_uint32),
    ]


class ProofMetadata(ctypes.Structure):
    _fields_ = [
        ("node_id", c_uint8 * 32),
        ("commitment_atx_id", c_uint8 * 32),
        ("challenge", c_uint8 * 32),
        ("num_units", c_uint32),
        ("labels_per_unit", c_uint64),
    ]


class VrfNonce(ctypes.Structure):
    _fields_ = [("index", c_uint64), ("label", c_uint8 * 32), ("found", c_int)]


def _build_if_needed() -> None:
    if not os.path.exists(_LIB_PATH):
        subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)


def load() -> ctypes.CDLL:
    _build_if_needed()
    lib = ctypes.CDLL(_LIB_PATH)
    u8p = POINTER(c_uint8)
    sigs = {
        "oracle_sha256": (None, [ctypes.c_char_p, c_size_t, ctypes.c_char_p]),
        "oracle_hmac_sha256": (None, [ctypes.c_char_p, c_size_t,
                                      ctypes.c_char_p, c_size_t,
                                      ctypes.c_char_p]),
        "oracle_pbkdf2_sha256": (None, [ctypes.c_char_p, c_size_t,
                                        ctypes.c_char_p, c_size_t, c_uint32,
                                        ctypes.c_char_p, c_size_t]),
        "oracle_salsa20_8": (None, [ctypes.c_char_p]),
        "oracle_scrypt": (c_int, [ctypes.c_char_p, c_size_t, ctypes.c_char_p,
                                  c_size_t, c_uint32, c_uint32, c_uint32,
                                  ctypes.c_char_p, c_size_t]),
        "oracle_blake3": (None, [ctypes.c_char_p, c_size_t, ctypes.c_char_p]),
        "oracle_blake3_xof": (None, [ctypes.c_char_p, c_size_t,
                                     ctypes.c_char_p, c_size_t]),
        "oracle_aes128_enc_block": (None, [ctypes.c_char_p, ctypes.c_char_p,
                                           ctypes.c_char_p]),
        "oracle_commitment": (None, [ctypes.c_char_p, ctypes.c_char_p,
                                     ctypes.c_char_p]),
        "oracle_label": (c_int, [ctypes.c_char_p, c_uint64, c_uint32,
                                 ctypes.c_char_p]),
        "oracle_init_range": (c_int, [ctypes.c_char_p, c_uint64, c_uint64,
                                      c_uint32, ctypes.c_char_p,
                                      ctypes.c_char_p, POINTER(VrfNonce)]),
        "oracle_vrf_difficulty": (None, [c_uint64, ctypes.c_char_p]),
        "oracle_proving_difficulty": (c_uint64, [c_uint32, c_uint64]),
        "oracle_k2pow": (c_uint64, [ctypes.c_char_p, c_uint32,
                                    ctypes.c_char_p]),
        "oracle_k2pow_verify": (c_int, [ctypes.c_char_p, c_uint32, c_uint64,
                                        ctypes.c_char_p]),
        "oracle_prove_cipher_key": (None, [ctypes.c_char_p, c_uint32,
                                           c_uint64, ctypes.c_char_p]),
        "oracle_prove": (c_int, [ctypes.c_char_p, c_uint64, ctypes.c_char_p,
                                 c_uint32, c_uint32, c_uint32,
                                 ctypes.c_char_p, POINTER(Proof)]),
        "oracle_verify": (c_int, [POINTER(Proof), POINTER(ProofMetadata),
                                  c_uint32, c_uint32, c_uint32, c_uint32,
                                  ctypes.c_char_p, c_size_t, c_int32,
                                  ctypes.c_char_p, POINTER(c_uint32)]),
        "oracle_verify_vrf_nonce": (c_int, [POINTER(ProofMetadata), c_uint64,
                                            c_uint32]),
        "oracle_bits_per_index": (c_uint32, [c_uint64]),
        "oracle_pack_indices": (c_uint32, [POINTER(c_uint64), c_uint32,
                                           c_uint32, ctypes.c_char_p]),
        "oracle_unpack_indices": (None, [ctypes.c_char_p, c_uint32, c_uint32,
                                         POINTER(c_uint64)]),
        "oracle_subset": (None, [c_uint32, c_uint32, ctypes.c_char_p,
                                 c_size_t, POINTER(c_uint32)]),
    }
    for name, (res, args) in sigs.items():
        fn = getattr(lib, name)
        fn.restype = res
        fn.argtypes = args
    return lib


class Oracle:
    """High-level helpers over the C oracle."""

    def __init__(self) -> None:
        self.lib = load()

    # primitives -----------------------------------------------------------
    def sha256(self, msg: bytes) -> bytes:
        out = create_string_buffer(32)
        self.lib.oracle_sha256(msg, len(msg), out)
        return out.raw

    def hmac_sha256(self, key: bytes, msg: bytes) -> bytes:
        out = create_string_buffer(32)
        self.lib.oracle_hmac_sha256(key, len(key), msg, len(msg), out)
        return out.raw

    def pbkdf2(self, pw: bytes, salt: bytes, iters: int, dklen: int) -> bytes:
        out = create_string_buffer(dklen)
        self.lib.oracle_pbkdf2_sha256(pw, len(pw), salt, len(salt), iters,
                                      out, dklen)
        return out.raw

    def scrypt(self, pw: bytes, salt: bytes, n: int, r: int, p: int,
               dklen: int) -> bytes:
        out = create_string_buffer(dklen)
        rc = self.lib.oracle_scrypt(pw, len(pw), salt, len(salt), n, r, p,
                                    out, dklen)
        if rc:
            raise ValueError("scrypt failed")
        return out.raw

    def blake3(self, msg: bytes, outlen: int = 32) -> bytes:
        out = create_string_buffer(outlen)
        self.lib.oracle_blake3_xof(msg, len(msg), out, outlen)
        return out.raw

    def aes128(self, key: bytes, block: bytes) -> bytes:
        out = create_string_buffer(16)
        self.lib.oracle_aes128_enc_block(key, block, out)
        return out.raw

    # POST ------------------------------------------------------------------
    def commitment(self, node_id: bytes, atx_id: bytes) -> bytes:
        out = create_string_buffer(32)
        self.lib.oracle_commitment(node_id, atx_id, out)
        return out.raw

    def label(self, commitment: bytes, index: int, scrypt_n: int) -> bytes:
        out = create_string_buffer(32)
        rc = self.lib.oracle_label(commitment, index, scrypt_n, out)
        if rc:
            raise ValueError("label failed")
        return out.raw

    def init_range(self, commitment: bytes, start: int, end: int,
                   scrypt_n: int, with_nonce: bool = True):
        n = end - start
        out = create_string_buffer(n * LABEL_SIZE)
        best = VrfNonce(0, (c_uint8 * 32)(), 0)
        diff = bytes([0xFF]) * 32 if with_nonce else None
        rc = self.lib.oracle_init_range(commitment, start, end, scrypt_n, out,
                                        diff, byref(best))
        if rc:
            raise ValueError("init failed")
        return out.raw, best

    def prove(self, labels: bytes, num_labels: int, challenge: bytes,
              k1: int, k2: int, nonces: int, pow_difficulty: bytes) -> Proof:
        proof = Proof()
        rc = self.lib.oracle_prove(labels, num_labels, challenge, k1, k2,
                                   nonces, pow_difficulty, byref(proof))
        if rc:
            raise ValueError("prove found no nonce")
        return proof

    def verify(self, proof: Proof, meta: ProofMetadata, scrypt_n: int,
               k1: int, k2: int, k3: int, subset_seed: bytes | None,
               selected_index: int, pow_difficulty: bytes):
        inv = c_uint32(0)
        rc = self.lib.oracle_verify(
            byref(proof), byref(meta), scrypt_n, k1, k2, k3,
            subset_seed, len(subset_seed) if subset_seed else 0,
            selected_index, pow_difficulty, byref(inv))
        return rc, inv.value

    def verify_vrf_nonce(self, meta: ProofMetadata, index: int,
                         scrypt_n: int) -> int:
        return self.lib.oracle_verify_vrf_nonce(byref(meta), index, scrypt_n)


def make_meta(node_id: bytes, atx_id: bytes, challenge: bytes,
              num_units: int, labels_per_unit: int) -> ProofMetadata:
    return ProofMetadata((c_uint8 * 32)(*node_id), (c_uint8 * 32)(*atx_id),
                         (c_uint8 * 32)(*challenge), num_units,
                         labels_per_unit)
