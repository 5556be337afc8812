"""Oracle POST semantics: init -> prove -> verify round trips and edge cases.

Mirrors the reference's own pinning strategy: real-compute
round-trip tests at shrunk parameters (scryptN=2, toy K1/K2/K3 — cf.
activation/e2e/nipost_test.go:76-84, post_test.go:356-357) plus the
adversarial invalid-index case
(systest/tests/distributed_post_verification_test.go:253-267).
"""
import random

import pytest

from oracle import Proof, make_meta

NODE = bytes([0xA5]) * 32
ATX = bytes([0x5A]) * 32
CHALLENGE = bytes(32)  # shared.ZeroChallenge semantics (activation/nipost.go:165)
POW_DIFF = bytes([0x0F]) + bytes([0xFF]) * 31
NU, LPU, N = 2, 64, 2
K1, K2, K3, NONCES = 12, 8, 4, 16


@pytest.fixture(scope="module")
def setup(oracle):
    commitment = oracle.commitment(NODE, ATX)
    labels, best = oracle.init_range(commitment, 0, NU * LPU, N)
    proof = oracle.prove(labels, NU * LPU, CHALLENGE, K1, K2, NONCES,
                         POW_DIFF)
    meta = make_meta(NODE, ATX, CHALLENGE, NU, LPU)
    return commitment, labels, best, proof, meta


def test_roundtrip_full_k2(oracle, setup):
    _, _, _, proof, meta = setup
    rc, _ = oracle.verify(proof, meta, N, K1, K2, K2, None, -1, POW_DIFF)
    assert rc == 0


def test_roundtrip_subset_k3(oracle, setup):
    # validation.go:206-209: Subset(K3, seed); different seeds, all pass
    _, _, _, proof, meta = setup
    for seed in [b"a", b"peer-id-0", bytes(32)]:
        rc, _ = oracle.verify(proof, meta, N, K1, K2, K3, seed, -1, POW_DIFF)
        assert rc == 0


def test_roundtrip_selected_index(oracle, setup):
    # malfeasance.go:161-169: verify exactly one chosen index position
    _, _, _, proof, meta = setup
    for pos in range(K2):
        rc, _ = oracle.verify(proof, meta, N, K1, K2, K2, None, pos, POW_DIFF)
        assert rc == 0


def test_corrupt_index_detected(oracle, setup):
    _, _, _, proof, meta = setup
    bad = Proof.from_buffer_copy(proof)
    bad.indices[0] ^= 0xFF
    rc, _ = oracle.verify(bad, meta, N, K1, K2, K2, None, -1, POW_DIFF)
    assert rc in (1, 3)  # invalid index (or malformed if out of range)


def test_bad_pow_detected(oracle, setup):
    _, _, _, proof, meta = setup
    bad = Proof.from_buffer_copy(proof)
    bad.pow += 1
    rc, _ = oracle.verify(bad, meta, N, K1, K2, K2, None, -1, POW_DIFF)
    assert rc == 2


def test_wrong_nonce_detected(oracle, setup):
    _, _, _, proof, meta = setup
    bad = Proof.from_buffer_copy(proof)
    bad.nonce = (bad.nonce + 1) % NONCES
    rc, _ = oracle.verify(bad, meta, N, K1, K2, K2, None, -1, POW_DIFF)
    assert rc != 0


def test_wrong_identity_detected(oracle, setup):
    _, _, _, proof, _ = setup
    other = make_meta(bytes([1]) * 32, ATX, CHALLENGE, NU, LPU)
    rc, _ = oracle.verify(proof, other, N, K1, K2, K2, None, -1, POW_DIFF)
    assert rc != 0


def test_vrf_nonce_roundtrip(oracle, setup):
    _, _, best, _, meta = setup
    assert best.found
    assert oracle.verify_vrf_nonce(meta, best.index, N) == 0


def test_vrf_nonce_is_global_min(oracle, setup):
    commitment, _, best, _, _ = setup
    # the nonce the initializer reports is the argmin full label
    full = [oracle.label(commitment, i, N) for i in range(NU * LPU)]
    argmin = min(range(NU * LPU), key=lambda i: full[i])
    assert best.index == argmin
    assert bytes(best.label) == full[argmin]


def test_init_range_resume_split(oracle, setup):
    # Resume semantics (activation/post.go:267-271): computing [0,T) in one
    # go equals computing [0,a) then [a,T) — labels and the tracked nonce.
    commitment, labels, best, _, _ = setup
    from oracle import VrfNonce
    import ctypes
    a = 37
    l1, b1 = oracle.init_range(commitment, 0, a, N)
    # continue with the same best-so-far state
    n2 = a
    total = NU * LPU
    out2 = ctypes.create_string_buffer((total - a) * 16)
    diff = bytes([0xFF]) * 32
    rc = oracle.lib.oracle_init_range(commitment, a, total, N, out2, diff,
                                      ctypes.byref(b1))
    assert rc == 0
    assert l1 + out2.raw == labels
    assert b1.index == best.index


def test_labels_match_file_truncation(oracle, setup):
    commitment, labels, _, _, _ = setup
    # stored label = first 16 bytes of the 32-byte scrypt output
    for i in [0, 1, NU * LPU - 1]:
        assert labels[i * 16:(i + 1) * 16] == \
            oracle.label(commitment, i, N)[:16]


def test_pack_unpack_roundtrip(oracle):
    import ctypes
    rng = random.Random(7)
    for num_labels in [64, 2**22, 2**36]:
        bpi = oracle.lib.oracle_bits_per_index(num_labels)
        assert bpi == max(1, (num_labels - 1).bit_length())
        k = 37
        idx = [rng.randrange(num_labels) for _ in range(k)]
        arr = (ctypes.c_uint64 * k)(*idx)
        out = ctypes.create_string_buffer(800)
        nbytes = oracle.lib.oracle_pack_indices(arr, k, bpi, out)
        assert nbytes == (k * bpi + 7) // 8
        assert nbytes <= 800  # wire_v1.go:43 cap
        back = (ctypes.c_uint64 * k)()
        oracle.lib.oracle_unpack_indices(out.raw, k, bpi, back)
        assert list(back) == idx


def test_subset_deterministic_and_in_range(oracle):
    import ctypes
    k2, k3 = 37, 5
    pos = (ctypes.c_uint32 * k3)()
    oracle.lib.oracle_subset(k2, k3, b"seed-x", 6, pos)
    first = list(pos)
    oracle.lib.oracle_subset(k2, k3, b"seed-x", 6, pos)
    assert list(pos) == first
    assert len(set(first)) == k3
    assert all(0 <= p < k2 for p in first)
    oracle.lib.oracle_subset(k2, k3, b"seed-y", 6, pos)
    assert list(pos) != first  # overwhelmingly likely


def test_proving_difficulty_formula(oracle):
    # floor(k1 * 2^64 / num_labels)
    assert oracle.lib.oracle_proving_difficulty(26, 1 << 36) == \
        (26 << 64) // (1 << 36)
    assert oracle.lib.oracle_proving_difficulty(1, 1) == 2**64 - 1  # clamp


def test_vrf_difficulty_formula(oracle):
    import ctypes
    out = ctypes.create_string_buffer(32)
    for n in [128, 1 << 22, 1 << 36]:
        oracle.lib.oracle_vrf_difficulty(n, out)
        assert int.from_bytes(out.raw, "big") == (16 << 256) // n
    oracle.lib.oracle_vrf_difficulty(16, out)
    assert out.raw == b"\xff" * 32


def test_k2pow_verify_matches_search(oracle):
    pow_ = oracle.lib.oracle_k2pow(CHALLENGE, 0, POW_DIFF)
    assert oracle.lib.oracle_k2pow_verify(CHALLENGE, 0, pow_, POW_DIFF) == 0
    assert oracle.lib.oracle_k2pow_verify(CHALLENGE, 0, pow_ + 10**6,
                                          POW_DIFF) in (0, -1)
    # minimality: all smaller pows fail
    for p in range(pow_):
        assert oracle.lib.oracle_k2pow_verify(CHALLENGE, 0, p, POW_DIFF) == -1


def test_prove_rejects_bad_nonce_count(oracle, setup):
    _, labels, _, _, _ = setup
    proof = Proof()
    import ctypes
    rc = oracle.lib.oracle_prove(labels, NU * LPU, CHALLENGE, K1, K2, 7,
                                 POW_DIFF, ctypes.byref(proof))
    assert rc != 0  # nonces must be a multiple of the nonce group
