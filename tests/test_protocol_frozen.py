"""The restated post-rs protocol semantics, frozen: every derivation the
oracle defines (k2pow, cipher keys, K3 subset, difficulties, a full toy
proof) must keep producing the committed fixtures.  This is the guard that
the RESTATED constants cannot drift between rounds."""
import ctypes
import hashlib
import json
import os

GOLDEN = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "golden.json")))
FX = GOLDEN["protocol_frozen"]


def test_k2pow_frozen(oracle):
    for v in FX["k2pow"]:
        got = oracle.lib.oracle_k2pow(bytes.fromhex(v["challenge"]),
                                      v["group"],
                                      bytes.fromhex(v["difficulty"]))
        assert got == v["pow"], v


def test_cipher_keys_frozen(oracle):
    for v in FX["cipher_keys"]:
        out = ctypes.create_string_buffer(16)
        oracle.lib.oracle_prove_cipher_key(bytes.fromhex(v["challenge"]),
                                           v["cipher"], v["group_pow"], out)
        assert out.raw.hex() == v["key"], v


def test_subset_frozen(oracle):
    for v in FX["subset"]:
        pos = (ctypes.c_uint32 * v["k3"])()
        seed = bytes.fromhex(v["seed"])
        oracle.lib.oracle_subset(v["k2"], v["k3"], seed, len(seed), pos)
        assert list(pos) == v["positions"], v


def test_difficulties_frozen(oracle):
    for v in FX["vrf_difficulty"]:
        out = ctypes.create_string_buffer(32)
        oracle.lib.oracle_vrf_difficulty(v["num_labels"], out)
        assert out.raw.hex() == v["difficulty"], v
    for v in FX["proving_difficulty"]:
        got = oracle.lib.oracle_proving_difficulty(v["k1"], v["num_labels"])
        assert got == v["difficulty"], v


def test_proof_fixture_frozen(oracle):
    fx = FX["proof_fixture"]
    commit = oracle.commitment(bytes.fromhex(fx["node_id"]),
                               bytes.fromhex(fx["atx_id"]))
    labels, best = oracle.init_range(commit, 0, fx["num_labels"],
                                     fx["scrypt_n"])
    assert hashlib.sha256(labels).hexdigest() == fx["labels_sha256"]
    assert best.index == fx["vrf_nonce_index"]
    assert bytes(best.label).hex() == fx["vrf_nonce_label"]
    proof = oracle.prove(labels, fx["num_labels"],
                         bytes.fromhex(fx["challenge"]), fx["k1"], fx["k2"],
                         fx["nonces"], bytes.fromhex(fx["pow_difficulty"]))
    assert proof.nonce == fx["proof_nonce"]
    assert proof.pow == fx["proof_pow"]
    assert bytes(proof.indices[:proof.indices_len]).hex() == \
        fx["proof_indices"]
