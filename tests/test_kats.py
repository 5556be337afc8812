"""Known-answer tests pinning the oracle's primitives.

scrypt/PBKDF2/SHA-256 are pinned against RFC 7914 vectors and Python
hashlib/hmac (OpenSSL); AES-128 against FIPS-197; BLAKE3 against the official
vectors for the two inputs embedded in tests/golden/golden.json.
"""
import hashlib
import hmac as hmac_mod
import json
import os
import random

GOLDEN = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                     "golden.json")))


def test_sha256_vs_hashlib(oracle):
    rng = random.Random(1)
    for n in [0, 1, 55, 56, 63, 64, 65, 127, 128, 1000]:
        msg = bytes(rng.randrange(256) for _ in range(n))
        assert oracle.sha256(msg) == hashlib.sha256(msg).digest()


def test_hmac_vs_hashlib(oracle):
    rng = random.Random(2)
    for klen in [0, 16, 40, 64, 65, 100]:
        key = bytes(rng.randrange(256) for _ in range(klen))
        msg = bytes(rng.randrange(256) for _ in range(77))
        assert oracle.hmac_sha256(key, msg) == hmac_mod.new(
            key, msg, hashlib.sha256).digest()


def test_pbkdf2_vs_hashlib(oracle):
    rng = random.Random(3)
    for iters, dklen in [(1, 32), (1, 128), (2, 64), (10, 100)]:
        pw = bytes(rng.randrange(256) for _ in range(40))
        salt = bytes(rng.randrange(256) for _ in range(rng.randrange(0, 64)))
        assert oracle.pbkdf2(pw, salt, iters, dklen) == hashlib.pbkdf2_hmac(
            "sha256", pw, salt, iters, dklen)


def test_scrypt_rfc7914(oracle):
    for v in GOLDEN["scrypt_rfc7914"]:
        got = oracle.scrypt(v["P"].encode(), v["S"].encode(), v["N"], v["r"],
                            v["p"], v["dkLen"])
        assert got.hex() == v["out"]


def test_scrypt_post_params_golden(oracle):
    for v in GOLDEN["scrypt_post_params_openssl"]:
        got = oracle.scrypt(bytes.fromhex(v["P_hex"]), b"", v["N"], 1, 1, 32)
        assert got.hex() == v["out"]


def test_scrypt_vs_openssl_random(oracle):
    rng = random.Random(4)
    for _ in range(4):
        pw = bytes(rng.randrange(256) for _ in range(40))
        for n, r, p in [(2, 1, 1), (16, 2, 2), (8192, 1, 1)]:
            got = oracle.scrypt(pw, b"", n, r, p, 32)
            ref = hashlib.scrypt(pw, salt=b"", n=n, r=r, p=p, dklen=32,
                                 maxmem=2**27)
            assert got == ref


def test_blake3_official_vectors(oracle):
    for v in GOLDEN["blake3_official"]:
        assert oracle.blake3(bytes.fromhex(v["input_hex"])).hex() == v["out"]


def test_blake3_xof_prefix_consistency(oracle):
    # XOF output must be prefix-consistent and extend the 32-byte hash.
    msg = b"prefix-consistency"
    h32 = oracle.blake3(msg)
    h96 = oracle.blake3(msg, outlen=96)
    assert h96[:32] == h32
    assert oracle.blake3(msg, outlen=64) == h96[:64]


def test_aes128_fips197(oracle):
    for v in GOLDEN["aes128_fips197"]:
        got = oracle.aes128(bytes.fromhex(v["key"]), bytes.fromhex(v["pt"]))
        assert got.hex() == v["ct"]


def test_label_golden_openssl(oracle):
    lv = GOLDEN["labels_openssl"]
    commitment = oracle.commitment(bytes.fromhex(lv["node_id"]),
                                   bytes.fromhex(lv["atx_id"]))
    assert commitment.hex() == lv["commitment"]
    for v in lv["labels"]:
        if v["N"] > 1024:
            continue  # keep the CPU suite fast; N=8192 covered in post tests
        assert oracle.label(commitment, v["index"], v["N"]).hex() == v["full"]


def test_label_golden_openssl_n8192(oracle):
    lv = GOLDEN["labels_openssl"]
    commitment = bytes.fromhex(lv["commitment"])
    vecs = [v for v in lv["labels"] if v["N"] == 8192][:3]
    for v in vecs:
        assert oracle.label(commitment, v["index"], v["N"]).hex() == v["full"]
