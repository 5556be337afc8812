"""PoET membership merkle validation (validation.go:109-174 restatement):
property tests — generated partial proofs validate, any corruption fails."""
import hashlib
import importlib
import random

import pytest

poet = importlib.import_module("go-spacemesh_amd.poet")


def _leaves(n, seed=0):
    return [hashlib.sha256(b"leaf-%d-%d" % (seed, i)).digest()
            for i in range(n)]


def test_single_leaf_proofs_all_positions():
    for n in [1, 2, 4, 8, 32]:
        layers = poet.build_tree(_leaves(n))
        root = layers[-1][0]
        for i in range(n):
            proof = poet.generate_partial_proof(layers, [i])
            assert poet.validate_merkle_proof(layers[0][i], i, proof, root, n)


def test_multi_leaf_proofs_random_subsets():
    rng = random.Random(21)
    for n in [4, 16, 64, 256]:
        layers = poet.build_tree(_leaves(n, seed=n))
        root = layers[-1][0]
        for _ in range(10):
            k = rng.randrange(1, min(n, 9) + 1)
            idx = sorted(rng.sample(range(n), k))
            proof = poet.generate_partial_proof(layers, idx)
            leaves = [layers[0][i] for i in idx]
            assert poet.validate_partial_tree(idx, leaves, proof, root, n)


def test_corruption_fails():
    rng = random.Random(22)
    n = 64
    layers = poet.build_tree(_leaves(n, seed=99))
    root = layers[-1][0]
    idx = [3, 17, 40]
    leaves = [layers[0][i] for i in idx]
    proof = poet.generate_partial_proof(layers, idx)
    # corrupt root
    assert not poet.validate_partial_tree(idx, leaves, proof,
                                          bytes(32), n)
    # corrupt a leaf
    bad = list(leaves)
    bad[1] = bytes(32)
    assert not poet.validate_partial_tree(idx, bad, proof, root, n)
    # corrupt a proof node
    for pos in range(len(proof)):
        badp = list(proof)
        badp[pos] = hashlib.sha256(b"x").digest()
        assert not poet.validate_partial_tree(idx, leaves, badp, root, n)
    # wrong index
    assert not poet.validate_partial_tree([4, 17, 40], leaves, proof, root, n)


def test_malformed_proofs_raise():
    n = 8
    layers = poet.build_tree(_leaves(n, seed=5))
    root = layers[-1][0]
    proof = poet.generate_partial_proof(layers, [2])
    with pytest.raises(poet.InvalidProof):
        poet.validate_partial_tree([2, 1], [layers[0][2], layers[0][1]],
                                   proof, root, n)  # not ascending
    with pytest.raises(poet.InvalidProof):
        poet.validate_partial_tree([2], [layers[0][2]], proof[:-1], root, n)
    with pytest.raises(poet.InvalidProof):
        poet.validate_partial_tree([2], [layers[0][2]], proof + [bytes(32)],
                                   root, n)
    with pytest.raises(poet.InvalidProof):
        poet.validate_partial_tree([9], [bytes(32)], proof, root, n)


def test_adjacent_pair_needs_no_proof():
    layers = poet.build_tree(_leaves(2, seed=7))
    root = layers[-1][0]
    assert poet.validate_partial_tree([0, 1], layers[0], [], root, 2)
