"""PoET membership merkle validation — the non-POST half of
Validator.NIPost.

Restates activation/validation.go:109-174: `validateMerkleProof` /
`validateMultiMerkleProof` delegate to the external spacemeshos/merkle-tree
`ValidatePartialTree` with the PoET membership node hash.  Neither dep is
in-container:
- node hash RESTATED as sha256(0x01 || left || right) (poetShared.
  HashMembershipTreeNode; the 0x01 internal-node domain prefix matches the
  in-repo atxTreeHash convention, wire_v2.go:307-313);
- partial-tree validation RESTATED as the canonical bottom-up algorithm:
  known nodes processed in ascending position per layer, proof nodes
  consumed in that order.  Self-consistency is property-tested against the
  generator in tests/test_poet.py; parity with the real library is
  unpinned until upstream vectors are imported.

CPU-only by design: sha256 merkle audits are microseconds of host work
(SURVEY §8(f)4 "CPU sha256 merkle, cheap")."""
from __future__ import annotations

import hashlib
from typing import Dict, List, Sequence


def hash_membership_tree_node(left: bytes, right: bytes) -> bytes:
    h = hashlib.sha256()
    h.update(b"\x01")
    h.update(left)
    h.update(right)
    return h.digest()


class InvalidProof(ValueError):
    pass


def validate_partial_tree(leaf_indices: Sequence[int],
                          leaves: Sequence[bytes],
                          proof_nodes: Sequence[bytes],
                          expected_root: bytes,
                          num_leaves: int) -> bool:
    """Reconstruct the root from `leaves` at `leaf_indices` (ascending)
    plus `proof_nodes` (consumed in ascending-position order per layer),
    compare with expected_root.  The tree has num_leaves leaves (power of
    two; the PoET membership tree is built padded)."""
    if len(leaf_indices) != len(leaves) or not leaf_indices:
        raise InvalidProof("leaf count mismatch")
    if sorted(set(leaf_indices)) != list(leaf_indices):
        raise InvalidProof("leaf indices must be ascending and unique")
    if num_leaves & (num_leaves - 1) or num_leaves <= 0:
        raise InvalidProof("num_leaves must be a power of two")
    if any(i >= num_leaves for i in leaf_indices):
        raise InvalidProof("leaf index out of range")

    nodes: Dict[int, bytes] = dict(zip(leaf_indices, leaves))
    proof = list(proof_nodes)
    width = num_leaves
    while width > 1:
        nxt: Dict[int, bytes] = {}
        for pos in sorted(nodes):
            if pos in nodes and (pos ^ 1) in nodes and pos & 1:
                continue  # right child handled with its left sibling
            sib = pos ^ 1
            if sib in nodes:
                left = nodes[min(pos, sib)]
                right = nodes[max(pos, sib)]
            else:
                if not proof:
                    raise InvalidProof("proof exhausted")
                sval = proof.pop(0)
                left, right = ((nodes[pos], sval) if pos % 2 == 0
                               else (sval, nodes[pos]))
            nxt[pos // 2] = hash_membership_tree_node(left, right)
        nodes = nxt
        width //= 2
    if proof:
        raise InvalidProof("unconsumed proof nodes")
    return nodes[0] == expected_root


def validate_merkle_proof(leaf: bytes, leaf_index: int,
                          proof_nodes: Sequence[bytes],
                          expected_root: bytes, num_leaves: int) -> bool:
    """Single-leaf form (validateMerkleProof, validation.go:142-149)."""
    return validate_partial_tree([leaf_index], [leaf], proof_nodes,
                                 expected_root, num_leaves)


# ---- reference builder/generator (used by tests and PoET-side tooling) ----

def build_tree(leaves: Sequence[bytes]) -> List[List[bytes]]:
    """Full tree layers, leaf layer first.  Leaves are used raw (the
    membership tree hashes members before insertion on the PoET side)."""
    n = len(leaves)
    if n & (n - 1) or n == 0:
        raise ValueError("power-of-two leaf count required")
    layers = [list(leaves)]
    while len(layers[-1]) > 1:
        prev = layers[-1]
        layers.append([hash_membership_tree_node(prev[i], prev[i + 1])
                       for i in range(0, len(prev), 2)])
    return layers


def generate_partial_proof(layers: List[List[bytes]],
                           leaf_indices: Sequence[int]) -> List[bytes]:
    """Proof nodes in the consumption order of validate_partial_tree."""
    proof: List[bytes] = []
    known = set(leaf_indices)
    for depth in range(len(layers) - 1):
        layer = layers[depth]
        nxt = set()
        for pos in sorted(known):
            if pos ^ 1 in known and pos & 1:
                continue
            if (pos ^ 1) not in known:
                proof.append(layer[pos ^ 1])
            nxt.add(pos // 2)
        known = nxt
    return proof
