"""Wire formats of the POST surface.

- PostV1 scale codec: restated from the reference's OWN generated encoder
  (activation/wire/wire_v1_scale.go:157-207): Compact32(nonce),
  ByteSliceWithLimit(indices, 800), Compact64(pow).  go-scale follows the
  SCALE compact-integer scheme (parity spec), pinned here by the spec's
  known answers in tests/test_wire.py.
- ATX merkle-leaf Root() of a PostV1 (activation/wire/wire_v1.go:47-63):
  three leaves (LE32 nonce, indices, LE64 pow), node hash =
  blake3(0x01 || left || right) (wire_v2.go:307-313).  The merkle-tree
  library itself (spacemeshos/merkle-tree, go.mod dep) is NOT in-container;
  its unbalanced-tree behavior is RESTATED as zero-padding the leaf level
  to the next power of two (the reference's own v2 code pads leaf counts
  manually to powers of two, wire_v2.go:300-302) — parity for Root() is
  therefore unpinned until upstream vectors are imported.
- postdata_metadata.json reader matching the engine's writer
  (shared.PostMetadata usage, activation/post_test.go:305-309).
"""
from __future__ import annotations

import base64
import dataclasses
import json
import os
import struct
from typing import List, Optional, Tuple


# ---------------- SCALE compact integers (parity spec) ----------------

def encode_compact(v: int) -> bytes:
    if v < 0:
        raise ValueError("negative")
    if v < (1 << 6):
        return bytes([v << 2])
    if v < (1 << 14):
        return struct.pack("<H", (v << 2) | 0b01)
    if v < (1 << 30):
        return struct.pack("<I", (v << 2) | 0b10)
    data = v.to_bytes((v.bit_length() + 7) // 8, "little")
    if len(data) < 4:
        data = data.ljust(4, b"\0")
    return bytes([0b11 | ((len(data) - 4) << 2)]) + data


def decode_compact(buf: bytes, off: int = 0) -> Tuple[int, int]:
    """Returns (value, bytes consumed)."""
    b0 = buf[off]
    mode = b0 & 0b11
    if mode == 0b00:
        return b0 >> 2, 1
    if mode == 0b01:
        return struct.unpack_from("<H", buf, off)[0] >> 2, 2
    if mode == 0b10:
        return struct.unpack_from("<I", buf, off)[0] >> 2, 4
    n = (b0 >> 2) + 4
    return int.from_bytes(buf[off + 1:off + 1 + n], "little"), 1 + n


# ---------------- PostV1 (wire_v1_scale.go:157-207) ----------------

@dataclasses.dataclass
class PostV1:
    nonce: int
    indices: bytes
    pow: int

    MAX_INDICES = 800  # wire_v1.go:43

    def encode(self) -> bytes:
        if len(self.indices) > self.MAX_INDICES:
            raise ValueError("indices exceed the 800-byte wire cap")
        return (encode_compact(self.nonce) +
                encode_compact(len(self.indices)) + self.indices +
                encode_compact(self.pow))

    @classmethod
    def decode(cls, buf: bytes) -> "PostV1":
        off = 0
        nonce, n = decode_compact(buf, off)
        off += n
        ln, n = decode_compact(buf, off)
        off += n
        if ln > cls.MAX_INDICES:
            raise ValueError("indices exceed the 800-byte wire cap")
        indices = bytes(buf[off:off + ln])
        off += ln
        pw, n = decode_compact(buf, off)
        off += n
        if off != len(buf):
            raise ValueError("trailing bytes")
        return cls(nonce=nonce, indices=indices, pow=pw)

    def root(self) -> bytes:
        """ATX merkle leaf (wire_v1.go:47-63).  See module docstring for
        the padding restatement status."""
        leaves = [struct.pack("<I", self.nonce), self.indices,
                  struct.pack("<Q", self.pow)]
        return merkle_root(leaves)


def atx_tree_hash(left: bytes, right: bytes) -> bytes:
    """blake3(0x01 || l || r) — wire_v2.go:307-313."""
    import gsm_amd
    return gsm_amd.Engine().selftest_blake3(b"\x01" + left + right)


def merkle_root(leaves: List[bytes]) -> bytes:
    nodes = list(leaves)
    size = 1
    while size < len(nodes):
        size *= 2
    nodes += [bytes(32)] * (size - len(nodes))  # RESTATED zero padding
    while len(nodes) > 1:
        nodes = [atx_tree_hash(nodes[i], nodes[i + 1])
                 for i in range(0, len(nodes), 2)]
    return nodes[0]


# ---------------- postdata_metadata.json ----------------

@dataclasses.dataclass
class PostMetadata:
    node_id: bytes
    commitment_atx_id: bytes
    labels_per_unit: int
    num_units: int
    max_file_size: int
    scrypt_n: int
    nonce: Optional[int] = None
    nonce_value: Optional[bytes] = None

    @classmethod
    def read(cls, data_dir: str) -> "PostMetadata":
        with open(os.path.join(data_dir, "postdata_metadata.json")) as f:
            d = json.load(f)
        return cls(
            node_id=base64.b64decode(d["NodeId"]),
            commitment_atx_id=base64.b64decode(d["CommitmentAtxId"]),
            labels_per_unit=d["LabelsPerUnit"],
            num_units=d["NumUnits"],
            max_file_size=d["MaxFileSize"],
            scrypt_n=d.get("Scrypt", {}).get("N", 8192),
            nonce=d.get("Nonce"),
            nonce_value=base64.b64decode(d["NonceValue"])
            if "NonceValue" in d else None,
        )

    def num_labels(self) -> int:
        return self.num_units * self.labels_per_unit
