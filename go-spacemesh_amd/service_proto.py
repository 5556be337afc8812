"""spacemesh.v1.PostService message codec (hand-rolled protobuf wire format).

The reference's node<->post-service boundary is the gRPC bidi stream
`spacemesh.v1.PostService/Register` (api v1.55.0, go.mod:42; server side
api/grpcserver/post_service.go:91-141, client protocol post_client.go:69-143):
the service dials the node and calls Register; the node sends
`NodeRequest{Metadata | GenProof{challenge}}` down the response stream and
the service answers with `ServiceResponse{Metadata | GenProof{status,
proof, metadata}}`.

The spacemeshos/api protobuf definitions are NOT in-container, so the FIELD
NUMBERS below are PROVISIONAL (self-consistent between our node shim and
service; swap this table for the real descriptors when spacemeshos/api
v1.55.0 is importable).  Message CONTENT follows the call sites cited above.
"""
from __future__ import annotations

import dataclasses
from typing import Optional, Tuple


# ---------------- protobuf wire primitives ----------------

def _varint(v: int) -> bytes:
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _read_varint(buf: bytes, off: int) -> Tuple[int, int]:
    v = 0
    shift = 0
    while True:
        b = buf[off]
        off += 1
        v |= (b & 0x7F) << shift
        if not b & 0x80:
            return v, off
        shift += 7


def _field_bytes(num: int, payload: bytes) -> bytes:
    return _varint((num << 3) | 2) + _varint(len(payload)) + payload


def _field_uint(num: int, v: int) -> bytes:
    return _varint(num << 3) + _varint(v)


def _iter_fields(buf: bytes):
    off = 0
    while off < len(buf):
        tag, off = _read_varint(buf, off)
        num, wt = tag >> 3, tag & 7
        if wt == 0:
            v, off = _read_varint(buf, off)
            yield num, v
        elif wt == 2:
            ln, off = _read_varint(buf, off)
            yield num, bytes(buf[off:off + ln])
            off += ln
        else:
            raise ValueError(f"unsupported wire type {wt}")


# ---------------- messages ----------------

GEN_PROOF_STATUS_OK = 1
GEN_PROOF_STATUS_IN_PROGRESS = 2
GEN_PROOF_STATUS_ERROR = 3


@dataclasses.dataclass
class Metadata:
    """ServiceResponse.Metadata (post_client.go:124-141 field set)."""
    node_id: bytes
    commitment_atx_id: bytes
    nonce: Optional[int]
    num_units: int
    labels_per_unit: int

    def encode(self) -> bytes:
        out = _field_bytes(1, self.node_id)
        out += _field_bytes(2, self.commitment_atx_id)
        if self.nonce is not None:
            out += _field_uint(3, self.nonce)
        out += _field_uint(4, self.num_units)
        out += _field_uint(5, self.labels_per_unit)
        return out

    @classmethod
    def decode(cls, buf: bytes) -> "Metadata":
        d = {"nonce": None}
        for num, v in _iter_fields(buf):
            if num == 1:
                d["node_id"] = v
            elif num == 2:
                d["commitment_atx_id"] = v
            elif num == 3:
                d["nonce"] = v
            elif num == 4:
                d["num_units"] = v
            elif num == 5:
                d["labels_per_unit"] = v
        return cls(**d)


@dataclasses.dataclass
class Proof:
    nonce: int
    indices: bytes
    pow: int

    def encode(self) -> bytes:
        return (_field_uint(1, self.nonce) + _field_bytes(2, self.indices) +
                _field_uint(3, self.pow))

    @classmethod
    def decode(cls, buf: bytes) -> "Proof":
        d = {}
        for num, v in _iter_fields(buf):
            d[{1: "nonce", 2: "indices", 3: "pow"}[num]] = v
        return cls(**d)


@dataclasses.dataclass
class NodeRequest:
    """metadata request (gen_proof None) or GenProof{challenge}."""
    gen_proof_challenge: Optional[bytes] = None

    def encode(self) -> bytes:
        if self.gen_proof_challenge is None:
            return _field_bytes(1, b"")
        return _field_bytes(2, _field_bytes(1, self.gen_proof_challenge))

    @classmethod
    def decode(cls, buf: bytes) -> "NodeRequest":
        for num, v in _iter_fields(buf):
            if num == 1:
                return cls()
            if num == 2:
                ch = b""
                for n2, v2 in _iter_fields(v):
                    if n2 == 1:
                        ch = v2
                return cls(gen_proof_challenge=ch)
        return cls()


@dataclasses.dataclass
class ServiceResponse:
    metadata: Optional[Metadata] = None
    gen_proof_status: Optional[int] = None
    gen_proof_proof: Optional[Proof] = None

    def encode(self) -> bytes:
        if self.metadata is not None:
            return _field_bytes(1, self.metadata.encode())
        gp = _field_uint(1, self.gen_proof_status or 0)
        if self.gen_proof_proof is not None:
            gp += _field_bytes(2, self.gen_proof_proof.encode())
        return _field_bytes(2, gp)

    @classmethod
    def decode(cls, buf: bytes) -> "ServiceResponse":
        for num, v in _iter_fields(buf):
            if num == 1:
                return cls(metadata=Metadata.decode(v))
            if num == 2:
                status = None
                proof = None
                for n2, v2 in _iter_fields(v):
                    if n2 == 1:
                        status = v2
                    elif n2 == 2:
                        proof = Proof.decode(v2)
                return cls(gen_proof_status=status, gen_proof_proof=proof)
        raise ValueError("empty ServiceResponse")


REGISTER_METHOD = "/spacemesh.v1.PostService/Register"
