"""Minimal protobuf wire-format codec.

The environment has no generated Envoy protos (no network, no protoc
envoy tree), so the ext-proc messages are encoded/decoded directly at
the wire level. Only what ext-proc needs: varints, length-delimited
fields, and a generic message parser returning (field, wire_type, value)
triples. Field numbers are pinned in extproc.py against
envoy/service/ext_proc/v3/external_processor.proto.
"""

from __future__ import annotations

from typing import Iterator, List, Tuple, Union


def encode_varint(n: int) -> bytes:
    out = bytearray()
    if n < 0:
        n += 1 << 64  # two's complement, 64-bit
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def decode_varint(data: bytes, i: int) -> Tuple[int, int]:
    shift = 0
    val = 0
    while True:
        b = data[i]
        i += 1
        val |= (b & 0x7F) << shift
        if not b & 0x80:
            return val, i
        shift += 7
        if shift > 63:
            raise ValueError("varint too long")


def tag(field: int, wire: int) -> bytes:
    return encode_varint((field << 3) | wire)


def len_field(field: int, payload: bytes) -> bytes:
    """Length-delimited field (submessage / string / bytes)."""
    return tag(field, 2) + encode_varint(len(payload)) + payload


def str_field(field: int, s: str) -> bytes:
    return len_field(field, s.encode("utf-8"))


def varint_field(field: int, n: int) -> bytes:
    """Varint field; 0 is omitted (proto3 default-elision)."""
    if n == 0:
        return b""
    return tag(field, 0) + encode_varint(n)


def bool_field(field: int, v: bool) -> bytes:
    return varint_field(field, 1 if v else 0)


Value = Union[int, bytes]


def parse(data: bytes) -> Iterator[Tuple[int, int, Value]]:
    """Yield (field_number, wire_type, value). Length-delimited values
    are bytes; varints are ints; fixed32/64 raw ints."""
    i = 0
    n = len(data)
    while i < n:
        key, i = decode_varint(data, i)
        field, wire = key >> 3, key & 7
        if wire == 0:
            val, i = decode_varint(data, i)
        elif wire == 2:
            ln, i = decode_varint(data, i)
            val = data[i : i + ln]
            i += ln
        elif wire == 5:
            val = int.from_bytes(data[i : i + 4], "little")
            i += 4
        elif wire == 1:
            val = int.from_bytes(data[i : i + 8], "little")
            i += 8
        else:
            raise ValueError(f"unsupported wire type {wire}")
        yield field, wire, val


def fields(data: bytes) -> List[Tuple[int, int, Value]]:
    return list(parse(data))
