"""Minimal protobuf wire-format encoder/decoder.

Hand-rolled because the image has the protobuf runtime but no protoc/
grpcio-tools codegen. Used for the pprof profile format, the Parca
ProfileStore/Debuginfo gRPC messages, and the Prometheus remote-write
payload (reference shapes: buf.build parca protos).

Wire types: 0 = varint, 1 = fixed64, 2 = length-delimited, 5 = fixed32.
"""

from __future__ import annotations

import struct
from typing import Iterator, List, Tuple, Union


# Single-byte varints cover the overwhelming majority of encoded values
# (tags, string-table/location indexes, small lengths); pre-building
# them removes the hot path's loop + bytes() allocation. The flush
# encoder emits millions of varints per batch on busy nodes.
_SINGLE = [bytes((i,)) for i in range(0x80)]
_TWO = {}


def encode_varint(value: int) -> bytes:
    """Encode a non-negative integer as a base-128 varint."""
    if 0 <= value < 0x80:
        return _SINGLE[value]
    if 0 <= value < 0x4000:
        cached = _TWO.get(value)
        if cached is None:
            cached = bytes(((value & 0x7F) | 0x80, value >> 7))
            _TWO[value] = cached
        return cached
    if value < 0:
        # Negative int64 values are encoded as their 64-bit two's complement,
        # which takes the full 10 bytes on the wire.
        value &= (1 << 64) - 1
    out = bytearray()
    while True:
        bits = value & 0x7F
        value >>= 7
        if value:
            out.append(bits | 0x80)
        else:
            out.append(bits)
            return bytes(out)


def decode_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    """Decode a varint at ``pos``; returns (value, new_pos)."""
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7
        if shift > 70:
            raise ValueError("varint too long")


def zigzag(value: int) -> int:
    return (value << 1) ^ (value >> 63)


def unzigzag(value: int) -> int:
    return (value >> 1) ^ -(value & 1)


def to_int64(value: int) -> int:
    """Interpret an unsigned varint value as a signed int64."""
    if value >= 1 << 63:
        value -= 1 << 64
    return value


class Writer:
    """Append-only protobuf message writer."""

    def __init__(self) -> None:
        self._parts: List[bytes] = []

    def tag(self, field: int, wire_type: int) -> None:
        self._parts.append(encode_varint((field << 3) | wire_type))

    def varint(self, field: int, value: int) -> None:
        if value == 0:
            return
        self.tag(field, 0)
        self._parts.append(encode_varint(value))

    def varint_keep_zero(self, field: int, value: int) -> None:
        self.tag(field, 0)
        self._parts.append(encode_varint(value))

    def bool(self, field: int, value: bool) -> None:
        if value:
            self.varint_keep_zero(field, 1)

    def fixed64(self, field: int, value: int) -> None:
        if value == 0:
            return
        self.tag(field, 1)
        self._parts.append(struct.pack("<Q", value & ((1 << 64) - 1)))

    def double(self, field: int, value: float) -> None:
        if value == 0.0:
            return
        self.tag(field, 1)
        self._parts.append(struct.pack("<d", value))

    def bytes(self, field: int, value: Union[bytes, bytearray, memoryview]) -> None:
        if not value:
            return
        self.tag(field, 2)
        self._parts.append(encode_varint(len(value)))
        self._parts.append(bytes(value))

    def string(self, field: int, value: str) -> None:
        if value:
            self.bytes(field, value.encode("utf-8"))

    def message(self, field: int, writer: "Writer") -> None:
        self.bytes(field, writer.getvalue())

    def packed_varints(self, field: int, values) -> None:
        if not values:
            return
        # Inline bulk encode: one bytearray, no per-value bytes objects.
        body = bytearray()
        append = body.append
        for v in values:
            if 0 <= v < 0x80:
                append(v)
                continue
            if v < 0:
                v += 1 << 64
            while True:
                bits = v & 0x7F
                v >>= 7
                if v:
                    append(bits | 0x80)
                else:
                    append(bits)
                    break
        self.bytes(field, body)

    def getvalue(self) -> bytes:
        return b"".join(self._parts)

    def __len__(self) -> int:
        return sum(len(p) for p in self._parts)


def iter_fields(buf: bytes) -> Iterator[Tuple[int, int, Union[int, bytes]]]:
    """Yield (field_number, wire_type, value) triples from a message body.

    Length-delimited fields yield the raw bytes; varints yield ints;
    fixed64/fixed32 yield ints.
    """
    pos = 0
    end = len(buf)
    while pos < end:
        key, pos = decode_varint(buf, pos)
        field, wire_type = key >> 3, key & 7
        if wire_type == 0:
            value, pos = decode_varint(buf, pos)
        elif wire_type == 1:
            (value,) = struct.unpack_from("<Q", buf, pos)
            pos += 8
        elif wire_type == 2:
            length, pos = decode_varint(buf, pos)
            value = buf[pos : pos + length]
            pos += length
        elif wire_type == 5:
            (value,) = struct.unpack_from("<I", buf, pos)
            pos += 4
        else:
            raise ValueError(f"unsupported wire type {wire_type}")
        yield field, wire_type, value


def decode_packed_varints(buf: bytes) -> List[int]:
    out = []
    pos = 0
    while pos < len(buf):
        v, pos = decode_varint(buf, pos)
        out.append(v)
    return out
