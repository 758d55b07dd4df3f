"""Minimal protobuf wire-format codec (no protobuf/onnx dependency).

ONNX models are protobuf messages; this module provides just enough of the
wire format (varint, length-delimited, fixed32, packed repeated scalars) to
serialize the ModelProto subset the exporter emits and to parse it back for
the self-contained evaluator/tests.

Wire format reference: protobuf encoding spec (public, stable since proto2).
"""

from __future__ import annotations

import struct
from typing import Dict, List, Tuple


def varint(n: int) -> bytes:
    """Unsigned LEB128. Negative ints are two's-complement 64-bit (protobuf
    int32/int64 semantics)."""
    if n < 0:
        n &= (1 << 64) - 1
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def tag(field: int, wire_type: int) -> bytes:
    return varint((field << 3) | wire_type)


def f_varint(field: int, value: int) -> bytes:
    return tag(field, 0) + varint(value)


def f_bytes(field: int, value: bytes) -> bytes:
    return tag(field, 2) + varint(len(value)) + value


def f_string(field: int, value: str) -> bytes:
    return f_bytes(field, value.encode("utf-8"))


def f_float(field: int, value: float) -> bytes:
    return tag(field, 5) + struct.pack("<f", value)


def f_packed_floats(field: int, values) -> bytes:
    payload = struct.pack(f"<{len(values)}f", *values)
    return f_bytes(field, payload)


def f_packed_varints(field: int, values) -> bytes:
    payload = b"".join(varint(int(v)) for v in values)
    return f_bytes(field, payload)


# ---------------------------------------------------------------------------
# reader (generic: field -> list of raw values in order)
# ---------------------------------------------------------------------------


def read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7


def parse_message(buf: bytes) -> Dict[int, List[Tuple[int, object]]]:
    """Parse one message into {field_number: [(wire_type, raw_value), ...]}.
    Length-delimited values stay bytes; varints stay unsigned ints."""
    fields: Dict[int, List[Tuple[int, object]]] = {}
    pos = 0
    while pos < len(buf):
        key, pos = read_varint(buf, pos)
        field, wt = key >> 3, key & 7
        if wt == 0:
            v, pos = read_varint(buf, pos)
        elif wt == 1:
            v = buf[pos:pos + 8]
            pos += 8
        elif wt == 2:
            ln, pos = read_varint(buf, pos)
            v = buf[pos:pos + ln]
            pos += ln
        elif wt == 5:
            v = buf[pos:pos + 4]
            pos += 4
        else:
            raise ValueError(f"unsupported wire type {wt}")
        fields.setdefault(field, []).append((wt, v))
    return fields


def as_int64(raw: int) -> int:
    """Interpret an unsigned varint as two's-complement int64."""
    return raw - (1 << 64) if raw >= (1 << 63) else raw


def unpack_floats(payload: bytes) -> List[float]:
    return list(struct.unpack(f"<{len(payload) // 4}f", payload))


def unpack_varints(payload: bytes) -> List[int]:
    out = []
    pos = 0
    while pos < len(payload):
        v, pos = read_varint(payload, pos)
        out.append(as_int64(v))
    return out
