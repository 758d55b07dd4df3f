"""Minimal Avro Object Container File reader/writer (no external deps).

Implements exactly what the model format needs (Avro 1.x spec):
zigzag-varint ints/longs, IEEE float/double, unions, arrays, records,
container framing (magic, metadata map, sync-marked blocks) with codecs
``null``, ``deflate`` (raw zlib) for read+write and ``snappy`` for read
(the reference's Spark writer emits snappy —
IsolationForestModelReadWrite.scala:248 via spark-avro defaults; a pure-
Python snappy decoder below lets us load those fixtures without the
python-snappy wheel).

The record schemas are the reference's on-disk compatibility surface
(SURVEY.md §2.1 "Serialized formats"):

standard:  {treeID:int, nodeData: union[{id,leftChild,rightChild,
            splitAttribute:int, splitValue:double, numInstances:long}, null]}
extended:  {treeID:int, extendedNodeData: union[{id,leftChild,rightChild,
            indices: union[array<int>,null], weights: union[array<float>,null],
            offset:double, numInstances:long}, null]}
"""

from __future__ import annotations

import io
import json
import os
import struct
import zlib

MAGIC = b"Obj\x01"

STANDARD_SCHEMA = {
    "type": "record",
    "name": "topLevelRecord",
    "fields": [
        {"name": "treeID", "type": "int"},
        {
            "name": "nodeData",
            "type": [
                {
                    "type": "record",
                    "name": "nodeData",
                    "namespace": ".nodeData",
                    "fields": [
                        {"name": "id", "type": "int"},
                        {"name": "leftChild", "type": "int"},
                        {"name": "rightChild", "type": "int"},
                        {"name": "splitAttribute", "type": "int"},
                        {"name": "splitValue", "type": "double"},
                        {"name": "numInstances", "type": "long"},
                    ],
                },
                "null",
            ],
        },
    ],
}

EXTENDED_SCHEMA = {
    "type": "record",
    "name": "topLevelRecord",
    "fields": [
        {"name": "treeID", "type": "int"},
        {
            "name": "extendedNodeData",
            "type": [
                {
                    "type": "record",
                    "name": "extendedNodeData",
                    "namespace": "topLevelRecord",
                    "fields": [
                        {"name": "id", "type": "int"},
                        {"name": "leftChild", "type": "int"},
                        {"name": "rightChild", "type": "int"},
                        {"name": "indices", "type": [{"type": "array", "items": "int"}, "null"]},
                        {"name": "weights", "type": [{"type": "array", "items": "float"}, "null"]},
                        {"name": "offset", "type": "double"},
                        {"name": "numInstances", "type": "long"},
                    ],
                },
                "null",
            ],
        },
    ],
}


# ---------------------------------------------------------------------------
# primitive codecs
# ---------------------------------------------------------------------------


def zigzag_encode(n: int) -> bytes:
    z = (n << 1) ^ (n >> 63) if n < 0 else (n << 1)
    out = bytearray()
    while True:
        b = z & 0x7F
        z >>= 7
        if z:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


class Reader:
    def __init__(self, buf: bytes):
        self.buf = buf
        self.pos = 0

    def read_long(self) -> int:
        shift = 0
        result = 0
        while True:
            b = self.buf[self.pos]
            self.pos += 1
            result |= (b & 0x7F) << shift
            if not (b & 0x80):
                break
            shift += 7
        return (result >> 1) ^ -(result & 1)

    def read_bytes(self, n: int) -> bytes:
        out = self.buf[self.pos : self.pos + n]
        self.pos += n
        return out

    def read_float(self) -> float:
        return struct.unpack("<f", self.read_bytes(4))[0]

    def read_double(self) -> float:
        return struct.unpack("<d", self.read_bytes(8))[0]

    def at_end(self) -> bool:
        return self.pos >= len(self.buf)


# ---------------------------------------------------------------------------
# snappy raw-format decompressor (read path only)
# ---------------------------------------------------------------------------


def snappy_decompress(data: bytes) -> bytes:
    r = Reader(data)
    # preamble: uncompressed length as plain (non-zigzag) varint
    shift = 0
    length = 0
    while True:
        b = data[r.pos]
        r.pos += 1
        length |= (b & 0x7F) << shift
        if not (b & 0x80):
            break
        shift += 7
    out = bytearray()
    buf = data
    pos = r.pos
    n = len(data)
    while pos < n:
        tag = buf[pos]
        pos += 1
        ttype = tag & 0x03
        if ttype == 0:  # literal
            ln = (tag >> 2) + 1
            if ln > 60:
                extra = ln - 60
                ln = int.from_bytes(buf[pos : pos + extra], "little") + 1
                pos += extra
            out += buf[pos : pos + ln]
            pos += ln
        else:
            if ttype == 1:
                ln = ((tag >> 2) & 0x07) + 4
                offset = ((tag >> 5) << 8) | buf[pos]
                pos += 1
            elif ttype == 2:
                ln = (tag >> 2) + 1
                offset = int.from_bytes(buf[pos : pos + 2], "little")
                pos += 2
            else:
                ln = (tag >> 2) + 1
                offset = int.from_bytes(buf[pos : pos + 4], "little")
                pos += 4
            if offset == 0:
                raise ValueError("corrupt snappy stream: zero copy offset")
            start = len(out) - offset
            for i in range(ln):  # may overlap; byte-by-byte is the semantic
                out.append(out[start + i])
    if len(out) != length:
        raise ValueError(f"snappy length mismatch: {len(out)} != {length}")
    return bytes(out)


def _decompress(codec: str, block: bytes) -> bytes:
    if codec == "null":
        return block
    if codec == "deflate":
        return zlib.decompress(block, -15)
    if codec == "snappy":
        # avro snappy codec: payload + 4-byte big-endian CRC32 of the
        # UNCOMPRESSED bytes
        payload = snappy_decompress(block[:-4])
        crc = struct.unpack(">I", block[-4:])[0]
        if zlib.crc32(payload) & 0xFFFFFFFF != crc:
            raise ValueError("snappy block CRC mismatch")
        return payload
    raise ValueError(f"unsupported avro codec {codec!r}")


def snappy_compress_literal(data: bytes) -> bytes:
    """Valid (uncompressed) snappy stream: length preamble + one literal
    element. Any snappy decoder accepts it; used for codec parity with the
    reference's Spark-written files (which default to snappy)."""
    out = bytearray()
    n = len(data)
    while True:  # LEB128 uncompressed-length preamble
        b = n & 0x7F
        n >>= 7
        out.append(b | (0x80 if n else 0))
        if not n:
            break
    ln = len(data) - 1
    if len(data) == 0:
        return bytes(out)
    if ln < 60:
        out.append(ln << 2)
    elif ln < (1 << 8):
        out.append(60 << 2)
        out += ln.to_bytes(1, "little")
    elif ln < (1 << 16):
        out.append(61 << 2)
        out += ln.to_bytes(2, "little")
    elif ln < (1 << 24):
        out.append(62 << 2)
        out += ln.to_bytes(3, "little")
    else:
        out.append(63 << 2)
        out += ln.to_bytes(4, "little")
    out += data
    return bytes(out)


def _compress(codec: str, block: bytes) -> bytes:
    if codec == "null":
        return block
    if codec == "deflate":
        c = zlib.compressobj(level=6, wbits=-15)
        return c.compress(block) + c.flush()
    if codec == "snappy":
        # avro snappy codec: payload + 4-byte big-endian CRC32 of the
        # UNCOMPRESSED bytes (mirror of _decompress)
        crc = zlib.crc32(block) & 0xFFFFFFFF
        return snappy_compress_literal(block) + struct.pack(">I", crc)
    raise ValueError(f"unsupported write codec {codec!r}")


# ---------------------------------------------------------------------------
# pre-encoded container write / raw read (vectorized path, avro_fast.py)
# ---------------------------------------------------------------------------


def write_container_encoded(path: str, schema: dict, payload, reclens,
                            codec: str = "deflate", extra_meta: dict = None,
                            block_records: int = 65536, sync: bytes = None):
    """Write a container from pre-encoded record bytes (numpy uint8) plus
    per-record byte lengths; records are split into blocks on record
    boundaries. ``extra_meta``: additional header metadata entries
    (e.g. avro_fast.RECLENS_KEY) — other Avro readers ignore unknown keys."""
    import numpy as _np

    sync = sync or os.urandom(16)
    schema_bytes = json.dumps(schema, separators=(",", ":")).encode()
    out = io.BytesIO()
    out.write(MAGIC)
    meta = {"avro.schema": schema_bytes, "avro.codec": codec.encode()}
    for k, v in (extra_meta or {}).items():
        meta[k] = v
    out.write(zigzag_encode(len(meta)))
    for k, v in meta.items():
        kb = k.encode()
        out.write(zigzag_encode(len(kb)))
        out.write(kb)
        out.write(zigzag_encode(len(v)))
        out.write(v)
    out.write(zigzag_encode(0))
    out.write(sync)

    payload = _np.ascontiguousarray(payload, dtype=_np.uint8)
    reclens = _np.ascontiguousarray(reclens, dtype=_np.int64)
    nrec = len(reclens)
    bounds = _np.zeros(nrec + 1, dtype=_np.int64)
    _np.cumsum(reclens, out=bounds[1:])
    i = 0
    while i < nrec:
        j = min(i + block_records, nrec)
        raw = payload[bounds[i]:bounds[j]].tobytes()
        block = _compress(codec, raw)
        out.write(zigzag_encode(j - i))
        out.write(zigzag_encode(len(block)))
        out.write(block)
        out.write(sync)
        i = j
    with open(path, "wb") as f:
        f.write(out.getvalue())
    return path


def read_container_raw(path: str):
    """Returns (schema_dict, meta_dict, payload_bytes, record_count) without
    decoding records — the caller decodes (avro_fast) or falls back."""
    data = open(path, "rb").read()
    if data[:4] != MAGIC:
        raise ValueError(f"{path} is not an Avro object container file")
    r = Reader(data)
    r.pos = 4
    meta = {}
    while True:
        cnt = r.read_long()
        if cnt == 0:
            break
        if cnt < 0:
            r.read_long()
            cnt = -cnt
        for _ in range(cnt):
            k = r.read_bytes(r.read_long()).decode()
            meta[k] = r.read_bytes(r.read_long())
    sync = r.read_bytes(16)
    schema = json.loads(meta["avro.schema"].decode())
    codec = meta.get("avro.codec", b"null").decode()
    chunks = []
    total = 0
    while not r.at_end():
        count = r.read_long()
        size = r.read_long()
        chunks.append(_decompress(codec, r.read_bytes(size)))
        if r.read_bytes(16) != sync:
            raise ValueError("avro sync marker mismatch")
        total += count
    return schema, meta, b"".join(chunks), total


# ---------------------------------------------------------------------------
# container read
# ---------------------------------------------------------------------------


def read_container(path: str):
    """Returns (schema_dict, list_of_record_dicts)."""
    data = open(path, "rb").read()
    if data[:4] != MAGIC:
        raise ValueError(f"{path} is not an Avro object container file")
    r = Reader(data)
    r.pos = 4
    meta = {}
    while True:
        cnt = r.read_long()
        if cnt == 0:
            break
        if cnt < 0:
            r.read_long()  # byte size, unused
            cnt = -cnt
        for _ in range(cnt):
            k = r.read_bytes(r.read_long()).decode()
            meta[k] = r.read_bytes(r.read_long())
    sync = r.read_bytes(16)
    schema = json.loads(meta["avro.schema"].decode())
    codec = meta.get("avro.codec", b"null").decode()

    records = []
    while not r.at_end():
        count = r.read_long()
        size = r.read_long()
        block = _decompress(codec, r.read_bytes(size))
        if r.read_bytes(16) != sync:
            raise ValueError("avro sync marker mismatch")
        br = Reader(block)
        for _ in range(count):
            records.append(_read_value(br, schema, schema))
    return schema, records


def _read_value(r: Reader, schema, root):
    if isinstance(schema, str):
        if schema == "int" or schema == "long":
            return r.read_long()
        if schema == "float":
            return r.read_float()
        if schema == "double":
            return r.read_double()
        if schema == "null":
            return None
        if schema == "string" or schema == "bytes":
            return r.read_bytes(r.read_long())
        if schema == "boolean":
            return bool(r.read_bytes(1)[0])
        raise ValueError(f"unsupported avro type {schema!r}")
    if isinstance(schema, list):  # union
        idx = r.read_long()
        return _read_value(r, schema[idx], root)
    t = schema["type"]
    if t == "record":
        return {f["name"]: _read_value(r, f["type"], root) for f in schema["fields"]}
    if t == "array":
        out = []
        while True:
            cnt = r.read_long()
            if cnt == 0:
                break
            if cnt < 0:
                r.read_long()
                cnt = -cnt
            for _ in range(cnt):
                out.append(_read_value(r, schema["items"], root))
        return out
    return _read_value(r, t, root)


# ---------------------------------------------------------------------------
# container write
# ---------------------------------------------------------------------------


def write_container(path: str, schema: dict, records, codec: str = "deflate",
                    sync: bytes = None, block_records: int = 4096):
    sync = sync or os.urandom(16)
    schema_bytes = json.dumps(schema, separators=(",", ":")).encode()
    out = io.BytesIO()
    out.write(MAGIC)
    meta = {"avro.schema": schema_bytes, "avro.codec": codec.encode()}
    out.write(zigzag_encode(len(meta)))
    for k, v in meta.items():
        kb = k.encode()
        out.write(zigzag_encode(len(kb)))
        out.write(kb)
        out.write(zigzag_encode(len(v)))
        out.write(v)
    out.write(zigzag_encode(0))
    out.write(sync)

    i = 0
    records = list(records)
    while i < len(records):
        chunk = records[i : i + block_records]
        i += block_records
        body = io.BytesIO()
        for rec in chunk:
            _write_value(body, schema, rec)
        block = _compress(codec, body.getvalue())
        out.write(zigzag_encode(len(chunk)))
        out.write(zigzag_encode(len(block)))
        out.write(block)
        out.write(sync)
    with open(path, "wb") as f:
        f.write(out.getvalue())


def _write_value(out, schema, value):
    if isinstance(schema, str):
        if schema in ("int", "long"):
            out.write(zigzag_encode(int(value)))
        elif schema == "float":
            out.write(struct.pack("<f", float(value)))
        elif schema == "double":
            out.write(struct.pack("<d", float(value)))
        elif schema == "null":
            pass
        elif schema in ("string", "bytes"):
            b = value.encode() if isinstance(value, str) else value
            out.write(zigzag_encode(len(b)))
            out.write(b)
        elif schema == "boolean":
            out.write(b"\x01" if value else b"\x00")
        else:
            raise ValueError(f"unsupported avro type {schema!r}")
        return
    if isinstance(schema, list):  # union: pick first branch matching None-ness
        if value is None:
            idx = schema.index("null")
            out.write(zigzag_encode(idx))
        else:
            idx = 0 if schema[0] != "null" else 1
            out.write(zigzag_encode(idx))
            _write_value(out, schema[idx], value)
        return
    t = schema["type"]
    if t == "record":
        for f in schema["fields"]:
            _write_value(out, f["type"], value[f["name"]])
        return
    if t == "array":
        value = list(value)
        if value:
            out.write(zigzag_encode(len(value)))
            for item in value:
                _write_value(out, schema["items"], item)
        out.write(zigzag_encode(0))
        return
    _write_value(out, t, value)
