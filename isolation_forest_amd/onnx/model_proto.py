"""ONNX ModelProto subset built on the hand-rolled protobuf codec.

Covers exactly what the Isolation Forest export graph needs: ModelProto,
GraphProto, NodeProto, AttributeProto (FLOAT/INT/STRING/TENSOR/FLOATS/INTS/
STRINGS), TensorProto (FLOAT/INT32 via float_data/int32_data or raw), and
ValueInfoProto with a [None, k] tensor shape. Field numbers follow the
public onnx.proto3 schema.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Sequence

from . import protowire as pw

# TensorProto.DataType
FLOAT = 1
BOOL = 9
INT32 = 6
INT64 = 7

# AttributeProto.AttributeType
ATTR_FLOAT = 1
ATTR_INT = 2
ATTR_STRING = 3
ATTR_TENSOR = 4
ATTR_FLOATS = 6
ATTR_INTS = 7
ATTR_STRINGS = 8


@dataclass
class Tensor:
    name: str = ""
    data_type: int = FLOAT
    dims: Sequence[int] = ()
    float_data: Sequence[float] = ()
    int32_data: Sequence[int] = ()

    def serialize(self) -> bytes:
        out = bytearray()
        for d in self.dims:
            out += pw.f_varint(1, d)
        out += pw.f_varint(2, self.data_type)
        if self.float_data:
            out += pw.f_packed_floats(4, self.float_data)
        if self.int32_data:
            out += pw.f_packed_varints(5, self.int32_data)
        if self.name:
            out += pw.f_string(8, self.name)
        return bytes(out)


@dataclass
class Attribute:
    name: str
    type: int
    f: float = 0.0
    i: int = 0
    s: bytes = b""
    t: Optional[Tensor] = None
    floats: Sequence[float] = ()
    ints: Sequence[int] = ()
    strings: Sequence[bytes] = ()

    def serialize(self) -> bytes:
        out = bytearray(pw.f_string(1, self.name))
        if self.type == ATTR_FLOAT:
            out += pw.f_float(2, self.f)
        elif self.type == ATTR_INT:
            out += pw.f_varint(3, self.i)
        elif self.type == ATTR_STRING:
            out += pw.f_bytes(4, self.s)
        elif self.type == ATTR_TENSOR:
            out += pw.f_bytes(5, self.t.serialize())
        elif self.type == ATTR_FLOATS:
            out += pw.f_packed_floats(7, self.floats)
        elif self.type == ATTR_INTS:
            out += pw.f_packed_varints(8, self.ints)
        elif self.type == ATTR_STRINGS:
            for s in self.strings:
                out += pw.f_bytes(9, s)
        else:
            raise ValueError(f"unsupported attribute type {self.type}")
        out += pw.f_varint(20, self.type)
        return bytes(out)


def attr_float(name, v):
    return Attribute(name, ATTR_FLOAT, f=float(v))


def attr_int(name, v):
    return Attribute(name, ATTR_INT, i=int(v))


def attr_string(name, v: str):
    return Attribute(name, ATTR_STRING, s=v.encode("utf-8"))


def attr_tensor(name, t: Tensor):
    return Attribute(name, ATTR_TENSOR, t=t)


def attr_floats(name, vs):
    return Attribute(name, ATTR_FLOATS, floats=[float(v) for v in vs])


def attr_ints(name, vs):
    return Attribute(name, ATTR_INTS, ints=[int(v) for v in vs])


def attr_strings(name, vs: Sequence[str]):
    return Attribute(name, ATTR_STRINGS, strings=[v.encode("utf-8") for v in vs])


@dataclass
class Node:
    op_type: str
    inputs: Sequence[str]
    outputs: Sequence[str]
    name: str = ""
    domain: str = ""
    doc_string: str = ""
    attributes: List[Attribute] = field(default_factory=list)

    def serialize(self) -> bytes:
        out = bytearray()
        for s in self.inputs:
            out += pw.f_string(1, s)
        for s in self.outputs:
            out += pw.f_string(2, s)
        if self.name:
            out += pw.f_string(3, self.name)
        out += pw.f_string(4, self.op_type)
        for a in self.attributes:
            out += pw.f_bytes(5, a.serialize())
        if self.doc_string:
            out += pw.f_string(6, self.doc_string)
        if self.domain:
            out += pw.f_string(7, self.domain)
        return bytes(out)


@dataclass
class ValueInfo:
    name: str
    elem_type: int
    shape: Sequence[Optional[int]]  # None => symbolic/batch dim

    def serialize(self) -> bytes:
        shape_out = bytearray()
        for i, d in enumerate(self.shape):
            if d is None:
                dim = pw.f_string(2, f"dim_{i}")
            else:
                dim = pw.f_varint(1, d)
            shape_out += pw.f_bytes(1, dim)
        tensor_type = (pw.f_varint(1, self.elem_type)
                       + pw.f_bytes(2, bytes(shape_out)))
        type_proto = pw.f_bytes(1, tensor_type)
        return bytes(pw.f_string(1, self.name) + pw.f_bytes(2, type_proto))


@dataclass
class Graph:
    name: str
    nodes: List[Node]
    inputs: List[ValueInfo]
    outputs: List[ValueInfo]

    def serialize(self) -> bytes:
        out = bytearray()
        for n in self.nodes:
            out += pw.f_bytes(1, n.serialize())
        out += pw.f_string(2, self.name)
        for v in self.inputs:
            out += pw.f_bytes(11, v.serialize())
        for v in self.outputs:
            out += pw.f_bytes(12, v.serialize())
        return bytes(out)


@dataclass
class OperatorSetId:
    domain: str
    version: int

    def serialize(self) -> bytes:
        return pw.f_string(1, self.domain) + pw.f_varint(2, self.version)


@dataclass
class Model:
    graph: Graph
    ir_version: int = 10
    producer_name: str = "isolation-forest-amd"
    producer_version: str = ""
    opset_imports: List[OperatorSetId] = field(default_factory=list)

    def serialize(self) -> bytes:
        out = bytearray(pw.f_varint(1, self.ir_version))
        for op in self.opset_imports:
            out += pw.f_bytes(8, op.serialize())
        out += pw.f_string(2, self.producer_name)
        if self.producer_version:
            out += pw.f_string(3, self.producer_version)
        out += pw.f_bytes(7, self.graph.serialize())
        return bytes(out)

    def save(self, path: str) -> None:
        with open(path, "wb") as f:
            f.write(self.serialize())
