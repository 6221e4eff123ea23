"""TensorFlow frozen-graph (.pb GraphDef) parser and writer — no TF import.

Hand-written schema for the subset of tensorflow/core/framework protos an
image-classification frozen graph uses (reference behavior: `gfile read +
tf.GraphDef.ParseFromString + tf.import_graph_def`). Field numbers follow
the public TF proto definitions:

  GraphDef        { repeated NodeDef node = 1; VersionDef versions = 4; }
  NodeDef         { string name = 1; string op = 2; repeated string input = 3;
                    string device = 4; map<string, AttrValue> attr = 5; }
  AttrValue       { ListValue list = 1; bytes s = 2; int64 i = 3; float f = 4;
                    bool b = 5; DataType type = 6; TensorShapeProto shape = 7;
                    TensorProto tensor = 8; }
  AttrValue.ListValue { repeated bytes s = 2; repeated int64 i = 3;
                    repeated float f = 4; repeated bool b = 5;
                    repeated DataType type = 6; repeated TensorShapeProto shape = 7; }
  TensorProto     { DataType dtype = 1; TensorShapeProto tensor_shape = 2;
                    bytes tensor_content = 4; repeated int half_val = 13*;
                    repeated float float_val = 5; repeated double double_val = 6;
                    repeated int32 int_val = 7; repeated bytes string_val = 8;
                    repeated int64 int64_val = 10; repeated bool bool_val = 11; }
  TensorShapeProto{ repeated Dim dim = 2 { int64 size = 1; string name = 2; }
                    bool unknown_rank = 3; }

(*half_val is field 13 in TF; bfloat16 uses half_val too.)
"""
from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np

from . import wire

# TF DataType enum values (tensorflow/core/framework/types.proto)
DT_FLOAT = 1
DT_DOUBLE = 2
DT_INT32 = 3
DT_UINT8 = 4
DT_INT16 = 5
DT_INT8 = 6
DT_STRING = 7
DT_INT64 = 9
DT_BOOL = 10
DT_BFLOAT16 = 14
DT_HALF = 19

_DT_TO_NUMPY = {
    DT_FLOAT: np.float32,
    DT_DOUBLE: np.float64,
    DT_INT32: np.int32,
    DT_UINT8: np.uint8,
    DT_INT16: np.int16,
    DT_INT8: np.int8,
    DT_INT64: np.int64,
    DT_BOOL: np.bool_,
    DT_HALF: np.float16,
}
_NUMPY_TO_DT = {
    np.dtype(np.float32): DT_FLOAT,
    np.dtype(np.float64): DT_DOUBLE,
    np.dtype(np.int32): DT_INT32,
    np.dtype(np.uint8): DT_UINT8,
    np.dtype(np.int16): DT_INT16,
    np.dtype(np.int8): DT_INT8,
    np.dtype(np.int64): DT_INT64,
    np.dtype(np.bool_): DT_BOOL,
    np.dtype(np.float16): DT_HALF,
}


@dataclass
class AttrValue:
    s: Optional[bytes] = None
    i: Optional[int] = None
    f: Optional[float] = None
    b: Optional[bool] = None
    type: Optional[int] = None
    shape: Optional[List[int]] = None
    tensor: Optional[np.ndarray] = None
    list_s: Optional[List[bytes]] = None
    list_i: Optional[List[int]] = None
    list_f: Optional[List[float]] = None
    list_type: Optional[List[int]] = None
    list_shape: Optional[List[List[int]]] = None


@dataclass
class NodeDef:
    name: str
    op: str
    input: List[str] = field(default_factory=list)
    attr: Dict[str, AttrValue] = field(default_factory=dict)
    device: str = ""


@dataclass
class GraphDef:
    node: List[NodeDef] = field(default_factory=list)


# ---------------------------------------------------------------- parse


def _parse_shape(buf: bytes) -> List[int]:
    dims: List[int] = []
    for f, wt, v in wire.iter_fields(buf):
        if f == 2 and wt == 2:  # Dim
            size = 0
            for f2, wt2, v2 in wire.iter_fields(v):
                if f2 == 1 and wt2 == 0:
                    size = wire.as_signed64(v2)
            dims.append(size)
        elif f == 3 and wt == 0 and v:
            return []  # unknown rank
    return dims


def _parse_tensor(buf: bytes) -> np.ndarray:
    dtype = DT_FLOAT
    shape: List[int] = []
    content = b""
    float_vals: List[float] = []
    int_vals: List[int] = []
    int64_vals: List[int] = []
    double_vals: List[float] = []
    bool_vals: List[bool] = []
    half_vals: List[int] = []
    string_vals: List[bytes] = []
    for f, wt, v in wire.iter_fields(buf):
        if f == 1 and wt == 0:
            dtype = v
        elif f == 2 and wt == 2:
            shape = _parse_shape(v)
        elif f == 4 and wt == 2:
            content = v
        elif f == 5:
            if wt == 2:
                float_vals.extend(wire.unpack_packed_floats(v))
            elif wt == 5:
                float_vals.append(wire.bits_to_float(v))
        elif f == 6:
            if wt == 1:
                double_vals.append(wire.bits_to_double(v))
            elif wt == 2:
                n = len(v) // 8
                double_vals.extend(struct.unpack(f"<{n}d", v))
        elif f == 7:
            if wt == 0:
                int_vals.append(wire.as_signed64(v))
            elif wt == 2:
                int_vals.extend(wire.unpack_packed_varints(v))
        elif f == 8 and wt == 2:
            string_vals.append(v)
        elif f == 10:
            if wt == 0:
                int64_vals.append(wire.as_signed64(v))
            elif wt == 2:
                int64_vals.extend(wire.unpack_packed_varints(v))
        elif f == 11:
            if wt == 0:
                bool_vals.append(bool(v))
            elif wt == 2:
                bool_vals.extend(bool(x) for x in wire.unpack_packed_varints(v))
        elif f == 13:
            if wt == 0:
                half_vals.append(v)
            elif wt == 2:
                half_vals.extend(wire.unpack_packed_varints(v))

    if dtype == DT_STRING:
        arr = np.array(string_vals, dtype=object)
        return arr.reshape(shape) if shape else arr

    nelem = int(np.prod(shape)) if shape else 1
    if dtype == DT_BFLOAT16:
        # store bf16 as float32 on the numpy side (upper 16 bits)
        if content:
            raw = np.frombuffer(content, dtype=np.uint16)
        else:
            raw = np.array(half_vals, dtype=np.uint16)
        as32 = raw.astype(np.uint32) << 16
        arr = as32.view(np.float32)
    elif content:
        arr = np.frombuffer(content, dtype=_DT_TO_NUMPY[dtype]).copy()
    else:
        np_dt = _DT_TO_NUMPY[dtype]
        if dtype == DT_FLOAT:
            vals = float_vals
        elif dtype == DT_DOUBLE:
            vals = double_vals
        elif dtype in (DT_INT32, DT_UINT8, DT_INT16, DT_INT8):
            vals = int_vals
        elif dtype == DT_INT64:
            vals = int64_vals
        elif dtype == DT_BOOL:
            vals = bool_vals
        elif dtype == DT_HALF:
            arr16 = np.array(half_vals, dtype=np.uint16).view(np.float16)
            vals = None
            arr = arr16.astype(np.float16)
        else:
            raise ValueError(f"unsupported TensorProto dtype {dtype}")
        if dtype != DT_HALF:
            arr = np.array(vals, dtype=np_dt)
        # TF semantics: a single value broadcasts to the full shape
        if arr.size == 1 and nelem > 1:
            arr = np.full(nelem, arr.item(), dtype=arr.dtype)
    if shape:
        arr = arr.reshape(shape)
    return arr


def _parse_attrvalue(buf: bytes) -> AttrValue:
    av = AttrValue()
    for f, wt, v in wire.iter_fields(buf):
        if f == 1 and wt == 2:  # ListValue
            for f2, wt2, v2 in wire.iter_fields(v):
                if f2 == 2 and wt2 == 2:
                    av.list_s = (av.list_s or []) + [v2]
                elif f2 == 3:
                    av.list_i = av.list_i or []
                    if wt2 == 0:
                        av.list_i.append(wire.as_signed64(v2))
                    else:
                        av.list_i.extend(wire.unpack_packed_varints(v2))
                elif f2 == 4:
                    av.list_f = av.list_f or []
                    if wt2 == 5:
                        av.list_f.append(wire.bits_to_float(v2))
                    else:
                        av.list_f.extend(wire.unpack_packed_floats(v2))
                elif f2 == 6:
                    av.list_type = av.list_type or []
                    if wt2 == 0:
                        av.list_type.append(v2)
                    else:
                        av.list_type.extend(wire.unpack_packed_varints(v2))
                elif f2 == 7 and wt2 == 2:
                    av.list_shape = (av.list_shape or []) + [_parse_shape(v2)]
        elif f == 2 and wt == 2:
            av.s = v
        elif f == 3 and wt == 0:
            av.i = wire.as_signed64(v)
        elif f == 4 and wt == 5:
            av.f = wire.bits_to_float(v)
        elif f == 5 and wt == 0:
            av.b = bool(v)
        elif f == 6 and wt == 0:
            av.type = v
        elif f == 7 and wt == 2:
            av.shape = _parse_shape(v)
        elif f == 8 and wt == 2:
            av.tensor = _parse_tensor(v)
    return av


def _parse_node(buf: bytes) -> NodeDef:
    node = NodeDef(name="", op="")
    for f, wt, v in wire.iter_fields(buf):
        if f == 1 and wt == 2:
            node.name = v.decode("utf-8")
        elif f == 2 and wt == 2:
            node.op = v.decode("utf-8")
        elif f == 3 and wt == 2:
            node.input.append(v.decode("utf-8"))
        elif f == 4 and wt == 2:
            node.device = v.decode("utf-8")
        elif f == 5 and wt == 2:  # map entry {key=1, value=2}
            key, val = None, None
            for f2, wt2, v2 in wire.iter_fields(v):
                if f2 == 1 and wt2 == 2:
                    key = v2.decode("utf-8")
                elif f2 == 2 and wt2 == 2:
                    val = _parse_attrvalue(v2)
            if key is not None and val is not None:
                node.attr[key] = val
    return node


def parse_graphdef(data: bytes) -> GraphDef:
    g = GraphDef()
    for f, wt, v in wire.iter_fields(data):
        if f == 1 and wt == 2:
            g.node.append(_parse_node(v))
    return g


def load_frozen_graph(path: str) -> GraphDef:
    with open(path, "rb") as fh:
        return parse_graphdef(fh.read())


# ---------------------------------------------------------------- write


def _encode_shape(dims: List[int]) -> bytes:
    out = b""
    for d in dims:
        out += wire.field_bytes(2, wire.field_varint(1, d))
    return out


def encode_tensor(arr: np.ndarray) -> bytes:
    arr = np.asarray(arr)
    dt = _NUMPY_TO_DT[arr.dtype]
    out = wire.field_varint(1, dt)
    out += wire.field_bytes(2, _encode_shape(list(arr.shape)))
    out += wire.field_bytes(4, arr.tobytes())
    return out


def encode_attrvalue(av: AttrValue) -> bytes:
    out = b""
    if av.s is not None:
        out += wire.field_bytes(2, av.s)
    if av.i is not None:
        out += wire.field_varint(3, av.i)
    if av.f is not None:
        out += wire.field_float(4, av.f)
    if av.b is not None:
        out += wire.field_varint(5, 1 if av.b else 0)
    if av.type is not None:
        out += wire.field_varint(6, av.type)
    if av.shape is not None:
        out += wire.field_bytes(7, _encode_shape(av.shape))
    if av.tensor is not None:
        out += wire.field_bytes(8, encode_tensor(av.tensor))
    lst = b""
    if av.list_s is not None:
        for s in av.list_s:
            lst += wire.field_bytes(2, s)
    if av.list_i is not None:
        for i in av.list_i:
            lst += wire.field_varint(3, i)
    if av.list_f is not None:
        for f in av.list_f:
            lst += wire.field_float(4, f)
    if av.list_type is not None:
        for t in av.list_type:
            lst += wire.field_varint(6, t)
    if av.list_shape is not None:
        for sh in av.list_shape:
            lst += wire.field_bytes(7, _encode_shape(sh))
    if lst:
        out += wire.field_bytes(1, lst)
    return out


def encode_node(node: NodeDef) -> bytes:
    out = wire.field_string(1, node.name)
    out += wire.field_string(2, node.op)
    for inp in node.input:
        out += wire.field_string(3, inp)
    if node.device:
        out += wire.field_string(4, node.device)
    for key, av in node.attr.items():
        entry = wire.field_string(1, key) + wire.field_bytes(2, encode_attrvalue(av))
        out += wire.field_bytes(5, entry)
    return out


def encode_graphdef(g: GraphDef) -> bytes:
    out = b""
    for n in g.node:
        out += wire.field_bytes(1, encode_node(n))
    return out


def save_frozen_graph(g: GraphDef, path: str) -> None:
    with open(path, "wb") as fh:
        fh.write(encode_graphdef(g))
