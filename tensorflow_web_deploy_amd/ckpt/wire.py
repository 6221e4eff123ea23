"""Minimal protobuf wire-format reader/writer (pure Python, no deps).

The reference loads TensorFlow checkpoints through the TF runtime's own
protobuf stack (reference: frozen-GraphDef / SavedModel loading via
`tf.import_graph_def` / `tf.saved_model.loader.load`). We have no TF
dependency by design (BASELINE.json north_star: "no TF runtime
dependency"), so the wire format is implemented by hand. Field numbers
are the stable public ABI of the TF protos.

Wire types: 0 = varint, 1 = 64-bit, 2 = length-delimited, 5 = 32-bit.
"""
from __future__ import annotations

import struct
from typing import Iterator, Tuple

# ---------------------------------------------------------------- decode


def read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    """Return (value, new_pos). Raises ValueError on truncation."""
    result = 0
    shift = 0
    while True:
        if pos >= len(buf):
            raise ValueError("truncated varint")
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7
        if shift > 70:
            raise ValueError("varint too long")


def zigzag_decode(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


def iter_fields(buf: bytes) -> Iterator[Tuple[int, int, object]]:
    """Yield (field_number, wire_type, value) for each field in buf.

    value is: int for varint, bytes for length-delimited, int (raw bits)
    for fixed32/fixed64.
    """
    pos = 0
    n = len(buf)
    while pos < n:
        tag, pos = read_varint(buf, pos)
        field = tag >> 3
        wt = tag & 7
        if wt == 0:
            v, pos = read_varint(buf, pos)
            yield field, wt, v
        elif wt == 1:
            if pos + 8 > n:
                raise ValueError("truncated fixed64")
            yield field, wt, struct.unpack_from("<Q", buf, pos)[0]
            pos += 8
        elif wt == 2:
            ln, pos = read_varint(buf, pos)
            if pos + ln > n:
                raise ValueError("truncated bytes field")
            yield field, wt, buf[pos : pos + ln]
            pos += ln
        elif wt == 5:
            if pos + 4 > n:
                raise ValueError("truncated fixed32")
            yield field, wt, struct.unpack_from("<I", buf, pos)[0]
            pos += 4
        else:
            raise ValueError(f"unsupported wire type {wt}")


def as_signed64(v: int) -> int:
    """Interpret a varint as a signed int64 (two's complement)."""
    return v - (1 << 64) if v >= (1 << 63) else v


def bits_to_float(v: int) -> float:
    return struct.unpack("<f", struct.pack("<I", v))[0]


def bits_to_double(v: int) -> float:
    return struct.unpack("<d", struct.pack("<Q", v))[0]


def unpack_packed_varints(buf: bytes) -> list:
    out = []
    pos = 0
    while pos < len(buf):
        v, pos = read_varint(buf, pos)
        out.append(as_signed64(v))
    return out


def unpack_packed_floats(buf: bytes) -> list:
    return list(struct.unpack(f"<{len(buf) // 4}f", buf))


# ---------------------------------------------------------------- encode


def write_varint(v: int) -> bytes:
    if v < 0:
        v += 1 << 64  # two's complement for negative int64
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def tag(field: int, wire_type: int) -> bytes:
    return write_varint((field << 3) | wire_type)


def field_varint(field: int, v: int) -> bytes:
    return tag(field, 0) + write_varint(v)


def field_bytes(field: int, b: bytes) -> bytes:
    return tag(field, 2) + write_varint(len(b)) + b


def field_string(field: int, s: str) -> bytes:
    return field_bytes(field, s.encode("utf-8"))


def field_float(field: int, f: float) -> bytes:
    return tag(field, 5) + struct.pack("<f", f)


def field_double(field: int, f: float) -> bytes:
    return tag(field, 1) + struct.pack("<d", f)


def field_packed_floats(field: int, fs) -> bytes:
    payload = struct.pack(f"<{len(fs)}f", *fs)
    return field_bytes(field, payload)
