"""OrderedCode codec — the byte encoding TF uses for TensorBundle
slice keys (partitioned variables).

A V2 checkpoint stores a partitioned variable as ONE metadata entry
under the plain tensor name (dtype + full shape + a `slices` list, no
data) plus one data entry PER SLICE whose key is
`encode_tensor_name_slice(name, slice)`:

    '\\x00' (write_num_increasing 0)      -- tag: "this is a slice key"
    write_string(name)                     -- escaped, terminated
    write_num_increasing(ndims)
    per dim: write_signed_num_increasing(start) then (length)
             (a full dim encodes start = length = -1)

(Behavioral spec: TF's saved_tensor_slice_util EncodeTensorNameSlice +
lib/strings ordered_code; the reference consumed this format through the
TF runtime — SURVEY §2a R4. Re-implemented from the format definition,
pure Python, no TF.)

Primitives (each preserves lexicographic==numeric order):

  write_num_increasing(v)    one length byte, then big-endian bytes of v
                             with leading zeros dropped.
  write_string(s)            0x00 -> 0x00 0xFF, 0xFF -> 0xFF 0x00,
                             terminator 0x00 0x01.
  write_signed_num_increasing(v)
      Sign-extended big-endian two's complement of v in L bytes, where
      L is minimal such that x = (v if v>=0 else ~v) < 2**(7L-1) (L<=9;
      else 10), with L "header" bits XORed into the leading byte(s):
      L<=8: top L bits of byte0; L=9: all of byte0 + top bit of byte1;
      L=10: all of byte0 + top 2 bits of byte1. For v>=0 the XOR sets
      the (zero) leading bits; for v<0 it clears the (one) leading
      bits, so negatives sort below positives and by magnitude.
"""
from __future__ import annotations

from typing import List, Tuple

# header bits for byte0/byte1, indexed by encoded length 1..10
_HEADER = [None, (0x80, 0x00), (0xC0, 0x00), (0xE0, 0x00), (0xF0, 0x00),
           (0xF8, 0x00), (0xFC, 0x00), (0xFE, 0x00), (0xFF, 0x00),
           (0xFF, 0x80), (0xFF, 0xC0)]


def write_num_increasing(v: int) -> bytes:
    if v < 0:
        raise ValueError("unsigned only")
    body = b""
    while v > 0:
        body = bytes([v & 0xFF]) + body
        v >>= 8
    return bytes([len(body)]) + body


def read_num_increasing(buf: bytes, pos: int) -> Tuple[int, int]:
    if pos >= len(buf):
        raise ValueError("truncated")
    n = buf[pos]
    pos += 1
    if pos + n > len(buf):
        raise ValueError("truncated")
    v = 0
    for i in range(n):
        v = (v << 8) | buf[pos + i]
    return v, pos + n


def write_string(s: bytes) -> bytes:
    out = bytearray()
    for b in s:
        if b == 0x00:
            out += b"\x00\xff"
        elif b == 0xFF:
            out += b"\xff\x00"
        else:
            out.append(b)
    out += b"\x00\x01"
    return bytes(out)


def read_string(buf: bytes, pos: int) -> Tuple[bytes, int]:
    out = bytearray()
    n = len(buf)
    while pos + 1 < n or (pos < n and buf[pos] not in (0x00, 0xFF)):
        b = buf[pos]
        if b == 0x00:
            nxt = buf[pos + 1]
            if nxt == 0x01:
                return bytes(out), pos + 2
            if nxt == 0xFF:
                out.append(0x00)
                pos += 2
                continue
            raise ValueError("bad 0x00 escape")
        if b == 0xFF:
            if buf[pos + 1] != 0x00:
                raise ValueError("bad 0xFF escape")
            out.append(0xFF)
            pos += 2
            continue
        out.append(b)
        pos += 1
    raise ValueError("unterminated string")


def _signed_len(x: int) -> int:
    for length in range(1, 10):
        if x < (1 << (7 * length - 1)):
            return length
    return 10


def write_signed_num_increasing(val: int) -> bytes:
    if not (-(1 << 63) <= val < (1 << 63)):
        raise ValueError("int64 range")
    x = ~val if val < 0 else val
    n = _signed_len(x)
    buf = bytearray((val & ((1 << (8 * n)) - 1)).to_bytes(n, "big"))
    h0, h1 = _HEADER[n]
    buf[0] ^= h0
    if n >= 2:
        buf[1] ^= h1
    return bytes(buf)


def read_signed_num_increasing(buf: bytes, pos: int) -> Tuple[int, int]:
    if pos >= len(buf):
        raise ValueError("truncated")
    b0 = buf[pos]
    neg = (b0 & 0x80) == 0
    norm0 = (~b0 & 0xFF) if neg else b0

    def _leading_ones(byte: int) -> int:
        c = 0
        m = 0x80
        while m and (byte & m):
            c += 1
            m >>= 1
        return c

    n = _leading_ones(norm0)
    if n == 8:
        if pos + 1 >= len(buf):
            raise ValueError("truncated")
        b1 = buf[pos + 1]
        norm1 = (~b1 & 0xFF) if neg else b1
        n += _leading_ones(norm1)
        if n > 10:
            raise ValueError("bad signed encoding")
    if n == 0:
        raise ValueError("bad signed encoding")
    if pos + n > len(buf):
        raise ValueError("truncated")
    raw = bytearray(buf[pos:pos + n])
    h0, h1 = _HEADER[n]
    raw[0] ^= h0
    if n >= 2:
        raw[1] ^= h1
    v = int.from_bytes(bytes(raw), "big")
    if neg:
        v -= 1 << (8 * n)  # sign-extend
    return v, pos + n


# ------------------------------------------------- slice keys (bundle)

def encode_tensor_name_slice(name: str,
                             extents: List[Tuple[int, int]]) -> bytes:
    """extents: per-dim (start, length); (-1, -1) = the full dimension."""
    out = write_num_increasing(0)
    out += write_string(name.encode("utf-8"))
    out += write_num_increasing(len(extents))
    for start, length in extents:
        out += write_signed_num_increasing(start)
        out += write_signed_num_increasing(length)
    return out


def decode_tensor_name_slice(key: bytes):
    """Inverse of encode_tensor_name_slice -> (name, extents)."""
    tag, pos = read_num_increasing(key, 0)
    if tag != 0:
        raise ValueError("not a slice key")
    name, pos = read_string(key, pos)
    ndims, pos = read_num_increasing(key, pos)
    extents = []
    for _ in range(ndims):
        s, pos = read_signed_num_increasing(key, pos)
        ln, pos = read_signed_num_increasing(key, pos)
        extents.append((s, ln))
    return name.decode("utf-8"), extents
