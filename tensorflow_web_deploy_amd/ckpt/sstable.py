"""Minimal LevelDB-format SSTable reader/writer (pure Python).

TensorFlow's TensorBundle `variables.index` file is a LevelDB table
(block-based SSTable: prefix-compressed key/value entries + restart
array per block, index block of BlockHandles, 48-byte footer with magic
0xdb4775248b80fb57). We read it without TF or leveldb; the writer
exists so tests can build real fixtures (round-trip against
our own writer).

Per-block trailer: 1 compression byte (0 = none, 1 = snappy) + 4-byte
masked crc32c. A small pure-Python snappy decoder covers compressed
blocks; block crc32c (Castagnoli, LevelDB-masked) is computed on write
and verified on read (a stored 0 — older fixtures — is tolerated).
"""
from __future__ import annotations

import struct
from typing import Dict, List, Tuple

MAGIC = 0xDB4775248B80FB57


# ------------------------------------------------------------- varint

def _rv(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7


def _wv(v: int) -> bytes:
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


# -------------------------------------------------------------- snappy

def snappy_decompress(data: bytes) -> bytes:
    """Pure-Python snappy raw-format decoder (no framing)."""
    n, pos = _rv(data, 0)
    out = bytearray()
    while pos < len(data):
        tag = data[pos]
        pos += 1
        t = tag & 3
        if t == 0:  # literal
            ln = (tag >> 2) + 1
            if ln > 60:
                nb = ln - 60
                ln = int.from_bytes(data[pos:pos + nb], "little") + 1
                pos += nb
            out += data[pos:pos + ln]
            pos += ln
        else:
            if t == 1:
                ln = ((tag >> 2) & 7) + 4
                off = ((tag >> 5) << 8) | data[pos]
                pos += 1
            elif t == 2:
                ln = (tag >> 2) + 1
                off = int.from_bytes(data[pos:pos + 2], "little")
                pos += 2
            else:
                ln = (tag >> 2) + 1
                off = int.from_bytes(data[pos:pos + 4], "little")
                pos += 4
            for _ in range(ln):  # may overlap; byte-by-byte
                out.append(out[-off])
    assert len(out) == n, f"snappy: expected {n} bytes, got {len(out)}"
    return bytes(out)


# -------------------------------------------------------------- reader

def _parse_block(block: bytes) -> List[Tuple[bytes, bytes]]:
    """Decode all (key, value) entries of one uncompressed block body."""
    if len(block) < 4:
        return []
    num_restarts = struct.unpack_from("<I", block, len(block) - 4)[0]
    data_end = len(block) - 4 - 4 * num_restarts
    entries = []
    pos = 0
    key = b""
    while pos < data_end:
        shared, pos = _rv(block, pos)
        unshared, pos = _rv(block, pos)
        vlen, pos = _rv(block, pos)
        key = key[:shared] + block[pos:pos + unshared]
        pos += unshared
        value = block[pos:pos + vlen]
        pos += vlen
        entries.append((key, value))
    return entries


# crc32c (Castagnoli 0x1EDC6F41, reflected) — table-driven; blocks in a
# TF bundle index are KB-sized so a Python loop is fine
_CRC_TABLE = []


def _crc_table():
    global _CRC_TABLE
    if not _CRC_TABLE:
        poly = 0x82F63B78
        t = []
        for i in range(256):
            c = i
            for _ in range(8):
                c = (c >> 1) ^ poly if c & 1 else c >> 1
            t.append(c)
        _CRC_TABLE = t
    return _CRC_TABLE


def crc32c(buf: bytes, crc: int = 0) -> int:
    t = _crc_table()
    c = crc ^ 0xFFFFFFFF
    for b in buf:
        c = t[(c ^ b) & 0xFF] ^ (c >> 8)
    return c ^ 0xFFFFFFFF


def _mask_crc(c: int) -> int:
    return (((c >> 15) | (c << 17)) + 0xA282EAD8) & 0xFFFFFFFF


def _read_block(data: bytes, offset: int, size: int,
                verify: bool = True) -> List[Tuple[bytes, bytes]]:
    body = data[offset:offset + size]
    ctype = data[offset + size]
    if verify:
        stored = struct.unpack(
            "<I", data[offset + size + 1:offset + size + 5])[0]
        if stored != 0:  # 0 = unchecksummed legacy fixture
            want = _mask_crc(crc32c(body + bytes([ctype])))
            if stored != want:
                raise ValueError(
                    f"SSTable block crc mismatch at {offset}: "
                    f"stored {stored:#x} != computed {want:#x}")
    if ctype == 1:
        body = snappy_decompress(body)
    elif ctype != 0:
        raise ValueError(f"unsupported block compression {ctype}")
    return _parse_block(body)


def read_table(data: bytes) -> Dict[bytes, bytes]:
    """Return every key->value in the table."""
    footer = data[-48:]
    magic = struct.unpack("<Q", footer[-8:])[0]
    if magic != MAGIC:
        raise ValueError("not an SSTable (bad magic)")
    pos = 0
    _mi_off, pos = _rv(footer, pos)
    _mi_size, pos = _rv(footer, pos)
    idx_off, pos = _rv(footer, pos)
    idx_size, pos = _rv(footer, pos)
    out: Dict[bytes, bytes] = {}
    for _k, handle in _read_block(data, idx_off, idx_size):
        hpos = 0
        off, hpos = _rv(handle, hpos)
        size, hpos = _rv(handle, hpos)
        for key, value in _read_block(data, off, size):
            out[key] = value
    return out


# -------------------------------------------------------------- writer

def _crc32c_masked(body: bytes, ctype: int = 0) -> int:
    """LevelDB record crc: crc32c over body + the type byte, masked."""
    return _mask_crc(crc32c(body + bytes([ctype])))


class TableWriter:
    """Minimal writer: all entries in ONE data block, restart_interval=1
    (no prefix compression), no block compression."""

    def __init__(self):
        self.items: List[Tuple[bytes, bytes]] = []

    def add(self, key: bytes, value: bytes):
        self.items.append((key, value))

    @staticmethod
    def _block(items: List[Tuple[bytes, bytes]]) -> bytes:
        body = bytearray()
        restarts = []
        for k, v in sorted(items):
            restarts.append(len(body))
            body += _wv(0) + _wv(len(k)) + _wv(len(v)) + k + v
        for r in restarts:
            body += struct.pack("<I", r)
        body += struct.pack("<I", len(restarts))
        return bytes(body)

    def finish(self) -> bytes:
        out = bytearray()
        datab = self._block(self.items)
        data_off, data_size = 0, len(datab)
        out += datab + bytes([0]) + struct.pack("<I", _crc32c_masked(datab))
        # metaindex (empty)
        metab = self._block([])
        mi_off, mi_size = len(out), len(metab)
        out += metab + bytes([0]) + struct.pack("<I", _crc32c_masked(metab))
        # index block: one entry key>=last key -> data handle
        last_key = max((k for k, _ in self.items), default=b"")
        handle = _wv(data_off) + _wv(data_size)
        idxb = self._block([(last_key + b"\x00", handle)])
        idx_off, idx_size = len(out), len(idxb)
        out += idxb + bytes([0]) + struct.pack("<I", _crc32c_masked(idxb))
        footer = _wv(mi_off) + _wv(mi_size) + _wv(idx_off) + _wv(idx_size)
        footer += b"\x00" * (40 - len(footer))
        footer += struct.pack("<Q", MAGIC)
        out += footer
        return bytes(out)
