"""TensorBundle reader/writer: TF's `variables.index` + `variables.data-*`.

Index = SSTable (ckpt/sstable.py) mapping:
  ""            -> BundleHeaderProto {1: num_shards, 2: endianness, 3: version}
  tensor name   -> BundleEntryProto  {1: dtype, 2: shape(TensorShapeProto),
                                      3: shard_id, 4: offset, 5: size,
                                      6: crc32c, 7: slices(TensorSliceProto)}
Data shards are raw little-endian tensor bytes at [offset, offset+size).

Partitioned/sliced variables: the full tensor's entry carries `slices`
(per-slice extents) and NO data of its own; each slice's bytes live in a
separate entry keyed by the OrderedCode form of (name, extents) — see
ckpt/ordered_code.py. `get(name)` stitches the slices back together
transparently.

The writer produces single-shard bundles for fixtures and for our own
SavedModel export (`tools/make_model.py --savedmodel`); `add_slice`
writes partitioned fixtures for the reader tests.
"""
from __future__ import annotations

import os
import struct
from typing import Dict, List, Optional, Tuple

import numpy as np

from . import graphdef as gd
from . import ordered_code as oc
from . import sstable, wire

# per-dim (start, length); (-1, -1) = the full dimension
Extent = Tuple[int, int]


class BundleEntry:
    def __init__(self, dtype: int, shape: List[int], shard_id: int,
                 offset: int, size: int,
                 slices: Optional[List[List[Extent]]] = None):
        self.dtype = dtype
        self.shape = shape
        self.shard_id = shard_id
        self.offset = offset
        self.size = size
        self.slices = slices or []


def _parse_slice(buf: bytes) -> List[Extent]:
    """TensorSliceProto { repeated Extent extent = 1; }
    Extent { int64 start = 1; int64 length = 2 (oneof: absent=full) }"""
    extents: List[Extent] = []
    for f, wt, v in wire.iter_fields(buf):
        if f == 1 and wt == 2:
            start, length = 0, None
            for f2, wt2, v2 in wire.iter_fields(v):
                if f2 == 1 and wt2 == 0:
                    start = wire.as_signed64(v2)
                elif f2 == 2 and wt2 == 0:
                    length = wire.as_signed64(v2)
            extents.append((-1, -1) if length is None else (start, length))
    return extents


def _encode_slice(extents: List[Extent]) -> bytes:
    out = b""
    for start, length in extents:
        ext = b""
        if (start, length) == (-1, -1):
            pass  # full dim: empty Extent
        else:
            if start:
                ext += wire.field_varint(1, start)
            ext += wire.field_varint(2, length)
        out += wire.field_bytes(1, ext)
    return out


def _parse_entry(buf: bytes) -> BundleEntry:
    dtype, shape, shard, off, size = gd.DT_FLOAT, [], 0, 0, 0
    slices: List[List[Extent]] = []
    for f, wt, v in wire.iter_fields(buf):
        if f == 1 and wt == 0:
            dtype = v
        elif f == 2 and wt == 2:
            shape = gd._parse_shape(v)
        elif f == 3 and wt == 0:
            shard = v
        elif f == 4 and wt == 0:
            off = v
        elif f == 5 and wt == 0:
            size = v
        elif f == 7 and wt == 2:
            slices.append(_parse_slice(v))
    return BundleEntry(dtype, shape, shard, off, size, slices)


def _encode_entry(e: BundleEntry) -> bytes:
    out = wire.field_varint(1, e.dtype)
    out += wire.field_bytes(2, gd._encode_shape(e.shape))
    if e.shard_id:
        out += wire.field_varint(3, e.shard_id)
    out += wire.field_varint(4, e.offset)
    out += wire.field_varint(5, e.size)
    for s in e.slices:
        out += wire.field_bytes(7, _encode_slice(s))
    return out


class BundleReader:
    def __init__(self, prefix: str):
        """prefix: e.g. <savedmodel>/variables/variables"""
        with open(prefix + ".index", "rb") as fh:
            table = sstable.read_table(fh.read())
        self.entries: Dict[str, BundleEntry] = {}
        self._raw: Dict[bytes, BundleEntry] = {}  # incl slice-data keys
        self.num_shards = 1
        for k, v in table.items():
            if k == b"":
                for f, wt, val in wire.iter_fields(v):
                    if f == 1 and wt == 0:
                        self.num_shards = val
            elif k[:1] == b"\x00":
                # OrderedCode slice-data key (not valid utf-8 in
                # general); looked up internally by get()
                self._raw[k] = _parse_entry(v)
            else:
                self.entries[k.decode("utf-8")] = _parse_entry(v)
        self.prefix = prefix
        self._shards: Dict[int, bytes] = {}

    def _shard(self, sid: int) -> bytes:
        if sid not in self._shards:
            path = (f"{self.prefix}.data-{sid:05d}-of-"
                    f"{self.num_shards:05d}")
            with open(path, "rb") as fh:
                self._shards[sid] = fh.read()
        return self._shards[sid]

    def keys(self):
        return self.entries.keys()

    def _decode_raw(self, e: BundleEntry, shape) -> np.ndarray:
        raw = self._shard(e.shard_id)[e.offset:e.offset + e.size]
        if e.dtype == gd.DT_BFLOAT16:
            arr = (np.frombuffer(raw, np.uint16).astype(np.uint32) << 16
                   ).view(np.float32)
        else:
            arr = np.frombuffer(raw, gd._DT_TO_NUMPY[e.dtype]).copy()
        return arr.reshape(shape)

    def get(self, name: str) -> np.ndarray:
        e = self.entries[name]
        if not e.slices:
            return self._decode_raw(e, e.shape)
        # partitioned variable: stitch the stored slices into the full
        # shape (each slice's data entry is keyed by the OrderedCode
        # encoding of its extents)
        if e.dtype == gd.DT_BFLOAT16:
            full = np.empty(e.shape, np.float32)
        else:
            full = np.empty(e.shape, gd._DT_TO_NUMPY[e.dtype])
        for extents in e.slices:
            if len(extents) != len(e.shape):
                raise ValueError(
                    f"slice rank {len(extents)} != tensor rank "
                    f"{len(e.shape)} for {name}")
            key = oc.encode_tensor_name_slice(name, extents)
            se = self._raw.get(key)
            if se is None:
                raise KeyError(f"missing slice entry for {name} "
                               f"extents {extents}")
            idx = tuple(
                slice(None) if (s, ln) == (-1, -1) else slice(s, s + ln)
                for (s, ln) in extents)
            sl_shape = [e.shape[d] if (s, ln) == (-1, -1) else ln
                        for d, (s, ln) in enumerate(extents)]
            full[idx] = self._decode_raw(se, sl_shape)
        return full


def write_bundle(prefix: str, tensors: Dict[str, np.ndarray],
                 sliced: Optional[Dict[str, Tuple[
                     List[int],
                     List[Tuple[List[Extent], np.ndarray]]]]] = None
                 ) -> None:
    """Single-shard bundle writer (fixtures / export tool).

    `sliced`: partitioned variables — name -> (full_shape,
    [(extents, slice_array), ...]). Writes the TF layout: a data-less
    full entry carrying `slices`, plus one data entry per slice under
    its OrderedCode key.
    """
    os.makedirs(os.path.dirname(prefix), exist_ok=True)
    data = bytearray()
    items: List[Tuple[bytes, bytes]] = []

    def _add_data(key: bytes, arr: np.ndarray,
                  slices=None, shape=None) -> None:
        arr = np.ascontiguousarray(arr)
        off = len(data)
        raw = arr.tobytes()
        data.extend(raw)
        e = BundleEntry(gd._NUMPY_TO_DT[arr.dtype],
                        list(arr.shape) if shape is None else shape,
                        0, off, len(raw), slices)
        items.append((key, _encode_entry(e)))

    for name in tensors:
        _add_data(name.encode("utf-8"), tensors[name])
    for name, (full_shape, parts) in (sliced or {}).items():
        dt = gd._NUMPY_TO_DT[np.ascontiguousarray(parts[0][1]).dtype]
        meta = BundleEntry(dt, list(full_shape), 0, 0, 0,
                           [ext for (ext, _) in parts])
        items.append((name.encode("utf-8"), _encode_entry(meta)))
        for extents, arr in parts:
            _add_data(oc.encode_tensor_name_slice(name, extents), arr)

    tw = sstable.TableWriter()
    tw.add(b"", wire.field_varint(1, 1))  # num_shards = 1
    for key, enc in sorted(items):  # SSTable keys must be ordered
        tw.add(key, enc)
    with open(prefix + ".data-00000-of-00001", "wb") as fh:
        fh.write(bytes(data))
    with open(prefix + ".index", "wb") as fh:
        fh.write(tw.finish())
