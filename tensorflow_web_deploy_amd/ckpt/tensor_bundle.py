"""TensorBundle reader/writer: TF's `variables.index` + `variables.data-*`.

Index = SSTable (ckpt/sstable.py) mapping:
  ""            -> BundleHeaderProto {1: num_shards, 2: endianness, 3: version}
  tensor name   -> BundleEntryProto  {1: dtype, 2: shape(TensorShapeProto),
                                      3: shard_id, 4: offset, 5: size,
                                      6: crc32c}
Data shards are raw little-endian tensor bytes at [offset, offset+size).

The writer produces single-shard bundles for fixtures and for our own
SavedModel export (`tools/make_model.py --savedmodel`).
"""
from __future__ import annotations

import os
import struct
from typing import Dict, List, Tuple

import numpy as np

from . import graphdef as gd
from . import sstable, wire


class BundleEntry:
    def __init__(self, dtype: int, shape: List[int], shard_id: int,
                 offset: int, size: int):
        self.dtype = dtype
        self.shape = shape
        self.shard_id = shard_id
        self.offset = offset
        self.size = size


def _parse_entry(buf: bytes) -> BundleEntry:
    dtype, shape, shard, off, size = gd.DT_FLOAT, [], 0, 0, 0
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
    return BundleEntry(dtype, shape, shard, off, size)


def _encode_entry(e: BundleEntry) -> bytes:
    out = wire.field_varint(1, e.dtype)
    out += wire.field_bytes(2, gd._encode_shape(e.shape))
    if e.shard_id:
        out += wire.field_varint(3, e.shard_id)
    out += wire.field_varint(4, e.offset)
    out += wire.field_varint(5, e.size)
    return out


class BundleReader:
    def __init__(self, prefix: str):
        """prefix: e.g. <savedmodel>/variables/variables"""
        with open(prefix + ".index", "rb") as fh:
            table = sstable.read_table(fh.read())
        self.entries: Dict[str, BundleEntry] = {}
        self.num_shards = 1
        for k, v in table.items():
            if k == b"":
                for f, wt, val in wire.iter_fields(v):
                    if f == 1 and wt == 0:
                        self.num_shards = val
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

    def get(self, name: str) -> np.ndarray:
        e = self.entries[name]
        raw = self._shard(e.shard_id)[e.offset:e.offset + e.size]
        if e.dtype == gd.DT_BFLOAT16:
            arr = (np.frombuffer(raw, np.uint16).astype(np.uint32) << 16
                   ).view(np.float32)
        else:
            arr = np.frombuffer(raw, gd._DT_TO_NUMPY[e.dtype]).copy()
        return arr.reshape(e.shape)


def write_bundle(prefix: str, tensors: Dict[str, np.ndarray]) -> None:
    """Single-shard bundle writer (fixtures / export tool)."""
    os.makedirs(os.path.dirname(prefix), exist_ok=True)
    data = bytearray()
    tw = sstable.TableWriter()
    header = wire.field_varint(1, 1)  # num_shards = 1
    tw.add(b"", header)
    for name in sorted(tensors):
        arr = np.ascontiguousarray(tensors[name])
        off = len(data)
        raw = arr.tobytes()
        data += raw
        e = BundleEntry(gd._NUMPY_TO_DT[arr.dtype], list(arr.shape), 0, off,
                        len(raw))
        tw.add(name.encode("utf-8"), _encode_entry(e))
    with open(prefix + ".data-00000-of-00001", "wb") as fh:
        fh.write(bytes(data))
    with open(prefix + ".index", "wb") as fh:
        fh.write(tw.finish())
