"""Property/fuzz tests (hypothesis): the hand-written codecs must
round-trip arbitrary data and reject garbage with exceptions, never
crashes or hangs (SURVEY §4 robustness class)."""
import io

import numpy as np
import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from tensorflow_web_deploy_amd.ckpt import graphdef as gd
from tensorflow_web_deploy_amd.ckpt import sstable, wire


@given(st.integers(min_value=0, max_value=2**64 - 1))
def test_varint_roundtrip_any(v):
    out, pos = wire.read_varint(wire.write_varint(v), 0)
    assert out == v


@given(st.binary(max_size=200))
@settings(max_examples=200)
def test_wire_iter_never_hangs(data):
    try:
        list(wire.iter_fields(data))
    except ValueError:
        pass  # malformed is fine; crashing/hanging is not


@given(st.lists(st.tuples(st.text(min_size=1, max_size=12),
                          st.binary(max_size=40)),
                min_size=0, max_size=8, unique_by=lambda kv: kv[0]))
def test_sstable_roundtrip_any(items):
    w = sstable.TableWriter()
    for k, v in items:
        w.add(k.encode(), v)
    table = sstable.read_table(w.finish())
    assert table == {k.encode(): v for k, v in items}


@given(st.binary(min_size=48, max_size=300))
@settings(max_examples=200)
def test_sstable_garbage_rejected(data):
    try:
        sstable.read_table(data)
    except Exception:
        pass  # any exception is acceptable; crash/hang is the failure


@given(st.lists(st.sampled_from([np.float32, np.int32, np.int64]),
                min_size=1, max_size=3),
       st.integers(min_value=0, max_value=999))
def test_tensorproto_roundtrip(dtypes, seed):
    rng = np.random.default_rng(seed)
    for dt in dtypes:
        shape = tuple(rng.integers(1, 5, size=rng.integers(0, 4)))
        arr = (rng.random(shape) * 100).astype(dt)
        out = gd._parse_tensor(gd.encode_tensor(arr))
        assert out.dtype == arr.dtype
        np.testing.assert_array_equal(out, arr)


@given(st.binary(max_size=400))
@settings(max_examples=300)
def test_jpeg_parser_rejects_garbage(data):
    """Random bytes (incl SOI-prefixed) must raise, not crash/OOB."""
    from tensorflow_web_deploy_amd import _jpegc
    for payload in (data, b"\xff\xd8" + data):
        try:
            _jpegc.parse(payload)
        except (ValueError, RuntimeError):
            pass


@given(st.integers(min_value=0, max_value=2**31 - 1),
       st.integers(min_value=1, max_value=120))
@settings(max_examples=100)
def test_jpeg_truncation_rejected(seed, cut_pct):
    """Truncating a REAL jpeg at any point raises cleanly (or, when the
    cut lands inside entropy data, decodes a partial image)."""
    from PIL import Image

    from tensorflow_web_deploy_amd import _jpegc
    rng = np.random.default_rng(seed % 7)
    arr = (rng.random((40, 56, 3)) * 255).astype(np.uint8)
    b = io.BytesIO()
    Image.fromarray(arr, "RGB").save(b, "JPEG", quality=85)
    blob = b.getvalue()
    cut = max(2, len(blob) * cut_pct // 120)
    try:
        _jpegc.parse(blob[:cut])
    except (ValueError, RuntimeError):
        pass


@given(st.binary(max_size=300))
@settings(max_examples=200)
def test_graphdef_garbage_rejected(data):
    try:
        gd.parse_graphdef(data)
    except Exception:
        pass  # reject is fine; crash/hang is the failure


@given(st.integers(min_value=0, max_value=99))
def test_graphdef_roundtrip_random(seed):
    rng = np.random.default_rng(seed)
    g = gd.GraphDef()
    nodes = []
    for i in range(int(rng.integers(1, 6))):
        n = gd.NodeDef(name=f"n{i}", op=rng.choice(
            ["Const", "Conv2D", "Relu", "Weird/Op-Name_2"]))
        if i:
            n.input = [f"n{rng.integers(0, i)}"]
        if rng.random() < 0.7:
            arr = rng.standard_normal(
                tuple(rng.integers(1, 4, size=2))).astype(np.float32)
            n.attr["value"] = gd.AttrValue(tensor=arr)
        n.attr["s"] = gd.AttrValue(s=bytes(rng.integers(
            0, 256, size=int(rng.integers(0, 10))).astype(np.uint8)))
        nodes.append(n)
    g.node = nodes
    g2 = gd.parse_graphdef(gd.encode_graphdef(g))
    assert [n.name for n in g2.node] == [n.name for n in nodes]
    assert [n.op for n in g2.node] == [n.op for n in nodes]
    for a, b in zip(nodes, g2.node):
        assert a.input == b.input
        assert b.attr["s"].s == a.attr["s"].s
        if "value" in a.attr:
            np.testing.assert_array_equal(b.attr["value"].tensor,
                                          a.attr["value"].tensor)


@given(st.binary(max_size=64))
@settings(max_examples=200, deadline=None)
def test_ordered_code_decode_never_hangs(data):
    """Slice-key decoders on arbitrary bytes: parse or raise ValueError
    (never hang / never index out of range)."""
    from tensorflow_web_deploy_amd.ckpt import ordered_code as oc
    for fn in (oc.read_signed_num_increasing, oc.read_num_increasing,
               oc.read_string):
        try:
            fn(data, 0)
        except (ValueError, IndexError):
            pass
    try:
        oc.decode_tensor_name_slice(data)
    except (ValueError, IndexError, UnicodeDecodeError):
        pass


@given(st.integers(min_value=0, max_value=2**32), st.integers(0, 100))
@settings(max_examples=100, deadline=None)
def test_sliced_bundle_truncation_rejected(seed, cut_pct):
    """Truncating either bundle file of a sliced checkpoint must raise,
    not return silently-wrong weights."""
    import struct
    import tempfile

    from tensorflow_web_deploy_amd.ckpt.tensor_bundle import (BundleReader,
                                                              write_bundle)
    rng = np.random.default_rng(seed)
    full = rng.standard_normal((6, 4)).astype(np.float32)
    with tempfile.TemporaryDirectory() as td:
        prefix = td + "/variables"
        write_bundle(prefix, {},
                     sliced={"w": ([6, 4], [
                         ([(0, 3), (-1, -1)], full[:3]),
                         ([(3, 3), (-1, -1)], full[3:]),
                     ])})
        for suffix in (".index", ".data-00000-of-00001"):
            path = prefix + suffix
            with open(path, "rb") as fh:
                blob = fh.read()
            cut = len(blob) * cut_pct // 100
            if cut == len(blob):
                continue
            with open(path, "wb") as fh:
                fh.write(blob[:cut])
            try:
                r = BundleReader(prefix)
                got = r.get("w")
                # ok only if the cut did not reach this tensor's bytes
                np.testing.assert_array_equal(got, full)
            except (ValueError, KeyError, EOFError, struct.error,
                    FileNotFoundError, IndexError):
                pass
            with open(path, "wb") as fh:
                fh.write(blob)
