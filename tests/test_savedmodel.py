"""SavedModel + TensorBundle round-trip tests (no TF, CPU)."""
import numpy as np
import pytest
import torch

from tensorflow_web_deploy_amd.ckpt import sstable
from tensorflow_web_deploy_amd.ckpt.savedmodel import (load_savedmodel,
                                                       save_savedmodel)
from tensorflow_web_deploy_amd.ckpt.tensor_bundle import (BundleReader,
                                                          write_bundle)
from tensorflow_web_deploy_amd.engine import ir
from tensorflow_web_deploy_amd.engine.executor import Executor
from tensorflow_web_deploy_amd.models.lenet import build_lenet


def test_sstable_roundtrip():
    tw = sstable.TableWriter()
    items = {b"": b"header", b"alpha": b"1", b"beta/gamma": b"x" * 1000,
             b"zeta": np.arange(10, dtype=np.float32).tobytes()}
    for k, v in items.items():
        tw.add(k, v)
    data = tw.finish()
    out = sstable.read_table(data)
    assert out == items


def test_snappy_decoder():
    # literal + copy patterns, hand-built raw snappy stream
    # "abcdabcdabcd": literal "abcd" then copy(off=4, len=8)
    payload = b"\x0c"                      # uncompressed length 12
    payload += b"\x0c" + b"abcd"           # literal len 4 (tag (4-1)<<2)
    payload += bytes([(8 - 4) << 2 | 1, 4])  # copy-1: len 8, off 4
    assert sstable.snappy_decompress(payload) == b"abcdabcdabcd"


def test_bundle_roundtrip(tmp_path):
    tensors = {
        "layer1/weights": np.random.randn(5, 5, 1, 32).astype(np.float32),
        "layer1/bias": np.random.randn(32).astype(np.float32),
        "step": np.array([7], np.int64),
    }
    prefix = str(tmp_path / "variables" / "variables")
    write_bundle(prefix, tensors)
    r = BundleReader(prefix)
    assert set(r.keys()) == set(tensors)
    for k, v in tensors.items():
        got = r.get(k)
        assert got.dtype == v.dtype and got.shape == v.shape
        np.testing.assert_array_equal(got, v)


def test_savedmodel_variables_roundtrip(tmp_path):
    """LeNet exported as a VARIABLE-backed SavedModel must load and match
    the frozen-graph version exactly."""
    g = build_lenet(seed=9)
    sm_dir = str(tmp_path / "sm")
    save_savedmodel(g, sm_dir, ["input"], ["softmax"], as_variables=True)
    # the export actually moved weights into the bundle
    import os
    assert os.path.exists(os.path.join(sm_dir, "variables",
                                       "variables.index"))

    frozen = ir.optimize(ir.from_graphdef(g, ["softmax"]))
    loaded = ir.load_model(sm_dir)
    x = torch.randn(3, 28, 28, 1)
    y0 = Executor(frozen).run({"input": x})[0]
    y1 = Executor(loaded).run({"input": x})[0]
    torch.testing.assert_close(y0, y1, rtol=1e-6, atol=1e-7)


def test_savedmodel_signature_outputs(tmp_path):
    g = build_lenet(seed=2)
    sm_dir = str(tmp_path / "sm2")
    save_savedmodel(g, sm_dir, ["input"], ["softmax"], as_variables=False)
    loaded = ir.load_model(sm_dir)
    assert loaded.outputs == ["softmax"]


def test_sstable_crc_roundtrip_and_corruption():
    """Writer emits real masked crc32c; reader verifies and rejects
    flipped bytes (LevelDB record format)."""
    from tensorflow_web_deploy_amd.ckpt import sstable

    w = sstable.TableWriter()
    w.add(b"alpha", b"1" * 100)
    w.add(b"beta", b"2" * 50)
    data = w.finish()
    out = sstable.read_table(data)
    assert out[b"alpha"] == b"1" * 100 and out[b"beta"] == b"2" * 50

    # flip one byte inside the data block -> crc must catch it
    bad = bytearray(data)
    bad[3] ^= 0xFF
    with pytest.raises(ValueError, match="crc"):
        sstable.read_table(bytes(bad))

    # known-answer check (RFC 3720 test vector): crc32c of 32 zeros
    assert sstable.crc32c(b"\x00" * 32) == 0x8A9136AA


def test_tf2_object_graph_key_resolution(tmp_path):
    """Object-based (TF2) checkpoint: variable node names resolve
    through the TrackableObjectGraph's full_name -> checkpoint_key
    mapping, not by name suffixes."""
    from tensorflow_web_deploy_amd.ckpt import object_graph as og
    from tensorflow_web_deploy_amd.ckpt import wire
    from tensorflow_web_deploy_amd.ckpt.savedmodel import \
        _materialize_variables
    from tensorflow_web_deploy_amd.ckpt import graphdef as gd
    from tensorflow_web_deploy_amd.ckpt.tensor_bundle import (BundleReader,
                                                              write_bundle)

    # object graph: root -> model -> dense, with kernel stored under a
    # path that does NOT contain the variable's graph name
    root = og.TrackableObject()
    root.children = [(1, "model")]
    model = og.TrackableObject()
    model.children = [(2, "blk")]
    dense = og.TrackableObject()
    st = og.SerializedTensor()
    st.name = "kernel"
    st.full_name = "dense_1/kernel"          # the graph node name
    st.checkpoint_key = "model/blk/kernel/.ATTRIBUTES/VARIABLE_VALUE"
    dense.attributes = [st]
    proto = og.encode_object_graph([root, model, dense])

    w = np.arange(6, dtype=np.float32).reshape(2, 3)
    prefix = str(tmp_path / "variables" / "variables")
    # string-tensor framing: varint length + proto bytes
    write_bundle(prefix, {
        "model/blk/kernel/.ATTRIBUTES/VARIABLE_VALUE": w,
    })
    # append the object-graph entry by rewriting the bundle with a
    # uint8 view (write_bundle stores raw bytes for uint8)
    framed = wire.write_varint(len(proto)) + proto
    write_bundle(prefix, {
        "model/blk/kernel/.ATTRIBUTES/VARIABLE_VALUE": w,
        og.OBJECT_GRAPH_KEY: np.frombuffer(framed, np.uint8),
    })

    g = gd.GraphDef()
    n = gd.NodeDef(name="dense_1/kernel", op="VarHandleOp")
    g.node = [n]
    bundle = BundleReader(prefix)
    _materialize_variables(g, bundle)
    assert g.node[0].op == "Const"
    np.testing.assert_array_equal(g.node[0].attr["value"].tensor, w)


def test_object_graph_roundtrip_and_paths():
    from tensorflow_web_deploy_amd.ckpt import object_graph as og
    root = og.TrackableObject()
    root.children = [(1, "net")]
    net = og.TrackableObject()
    st = og.SerializedTensor()
    st.name = "VARIABLE_VALUE"
    st.checkpoint_key = "net/w/.ATTRIBUTES/VARIABLE_VALUE"
    net.attributes = [st]
    nodes = og.parse_object_graph(og.encode_object_graph([root, net]))
    assert len(nodes) == 2
    idx = og.ObjectGraphIndex(nodes)
    assert idx.by_path["net/VARIABLE_VALUE"] == \
        "net/w/.ATTRIBUTES/VARIABLE_VALUE"


def test_frozen_pb_roundtrip_full_model(tmp_path):
    """Full-model checkpoint round trip: build MobileNetV2, save as a
    frozen .pb via our writer, load via load_model, and verify the
    optimized loaded graph computes the SAME probs as the directly
    built one (weights, BN folding and fusion all survive the wire)."""
    from tensorflow_web_deploy_amd.ckpt import graphdef as gd
    from tensorflow_web_deploy_amd.engine import ir
    from tensorflow_web_deploy_amd.engine.executor import Executor
    from tensorflow_web_deploy_amd.engine.ir import load_model
    from tensorflow_web_deploy_amd.models.mobilenet_v2 import \
        build_mobilenet_v2

    g = build_mobilenet_v2(seed=7)
    path = str(tmp_path / "mnv2.pb")
    gd.save_frozen_graph(g, path)

    direct = ir.optimize(ir.from_graphdef(g, outputs=["softmax"]))
    loaded = load_model(path, outputs=["softmax"])

    x = torch.from_numpy(
        np.random.default_rng(0).random((1, 224, 224, 3),
                                        dtype=np.float32))
    p1 = Executor(direct).run({"input": x})[0]
    p2 = Executor(loaded).run({"input": x})[0]
    torch.testing.assert_close(p1, p2, atol=1e-5, rtol=1e-5)
