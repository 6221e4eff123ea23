"""Round-trip tests for the hand-written TF protobuf wire format."""
import numpy as np
import pytest

from tensorflow_web_deploy_amd.ckpt import graphdef as gd
from tensorflow_web_deploy_amd.ckpt import wire


def test_varint_roundtrip():
    for v in [0, 1, 127, 128, 300, 2**32, 2**63 - 1]:
        buf = wire.write_varint(v)
        out, pos = wire.read_varint(buf, 0)
        assert out == v and pos == len(buf)


def test_negative_int64_roundtrip():
    buf = wire.field_varint(3, -5)
    fields = list(wire.iter_fields(buf))
    assert fields[0][0] == 3
    assert wire.as_signed64(fields[0][2]) == -5


def test_tensor_roundtrip_dtypes():
    for arr in [
        np.random.randn(3, 4).astype(np.float32),
        np.arange(12, dtype=np.int32).reshape(3, 4),
        np.arange(6, dtype=np.int64),
        np.array([True, False, True]),
        np.random.randn(2, 2, 2, 2).astype(np.float32),
    ]:
        enc = gd.encode_tensor(arr)
        out = gd._parse_tensor(enc)
        assert out.dtype == arr.dtype
        np.testing.assert_array_equal(out, arr)


def test_scalar_broadcast_tensor():
    # TF semantics: single float_val broadcasts over the shape
    buf = (wire.field_varint(1, gd.DT_FLOAT)
           + wire.field_bytes(2, gd._encode_shape([2, 3]))
           + wire.field_float(5, 2.5))
    out = gd._parse_tensor(buf)
    assert out.shape == (2, 3)
    assert np.all(out == 2.5)


def test_graphdef_roundtrip():
    g = gd.GraphDef()
    n1 = gd.NodeDef(name="x", op="Placeholder")
    n1.attr["dtype"] = gd.AttrValue(type=gd.DT_FLOAT)
    n1.attr["shape"] = gd.AttrValue(shape=[-1, 28, 28, 1])
    n2 = gd.NodeDef(name="w", op="Const")
    w = np.random.randn(5, 5, 1, 32).astype(np.float32)
    n2.attr["value"] = gd.AttrValue(tensor=w)
    n2.attr["dtype"] = gd.AttrValue(type=gd.DT_FLOAT)
    n3 = gd.NodeDef(name="conv", op="Conv2D", input=["x", "w"])
    n3.attr["strides"] = gd.AttrValue(list_i=[1, 1, 1, 1])
    n3.attr["padding"] = gd.AttrValue(s=b"SAME")
    g.node = [n1, n2, n3]

    data = gd.encode_graphdef(g)
    g2 = gd.parse_graphdef(data)
    assert [n.name for n in g2.node] == ["x", "w", "conv"]
    assert g2.node[2].op == "Conv2D"
    assert g2.node[2].input == ["x", "w"]
    assert g2.node[2].attr["padding"].s == b"SAME"
    assert g2.node[2].attr["strides"].list_i == [1, 1, 1, 1]
    np.testing.assert_array_equal(g2.node[1].attr["value"].tensor, w)
    assert g2.node[0].attr["shape"].shape == [-1, 28, 28, 1]


def test_frozen_pb_file_roundtrip(tmp_path, lenet_pb):
    g = gd.load_frozen_graph(lenet_pb)
    assert len(g.node) > 10
    ops = {n.op for n in g.node}
    assert {"Placeholder", "Const", "Conv2D", "MaxPool", "MatMul",
            "Softmax"} <= ops


def test_truncated_raises():
    with pytest.raises(ValueError):
        list(wire.iter_fields(b"\x0a\xff"))


def test_bf16_frozen_graph_roundtrip(tmp_path):
    """bf16-stored weights in a frozen graph parse to f32 numpy."""
    import torch

    from tensorflow_web_deploy_amd.ckpt import wire
    w = np.random.randn(4, 4).astype(np.float32)
    wbf = torch.from_numpy(w).bfloat16()
    raw = wbf.view(torch.uint16).numpy().tobytes()
    enc = (wire.field_varint(1, gd.DT_BFLOAT16)
           + wire.field_bytes(2, gd._encode_shape([4, 4]))
           + wire.field_bytes(4, raw))
    out = gd._parse_tensor(enc)
    assert out.dtype == np.float32 and out.shape == (4, 4)
    np.testing.assert_allclose(out, wbf.float().numpy(), rtol=0, atol=0)
