"""Model-family builders: structure + CPU forward sanity."""
import numpy as np
import torch
from collections import Counter

from tensorflow_web_deploy_amd.conf.config import Config
from tensorflow_web_deploy_amd.engine.service import LocalEngine


def _run(model, h, w, c, nc, batch=1):
    cfg = Config(model_name=model, device="cpu", input_height=h,
                 input_width=w, input_channels=c, num_classes=nc)
    eng = LocalEngine(cfg)
    x = np.random.rand(batch, h, w, c).astype(np.float32)
    p = eng.infer(x)
    assert p.shape == (batch, nc)
    np.testing.assert_allclose(p.sum(1), 1.0, atol=1e-4)
    return eng, Counter(n.op for n in eng.graph.nodes)


def test_inception_v3():
    eng, ops = _run("inception_v3", 299, 299, 3, 1000)
    assert ops["ConvBlock"] == 94
    assert ops["ConcatV2"] == 11
    assert ops["AvgPool"] == 9 and ops["MaxPool"] == 4


def test_mobilenet_v2():
    eng, ops = _run("mobilenet_v2", 224, 224, 3, 1000)
    # 17 inverted residuals; all BN/Relu6 absorbed into the blocks
    assert ops["DepthwiseConvBlock"] == 17
    assert ops["ConvBlock"] == 35
    assert "Relu6" not in ops and "FusedBatchNormV3" not in ops
    # depthwise blocks carry the relu6 epilogue
    dw = [n for n in eng.graph.nodes if n.op == "DepthwiseConvBlock"]
    assert all(n.attrs.get("relu6") and n.attrs.get("relu") for n in dw)
    # project convs are linear (no act), expand convs are relu6
    cb = [n for n in eng.graph.nodes if n.op == "ConvBlock"]
    assert any(not n.attrs.get("relu") for n in cb)
    assert any(n.attrs.get("relu6") for n in cb)


def test_vit_b16():
    eng, ops = _run("vit_b16", 224, 224, 3, 1000)
    # the primitive-op graph must fully fuse
    assert ops["LayerNorm"] == 25   # 2/layer + final
    assert ops["DenseND"] == 72     # qkv/proj/mlp with fused bias+act
    assert ops["Softmax"] == 13     # 12 attention + classifier
    assert "SquaredDifference" not in ops and "Erf" not in ops
    assert "Gelu" not in ops        # absorbed into DenseND epilogue


def test_vit_layernorm_fusion_equivalence():
    """Fused LayerNorm/Gelu must match the raw primitive graph."""
    from tensorflow_web_deploy_amd.engine import ir
    from tensorflow_web_deploy_amd.engine.executor import Executor
    from tensorflow_web_deploy_amd.models.builder import GraphBuilder
    from tensorflow_web_deploy_amd.models.vit import _gelu, _layernorm
    b = GraphBuilder(seed=4)
    x = b.placeholder("input", [-1, 6, 32])
    y = _layernorm(b, x, "ln", 32)
    y = _gelu(b, y, "gelu")
    b.node("Identity", "out", [y])
    g = b.graphdef()
    raw = ir.toposort(ir.strip_identity(ir.from_graphdef(g, ["out"])))
    opt = ir.optimize(ir.from_graphdef(g, ["out"]))
    ops = [n.op for n in opt.nodes]
    assert ops == ["LayerNorm", "Gelu"], ops
    xv = torch.randn(3, 6, 32)
    y_raw = Executor(raw).run({"input": xv})[0]
    y_opt = Executor(opt).run({"input": xv})[0]
    torch.testing.assert_close(y_raw, y_opt, rtol=1e-5, atol=1e-6)


def test_vgg16():
    eng, ops = _run("vgg16", 224, 224, 3, 1000)
    assert ops["ConvBlock"] == 13
    assert ops["MaxPool"] == 5
    # fc6/fc7/fc8 fuse to Dense blocks
    assert ops.get("Dense", 0) + ops.get("DenseBlock", 0) == 3


def test_densenet121():
    eng, ops = _run("densenet121", 224, 224, 3, 1000)
    # 120 convs (stem + 58x2 dense-layer convs + 3 transitions), all
    # pre-activation: BN runs standalone as BnScaleBias
    assert ops["ConvBlock"] == 120
    assert ops["BnScaleBias"] == 62
    assert ops["ConcatV2"] == 58
    assert ops["AvgPool"] == 3
