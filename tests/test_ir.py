"""IR lowering-pass tests: optimized graph must match the raw graph."""
import numpy as np
import pytest
import torch

from tensorflow_web_deploy_amd.ckpt import graphdef as gd
from tensorflow_web_deploy_amd.engine import ir
from tensorflow_web_deploy_amd.engine.executor import Executor
from tensorflow_web_deploy_amd.models.lenet import build_lenet
from tensorflow_web_deploy_amd.models.resnet50 import build_resnet50


def _run(graph, feeds):
    return Executor(graph).run(feeds)[0]


def test_lenet_fusion_equivalence():
    g = build_lenet(seed=3)
    raw = ir.toposort(ir.strip_identity(ir.from_graphdef(g, ["softmax"])))
    opt = ir.optimize(ir.from_graphdef(g, ["softmax"]))
    x = torch.randn(4, 28, 28, 1)
    y_raw = _run(raw, {"input": x})
    y_opt = _run(opt, {"input": x})
    torch.testing.assert_close(y_raw, y_opt, rtol=1e-5, atol=1e-6)
    ops = [n.op for n in opt.nodes]
    assert "Conv2D" not in ops and "BiasAdd" not in ops
    assert ops.count("ConvBlock") == 2 and ops.count("Dense") == 2


def test_resnet_block_bn_fold_equivalence():
    # one-bottleneck mini-resnet: BN fold + residual fusion must be exact
    from tensorflow_web_deploy_amd.models.builder import GraphBuilder
    from tensorflow_web_deploy_amd.models.resnet50 import _bottleneck
    b = GraphBuilder(seed=11)
    x = b.placeholder("input", [-1, 16, 16, 8])
    y = _bottleneck(b, x, "blk", 8, 4, 1)
    b.node("Softmax", "softmax", [b.global_mean(y, "gp")])
    g = b.graphdef()
    raw = ir.toposort(ir.strip_identity(ir.from_graphdef(g, ["softmax"])))
    opt = ir.optimize(ir.from_graphdef(g, ["softmax"]))
    xv = torch.randn(2, 16, 16, 8)
    y_raw = _run(raw, {"input": xv})
    y_opt = _run(opt, {"input": xv})
    torch.testing.assert_close(y_raw, y_opt, rtol=1e-4, atol=1e-5)
    ops = [n.op for n in opt.nodes]
    assert "FusedBatchNormV3" not in ops
    assert "AddV2" not in ops  # absorbed into the main-branch ConvBlock


def test_resnet50_structure():
    g = build_resnet50(seed=0)
    opt = ir.optimize(ir.from_graphdef(g, ["softmax"]))
    from collections import Counter
    c = Counter(n.op for n in opt.nodes)
    assert c["ConvBlock"] == 53
    assert c["MaxPool"] == 1 and c["Dense"] == 1 and c["Softmax"] == 1
    # exactly 16 residual-fused blocks
    res = [n for n in opt.nodes if n.op == "ConvBlock" and n.attrs.get("residual")]
    assert len(res) == 16


def test_same_padding_asymmetric():
    # TF SAME on even input with stride 2: pad goes bottom/right
    from tensorflow_web_deploy_amd.engine.executor import tf_same_pad
    x = torch.arange(16, dtype=torch.float32).reshape(1, 1, 4, 4)
    p = tf_same_pad(x, 3, 3, 2, 2)
    assert p.shape == (1, 1, 5, 5)
    assert p[0, 0, 4].sum() == 0 and p[0, 0, :, 4].sum() == 0  # pad at end
    assert p[0, 0, 0, 0] == 0.0 and p[0, 0, 0, 1] == 1.0  # no pad at start


def test_glue_ops_cpu_reference():
    """Rare glue ops from real TF exports run through the CPU executor
    with TF semantics (Slice with size=-1, Fill, activations)."""
    import torch

    from tensorflow_web_deploy_amd.engine.executor import _CPU_OPS as CPU_OPS
    from tensorflow_web_deploy_amd.engine.ir import Node

    x = torch.tensor([[-2.0, -0.5, 0.0, 1.5]])

    def run(op, ins, **attrs):
        return CPU_OPS[op](Node("n", op, [], attrs), ins, {})

    assert torch.equal(run("Maximum", [x, torch.zeros_like(x)]),
                       torch.clamp(x, min=0))
    assert torch.equal(run("Minimum", [x, torch.zeros_like(x)]),
                       torch.clamp(x, max=0))
    lr = run("LeakyRelu", [x], alpha=0.1)
    assert torch.allclose(lr, torch.nn.functional.leaky_relu(x, 0.1))
    assert torch.allclose(run("Square", [x]), x * x)
    assert torch.allclose(run("Abs", [x]), x.abs())
    y = torch.arange(24.0).reshape(2, 3, 4)
    s = run("Slice", [y, torch.tensor([0, 1, 2]),
                      torch.tensor([2, -1, 2])])
    assert torch.equal(s, y[0:2, 1:, 2:4])
    f = run("Fill", [torch.tensor([2, 2]), torch.tensor(3.5)])
    assert torch.equal(f, torch.full((2, 2), 3.5))


def test_resize_bilinear_cpu():
    import torch

    from tensorflow_web_deploy_amd.engine.executor import _CPU_OPS
    from tensorflow_web_deploy_amd.engine.ir import Node
    x = torch.rand(1, 6, 8, 3)
    # half_pixel_centers=True == torch align_corners=False
    out = _CPU_OPS["ResizeBilinear"](
        Node("r", "ResizeBilinear", [],
             {"align_corners": False, "half_pixel_centers": True}),
        [x, torch.tensor([12, 16])], {})
    assert out.shape == (1, 12, 16, 3)
    ref = torch.nn.functional.interpolate(
        x.permute(0, 3, 1, 2), size=(12, 16), mode="bilinear",
        align_corners=False).permute(0, 2, 3, 1)
    assert torch.allclose(out, ref)


def test_resize_bilinear_tf1_default_asymmetric():
    """TF1 default (align_corners=False, half_pixel_centers=False) maps
    src = dst * in/out — NOT half-pixel (ADVICE round-1 low). Check
    against a scalar reference implementation."""
    import torch

    from tensorflow_web_deploy_amd.engine.executor import _CPU_OPS
    from tensorflow_web_deploy_amd.engine.ir import Node
    torch.manual_seed(0)
    x = torch.rand(2, 5, 7, 3)
    out = _CPU_OPS["ResizeBilinear"](
        Node("r", "ResizeBilinear", [], {"align_corners": False}),
        [x, torch.tensor([9, 4])], {})
    H, W, oh, ow = 5, 7, 9, 4
    ref = torch.empty(2, oh, ow, 3)
    for oy in range(oh):
        fy = oy * H / oh
        y0 = min(int(fy), H - 1)
        y1 = min(y0 + 1, H - 1)
        wy = fy - y0
        for ox in range(ow):
            fx = ox * W / ow
            x0 = min(int(fx), W - 1)
            x1 = min(x0 + 1, W - 1)
            wx = fx - x0
            top = x[:, y0, x0] + wx * (x[:, y0, x1] - x[:, y0, x0])
            bot = x[:, y1, x0] + wx * (x[:, y1, x1] - x[:, y1, x0])
            ref[:, oy, ox] = top + wy * (bot - top)
    assert torch.allclose(out, ref, atol=1e-5)


def test_multi_output_split_through_executor():
    """Split/Unpack produce name:k tensors that downstream nodes can
    reference (real exported graphs use this)."""
    import numpy as np
    import torch

    from tensorflow_web_deploy_amd.engine.executor import Executor
    from tensorflow_web_deploy_amd.engine.ir import Graph, Node
    consts = {"axis": np.array(1, np.int32)}
    nodes = [
        Node("x", "Placeholder", [], {}),
        Node("sp", "Split", ["axis", "x"], {"num_split": 2}),
        Node("sum", "Add", ["sp", "sp:1"], {}),  # sp == sp:0
        Node("packed", "Pack", ["sp:0", "sp:1"], {"axis": 0}),
    ]
    g = Graph(nodes, consts, ["x"], ["sum", "packed"])
    ex = Executor(g)
    x = torch.arange(8.0).reshape(2, 4)
    out, packed = ex.run({"x": x})
    assert torch.equal(out, x[:, :2] + x[:, 2:])
    assert packed.shape == (2, 2, 2)


@pytest.mark.parametrize("family", ["lenet", "resnet50", "mobilenet_v2",
                                    "inception_v3", "vit_b16", "vgg16",
                                    "densenet121"])
def test_optimize_preserves_semantics(family):
    """optimize() (BN fold, block fusion, concat elimination, ...) must
    not change the function: raw-graph execution == optimized-graph
    execution on the same random input."""
    import numpy as np
    import torch

    from tensorflow_web_deploy_amd.engine import ir
    from tensorflow_web_deploy_amd.engine.executor import Executor
    from tensorflow_web_deploy_amd.models import (densenet121,
                                                  inception_v3, lenet,
                                                  mobilenet_v2, resnet50,
                                                  vgg16, vit)
    build = {"lenet": lenet.build_lenet,
             "resnet50": resnet50.build_resnet50,
             "mobilenet_v2": mobilenet_v2.build_mobilenet_v2,
             "inception_v3": inception_v3.build_inception_v3,
             "vit_b16": vit.build_vit_b16,
             "vgg16": vgg16.build_vgg16,
             "densenet121": densenet121.build_densenet121}[family]
    g = build(seed=3)
    raw = ir.from_graphdef(g, outputs=["softmax"])
    opt = ir.optimize(ir.from_graphdef(g, outputs=["softmax"]))
    shape = {"lenet": (1, 28, 28, 1),
             "inception_v3": (1, 299, 299, 3)}.get(family,
                                                   (1, 224, 224, 3))
    x = torch.from_numpy(
        np.random.default_rng(1).random(shape, dtype=np.float32))
    p_raw = Executor(raw).run({"input": x})[0]
    p_opt = Executor(opt).run({"input": x})[0]
    torch.testing.assert_close(p_raw, p_opt, atol=2e-4, rtol=2e-4)


def test_assign_streams_inception_branches():
    """Inter-branch stream assignment (gpu_executor.assign_streams is a
    pure function): Inception blocks fork 2-4 independent conv chains —
    sibling chains must land on different streams, joins and trunk stay
    on stream 0, and chain membership never crosses a multi-input op."""
    from tensorflow_web_deploy_amd.conf.config import Config
    from tensorflow_web_deploy_amd.engine.gpu_executor import assign_streams
    from tensorflow_web_deploy_amd.engine.service import build_graph
    cfg = Config(model_name="inception_v3", device="cpu",
                 input_height=299, input_width=299, input_channels=3,
                 num_classes=1000)
    g = build_graph(cfg)
    sid = assign_streams(g.nodes, g.consts, 2)
    n_side = sum(1 for v in sid.values() if v)
    assert n_side >= 20, f"expected many side-chain nodes, got {n_side}"
    prods = {n.name: n for n in g.nodes}
    # no multi-input (join) node off stream 0
    for n in g.nodes:
        ins = {i.split(":")[0] for i in n.inputs if i.split(":")[0] in prods}
        if len(ins) > 1:
            assert sid[n.name] == 0, n.name
    # ViT / ResNet-style linear trunks: residual Adds are joins -> 0;
    # spot-check ResNet stays effectively single-stream
    cfg2 = Config(model_name="resnet50", device="cpu", input_height=224,
                  input_width=224, input_channels=3, num_classes=1000)
    g2 = build_graph(cfg2)
    sid2 = assign_streams(g2.nodes, g2.consts, 2)
    frac = sum(1 for v in sid2.values() if v) / max(1, len(sid2))
    assert frac < 0.5, frac


def test_tile_cluster_remap_bijective():
    """Mirror of the 4x4 tile-clustering + XCD remap in gemm_conv.hip:
    every (tm, tn) tile must be produced exactly once for any grid with
    both dims divisible by 4 (a hole would silently skip output tiles)."""
    def remap(bid, nwg):
        q, r = nwg >> 3, nwg & 7
        xcd, off = bid & 7, bid >> 3
        return (xcd * (q + 1) if xcd < r
                else r * (q + 1) + (xcd - r) * q) + off

    for tiles_m, tiles_n in [(16, 16), (32, 32), (4, 128), (128, 4),
                             (20, 8), (8, 20), (196, 4)]:
        nwg = tiles_m * tiles_n
        seen = set()
        bpr = tiles_m >> 2
        for hw_bid in range(nwg):
            bid = remap(hw_bid, nwg)
            blk, wi = bid >> 4, bid & 15
            bm, bn = blk % bpr, blk // bpr
            tm = bm * 4 + (wi & 3)
            tn = bn * 4 + (wi >> 2)
            assert 0 <= tm < tiles_m and 0 <= tn < tiles_n, \
                (tiles_m, tiles_n, hw_bid, tm, tn)
            seen.add((tm, tn))
        assert len(seen) == nwg, (tiles_m, tiles_n, len(seen))


def test_stream_schedule_event_wiring():
    """_build_stream_schedule (pure logic): cross-stream edges must get
    record/wait events, deps must pass THROUGH step-less nodes
    (eliminated concats / view reshapes), and fused-residual attrs
    count as edges — the two round-2 bug classes."""
    from tensorflow_web_deploy_amd.engine.gpu_executor import GpuExecutor
    from tensorflow_web_deploy_amd.engine.ir import Graph, Node

    # fork: x -> (a-chain | b-chain) -> cat(step-less) -> tail
    # tail also takes a 'residual' attr ref to b.
    nodes = [
        Node("x", "Placeholder", []),
        Node("a", "ConvBlock", ["x"], {}),
        Node("b", "ConvBlock", ["x"], {}),
        Node("cat", "ConcatV2", ["a", "b"], {}),
        Node("tail", "ConvBlock", ["cat"], {"residual": "b"}),
    ]
    g = Graph(nodes, {}, ["x"], ["tail"])

    class _P:
        pass

    plan = _P()
    # spans: x steps [0,1), a [1,2), b [2,3), cat NO steps, tail [3,4)
    spans = [("x", 0), ("a", 1), ("b", 2), ("cat", 3), ("tail", 3)]
    ex = GpuExecutor.__new__(GpuExecutor)  # no GPU init needed
    ex._build_stream_schedule(plan, g, spans, 4)
    assert plan.sched is not None
    sched = {i: (sid, waits, rec) for (i, sid, waits, rec) in plan.sched}
    # a and b are sibling heavy consumers of x -> different streams
    sid_a, sid_b = sched[1][0], sched[2][0]
    assert {sid_a, sid_b} == {0, 1}
    # whichever chain is on stream 1 must RECORD an event, and tail
    # (stream 0, reading through the step-less cat AND the residual)
    # must WAIT on it
    side_step = 1 if sid_a == 1 else 2
    rec = sched[side_step][2]
    assert rec is not None
    assert rec in sched[3][1], (plan.sched,)
    # the stream-0 sibling needs no event for the stream-0 consumer
    main_step = 2 if side_step == 1 else 1
    # x (stream 0) -> side chain start must wait on x's event only if
    # x recorded one; x is a producer consumed cross-stream -> event
    assert sched[0][2] is not None
    assert sched[0][2] in sched[side_step][1]
    _ = main_step
