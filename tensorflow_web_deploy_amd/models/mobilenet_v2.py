"""MobileNetV2 frozen-graph builder (NHWC, random-init weights).

Standard 1.0/224 topology: 3x3s2 stem (32ch) -> 17 inverted-residual
bottlenecks (expand 1x1 + BN + Relu6, depthwise 3x3 + BN + Relu6,
project 1x1 + BN linear, residual when stride=1 and cin==cout) ->
1x1 conv to 1280 -> global mean -> fc1000 -> softmax. Emitted as plain
TF ops (Conv2D / DepthwiseConv2dNative / FusedBatchNormV3 / Relu6 /
AddV2); the IR passes fold BN into the (depthwise) conv weights and
fuse the blocks into ConvBlock / DepthwiseConvBlock with relu6
epilogues.
"""
from __future__ import annotations

from .builder import GraphBuilder

INPUT_NAME = "input"
OUTPUT_NAME = "softmax"
NUM_CLASSES = 1000

# (expansion t, cout, repeats, first-stride)
_CFG = [
    (1, 16, 1, 1),
    (6, 24, 2, 2),
    (6, 32, 3, 2),
    (6, 64, 4, 2),
    (6, 96, 3, 1),
    (6, 160, 3, 2),
    (6, 320, 1, 1),
]


def _conv_bn6(b: GraphBuilder, x: str, name: str, k: int, cin: int,
              cout: int, stride: int = 1, act: bool = True) -> str:
    y = b.conv2d(x, name, k, k, cin, cout, stride)
    y = b.batchnorm(y, f"{name}/bn", cout)
    if act:
        y = b.relu6(y, f"{name}/relu6")
    return y


def _inverted_residual(b: GraphBuilder, x: str, name: str, cin: int,
                       cout: int, stride: int, t: int) -> str:
    y = x
    cmid = cin * t
    if t != 1:
        y = _conv_bn6(b, y, f"{name}/expand", 1, cin, cmid)
    y = b.depthwise_conv2d(y, f"{name}/depthwise", 3, cmid, stride)
    y = b.batchnorm(y, f"{name}/depthwise/bn", cmid)
    y = b.relu6(y, f"{name}/depthwise/relu6")
    y = _conv_bn6(b, y, f"{name}/project", 1, cmid, cout, act=False)
    if stride == 1 and cin == cout:
        y = b.node("AddV2", f"{name}/add", [y, x])
    return y


def build_mobilenet_v2(seed: int = 0, num_classes: int = NUM_CLASSES):
    b = GraphBuilder(seed=seed)
    x = b.placeholder(INPUT_NAME, [-1, 224, 224, 3])
    y = _conv_bn6(b, x, "stem", 3, 3, 32, 2)
    cin = 32
    for bi, (t, cout, reps, stride) in enumerate(_CFG):
        for r in range(reps):
            s = stride if r == 0 else 1
            y = _inverted_residual(b, y, f"block{bi + 1}_{r + 1}",
                                   cin, cout, s, t)
            cin = cout
    y = _conv_bn6(b, y, "head", 1, cin, 1280)
    y = b.global_mean(y, "global_pool")
    y = b.dense(y, "fc", 1280, num_classes)
    b.node("Softmax", OUTPUT_NAME, [y])
    return b.graphdef()
