"""ResNet-50 v1.5 frozen-graph builder (NHWC, random-init weights).

BASELINE config 2/3: "ResNet-50 224x224 bf16, batch=64" — the headline
benchmark model. Standard v1.5 topology (stride-2 in the 3x3 of each
downsampling bottleneck): 7x7s2 stem -> maxpool3x3s2 -> [3,4,6,3]
bottlenecks (64/128/256/512 base widths) -> global mean -> fc1000 ->
softmax. Conv+BN+ReLU emitted as separate TF nodes; the IR optimizer
folds BN and fuses the blocks (engine/ir.py).
"""
from __future__ import annotations

from .builder import GraphBuilder

INPUT_NAME = "input"
OUTPUT_NAME = "softmax"
NUM_CLASSES = 1000


def _conv_bn(b: GraphBuilder, x: str, name: str, k: int, cin: int, cout: int,
             stride: int = 1, relu: bool = True, padding: str = "SAME") -> str:
    y = b.conv2d(x, name, k, k, cin, cout, stride, padding)
    y = b.batchnorm(y, f"{name}/bn", cout)
    if relu:
        y = b.relu(y, f"{name}/relu")
    return y


def _bottleneck(b: GraphBuilder, x: str, name: str, cin: int, width: int,
                stride: int) -> str:
    cout = width * 4
    if stride != 1 or cin != cout:
        shortcut = _conv_bn(b, x, f"{name}/shortcut", 1, cin, cout,
                            stride, relu=False)
    else:
        shortcut = x
    y = _conv_bn(b, x, f"{name}/conv1", 1, cin, width, 1)
    y = _conv_bn(b, y, f"{name}/conv2", 3, width, width, stride)
    y = _conv_bn(b, y, f"{name}/conv3", 1, width, cout, 1, relu=False)
    y = b.node("AddV2", f"{name}/add", [y, shortcut])
    y = b.relu(y, f"{name}/out")
    return y


def build_resnet50(seed: int = 0, num_classes: int = NUM_CLASSES):
    b = GraphBuilder(seed=seed)
    x = b.placeholder(INPUT_NAME, [-1, 224, 224, 3])
    y = _conv_bn(b, x, "stem", 7, 3, 64, 2)
    y = b.maxpool(y, "stem/pool", 3, 2)
    cin = 64
    for stage, (blocks, width, stride) in enumerate(
            [(3, 64, 1), (4, 128, 2), (6, 256, 2), (3, 512, 2)]):
        for blk in range(blocks):
            s = stride if blk == 0 else 1
            y = _bottleneck(b, y, f"stage{stage + 1}/block{blk + 1}", cin, width, s)
            cin = width * 4
    y = b.global_mean(y, "global_pool")
    y = b.dense(y, "fc", 2048, num_classes)
    b.node("Softmax", OUTPUT_NAME, [y])
    return b.graphdef()


def save_resnet50(path: str, seed: int = 0) -> None:
    from ..ckpt import graphdef as gd
    gd.save_frozen_graph(build_resnet50(seed), path)
