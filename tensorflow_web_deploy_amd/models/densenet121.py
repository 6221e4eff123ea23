"""DenseNet-121 frozen-graph builder (NHWC, random-init weights).

Seventh built-in family: PRE-ACTIVATION composites (BN -> ReLU -> conv,
so the BN cannot fold into a producing conv and runs as the standalone
BnScaleBias/scale_shift kernel) and the deepest concat chains of any
family (each dense layer concatenates onto a growing feature map) —
stresses exactly the graph patterns the other six families don't.
"""
from __future__ import annotations

from .builder import GraphBuilder

INPUT_NAME = "input"
OUTPUT_NAME = "softmax"
NUM_CLASSES = 1000
GROWTH = 32
_BLOCKS = [6, 12, 24, 16]


def _bn_relu(b: GraphBuilder, x: str, name: str, c: int) -> str:
    y = b.batchnorm(x, f"{name}/bn", c)
    return b.relu(y, f"{name}/relu")


def _dense_layer(b: GraphBuilder, x: str, name: str, cin: int) -> str:
    """BN-ReLU-conv1x1(4k) -> BN-ReLU-conv3x3(k); returns the k new
    channels (caller concatenates)."""
    y = _bn_relu(b, x, f"{name}/pre1", cin)
    y = b.conv2d(y, f"{name}/conv1", 1, 1, cin, 4 * GROWTH, 1, "SAME")
    y = _bn_relu(b, y, f"{name}/pre2", 4 * GROWTH)
    y = b.conv2d(y, f"{name}/conv2", 3, 3, 4 * GROWTH, GROWTH, 1, "SAME")
    return y


def build_densenet121(seed: int = 0, num_classes: int = NUM_CLASSES):
    b = GraphBuilder(seed=seed)
    x = b.placeholder(INPUT_NAME, [-1, 224, 224, 3])
    y = b.conv2d(x, "stem", 7, 7, 3, 64, 2, "SAME")
    y = b.batchnorm(y, "stem/bn", 64)
    y = b.relu(y, "stem/relu")
    y = b.maxpool(y, "stem/pool", 3, 2)
    cin = 64
    for bi, nlayers in enumerate(_BLOCKS):
        for li in range(nlayers):
            new = _dense_layer(b, y, f"db{bi + 1}/l{li + 1}", cin)
            y = b.concat([y, new], f"db{bi + 1}/l{li + 1}/cat")
            cin += GROWTH
        if bi < len(_BLOCKS) - 1:
            # transition: BN-ReLU-conv1x1(c/2)-avgpool2s2
            y = _bn_relu(b, y, f"t{bi + 1}/pre", cin)
            cin //= 2
            y = b.conv2d(y, f"t{bi + 1}/conv", 1, 1, cin * 2, cin, 1,
                         "SAME")
            y = b.avgpool(y, f"t{bi + 1}/pool", 2, 2)
    y = _bn_relu(b, y, "head/pre", cin)
    y = b.global_mean(y, "head/gap")
    y = b.dense(y, "fc", cin, num_classes)
    b.softmax(y, OUTPUT_NAME)
    return b.graphdef()
