"""VGG-16 frozen-graph builder (NHWC, random-init weights).

Sixth built-in family (beyond the BASELINE configs — breadth check for
the checkpoint/executor surface): the classic plain-conv topology
exercises the biggest implicit-GEMM shapes in the suite (3x3 convs up
to C=512 at 56²-224² maps, plus the 25088->4096 dense head, the
largest single GEMM of any built-in model). Conv+BiasAdd+ReLU emitted
as separate TF nodes; the IR optimizer fuses the blocks.
"""
from __future__ import annotations

from .builder import GraphBuilder

INPUT_NAME = "input"
OUTPUT_NAME = "softmax"
NUM_CLASSES = 1000

# (convs per stage, channels)
_STAGES = [(2, 64), (2, 128), (3, 256), (3, 512), (3, 512)]


def build_vgg16(seed: int = 0, num_classes: int = NUM_CLASSES):
    b = GraphBuilder(seed=seed)
    x = b.placeholder(INPUT_NAME, [-1, 224, 224, 3])
    y = x
    cin = 3
    for si, (reps, ch) in enumerate(_STAGES):
        for ri in range(reps):
            name = f"conv{si + 1}_{ri + 1}"
            y = b.conv2d(y, name, 3, 3, cin, ch, 1, "SAME")
            y = b.bias_add(y, f"{name}/bias", ch)
            y = b.relu(y, f"{name}/relu")
            cin = ch
        y = b.maxpool(y, f"pool{si + 1}", 2, 2)
    # 7x7x512 -> flatten -> fc4096 -> fc4096 -> fc1000
    y = b.reshape(y, "flatten", [-1, 7 * 7 * 512])
    y = b.dense(y, "fc6", 7 * 7 * 512, 4096)
    y = b.relu(y, "fc6/relu")
    y = b.dense(y, "fc7", 4096, 4096)
    y = b.relu(y, "fc7/relu")
    y = b.dense(y, "fc8", 4096, num_classes)
    b.softmax(y, OUTPUT_NAME)
    return b.graphdef()
