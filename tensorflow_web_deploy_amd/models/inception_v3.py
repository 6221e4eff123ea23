"""Inception-v3 frozen-graph builder (299x299 NHWC, random-init).

BASELINE config 4: "Inception-v3 299x299 (deeper conv stack, mixed
kernel sizes)". Standard TF-slim v3 topology: stem convs -> 3x
inception-A (35x35) -> reduction-A -> 4x inception-B (17x17, 1x7/7x1
factorized convs) -> reduction-B -> 2x inception-C (8x8) -> global
avgpool -> fc -> softmax. Conv+BN+ReLU emitted as separate TF nodes
(IR folds/fuses); every branch concat is ConcatV2 over channels.
"""
from __future__ import annotations

from .builder import GraphBuilder

INPUT_NAME = "input"
OUTPUT_NAME = "softmax"
NUM_CLASSES = 1000


def _cbr(b, x, name, kh, kw, cin, cout, stride=1, padding="SAME"):
    y = b.conv2d_hw(x, name, kh, kw, cin, cout, stride, padding)
    y = b.batchnorm(y, f"{name}/bn", cout, eps=1e-3)
    return b.relu(y, f"{name}/relu")


def _inception_a(b, x, name, cin, pool_c):
    b1 = _cbr(b, x, f"{name}/b1x1", 1, 1, cin, 64)
    b5 = _cbr(b, x, f"{name}/b5x5_1", 1, 1, cin, 48)
    b5 = _cbr(b, b5, f"{name}/b5x5_2", 5, 5, 48, 64)
    b3 = _cbr(b, x, f"{name}/b3x3_1", 1, 1, cin, 64)
    b3 = _cbr(b, b3, f"{name}/b3x3_2", 3, 3, 64, 96)
    b3 = _cbr(b, b3, f"{name}/b3x3_3", 3, 3, 96, 96)
    bp = b.avgpool(x, f"{name}/pool", 3, 1, "SAME")
    bp = _cbr(b, bp, f"{name}/bpool", 1, 1, cin, pool_c)
    return b.concat([b1, b5, b3, bp], f"{name}/concat"), 64 + 64 + 96 + pool_c


def _reduction_a(b, x, name, cin):
    b3 = _cbr(b, x, f"{name}/b3x3", 3, 3, cin, 384, 2, "VALID")
    bd = _cbr(b, x, f"{name}/bd_1", 1, 1, cin, 64)
    bd = _cbr(b, bd, f"{name}/bd_2", 3, 3, 64, 96)
    bd = _cbr(b, bd, f"{name}/bd_3", 3, 3, 96, 96, 2, "VALID")
    bp = b.maxpool(x, f"{name}/pool", 3, 2, "VALID")
    return b.concat([b3, bd, bp], f"{name}/concat"), 384 + 96 + cin


def _inception_b(b, x, name, cin, c7):
    b1 = _cbr(b, x, f"{name}/b1x1", 1, 1, cin, 192)
    b7 = _cbr(b, x, f"{name}/b7_1", 1, 1, cin, c7)
    b7 = _cbr(b, b7, f"{name}/b7_2", 1, 7, c7, c7)
    b7 = _cbr(b, b7, f"{name}/b7_3", 7, 1, c7, 192)
    bd = _cbr(b, x, f"{name}/bd_1", 1, 1, cin, c7)
    bd = _cbr(b, bd, f"{name}/bd_2", 7, 1, c7, c7)
    bd = _cbr(b, bd, f"{name}/bd_3", 1, 7, c7, c7)
    bd = _cbr(b, bd, f"{name}/bd_4", 7, 1, c7, c7)
    bd = _cbr(b, bd, f"{name}/bd_5", 1, 7, c7, 192)
    bp = b.avgpool(x, f"{name}/pool", 3, 1, "SAME")
    bp = _cbr(b, bp, f"{name}/bpool", 1, 1, cin, 192)
    return b.concat([b1, b7, bd, bp], f"{name}/concat"), 768


def _reduction_b(b, x, name, cin):
    b3 = _cbr(b, x, f"{name}/b3_1", 1, 1, cin, 192)
    b3 = _cbr(b, b3, f"{name}/b3_2", 3, 3, 192, 320, 2, "VALID")
    b7 = _cbr(b, x, f"{name}/b7_1", 1, 1, cin, 192)
    b7 = _cbr(b, b7, f"{name}/b7_2", 1, 7, 192, 192)
    b7 = _cbr(b, b7, f"{name}/b7_3", 7, 1, 192, 192)
    b7 = _cbr(b, b7, f"{name}/b7_4", 3, 3, 192, 192, 2, "VALID")
    bp = b.maxpool(x, f"{name}/pool", 3, 2, "VALID")
    return b.concat([b3, b7, bp], f"{name}/concat"), 320 + 192 + cin


def _inception_c(b, x, name, cin):
    b1 = _cbr(b, x, f"{name}/b1x1", 1, 1, cin, 320)
    b3 = _cbr(b, x, f"{name}/b3_1", 1, 1, cin, 384)
    b3a = _cbr(b, b3, f"{name}/b3_2a", 1, 3, 384, 384)
    b3b = _cbr(b, b3, f"{name}/b3_2b", 3, 1, 384, 384)
    bd = _cbr(b, x, f"{name}/bd_1", 1, 1, cin, 448)
    bd = _cbr(b, bd, f"{name}/bd_2", 3, 3, 448, 384)
    bda = _cbr(b, bd, f"{name}/bd_3a", 1, 3, 384, 384)
    bdb = _cbr(b, bd, f"{name}/bd_3b", 3, 1, 384, 384)
    bp = b.avgpool(x, f"{name}/pool", 3, 1, "SAME")
    bp = _cbr(b, bp, f"{name}/bpool", 1, 1, cin, 192)
    return (b.concat([b1, b3a, b3b, bda, bdb, bp], f"{name}/concat"),
            320 + 4 * 384 + 192)


def build_inception_v3(seed: int = 0, num_classes: int = NUM_CLASSES):
    b = GraphBuilder(seed=seed)
    x = b.placeholder(INPUT_NAME, [-1, 299, 299, 3])
    y = _cbr(b, x, "stem/conv1", 3, 3, 3, 32, 2, "VALID")    # 149
    y = _cbr(b, y, "stem/conv2", 3, 3, 32, 32, 1, "VALID")   # 147
    y = _cbr(b, y, "stem/conv3", 3, 3, 32, 64, 1, "SAME")    # 147
    y = b.maxpool(y, "stem/pool1", 3, 2, "VALID")            # 73
    y = _cbr(b, y, "stem/conv4", 1, 1, 64, 80, 1, "VALID")
    y = _cbr(b, y, "stem/conv5", 3, 3, 80, 192, 1, "VALID")  # 71
    y = b.maxpool(y, "stem/pool2", 3, 2, "VALID")            # 35
    cin = 192
    y, cin = _inception_a(b, y, "mixed0", cin, 32)
    y, cin = _inception_a(b, y, "mixed1", cin, 64)
    y, cin = _inception_a(b, y, "mixed2", cin, 64)
    y, cin = _reduction_a(b, y, "mixed3", cin)               # 17x17x768
    for i, c7 in enumerate((128, 160, 160, 192)):
        y, cin = _inception_b(b, y, f"mixed{4 + i}", cin, c7)
    y, cin = _reduction_b(b, y, "mixed8", cin)               # 8x8x1280
    y, cin = _inception_c(b, y, "mixed9", cin)
    y, cin = _inception_c(b, y, "mixed10", cin)              # 8x8x2048
    y = b.global_mean(y, "global_pool")
    y = b.dense(y, "fc", 2048, num_classes)
    b.node("Softmax", OUTPUT_NAME, [y])
    return b.graphdef()


def save_inception_v3(path: str, seed: int = 0) -> None:
    from ..ckpt import graphdef as gd
    gd.save_frozen_graph(build_inception_v3(seed), path)
