"""MNIST LeNet frozen-graph builder — BASELINE config 1 ("MNIST LeNet
frozen .pb served over Flask on CPU"). Mirrors the classic LeNet-5 shape
used by TF-1.x MNIST demos: 28x28x1 -> conv5x5x32 -> pool -> conv5x5x64
-> pool -> fc1024 -> fc10 -> softmax.
"""
from __future__ import annotations

from .builder import GraphBuilder

INPUT_NAME = "input"
OUTPUT_NAME = "softmax"


def build_lenet(seed: int = 0):
    b = GraphBuilder(seed=seed)
    x = b.placeholder(INPUT_NAME, [-1, 28, 28, 1])
    y = b.conv2d(x, "conv1", 5, 5, 1, 32)
    y = b.bias_add(y, "conv1/bias_add", 32)
    y = b.relu(y, "conv1/relu")
    y = b.maxpool(y, "pool1", 2, 2)
    y = b.conv2d(y, "conv2", 5, 5, 32, 64)
    y = b.bias_add(y, "conv2/bias_add", 64)
    y = b.relu(y, "conv2/relu")
    y = b.maxpool(y, "pool2", 2, 2)
    y = b.reshape(y, "flatten", [-1, 7 * 7 * 64])
    y = b.dense(y, "fc1", 7 * 7 * 64, 1024)
    y = b.relu(y, "fc1/relu")
    y = b.dense(y, "fc2", 1024, 10)
    b.node("Softmax", OUTPUT_NAME, [y])
    return b.graphdef()


def save_lenet(path: str, seed: int = 0) -> None:
    from ..ckpt import graphdef as gd
    gd.save_frozen_graph(build_lenet(seed), path)
