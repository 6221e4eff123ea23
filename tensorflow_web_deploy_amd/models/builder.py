"""GraphDef builder: construct TF-format frozen graphs with our own proto
writer (ckpt/graphdef.py). Used to generate model fixtures and the
random-init benchmark models (BASELINE.json: "synthetic 224x224 images /
random-init weights") — no TF and no network required.
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import numpy as np

from ..ckpt import graphdef as gd


class GraphBuilder:
    def __init__(self, seed: int = 0):
        self.g = gd.GraphDef()
        self.rng = np.random.default_rng(seed)
        self._names = set()

    def _uniq(self, name: str) -> str:
        if name not in self._names:
            self._names.add(name)
            return name
        i = 1
        while f"{name}_{i}" in self._names:
            i += 1
        self._names.add(f"{name}_{i}")
        return f"{name}_{i}"

    def node(self, op: str, name: str, inputs: Sequence[str] = (),
             **attrs) -> str:
        name = self._uniq(name)
        nd = gd.NodeDef(name=name, op=op, input=list(inputs))
        for k, v in attrs.items():
            nd.attr[k] = self._attr(v)
        self.g.node.append(nd)
        return name

    @staticmethod
    def _attr(v) -> gd.AttrValue:
        if isinstance(v, gd.AttrValue):
            return v
        if isinstance(v, bool):
            return gd.AttrValue(b=v)
        if isinstance(v, int):
            return gd.AttrValue(i=v)
        if isinstance(v, float):
            return gd.AttrValue(f=v)
        if isinstance(v, str):
            return gd.AttrValue(s=v.encode())
        if isinstance(v, bytes):
            return gd.AttrValue(s=v)
        if isinstance(v, np.ndarray):
            return gd.AttrValue(tensor=v)
        if isinstance(v, (list, tuple)):
            if all(isinstance(x, int) for x in v):
                return gd.AttrValue(list_i=list(v))
            if all(isinstance(x, float) for x in v):
                return gd.AttrValue(list_f=list(v))
        raise TypeError(f"cannot encode attr {v!r}")

    def placeholder(self, name: str, shape: List[int],
                    dtype: int = gd.DT_FLOAT) -> str:
        name = self._uniq(name)
        nd = gd.NodeDef(name=name, op="Placeholder")
        nd.attr["dtype"] = gd.AttrValue(type=dtype)
        nd.attr["shape"] = gd.AttrValue(shape=shape)
        self.g.node.append(nd)
        return name

    def const(self, name: str, arr: np.ndarray) -> str:
        name = self._uniq(name)
        arr = np.asarray(arr)
        nd = gd.NodeDef(name=name, op="Const")
        nd.attr["dtype"] = gd.AttrValue(type=gd._NUMPY_TO_DT[arr.dtype])
        nd.attr["value"] = gd.AttrValue(tensor=arr)
        self.g.node.append(nd)
        return name

    # -------------------------------------------------- layer helpers

    def weight(self, name: str, shape, scale: Optional[float] = None) -> str:
        fan_in = int(np.prod(shape[:-1])) if len(shape) > 1 else shape[0]
        if scale is None:
            scale = float(np.sqrt(2.0 / max(fan_in, 1)))
        w = self.rng.standard_normal(shape, dtype=np.float32) * scale
        return self.const(name, w)

    def conv2d(self, x: str, name: str, kh: int, kw: int, cin: int,
               cout: int, stride: int = 1, padding: str = "SAME") -> str:
        w = self.weight(f"{name}/weights", (kh, kw, cin, cout))
        return self.node("Conv2D", name, [x, w],
                         strides=[1, stride, stride, 1], padding=padding,
                         data_format="NHWC", T=gd.AttrValue(type=gd.DT_FLOAT))

    def conv2d_hw(self, x: str, name: str, kh: int, kw: int, cin: int,
                  cout: int, stride: int = 1, padding: str = "SAME") -> str:
        """Asymmetric-kernel conv (Inception 1x7/7x1 factorizations)."""
        return self.conv2d(x, name, kh, kw, cin, cout, stride, padding)

    def concat(self, xs, name: str, axis: int = 3) -> str:
        ax = self.const(f"{name}/axis", np.array(axis, np.int32))
        return self.node("ConcatV2", name, list(xs) + [ax], N=len(xs))

    def bias_add(self, x: str, name: str, c: int, zero: bool = False) -> str:
        b = self.const(f"{name}/bias",
                       np.zeros(c, np.float32) if zero else
                       (self.rng.standard_normal(c).astype(np.float32) * 0.01))
        return self.node("BiasAdd", name, [x, b], data_format="NHWC")

    def batchnorm(self, x: str, name: str, c: int, eps: float = 1.001e-5) -> str:
        gamma = self.const(f"{name}/gamma",
                           (1.0 + 0.1 * self.rng.standard_normal(c)).astype(np.float32))
        beta = self.const(f"{name}/beta",
                          (0.1 * self.rng.standard_normal(c)).astype(np.float32))
        mean = self.const(f"{name}/moving_mean",
                          (0.1 * self.rng.standard_normal(c)).astype(np.float32))
        var = self.const(f"{name}/moving_variance",
                         (1.0 + 0.1 * self.rng.random(c)).astype(np.float32))
        return self.node("FusedBatchNormV3", name, [x, gamma, beta, mean, var],
                         epsilon=eps, is_training=False, data_format="NHWC")

    def relu(self, x: str, name: str) -> str:
        return self.node("Relu", name, [x])

    def relu6(self, x: str, name: str) -> str:
        return self.node("Relu6", name, [x])

    def depthwise_conv2d(self, x: str, name: str, k: int, cin: int,
                         stride: int = 1, padding: str = "SAME",
                         mult: int = 1) -> str:
        """TF DepthwiseConv2dNative: filter [kh, kw, C, mult]."""
        w = self.weight(f"{name}/depthwise_weights", (k, k, cin, mult))
        return self.node("DepthwiseConv2dNative", name, [x, w],
                         strides=[1, stride, stride, 1], padding=padding,
                         data_format="NHWC", T=gd.AttrValue(type=gd.DT_FLOAT))

    def maxpool(self, x: str, name: str, k: int, stride: int,
                padding: str = "SAME") -> str:
        return self.node("MaxPool", name, [x], ksize=[1, k, k, 1],
                         strides=[1, stride, stride, 1], padding=padding,
                         data_format="NHWC")

    def avgpool(self, x: str, name: str, k: int, stride: int,
                padding: str = "VALID") -> str:
        return self.node("AvgPool", name, [x], ksize=[1, k, k, 1],
                         strides=[1, stride, stride, 1], padding=padding,
                         data_format="NHWC")

    def global_mean(self, x: str, name: str) -> str:
        ax = self.const(f"{name}/axes", np.array([1, 2], np.int32))
        return self.node("Mean", name, [x, ax], keep_dims=False)

    def dense(self, x: str, name: str, cin: int, cout: int,
              bias: bool = True) -> str:
        w = self.weight(f"{name}/weights", (cin, cout),
                        scale=float(np.sqrt(1.0 / cin)))
        y = self.node("MatMul", name, [x, w], transpose_a=False,
                      transpose_b=False)
        if bias:
            y = self.bias_add(y, f"{name}/bias_add", cout)
        return y

    def softmax(self, x: str, name: str = "softmax") -> str:
        return self.node("Softmax", name, [x])

    def reshape(self, x: str, name: str, shape: List[int]) -> str:
        s = self.const(f"{name}/shape", np.array(shape, np.int32))
        return self.node("Reshape", name, [x, s])

    def graphdef(self) -> gd.GraphDef:
        return self.g

    def save(self, path: str) -> None:
        gd.save_frozen_graph(self.g, path)
