"""Graph executor: topo-ordered dispatch of the IR over a backend.

Reference behavior replaced: `tf.Session.run(output, {input: batch})` —
the TF C++ executor's scheduling/dispatch loop. Here the plan is an
explicit Python list of steps compiled once per (graph, input signature);
the GPU backend (engine/gpu_executor.py) additionally captures the
kernel-launch sequence into a HIP graph so the per-request hot loop
launches with no Python in the per-kernel path.

TF op semantics notes:
  * tensors are NHWC; filters HWIO; strides attr is [1, sh, sw, 1]
  * padding SAME is asymmetric (extra row/col goes bottom/right)
"""
from __future__ import annotations

import math
from typing import Callable, Dict, List, Optional

import numpy as np
import torch
import torch.nn.functional as F

from .ir import Graph, Node

OpFn = Callable[..., torch.Tensor]
_CPU_OPS: Dict[str, OpFn] = {}


def cpu_op(name):
    def deco(fn):
        _CPU_OPS[name] = fn
        return fn
    return deco


def tf_same_pad(x_nchw: torch.Tensor, kh: int, kw: int, sh: int, sw: int,
                value: float = 0.0) -> torch.Tensor:
    """TF 'SAME' padding (asymmetric: extra on bottom/right)."""
    ih, iw = x_nchw.shape[-2], x_nchw.shape[-1]
    oh = math.ceil(ih / sh)
    ow = math.ceil(iw / sw)
    ph = max((oh - 1) * sh + kh - ih, 0)
    pw = max((ow - 1) * sw + kw - iw, 0)
    pt, pb = ph // 2, ph - ph // 2
    pl, pr = pw // 2, pw - pw // 2
    if ph or pw:
        x_nchw = F.pad(x_nchw, (pl, pr, pt, pb), value=value)
    return x_nchw


def _conv2d(x, w, strides, padding, explicit=None, groups=1):
    """x NHWC, w HWIO -> NHWC (fp32 CPU reference)."""
    sh, sw = int(strides[1]), int(strides[2])
    xc = x.permute(0, 3, 1, 2).contiguous()
    wc = w.permute(3, 2, 0, 1).contiguous()  # HWIO -> OIHW
    kh, kw = w.shape[0], w.shape[1]
    if padding == "SAME":
        xc = tf_same_pad(xc, kh, kw, sh, sw)
    elif padding == "EXPLICIT" and explicit is not None:
        # TF explicit_paddings: [n0,n1, h0,h1, w0,w1, c0,c1] for NHWC
        pt, pb, pl, pr = explicit[2], explicit[3], explicit[4], explicit[5]
        xc = F.pad(xc, (pl, pr, pt, pb))
    y = F.conv2d(xc, wc, stride=(sh, sw), groups=groups)
    return y.permute(0, 2, 3, 1).contiguous()


@cpu_op("ConvBlock")
def _convblock(node, ins, consts):
    x, w = ins[0], ins[1]
    i = 2
    bias = ins[i] if node.attrs.get("bias") else None
    i += 1 if bias is not None else 0
    residual = ins[i] if node.attrs.get("residual") else None
    y = _conv2d(x, w, node.attrs["strides"],
                node.attrs.get("padding", "VALID"),
                node.attrs.get("explicit_paddings"))
    if bias is not None:
        y = y + bias
    if residual is not None:
        y = y + residual
    if node.attrs.get("relu"):
        y = (torch.clamp(y, 0.0, 6.0) if node.attrs.get("relu6")
             else F.relu(y))
    return y


@cpu_op("DepthwiseConv2dNative")
def _dwconv_raw(node, ins, consts):
    # unfused form (graphs where block fusion did not engage)
    x, w = ins[0], ins[1]
    kh, kw, c, mult = w.shape
    wt = w.permute(2, 3, 0, 1).reshape(c * mult, 1, kh, kw)
    sh, sw = int(node.attrs["strides"][1]), int(node.attrs["strides"][2])
    xc = x.permute(0, 3, 1, 2).contiguous()
    if node.attrs.get("padding", "VALID") == "SAME":
        xc = tf_same_pad(xc, kh, kw, sh, sw)
    y = F.conv2d(xc, wt, stride=(sh, sw), groups=c)
    return y.permute(0, 2, 3, 1).contiguous()


@cpu_op("DepthwiseConvBlock")
def _dwconvblock(node, ins, consts):
    x, w = ins[0], ins[1]
    i = 2
    bias = ins[i] if node.attrs.get("bias") else None
    i += 1 if bias is not None else 0
    residual = ins[i] if node.attrs.get("residual") else None
    # TF depthwise filter: [H, W, C, mult] -> torch groups=C, OIHW with O=C*mult
    kh, kw, c, mult = w.shape
    wt = w.permute(2, 3, 0, 1).reshape(c * mult, 1, kh, kw)
    sh, sw = int(node.attrs["strides"][1]), int(node.attrs["strides"][2])
    xc = x.permute(0, 3, 1, 2).contiguous()
    if node.attrs.get("padding", "VALID") == "SAME":
        xc = tf_same_pad(xc, kh, kw, sh, sw)
    y = F.conv2d(xc, wt, stride=(sh, sw), groups=c)
    y = y.permute(0, 2, 3, 1).contiguous()
    if bias is not None:
        y = y + bias
    if residual is not None:
        y = y + residual
    if node.attrs.get("relu"):
        y = (torch.clamp(y, 0.0, 6.0) if node.attrs.get("relu6")
             else F.relu(y))
    return y


@cpu_op("Dense")
def _dense(node, ins, consts):
    x, w = ins[0], ins[1]
    if node.attrs.get("transpose_a"):
        x = x.t()
    if node.attrs.get("transpose_b"):
        w = w.t()
    y = x @ w
    if node.attrs.get("bias"):
        y = y + ins[2]
    if node.attrs.get("relu"):
        y = F.relu(y)
    return y


@cpu_op("DenseND")
def _dense_nd(node, ins, consts):
    y = torch.matmul(ins[0], ins[1])
    if node.attrs.get("bias"):
        y = y + ins[2]
    act = node.attrs.get("act")
    if act == "relu":
        y = F.relu(y)
    elif act == "gelu":
        y = F.gelu(y)
    return y


@cpu_op("Conv2D")
def _conv_raw(node, ins, consts):
    return _conv2d(ins[0], ins[1], node.attrs["strides"],
                   node.attrs.get("padding", "VALID"),
                   node.attrs.get("explicit_paddings"))


@cpu_op("MatMul")
def _matmul(node, ins, consts):
    a, b = ins
    if node.attrs.get("transpose_a"):
        a = a.t()
    if node.attrs.get("transpose_b"):
        b = b.t()
    return a @ b


@cpu_op("BatchMatMul")
@cpu_op("BatchMatMulV2")
def _bmm(node, ins, consts):
    a, b = ins
    if node.attrs.get("adj_x"):
        a = a.transpose(-1, -2)
    if node.attrs.get("adj_y"):
        b = b.transpose(-1, -2)
    return torch.matmul(a, b)


@cpu_op("BiasAdd")
def _biasadd(node, ins, consts):
    return ins[0] + ins[1]


@cpu_op("BnScaleBias")
def _bnscale(node, ins, consts):
    return ins[0] * ins[1] + ins[2]


@cpu_op("FusedBatchNorm")
@cpu_op("FusedBatchNormV2")
@cpu_op("FusedBatchNormV3")
def _fused_bn(node, ins, consts):
    x, gamma, beta, mean, var = ins[:5]
    eps = float(node.attrs.get("epsilon", 1e-3) or 1e-3)
    scale = gamma * torch.rsqrt(var + eps)
    return x * scale + (beta - mean * scale)


@cpu_op("Relu")
def _relu(node, ins, consts):
    return F.relu(ins[0])


@cpu_op("Relu6")
def _relu6(node, ins, consts):
    return torch.clamp(ins[0], 0.0, 6.0)


def _pool(node, ins, kind):
    x = ins[0]
    ks = node.attrs["ksize"]
    st = node.attrs["strides"]
    kh, kw = int(ks[1]), int(ks[2])
    sh, sw = int(st[1]), int(st[2])
    xc = x.permute(0, 3, 1, 2).contiguous()
    if node.attrs.get("padding", "VALID") == "SAME":
        pad_val = float("-inf") if kind == "max" else 0.0
        xc = tf_same_pad(xc, kh, kw, sh, sw, value=pad_val)
    if kind == "max":
        y = F.max_pool2d(xc, (kh, kw), (sh, sw))
    else:
        # TF AvgPool SAME divides by the count of VALID elements only;
        # emulate via ones-mask normalization.
        y = F.avg_pool2d(xc, (kh, kw), (sh, sw), count_include_pad=True)
        if node.attrs.get("padding", "VALID") == "SAME":
            ones = torch.ones_like(x[..., :1].permute(0, 3, 1, 2))
            cnt = F.avg_pool2d(tf_same_pad(ones, kh, kw, sh, sw), (kh, kw), (sh, sw),
                               count_include_pad=True)
            y = y / cnt.clamp_min(1e-12)
    return y.permute(0, 2, 3, 1).contiguous()


@cpu_op("MaxPool")
def _maxpool(node, ins, consts):
    return _pool(node, ins, "max")


@cpu_op("AvgPool")
def _avgpool(node, ins, consts):
    return _pool(node, ins, "avg")


@cpu_op("Mean")
def _mean(node, ins, consts):
    axes = ins[1]
    dims = [int(a) for a in axes.flatten().tolist()] if torch.is_tensor(axes) else [int(axes)]
    keep = bool(node.attrs.get("keep_dims", node.attrs.get("keepdims", False)))
    return ins[0].mean(dim=dims, keepdim=keep)


@cpu_op("Softmax")
def _softmax(node, ins, consts):
    return F.softmax(ins[0], dim=-1)


@cpu_op("Reshape")
def _reshape(node, ins, consts):
    shape = [int(s) for s in ins[1].flatten().tolist()]
    return ins[0].reshape(shape)


@cpu_op("Squeeze")
def _squeeze(node, ins, consts):
    dims = node.attrs.get("squeeze_dims") or node.attrs.get("axis")
    if dims:
        y = ins[0]
        for d in sorted((int(d) for d in dims), reverse=True):
            y = y.squeeze(d)
        return y
    return ins[0].squeeze()


@cpu_op("ExpandDims")
def _expand(node, ins, consts):
    return ins[0].unsqueeze(int(ins[1].item()))


@cpu_op("ConcatV2")
def _concat(node, ins, consts):
    axis = int(ins[-1].item())
    return torch.cat(ins[:-1], dim=axis)


@cpu_op("Pad")
def _pad(node, ins, consts):
    pads = ins[1].to(torch.int64)
    # TF paddings [[d0lo,d0hi],...] -> torch pad (last dim first)
    flat = []
    for d in range(pads.shape[0] - 1, -1, -1):
        flat.extend([int(pads[d, 0]), int(pads[d, 1])])
    return F.pad(ins[0], flat)


@cpu_op("Add")
@cpu_op("AddV2")
def _add(node, ins, consts):
    return ins[0] + ins[1]


@cpu_op("Sub")
def _sub(node, ins, consts):
    return ins[0] - ins[1]


@cpu_op("Mul")
def _mul(node, ins, consts):
    return ins[0] * ins[1]


@cpu_op("RealDiv")
def _div(node, ins, consts):
    return ins[0] / ins[1]


@cpu_op("Rsqrt")
def _rsqrt(node, ins, consts):
    return torch.rsqrt(ins[0])


@cpu_op("Sqrt")
def _sqrt(node, ins, consts):
    return torch.sqrt(ins[0])


@cpu_op("SquaredDifference")
def _sqdiff(node, ins, consts):
    d = ins[0] - ins[1]
    return d * d


@cpu_op("Erf")
def _erf(node, ins, consts):
    return torch.erf(ins[0])


@cpu_op("Tanh")
def _tanh(node, ins, consts):
    return torch.tanh(ins[0])


@cpu_op("Sigmoid")
def _sigmoid(node, ins, consts):
    return torch.sigmoid(ins[0])


@cpu_op("Exp")
def _exp(node, ins, consts):
    return torch.exp(ins[0])


@cpu_op("Gelu")
def _gelu(node, ins, consts):
    approx = "tanh" if node.attrs.get("approximate") else "none"
    return F.gelu(ins[0], approximate=approx)


@cpu_op("LayerNorm")
def _layernorm(node, ins, consts):
    x, gamma, beta = ins
    eps = float(node.attrs.get("epsilon", 1e-6) or 1e-6)
    return F.layer_norm(x, (x.shape[-1],), gamma, beta, eps)


@cpu_op("Transpose")
def _transpose(node, ins, consts):
    perm = [int(p) for p in ins[1].flatten().tolist()]
    return ins[0].permute(perm).contiguous()


@cpu_op("StridedSlice")
def _stridedslice(node, ins, consts):
    x, begin, end, strides = ins
    begin = [int(v) for v in begin.flatten().tolist()]
    end = [int(v) for v in end.flatten().tolist()]
    strides = [int(v) for v in strides.flatten().tolist()]
    bm = int(node.attrs.get("begin_mask", 0) or 0)
    em = int(node.attrs.get("end_mask", 0) or 0)
    sm = int(node.attrs.get("shrink_axis_mask", 0) or 0)
    idx = []
    for d in range(len(begin)):
        b = None if bm & (1 << d) else begin[d]
        e = None if em & (1 << d) else end[d]
        if sm & (1 << d):
            idx.append(begin[d])
        else:
            idx.append(slice(b, e, strides[d]))
    return x[tuple(idx)]


@cpu_op("ArgMax")
def _argmax(node, ins, consts):
    return torch.argmax(ins[0], dim=int(ins[1].item()))


@cpu_op("Shape")
def _shape(node, ins, consts):
    return torch.tensor(list(ins[0].shape), dtype=torch.int32)


@cpu_op("Maximum")
def _maximum(node, ins, consts):
    return torch.maximum(ins[0], ins[1])


@cpu_op("Minimum")
def _minimum(node, ins, consts):
    return torch.minimum(ins[0], ins[1])


@cpu_op("LeakyRelu")
def _leaky_relu(node, ins, consts):
    return F.leaky_relu(ins[0], float(node.attrs.get("alpha", 0.2)))


@cpu_op("Elu")
def _elu(node, ins, consts):
    return F.elu(ins[0])


@cpu_op("Softplus")
def _softplus(node, ins, consts):
    return F.softplus(ins[0])


@cpu_op("Square")
def _square(node, ins, consts):
    return ins[0] * ins[0]


@cpu_op("Neg")
def _neg(node, ins, consts):
    return -ins[0]


@cpu_op("Abs")
def _abs(node, ins, consts):
    return torch.abs(ins[0])


@cpu_op("Log")
def _log(node, ins, consts):
    return torch.log(ins[0])


@cpu_op("Slice")
def _slice(node, ins, consts):
    # inputs: x, begin, size (size -1 = to end), both const or computed
    x = ins[0]
    begin = [int(v) for v in ins[1].tolist()]
    size = [int(v) for v in ins[2].tolist()]
    idx = tuple(slice(b, x.shape[d] if sz == -1 else b + sz)
                for d, (b, sz) in enumerate(zip(begin, size)))
    return x[idx]


@cpu_op("Fill")
def _fill(node, ins, consts):
    shape = [int(v) for v in ins[0].tolist()]
    return torch.full(shape, float(ins[1].reshape(-1)[0].item()))


@cpu_op("Split")
def _split(node, ins, consts):
    # inputs: axis (scalar const), value; num_split attr
    axis = int(ins[0].reshape(-1)[0].item())
    n = int(node.attrs.get("num_split", 1))
    return tuple(torch.chunk(ins[1], n, dim=axis))


@cpu_op("SplitV")
def _splitv(node, ins, consts):
    # inputs: value, size_splits, axis
    sizes = [int(v) for v in ins[1].tolist()]
    axis = int(ins[2].reshape(-1)[0].item())
    return tuple(torch.split(ins[0], sizes, dim=axis))


@cpu_op("Unpack")
def _unpack(node, ins, consts):
    axis = int(node.attrs.get("axis", 0))
    return tuple(t.squeeze(axis) for t in
                 torch.split(ins[0], 1, dim=axis))


@cpu_op("Pack")
def _pack(node, ins, consts):
    axis = int(node.attrs.get("axis", 0))
    return torch.stack(list(ins), dim=axis)


def tf_resize_bilinear(x_nchw, size, align_corners, half_pixel):
    """TF-semantics bilinear resize on an NCHW tensor.

    TF1's three coordinate mappings (ref: tf.image.resize_bilinear):
      align_corners=True          src = dst * (in-1)/(out-1)   == torch ac=True
      half_pixel_centers=True     src = (dst+.5)*scale - .5    == torch ac=False
      default (both False)        src = dst * scale            (asymmetric;
                                  no torch equivalent — computed explicitly)
    """
    oh, ow = int(size[0]), int(size[1])
    if align_corners:
        return F.interpolate(x_nchw, size=(oh, ow), mode="bilinear",
                             align_corners=True)
    if half_pixel:
        return F.interpolate(x_nchw, size=(oh, ow), mode="bilinear",
                             align_corners=False)
    N, C, H, W = x_nchw.shape
    dev = x_nchw.device
    x = x_nchw.float()

    def _axis(n_in, n_out):
        f = torch.arange(n_out, device=dev, dtype=torch.float32) \
            * (n_in / n_out)
        i0 = f.floor().long().clamp_(0, n_in - 1)
        i1 = (i0 + 1).clamp_(max=n_in - 1)
        return i0, i1, (f - i0.float())

    y0, y1, wy = _axis(H, oh)
    x0, x1, wx = _axis(W, ow)
    top = x.index_select(2, y0)
    bot = x.index_select(2, y1)
    xr = top + wy.view(1, 1, -1, 1) * (bot - top)
    left = xr.index_select(3, x0)
    right = xr.index_select(3, x1)
    return left + wx.view(1, 1, 1, -1) * (right - left)


@cpu_op("ResizeBilinear")
def _resize_bilinear(node, ins, consts):
    x = ins[0].permute(0, 3, 1, 2)
    size = [int(v) for v in ins[1].tolist()]
    ac = bool(node.attrs.get("align_corners"))
    hp = bool(node.attrs.get("half_pixel_centers"))
    return tf_resize_bilinear(x, size, ac, hp).permute(0, 2, 3, 1)


@cpu_op("ResizeNearestNeighbor")
def _resize_nearest(node, ins, consts):
    x = ins[0].permute(0, 3, 1, 2)
    size = [int(v) for v in ins[1].tolist()]
    hp = bool(node.attrs.get("half_pixel_centers"))
    y = F.interpolate(x, size=tuple(size),
                      mode="nearest-exact" if hp else "nearest")
    return y.permute(0, 2, 3, 1)


@cpu_op("Cast")
def _cast(node, ins, consts):
    from ..ckpt import graphdef as g
    dst = node.attrs.get("DstT", g.DT_FLOAT)
    m = {g.DT_FLOAT: torch.float32, g.DT_INT32: torch.int32,
         g.DT_INT64: torch.int64, g.DT_BOOL: torch.bool,
         g.DT_HALF: torch.float16, g.DT_BFLOAT16: torch.bfloat16}
    return ins[0].to(m.get(dst, torch.float32))


class Executor:
    """CPU reference executor (fp32, torch ops).

    The GPU executor subclasses this and overrides `run_node` dispatch
    with HIP kernels + HIP-graph capture.
    """

    def __init__(self, graph: Graph, device: str = "cpu"):
        self.graph = graph
        self.device = device
        self.consts: Dict[str, torch.Tensor] = {
            k: torch.from_numpy(np.ascontiguousarray(v)).to(device)
            if v.dtype != object else v
            for k, v in graph.consts.items()
        }
        self.ops = dict(_CPU_OPS)

    def run(self, feeds: Dict[str, torch.Tensor],
            outputs: Optional[List[str]] = None) -> List[torch.Tensor]:
        outputs = outputs or self.graph.outputs
        env: Dict[str, torch.Tensor] = {}
        env.update(self.consts)
        for k, v in feeds.items():
            env[k] = v.to(self.device)
        for node in self.graph.nodes:
            if node.name in env:
                continue
            try:
                ins = [env[i] for i in node.inputs]
            except KeyError as e:
                raise KeyError(f"node {node.name}({node.op}): missing input {e}")
            fn = self.ops.get(node.op)
            if fn is None:
                raise NotImplementedError(f"op {node.op} (node {node.name})")
            out = fn(node, ins, self.consts)
            if isinstance(out, (tuple, list)):
                # multi-output op (Split/Unstack): TF refs are name:k
                env[node.name] = out[0]
                for k, o in enumerate(out):
                    env[f"{node.name}:{k}"] = o
            else:
                env[node.name] = out
        return [env[o] for o in outputs]
