"""Framework IR: the executable graph lowered from a TF GraphDef.

The reference keeps the TF graph opaque and runs it through
`tf.Session.run`; here the GraphDef is lowered into a small explicit IR
(topo-sorted node list + weight dict) that the executor dispatches over
MI355X HIP kernels (hot ops) or PyTorch-ROCm tensor ops (glue ops).

Lowering passes (inference-oriented):
  - strip Identity / NoOp / control inputs
  - fold FusedBatchNorm(V2/V3) into the producing Conv2D's weight+bias
  - fuse Conv2D + BiasAdd + Relu (+ residual AddV2) into one ConvBlock
  - fuse MatMul + BiasAdd (+ activation) into Dense
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np

from ..ckpt import graphdef as gd


@dataclass
class Node:
    name: str
    op: str
    inputs: List[str]
    attrs: dict = field(default_factory=dict)


@dataclass
class Graph:
    nodes: List[Node]
    consts: Dict[str, np.ndarray]
    inputs: List[str]
    outputs: List[str]

    def node_map(self) -> Dict[str, Node]:
        return {n.name: n for n in self.nodes}


def _canon(name: str) -> str:
    """Canonical tensor name: strip the ':0' output index and '^' ctrl marker."""
    if name.startswith("^"):
        return ""
    if ":" in name:
        base, idx = name.rsplit(":", 1)
        if idx == "0":
            return base
        return name
    return name


def _attr_plain(av: gd.AttrValue):
    for v in (av.s, av.i, av.f, av.b, av.type, av.shape, av.tensor,
              av.list_i, av.list_f, av.list_s, av.list_type, av.list_shape):
        if v is not None:
            return v.decode() if isinstance(v, bytes) else v
    return None


def from_graphdef(g: gd.GraphDef,
                  outputs: Optional[List[str]] = None) -> Graph:
    """Lower a parsed GraphDef to the IR (no fusion yet — see `optimize`)."""
    nodes: List[Node] = []
    consts: Dict[str, np.ndarray] = {}
    inputs: List[str] = []
    for nd in g.node:
        attrs = {k: _attr_plain(v) for k, v in nd.attr.items()}
        if nd.op == "Const":
            consts[nd.name] = nd.attr["value"].tensor
            continue
        if nd.op == "Placeholder":
            inputs.append(nd.name)
            continue
        ins = [c for c in (_canon(i) for i in nd.input) if c]
        nodes.append(Node(nd.name, nd.op, ins, attrs))

    if outputs is None:
        produced = {n.name for n in nodes}
        consumed = set()
        for n in nodes:
            consumed.update(n.inputs)
        outputs = [n for n in produced if n not in consumed]
    return Graph(nodes, consts, inputs, outputs)


# ------------------------------------------------------------- passes


def strip_identity(graph: Graph) -> Graph:
    """Remove Identity/NoOp nodes, rewiring consumers."""
    alias: Dict[str, str] = {}
    keep: List[Node] = []
    for n in graph.nodes:
        if n.op in ("Identity", "NoOp", "StopGradient", "CheckNumerics") and n.inputs:
            alias[n.name] = n.inputs[0]
        else:
            keep.append(n)

    def resolve(name: str) -> str:
        seen = set()
        while name in alias and name not in seen:
            seen.add(name)
            name = alias[name]
        return name

    for n in keep:
        n.inputs = [resolve(i) for i in n.inputs]
    outputs = [resolve(o) for o in graph.outputs]
    return Graph(keep, graph.consts, graph.inputs, outputs)


def fold_batchnorm(graph: Graph) -> Graph:
    """Fold inference FusedBatchNorm(V2/V3) into the upstream Conv2D.

    y = gamma * (x - mean) / sqrt(var + eps) + beta
      = x * (gamma * rsqrt(var+eps)) + (beta - mean * gamma * rsqrt(var+eps))

    When x is a Conv2D output: scale folds into the HWIO filter's O axis,
    shift becomes a bias. When x is not a conv (or the conv has >1
    consumer), the BN is rewritten to a Mul+Add pair via a 'BnScaleBias'
    node instead.
    """
    consumers: Dict[str, int] = {}
    for n in graph.nodes:
        for i in n.inputs:
            consumers[i] = consumers.get(i, 0) + 1
    nmap = graph.node_map()
    new_nodes: List[Node] = []
    replaced: Dict[str, str] = {}

    for n in graph.nodes:
        if n.op in ("FusedBatchNorm", "FusedBatchNormV2", "FusedBatchNormV3"):
            x, gamma_n, beta_n, mean_n, var_n = n.inputs[:5]
            if not all(k in graph.consts for k in (gamma_n, beta_n, mean_n, var_n)):
                new_nodes.append(n)
                continue
            eps = float(n.attrs.get("epsilon", 1e-3) or 1e-3)
            gamma = graph.consts[gamma_n].astype(np.float64)
            beta = graph.consts[beta_n].astype(np.float64)
            mean = graph.consts[mean_n].astype(np.float64)
            var = graph.consts[var_n].astype(np.float64)
            scale = gamma / np.sqrt(var + eps)
            shift = beta - mean * scale
            prod = nmap.get(x)
            if (prod is not None
                    and prod.op in ("Conv2D", "DepthwiseConv2dNative")
                    and consumers.get(x, 0) == 1
                    and prod.inputs[1] in graph.consts):
                w = graph.consts[prod.inputs[1]]
                if prod.op == "Conv2D":  # HWIO: scale over O
                    wf = w.astype(np.float64) * scale[None, None, None, :]
                else:  # depthwise [kh,kw,C,mult]: out channel = (c,mult)
                    wf = (w.astype(np.float64) *
                          scale.reshape(w.shape[2], w.shape[3]))
                graph.consts[prod.inputs[1]] = wf.astype(w.dtype)
                bias_name = n.name + "/folded_bias"
                graph.consts[bias_name] = shift.astype(np.float32)
                bn_bias = Node(n.name, "BiasAdd", [x, bias_name],
                               {"data_format": "NHWC"})
                new_nodes.append(bn_bias)
            else:
                sc_name = n.name + "/bn_scale"
                sh_name = n.name + "/bn_shift"
                graph.consts[sc_name] = scale.astype(np.float32)
                graph.consts[sh_name] = shift.astype(np.float32)
                new_nodes.append(Node(n.name, "BnScaleBias", [x, sc_name, sh_name], {}))
            replaced[n.name] = n.name
        else:
            new_nodes.append(n)
    return Graph(new_nodes, graph.consts, graph.inputs, graph.outputs)


def fuse_conv_blocks(graph: Graph) -> Graph:
    """Fuse Conv2D [+BiasAdd] [+AddV2 residual] [+Relu] -> ConvBlock,
    and MatMul [+BiasAdd] [+Relu|Softmax kept separate] -> Dense.

    ConvBlock attrs: strides, padding, plus fusion fields:
      bias: const name or None; residual: tensor name or None; relu: bool
      (relu6=True refines relu to the 6-clamped form, MobileNet).
    """
    consumers: Dict[str, List[Node]] = {}
    for n in graph.nodes:
        for i in n.inputs:
            consumers.setdefault(i, []).append(n)
    out_set = set(graph.outputs)
    drop = set()
    created = set()  # id() of fused nodes (they reuse the tail's name)
    new_nodes: List[Node] = []
    pos = {n.name: i for i, n in enumerate(graph.nodes)}

    def sole_consumer(name: str) -> Optional[Node]:
        cs = consumers.get(name, [])
        if len(cs) == 1 and name not in out_set:
            return cs[0]
        return None

    for n in graph.nodes:
        if n.name in drop:
            continue
        if n.op in ("Conv2D", "DepthwiseConv2dNative"):
            bias = None
            residual = None
            relu = False
            tail = n
            c = sole_consumer(tail.name)
            if c is not None and c.op == "BiasAdd" and c.inputs[1] in graph.consts:
                bias = c.inputs[1]
                drop.add(c.name)
                tail = c
                c = sole_consumer(tail.name)
            if c is not None and c.op in ("Add", "AddV2"):
                other = [i for i in c.inputs if i != tail.name]
                # Only the MAIN branch absorbs the residual Add: the other
                # operand must already be produced (earlier in node order),
                # else both branches would claim the Add (duplicate nodes).
                if (len(other) == 1 and other[0] not in graph.consts
                        and pos.get(other[0], 1 << 30) < pos[n.name]):
                    residual = other[0]
                    drop.add(c.name)
                    tail = c
                    c = sole_consumer(tail.name)
            relu6 = False
            if c is not None and c.op in ("Relu", "Relu6"):
                relu = True
                relu6 = c.op == "Relu6"
                drop.add(c.name)
                tail = c
            op = "ConvBlock" if n.op == "Conv2D" else "DepthwiseConvBlock"
            attrs = dict(n.attrs)
            attrs.update({"bias": bias, "residual": residual, "relu": relu,
                          "relu6": relu6})
            ins = [n.inputs[0], n.inputs[1]]
            if bias:
                ins.append(bias)
            if residual:
                ins.append(residual)
            nb = Node(tail.name, op, ins, attrs)
            created.add(id(nb))
            new_nodes.append(nb)
        elif n.op == "MatMul":
            bias = None
            relu = False
            tail = n
            c = sole_consumer(tail.name)
            if c is not None and c.op == "BiasAdd" and c.inputs[1] in graph.consts:
                bias = c.inputs[1]
                drop.add(c.name)
                tail = c
                c = sole_consumer(tail.name)
            if c is not None and c.op == "Relu":
                relu = True
                drop.add(c.name)
                tail = c
            attrs = dict(n.attrs)
            attrs.update({"bias": bias, "relu": relu})
            ins = [n.inputs[0], n.inputs[1]]
            if bias:
                ins.append(bias)
            dn = Node(tail.name, "Dense", ins, attrs)
            created.add(id(dn))
            new_nodes.append(dn)
        else:
            new_nodes.append(n)

    new_nodes = [n for n in new_nodes
                 if id(n) in created or n.name not in drop]
    return Graph(new_nodes, graph.consts, graph.inputs, graph.outputs)


def fuse_layernorm(graph: Graph) -> Graph:
    """Recognize the canonical TF LayerNorm decomposition and fuse it:

        m    = Mean(x, [-1], keep_dims=True)
        sq   = SquaredDifference(x, m)
        v    = Mean(sq, [-1], keep_dims=True)
        veps = AddV2(v, eps)          # eps scalar const
        rstd = Rsqrt(veps)
        xm   = Sub(x, m)
        norm = Mul(xm, rstd)
        scl  = Mul(norm, gamma)
        out  = AddV2(scl, beta)

    -> LayerNorm(x, gamma, beta) {epsilon}.
    """
    nmap = graph.node_map()

    def producer(name):
        return nmap.get(name)

    drop = set()
    new_nodes: List[Node] = []
    for n in graph.nodes:
        if n.name in drop:
            pass
        matched = False
        if n.op in ("Add", "AddV2"):
            # out = AddV2(scl, beta)
            scl = producer(n.inputs[0])
            beta = n.inputs[1]
            if (scl is not None and scl.op == "Mul"
                    and beta in graph.consts):
                norm = producer(scl.inputs[0])
                gamma = scl.inputs[1]
                if (norm is not None and norm.op == "Mul"
                        and gamma in graph.consts):
                    xm = producer(norm.inputs[0])
                    rstd = producer(norm.inputs[1])
                    if (xm is not None and xm.op == "Sub"
                            and rstd is not None and rstd.op == "Rsqrt"):
                        x_name, m_name = xm.inputs
                        veps = producer(rstd.inputs[0])
                        if veps is not None and veps.op in ("Add", "AddV2"):
                            v = producer(veps.inputs[0])
                            eps_c = veps.inputs[1]
                            if (v is not None and v.op == "Mean"
                                    and eps_c in graph.consts):
                                sq = producer(v.inputs[0])
                                if (sq is not None
                                        and sq.op == "SquaredDifference"
                                        and sq.inputs[0] == x_name
                                        and sq.inputs[1] == m_name):
                                    m = producer(m_name)
                                    if m is not None and m.op == "Mean" \
                                            and m.inputs[0] == x_name:
                                        eps = float(
                                            graph.consts[eps_c].reshape(-1)[0])
                                        drop.update({scl.name, norm.name,
                                                     xm.name, rstd.name,
                                                     veps.name, v.name,
                                                     sq.name, m.name})
                                        new_nodes.append(Node(
                                            n.name, "LayerNorm",
                                            [x_name, gamma, beta],
                                            {"epsilon": eps}))
                                        matched = True
        if not matched:
            new_nodes.append(n)
    new_nodes = [x for x in new_nodes if x.name not in drop]
    return Graph(new_nodes, graph.consts, graph.inputs, graph.outputs)


def fuse_gelu(graph: Graph) -> Graph:
    """Recognize erf-GELU:  out = Mul(Mul(x, AddV2(Erf(Mul(x, 1/sqrt2)), 1)),
    0.5)  (and the 0.5-first variant) -> Gelu(x)."""
    nmap = graph.node_map()
    drop = set()
    new_nodes: List[Node] = []
    for n in graph.nodes:
        matched = False
        if n.op == "Mul":
            a, b_name = n.inputs
            half = graph.consts.get(b_name)
            inner = nmap.get(a)
            if (half is not None and half.size == 1
                    and abs(float(half.reshape(-1)[0]) - 0.5) < 1e-6
                    and inner is not None and inner.op == "Mul"):
                x_name, e1_name = inner.inputs
                e1 = nmap.get(e1_name)
                if e1 is not None and e1.op in ("Add", "AddV2"):
                    erf = nmap.get(e1.inputs[0])
                    one = graph.consts.get(e1.inputs[1])
                    if (erf is not None and erf.op == "Erf"
                            and one is not None and one.size == 1
                            and abs(float(one.reshape(-1)[0]) - 1.0) < 1e-6):
                        scale = nmap.get(erf.inputs[0])
                        if (scale is not None and scale.op == "Mul"
                                and scale.inputs[0] == x_name):
                            drop.update({inner.name, e1.name, erf.name,
                                         scale.name})
                            new_nodes.append(Node(n.name, "Gelu", [x_name],
                                                  {"approximate": False}))
                            matched = True
        if not matched:
            new_nodes.append(n)
    new_nodes = [x for x in new_nodes if x.name not in drop]
    return Graph(new_nodes, graph.consts, graph.inputs, graph.outputs)


def fuse_dense_bmm(graph: Graph) -> Graph:
    """Fuse BatchMatMulV2(x, W[2D const]) [+ AddV2 bias] [+ Gelu|Relu]
    -> DenseND (rows-flattened GEMM with fused bias/activation epilogue).
    This is the ViT projection/MLP pattern (bias and GELU were separate
    kernel launches: measured 12.5% of the ViT forward)."""
    consumers: Dict[str, List[Node]] = {}
    for n in graph.nodes:
        for i in n.inputs:
            consumers.setdefault(i, []).append(n)
    out_set = set(graph.outputs)

    def sole_consumer(name):
        cs = consumers.get(name, [])
        return cs[0] if len(cs) == 1 and name not in out_set else None

    drop = set()
    created = set()
    new_nodes: List[Node] = []
    for n in graph.nodes:
        if n.name in drop:
            continue
        if (n.op in ("BatchMatMul", "BatchMatMulV2")
                and n.inputs[1] in graph.consts
                and graph.consts[n.inputs[1]].ndim == 2
                and not n.attrs.get("adj_x") and not n.attrs.get("adj_y")):
            bias = None
            act = None
            tail = n
            c = sole_consumer(tail.name)
            if (c is not None and c.op in ("Add", "AddV2")
                    and c.inputs[0] == tail.name
                    and c.inputs[1] in graph.consts
                    and graph.consts[c.inputs[1]].ndim <= 1):
                bias = c.inputs[1]
                drop.add(c.name)
                tail = c
                c = sole_consumer(tail.name)
            if c is not None and c.op in ("Relu", "Gelu"):
                act = c.op.lower()
                drop.add(c.name)
                tail = c
            ins = [n.inputs[0], n.inputs[1]]
            if bias:
                ins.append(bias)
            nd = Node(tail.name, "DenseND", ins,
                      {"bias": bias, "act": act})
            created.add(id(nd))
            new_nodes.append(nd)
        else:
            new_nodes.append(n)
    new_nodes = [x for x in new_nodes
                 if id(x) in created or x.name not in drop]
    return Graph(new_nodes, graph.consts, graph.inputs, graph.outputs)


def fold_scalar_mul(graph: Graph) -> Graph:
    """Fold Mul(x, scalar_const) into an upstream DenseND/Dense weight
    (tracing through Transpose/Reshape views). The ViT attention scale
    (q * 1/sqrt(d)) becomes part of the q-projection weights: the Mul
    node (a full-tensor pass) disappears."""
    nmap = graph.node_map()
    consumers: Dict[str, int] = {}
    for n in graph.nodes:
        for i in n.inputs:
            consumers[i] = consumers.get(i, 0) + 1
    drop = set()
    alias: Dict[str, str] = {}
    for n in graph.nodes:
        if n.op != "Mul" or len(n.inputs) != 2:
            continue
        c = graph.consts.get(n.inputs[1])
        if c is None or c.size != 1:
            continue
        scale = float(c.reshape(-1)[0])
        # walk to the producing dense through view ops (each sole-consumer)
        cur = nmap.get(n.inputs[0])
        chain_ok = True
        while cur is not None and cur.op in ("Reshape", "Transpose"):
            if consumers.get(cur.name, 0) != 1:
                chain_ok = False
                break
            cur = nmap.get(cur.inputs[0])
        if (not chain_ok or cur is None
                or cur.op not in ("DenseND", "Dense")
                or consumers.get(cur.name, 0) != 1
                or cur.inputs[1] not in graph.consts):
            continue
        graph.consts[cur.inputs[1]] = (
            graph.consts[cur.inputs[1]] * scale).astype(
                graph.consts[cur.inputs[1]].dtype)
        bname = cur.attrs.get("bias")
        if bname and bname in graph.consts:
            graph.consts[bname] = (graph.consts[bname] * scale).astype(
                graph.consts[bname].dtype)
        alias[n.name] = n.inputs[0]
        drop.add(n.name)
    if not drop:
        return graph
    new_nodes = [n for n in graph.nodes if n.name not in drop]
    for n in new_nodes:
        n.inputs = [alias.get(i, i) for i in n.inputs]
    outputs = [alias.get(o, o) for o in graph.outputs]
    return Graph(new_nodes, graph.consts, graph.inputs, outputs)


def toposort(graph: Graph) -> Graph:
    nmap = graph.node_map()
    order: List[Node] = []
    state: Dict[str, int] = {}

    def visit(name: str):
        n = nmap.get(name)
        if n is None:
            return
        st = state.get(name, 0)
        if st == 1:
            raise ValueError(f"cycle at {name}")
        if st == 2:
            return
        state[name] = 1
        for i in n.inputs:
            visit(i)
        state[name] = 2
        order.append(n)

    import sys
    old = sys.getrecursionlimit()
    sys.setrecursionlimit(max(old, 10 * len(graph.nodes) + 1000))
    try:
        for o in graph.outputs:
            visit(o)
    finally:
        sys.setrecursionlimit(old)
    return Graph(order, graph.consts, graph.inputs, graph.outputs)


def optimize(graph: Graph) -> Graph:
    g = strip_identity(graph)
    g = fold_batchnorm(g)
    g = fuse_layernorm(g)
    g = fuse_gelu(g)
    g = fuse_dense_bmm(g)
    g = fold_scalar_mul(g)
    g = fuse_conv_blocks(g)
    g = toposort(g)
    return g


def load_model(path: str, outputs: Optional[List[str]] = None) -> Graph:
    """Load a frozen .pb file or a SavedModel directory into optimized IR."""
    import os
    if os.path.isdir(path):
        from ..ckpt.savedmodel import load_savedmodel
        g, sig = load_savedmodel(path)
        if outputs is None and sig is not None and sig.outputs:
            outputs = [_canon(n) for n in sig.outputs.values()]
        return optimize(from_graphdef(g, outputs))
    g = gd.load_frozen_graph(path)
    return optimize(from_graphdef(g, outputs))
