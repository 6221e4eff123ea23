"""Layer-by-layer numerics harness (VERDICT round-1 item 6).

Runs the CPU fp32 Executor and the GPU bf16 GpuExecutor over the SAME
optimized IR and input, then compares every node the GPU plan
materializes against its fp32 reference:

    rel(node) = max|gpu - cpu| / (max|cpu| + eps)

and asserts a per-op-class error budget. This catches a mid-network
regression (wrong tile config, bad epilogue, stale buffer) that the old
whole-model bounds (top-5 agreement / corr > 0.95) would hide: a broken
kernel produces rel ~ O(1) at its node, orders of magnitude above any
budget, while healthy bf16 drift stays in the few-percent range even
50 layers deep.

Budgets are per op CLASS (matmul-shaped work accumulates bf16
rounding with depth; pooling/elementwise are near-exact), scaled by
sqrt(depth_index+1) to model error accumulation through the network.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

# base budgets per op class at depth 0 (bf16 vs fp32, random-init
# weights, unit-scale activations); a class not listed uses DEFAULT
BUDGETS: Dict[str, float] = {
    "ConvBlock": 0.02,
    "MatMul": 0.02,
    "DenseBlock": 0.02,
    "DepthwiseBlock": 0.02,
    "AttentionFused": 0.03,
    "LayerNorm": 0.02,
    "Softmax": 0.01,
    "MaxPool": 0.01,
    "AvgPool": 0.015,
    "Mean": 0.015,
    "Add": 0.015,
    "AddV2": 0.015,
    "Relu": 0.015,
    "Relu6": 0.015,
    "BiasAdd": 0.015,
    "ConcatV2": 0.02,
}
DEFAULT_BUDGET = 0.02


class LayerReport:
    def __init__(self):
        # (name, op, depth, local_err, budget, global_err)
        self.rows: List[Tuple[str, str, int, float, float, float]] = []

    def worst(self, k: int = 10):
        return sorted(self.rows, key=lambda r: r[3] / r[4],
                      reverse=True)[:k]

    def violations(self):
        return [r for r in self.rows if r[3] > r[4]]

    def summary(self) -> str:
        out = [f"{len(self.rows)} nodes compared; "
               f"{len(self.violations())} over budget "
               f"(local-err gate; global shown for context)"]
        for name, op, depth, err, budget, gerr in self.worst(8):
            out.append(f"  local {err:8.5f} (budget {budget:.4f}) "
                       f"global {gerr:8.5f} d{depth:<3} {op:<16} {name}")
        return "\n".join(out)


@torch.inference_mode()
def per_layer_compare(graph, cfg, x: np.ndarray, device: str = "cuda",
                      budgets: Optional[Dict[str, float]] = None,
                      depth_scale: bool = True) -> LayerReport:
    """x: fp32 NHWC input batch. Returns a LayerReport; callers assert
    `not report.violations()`."""
    from .executor import Executor
    from .gpu_executor import GpuExecutor

    budgets = budgets or BUDGETS
    xt = torch.from_numpy(np.ascontiguousarray(x, dtype=np.float32))

    gx = GpuExecutor(graph, device=device, cfg=cfg)
    gx.use_graphs = False  # eager steps: buffers hold per-node values
    gx.run({graph.inputs[0]: xt})
    plan = gx._plans[x.shape[0]]

    cx = Executor(graph, device="cpu")
    node_by_name = {n.name: n for n in graph.nodes}
    want = [n for n in plan.bufs
            if n in node_by_name and n not in graph.consts
            and n not in graph.inputs]
    refs = cx.run({graph.inputs[0]: xt}, outputs=want)

    report = LayerReport()
    depth_of: Dict[str, int] = {}
    d = 0
    for n in graph.nodes:
        if n.op in ("ConvBlock", "MatMul", "DenseBlock",
                    "DepthwiseBlock", "AttentionFused"):
            d += 1
        depth_of[n.name] = d

    def _logical(name: str) -> Optional[np.ndarray]:
        """GPU buffer -> fp32 numpy at the node's LOGICAL shape
        (buffers may be channel/row padded)."""
        t = plan.bufs.get(name)
        if t is None:
            return None
        a = t.float().cpu().numpy()
        sh = plan.shapes.get(name)
        if sh is not None and tuple(a.shape) != tuple(sh):
            if a.size == int(np.prod(sh)):
                a = a.reshape(sh)
            else:
                a = a[tuple(slice(0, s) for s in sh)]
        return a

    from .executor import _CPU_OPS

    for name, ref in zip(want, refs):
        got = _logical(name)
        refn = ref.float().cpu().numpy()
        if got is None or got.size == 0:
            continue
        if got.shape != refn.shape:
            sl = tuple(slice(0, min(a, b))
                       for a, b in zip(got.shape, refn.shape))
            got, refn = got[sl], refn[sl]
        scale = float(np.abs(refn).max()) + 1e-6
        gerr = float(np.abs(got - refn).max()) / scale

        # LOCAL error — the gate: rerun THIS op in fp32 on the GPU's
        # own inputs; isolates the kernel from accumulated upstream
        # drift (which e.g. softmax amplifies exponentially)
        node = node_by_name[name]
        fn = _CPU_OPS.get(node.op)
        err = gerr
        have_local = False
        if fn is not None:
            ins, ok = [], True
            for i in node.inputs:
                if i in graph.consts:
                    ins.append(cx.consts[i])
                elif i in plan.bufs:
                    a = _logical(i)
                    ins.append(torch.from_numpy(np.ascontiguousarray(a)))
                elif i in graph.inputs:
                    ins.append(xt)
                else:
                    ok = False
                    break
            if ok:
                try:
                    lref = fn(node, ins, cx.consts)
                    if isinstance(lref, (tuple, list)):
                        lref = lref[0]
                    lr = lref.float().cpu().numpy()
                    g2, l2 = got, lr
                    if g2.shape != l2.shape:
                        sl = tuple(slice(0, min(a, b))
                                   for a, b in zip(g2.shape, l2.shape))
                        g2, l2 = g2[sl], l2[sl]
                    lscale = float(np.abs(l2).max()) + 1e-6
                    err = float(np.abs(g2 - l2).max()) / lscale
                    have_local = True
                except Exception:
                    pass  # fall back to the global metric for this node
        op = node.op
        depth = depth_of.get(name, 0)
        budget = budgets.get(op, DEFAULT_BUDGET)
        if depth_scale and not have_local:
            # global metric accumulates drift with depth
            budget *= float(np.sqrt(depth + 1.0))
        report.rows.append((name, op, depth, err, budget, gerr))
    return report
