"""ViT-B/16 frozen-graph builder (224x224 NHWC, random-init).

BASELINE config 5: "ViT-B/16 224x224 (attention path: MFMA QKV GEMM +
softmax)". Standard ViT-Base/16: 16x16 patch embed (conv s16) -> 196
tokens x 768 -> +positional embedding -> 12 transformer layers
(pre-LN, 12 heads x 64, MLP 3072 with erf-GELU) -> final LN -> global
token mean -> fc 1000 -> softmax.

Uses the GAP (global-average-pool) classifier head variant (no CLS
token) so the graph needs no dynamic-batch Tile; sequence length is 196.

The graph is emitted in primitive TF ops (Mean/SquaredDifference/Rsqrt/
Erf decompositions, BatchMatMulV2 attention) — the IR passes
`fuse_layernorm` / `fuse_gelu` recover the fused kernels exactly as they
would for a real exported TF ViT graph.
"""
from __future__ import annotations

import numpy as np

from .builder import GraphBuilder

INPUT_NAME = "input"
OUTPUT_NAME = "softmax"
NUM_CLASSES = 1000
DIM = 768
HEADS = 12
HEAD_DIM = 64
LAYERS = 12
MLP = 3072
TOKENS = 196  # 14x14 patches


def _layernorm(b: GraphBuilder, x: str, name: str, c: int,
               eps: float = 1e-6) -> str:
    gamma = b.const(f"{name}/gamma",
                    (1.0 + 0.02 * b.rng.standard_normal(c)).astype(np.float32))
    beta = b.const(f"{name}/beta",
                   (0.02 * b.rng.standard_normal(c)).astype(np.float32))
    eps_c = b.const(f"{name}/eps", np.float32(eps))
    one_ax = b.const(f"{name}/axis", np.array([-1], np.int32))
    m = b.node("Mean", f"{name}/mean", [x, one_ax], keep_dims=True)
    sq = b.node("SquaredDifference", f"{name}/sqdiff", [x, m])
    ax2 = b.const(f"{name}/axis2", np.array([-1], np.int32))
    v = b.node("Mean", f"{name}/var", [sq, ax2], keep_dims=True)
    veps = b.node("AddV2", f"{name}/addeps", [v, eps_c])
    rstd = b.node("Rsqrt", f"{name}/rstd", [veps])
    xm = b.node("Sub", f"{name}/sub", [x, m])
    norm = b.node("Mul", f"{name}/norm", [xm, rstd])
    scl = b.node("Mul", f"{name}/scale", [norm, gamma])
    return b.node("AddV2", f"{name}/out", [scl, beta])


def _gelu(b: GraphBuilder, x: str, name: str) -> str:
    inv = b.const(f"{name}/invsqrt2", np.float32(0.7071067811865476))
    half = b.const(f"{name}/half", np.float32(0.5))
    one = b.const(f"{name}/one", np.float32(1.0))
    s = b.node("Mul", f"{name}/s", [x, inv])
    e = b.node("Erf", f"{name}/erf", [s])
    e1 = b.node("AddV2", f"{name}/e1", [e, one])
    h = b.node("Mul", f"{name}/h", [x, e1])
    return b.node("Mul", f"{name}/out", [h, half])


def _transpose(b: GraphBuilder, x: str, name: str, perm) -> str:
    p = b.const(f"{name}/perm", np.array(perm, np.int32))
    return b.node("Transpose", name, [x, p])


def _attention(b: GraphBuilder, x: str, name: str) -> str:
    """x: [-1, TOKENS, DIM] -> [-1, TOKENS, DIM]"""
    # qkv projections as separate Dense ops (TF exports vary; separate
    # ops keep the per-head reshapes static)
    outs = []
    for which in ("q", "k", "v"):
        w = b.weight(f"{name}/{which}/w", (DIM, DIM),
                     scale=float(np.sqrt(1.0 / DIM)))
        y = b.node("BatchMatMulV2", f"{name}/{which}/mm", [x, w])
        bias = b.const(f"{name}/{which}/b",
                       (0.01 * b.rng.standard_normal(DIM)).astype(np.float32))
        y = b.node("AddV2", f"{name}/{which}/bias", [y, bias])
        y = b.reshape(y, f"{name}/{which}/r", [-1, TOKENS, HEADS, HEAD_DIM])
        y = _transpose(b, y, f"{name}/{which}/t", [0, 2, 1, 3])
        outs.append(y)
    q, k, v = outs
    scale = b.const(f"{name}/scale", np.float32(1.0 / np.sqrt(HEAD_DIM)))
    qs = b.node("Mul", f"{name}/qscale", [q, scale])
    scores = b.node("BatchMatMulV2", f"{name}/qk", [qs, k], adj_y=True)
    probs = b.node("Softmax", f"{name}/probs", [scores])
    ctx = b.node("BatchMatMulV2", f"{name}/pv", [probs, v])
    ctx = _transpose(b, ctx, f"{name}/ctx_t", [0, 2, 1, 3])
    ctx = b.reshape(ctx, f"{name}/ctx_r", [-1, TOKENS, DIM])
    wo = b.weight(f"{name}/o/w", (DIM, DIM), scale=float(np.sqrt(1.0 / DIM)))
    y = b.node("BatchMatMulV2", f"{name}/o/mm", [ctx, wo])
    bo = b.const(f"{name}/o/b",
                 (0.01 * b.rng.standard_normal(DIM)).astype(np.float32))
    return b.node("AddV2", f"{name}/o/bias", [y, bo])


def _mlp(b: GraphBuilder, x: str, name: str) -> str:
    w1 = b.weight(f"{name}/fc1/w", (DIM, MLP), scale=float(np.sqrt(1.0 / DIM)))
    y = b.node("BatchMatMulV2", f"{name}/fc1/mm", [x, w1])
    b1 = b.const(f"{name}/fc1/b",
                 (0.01 * b.rng.standard_normal(MLP)).astype(np.float32))
    y = b.node("AddV2", f"{name}/fc1/bias", [y, b1])
    y = _gelu(b, y, f"{name}/gelu")
    w2 = b.weight(f"{name}/fc2/w", (MLP, DIM), scale=float(np.sqrt(1.0 / MLP)))
    y = b.node("BatchMatMulV2", f"{name}/fc2/mm", [y, w2])
    b2 = b.const(f"{name}/fc2/b",
                 (0.01 * b.rng.standard_normal(DIM)).astype(np.float32))
    return b.node("AddV2", f"{name}/fc2/bias", [y, b2])


def build_vit_b16(seed: int = 0, num_classes: int = NUM_CLASSES):
    b = GraphBuilder(seed=seed)
    x = b.placeholder(INPUT_NAME, [-1, 224, 224, 3])
    y = b.conv2d(x, "patch_embed", 16, 16, 3, DIM, 16, "VALID")
    y = b.bias_add(y, "patch_embed/bias", DIM)          # [-1,14,14,768]
    y = b.reshape(y, "tokens", [-1, TOKENS, DIM])
    pos = b.const("pos_embed",
                  (0.02 * b.rng.standard_normal((1, TOKENS, DIM))
                   ).astype(np.float32))
    y = b.node("AddV2", "add_pos", [y, pos])
    for i in range(LAYERS):
        ln1 = _layernorm(b, y, f"layer{i}/ln1", DIM)
        att = _attention(b, ln1, f"layer{i}/attn")
        y = b.node("AddV2", f"layer{i}/res1", [att, y])
        ln2 = _layernorm(b, y, f"layer{i}/ln2", DIM)
        mlp = _mlp(b, ln2, f"layer{i}/mlp")
        y = b.node("AddV2", f"layer{i}/res2", [mlp, y])
    y = _layernorm(b, y, "final_ln", DIM)
    ax = b.const("gap/axis", np.array([1], np.int32))
    y = b.node("Mean", "gap", [y, ax], keep_dims=False)  # [-1, 768]
    y = b.dense(y, "head", DIM, num_classes)
    b.node("Softmax", OUTPUT_NAME, [y])
    return b.graphdef()


def save_vit_b16(path: str, seed: int = 0) -> None:
    from ..ckpt import graphdef as gd
    gd.save_frozen_graph(build_vit_b16(seed), path)
