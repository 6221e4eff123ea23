"""Single-shape GEMM loop for clean PMC counter collection.
   python bench/gemm_pmc.py [--n 4096] [--iters 20] [--shape conv]"""
import argparse
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

p = argparse.ArgumentParser()
p.add_argument("--n", type=int, default=4096)
p.add_argument("--iters", type=int, default=20)
p.add_argument("--shape", default="square")  # square | s43x3
args = p.parse_args()
from tensorflow_web_deploy_amd.ops.native import ext
E = ext()
if args.shape == "square":
    n = args.n
    A = torch.randn(n, n, device="cuda").bfloat16()
    B = torch.randn(n, n, device="cuda").bfloat16()
    C = torch.zeros(n, n, dtype=torch.bfloat16, device="cuda")
    for _ in range(args.iters):
        E.gemm_bf16(A, B, C, None, None, n, False)
elif args.shape == "8p":
    n = args.n
    A = torch.randn(n, n, device="cuda").bfloat16()
    B = torch.randn(n, n, device="cuda").bfloat16()
    C = torch.zeros(n, n, dtype=torch.bfloat16, device="cuda")
    for _ in range(args.iters):
        E.gemm8p_bf16(A, B, C, None, None, n, 0)
elif args.shape == "ln":
    X = torch.randn(12544, 768, device="cuda").bfloat16()
    g = torch.rand(768, device="cuda")
    b = torch.randn(768, device="cuda")
    Y = torch.zeros_like(X)
    for _ in range(args.iters):
        E.layernorm_bf16(X, g, b, Y, 1e-6)
elif args.shape == "attn":
    B, H, S, D = 64, 12, 196, 64
    q = torch.randn(B, S, H, D, device="cuda").bfloat16()
    k = torch.randn(B, S, H, D, device="cuda").bfloat16()
    v = torch.randn(B, S, H, D, device="cuda").bfloat16()
    o = torch.zeros_like(q)
    for _ in range(args.iters):
        E.attn_fused_bf16(q, k, v, o, B, H, S,
                          q.stride(0), q.stride(2), q.stride(1),
                          k.stride(0), k.stride(2), k.stride(1),
                          v.stride(0), v.stride(2), v.stride(1),
                          o.stride(0), o.stride(2), o.stride(1), 0.125)
elif args.shape == "stem":
    # Inception conv1: 149^2 x32 -> 3x3 VALID -> 147^2 x32 (120 us, 213 TF)
    X = torch.randn(64, 149, 149, 32, device="cuda").bfloat16()
    W2 = torch.randn(32, 320, device="cuda").bfloat16()  # K=288 pad 320
    Y = torch.zeros(64, 147, 147, 32, dtype=torch.bfloat16, device="cuda")
    for _ in range(args.iters):
        E.conv2d_bf16(X, W2, Y, None, None, 3, 3, 1, 1, 0, 0, False)
elif args.shape == "inc1x7":
    # Inception-v3 17x17 factorized conv class (~300 TF round 1)
    X = torch.randn(64, 17, 17, 160, device="cuda").bfloat16()
    W2 = torch.randn(160, 1152, device="cuda").bfloat16()  # K=1120 pad 1152
    Y = torch.zeros(64, 17, 17, 160, dtype=torch.bfloat16, device="cuda")
    for _ in range(args.iters):
        E.conv2d_bf16(X, W2, Y, None, None, 1, 7, 1, 1, 0, 3, False)
elif args.shape == "inc7x1":
    X = torch.randn(64, 17, 17, 160, device="cuda").bfloat16()
    W2 = torch.randn(160, 1152, device="cuda").bfloat16()
    Y = torch.zeros(64, 17, 17, 160, dtype=torch.bfloat16, device="cuda")
    for _ in range(args.iters):
        E.conv2d_bf16(X, W2, Y, None, None, 7, 1, 1, 1, 3, 0, False)
elif args.shape == "s13x3":
    X = torch.randn(64, 56, 56, 64, device="cuda").bfloat16()
    W2 = torch.randn(64, 576, device="cuda").bfloat16()
    Y = torch.zeros(64, 56, 56, 64, dtype=torch.bfloat16, device="cuda")
    for _ in range(args.iters):
        E.conv2d_bf16(X, W2, Y, None, None, 3, 3, 1, 1, 1, 1, False)
else:
    X = torch.randn(64, 7, 7, 512, device="cuda").bfloat16()
    W2 = torch.randn(512, 4608, device="cuda").bfloat16()
    Y = torch.zeros(64, 7, 7, 512, dtype=torch.bfloat16, device="cuda")
    for _ in range(args.iters):
        E.conv2d_bf16(X, W2, Y, None, None, 3, 3, 1, 1, 1, 1, False)
torch.cuda.synchronize()
print("done")
