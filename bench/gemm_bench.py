"""Microbenchmark: our MFMA GEMM/conv kernels on ResNet-50 layer shapes
plus square GEMMs, reporting TF/s. Run on a GPU box:
    python bench/gemm_bench.py [--iters 50]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def bench_gemm(E, M, N, K, iters):
    A = torch.randn(M, K, device="cuda").bfloat16()
    Kp = (K + 63) // 64 * 64
    B = torch.zeros(N, Kp, device="cuda").bfloat16()
    B[:, :K] = torch.randn(N, K, device="cuda").bfloat16()
    C = torch.zeros(M, N, dtype=torch.bfloat16, device="cuda")
    for _ in range(3):
        E.gemm_bf16(A, B, C, None, None, K, False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        E.gemm_bf16(A, B, C, None, None, K, False)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return 2.0 * M * N * K / dt / 1e12, dt * 1e6


def bench_conv(E, Nimg, H, W, C, R, O, stride, iters):
    from tensorflow_web_deploy_amd.engine.gpu_executor import _same_pads
    X = torch.randn(Nimg, H, W, C, device="cuda").bfloat16()
    K = R * R * C
    Kp = (K + 63) // 64 * 64
    W2 = torch.zeros(O, Kp, device="cuda").bfloat16()
    W2[:, :K] = torch.randn(O, K, device="cuda").bfloat16()
    HO, WO, pt, pl = _same_pads(H, W, R, R, stride, stride)
    Y = torch.zeros(Nimg, HO, WO, O, dtype=torch.bfloat16, device="cuda")

    def run():
        if R == 1 and stride == 1:
            E.gemm_bf16(X.view(-1, C), W2, Y.view(-1, O), None, None, K,
                        False)
        else:
            E.conv2d_bf16(X, W2, Y, None, None, R, R, stride, stride, pt, pl,
                          False)
    for _ in range(3):
        run()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        run()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    M = Nimg * HO * WO
    return 2.0 * M * O * K / dt / 1e12, dt * 1e6


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=50)
    args = p.parse_args()
    from tensorflow_web_deploy_amd.ops.native import ext
    E = ext()

    print("== ViT-B/16 b64 GEMM shapes ==")
    for (m, n, k) in [(12544, 768, 768), (12544, 3072, 768),
                      (12544, 768, 3072), (3136, 512, 2048)]:
        tf, us = bench_gemm(E, m, n, k, args.iters)
        print(f"  {m}x{n}x{k}: {tf:8.1f} TF/s  ({us:9.1f} us)")

    print("== square GEMMs (A[M,K] @ B[N,K]^T bf16) ==")
    for n in (2048, 4096, 8192):
        tf, us = bench_gemm(E, n, n, n, args.iters)
        print(f"  {n}^3: {tf:8.1f} TF/s  ({us:9.1f} us)")

    print("== ResNet-50 b64 layer shapes ==")
    layers = [
        # (tag, N,H,W,C,R,O,stride)
        ("stem 7x7s2 C8", 64, 224, 224, 8, 7, 64, 2),
        ("s1 1x1 64->64", 64, 56, 56, 64, 1, 64, 1),
        ("s1 3x3 64", 64, 56, 56, 64, 3, 64, 1),
        ("s1 1x1 64->256", 64, 56, 56, 64, 1, 256, 1),
        ("s1 1x1 256->64", 64, 56, 56, 256, 1, 64, 1),
        ("s2 3x3 128", 64, 28, 28, 128, 3, 128, 1),
        ("s2 1x1 128->512", 64, 28, 28, 128, 1, 512, 1),
        ("s3 3x3 256", 64, 14, 14, 256, 3, 256, 1),
        ("s3 1x1 256->1024", 64, 14, 14, 256, 1, 1024, 1),
        ("s4 3x3 512", 64, 7, 7, 512, 3, 512, 1),
        ("s4 1x1 512->2048", 64, 7, 7, 512, 1, 2048, 1),
        ("s4 1x1 2048->512", 64, 7, 7, 2048, 1, 512, 1),
    ]
    total_t = 0.0
    for tag, *sh in layers:
        tf, us = bench_conv(E, *sh, args.iters)
        total_t += us
        print(f"  {tag:20s}: {tf:8.1f} TF/s  ({us:9.1f} us)")
    print(f"  (sum of sampled layers: {total_t:.0f} us)")


if __name__ == "__main__":
    main()
