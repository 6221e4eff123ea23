"""Numerics + perf harness for the experimental 8-phase 256^2 GEMM
(csrc/gemm_8p.hip, unrouted — see NOTES_ROUND2.md item 2)."""
import sys
sys.path.insert(0, "/root/repo")
import torch
from tensorflow_web_deploy_amd.ops.native import ext

def main():
    E = ext()
    torch.manual_seed(0)
    dev = "cuda"
    for (M, N, K) in [(256, 256, 64), (256, 256, 256), (512, 512, 512),
                      (300, 500, 192), (1024, 768, 3072), (4096, 4096, 4096)]:
        Kp = (K + 63) // 64 * 64
        A = (torch.randn(M, K) * 0.5).bfloat16().cuda()
        B = (torch.randn(N, K) * 0.5).bfloat16().cuda()
        Ap = torch.zeros(M, Kp, dtype=torch.bfloat16, device=dev); Ap[:, :K] = A
        Bp = torch.zeros(N, Kp, dtype=torch.bfloat16, device=dev); Bp[:, :K] = B
        bias = torch.randn(N, device=dev)
        res = (torch.randn(M, N) * 0.1).bfloat16().cuda()
        C = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
        for (b, r, act) in [(None, None, 0), (bias, None, 1), (bias, res, 0)]:
            E.gemm8p_bf16(Ap, Bp, C, b, r, K, act)
            ref = A.float() @ B.float().t()
            if b is not None: ref = ref + b
            if r is not None: ref = ref + r.float()
            if act == 1: ref = torch.relu(ref)
            err = (C.float() - ref).abs().max().item()
            scale = ref.abs().max().item() + 1
            ok = err / scale < 2e-2
            print(f"M{M} N{N} K{K} bias={b is not None} res={r is not None} act={act}: relerr {err/scale:.2e} {'OK' if ok else 'FAIL'}")
            if not ok:
                return
    # perf
    import time
    for (M, N, K) in [(4096, 4096, 4096), (8192, 8192, 8192), (12544, 3072, 768)]:
        A = torch.randn(M, K).bfloat16().cuda()
        B = torch.randn(N, K).bfloat16().cuda()
        C = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
        for _ in range(3):
            E.gemm8p_bf16(A, B, C, None, None, K, 0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 30
        for _ in range(iters):
            E.gemm8p_bf16(A, B, C, None, None, K, 0)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        tf = 2 * M * N * K / dt / 1e12
        print(f"8p {M}x{N}x{K}: {tf:.1f} TF/s ({dt*1e6:.1f} us)")

if __name__ == "__main__":
    main()
