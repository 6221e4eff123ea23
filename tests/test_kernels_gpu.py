"""HIP kernel numerics vs plain PyTorch fp32 references (gpu-marked).

Every kernel is compared against a torch fp32 computation of the same op
on random data; tolerances account for bf16 inputs/outputs (f32 accum).
B operands are asymmetric random (guide §5.4 rule 16: transpose-detecting).
"""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def E():
    from tensorflow_web_deploy_amd.ops.native import ext
    return ext()


def _bf(x):
    return x.to("cuda", torch.bfloat16).contiguous()


def _pad64_cols(w):
    n, k = w.shape
    kp = (k + 63) // 64 * 64
    if kp != k:
        w = torch.cat([w, torch.zeros(n, kp - k, dtype=w.dtype,
                                      device=w.device)], 1)
    return w


@pytest.mark.parametrize("M,N,K", [(128, 128, 64), (256, 512, 256),
                                   (200, 1000, 2048), (64, 64, 72),
                                   (1, 16, 64), (3136, 2048, 512),
                                   # 256^2 pipelined-kernel shapes
                                   (8192, 2048, 128), (4096, 4096, 192),
                                   (8200, 2048, 64),
                                   # split-K shapes (small M*N, large K)
                                   (3136, 512, 2048), (196, 128, 4608)])
def test_gemm_bf16(E, M, N, K):
    torch.manual_seed(0)
    A = torch.randn(M, K) * 0.5
    B = torch.randn(N, K) * 0.5
    Ag, Bg = _bf(A), _pad64_cols(_bf(B))
    C = torch.zeros(M, N, dtype=torch.bfloat16, device="cuda")
    E.gemm_bf16(Ag, Bg, C, None, None, K, False)
    ref = (Ag.float() @ Bg.float()[:, :K].t())
    got = C.float()
    torch.testing.assert_close(got, ref, rtol=2e-2, atol=2e-2 * K ** 0.5)


@pytest.mark.parametrize("M,N,K", [(256, 128, 128), (3136, 512, 2048)])
def test_gemm_bias_relu_residual(E, M, N, K):
    torch.manual_seed(1)
    A = torch.randn(M, K)
    B = torch.randn(N, K)
    bias = torch.randn(N)
    res = torch.randn(M, N)
    Ag, Bg = _bf(A), _bf(B)
    biasg = bias.cuda()
    resg = _bf(res)
    C = torch.zeros(M, N, dtype=torch.bfloat16, device="cuda")
    E.gemm_bf16(Ag, Bg, C, biasg, resg, K, True)
    ref = torch.relu(Ag.float() @ Bg.float().t() + biasg + resg.float())
    torch.testing.assert_close(C.float(), ref, rtol=2e-2,
                               atol=0.1 * K ** 0.5)


def _tf_conv_ref(X, W, stride, pad_same):
    # X NHWC f32, W HWIO f32 -> NHWC f32 (same math as engine/executor.py)
    import torch.nn.functional as F
    xc = X.permute(0, 3, 1, 2)
    wc = W.permute(3, 2, 0, 1)
    kh, kw = W.shape[0], W.shape[1]
    if pad_same:
        from tensorflow_web_deploy_amd.engine.executor import tf_same_pad
        xc = tf_same_pad(xc, kh, kw, stride, stride)
    y = F.conv2d(xc, wc, stride=stride)
    return y.permute(0, 2, 3, 1).contiguous()


@pytest.mark.parametrize("N,H,W,C,R,O,stride,same", [
    (2, 16, 16, 64, 3, 64, 1, True),       # resnet 3x3 s1
    (2, 16, 16, 64, 3, 128, 2, True),      # resnet 3x3 s2
    (2, 14, 14, 256, 1, 64, 1, True),      # 1x1 s1 (gemm path)
    (2, 28, 28, 128, 1, 256, 2, True),     # 1x1 s2 shortcut
    (1, 30, 30, 32, 5, 48, 1, False),      # inception 5x5 VALID
    (2, 17, 17, 8, 7, 64, 2, True),        # 7x7 s2 stem-like (Cpad=8)
    (64, 56, 56, 64, 1, 256, 1, True),     # 256^2-kernel 1x1 (gemm path)
    (64, 56, 56, 64, 3, 256, 1, True),     # 256^2-kernel im2col path
    (2, 7, 7, 512, 3, 512, 1, True),       # split-K conv (s4 3x3)
])
def test_conv2d_bf16(E, N, H, W, C, R, O, stride, same):
    torch.manual_seed(2)
    X = torch.randn(N, H, W, C) * 0.5
    Wt = torch.randn(R, R, C, O) * (2.0 / (R * R * C)) ** 0.5
    ref = _tf_conv_ref(_bf(X).float().cpu(), Wt, stride, same)

    Xg = _bf(X)
    K = R * R * C
    Kp = (K + 63) // 64 * 64
    W2 = torch.zeros(O, Kp)
    W2[:, :K] = Wt.permute(3, 0, 1, 2).reshape(O, K)
    W2g = _bf(W2)
    if same:
        from tensorflow_web_deploy_amd.engine.gpu_executor import _same_pads
        HO, WO, pt, pl = _same_pads(H, W, R, R, stride, stride)
    else:
        HO, WO = (H - R) // stride + 1, (W - R) // stride + 1
        pt = pl = 0
    Y = torch.zeros(N, HO, WO, O, dtype=torch.bfloat16, device="cuda")
    if R == 1 and stride == 1:
        E.gemm_bf16(Xg.view(-1, C), W2g, Y.view(-1, O), None, None, K, False)
    else:
        E.conv2d_bf16(Xg, W2g, Y, None, None, R, R, stride, stride, pt, pl,
                      False)
    torch.testing.assert_close(Y.float().cpu(), ref, rtol=3e-2,
                               atol=2e-2 * K ** 0.5)


def test_maxpool_avgpool(E):
    torch.manual_seed(3)
    X = torch.randn(2, 15, 15, 64)
    Xg = _bf(X)
    from tensorflow_web_deploy_amd.engine.ir import Node
    from tensorflow_web_deploy_amd.engine.executor import _CPU_OPS
    node = Node("p", "MaxPool", ["x"],
                {"ksize": [1, 3, 3, 1], "strides": [1, 2, 2, 1],
                 "padding": "SAME"})
    ref = _CPU_OPS["MaxPool"](node, [Xg.float().cpu()], {})
    from tensorflow_web_deploy_amd.engine.gpu_executor import _same_pads
    HO, WO, pt, pl = _same_pads(15, 15, 3, 3, 2, 2)
    Y = torch.zeros(2, HO, WO, 64, dtype=torch.bfloat16, device="cuda")
    E.maxpool_bf16(Xg, Y, 3, 3, 2, 2, pt, pl)
    torch.testing.assert_close(Y.float().cpu(), ref, rtol=1e-2, atol=1e-2)

    node2 = Node("p", "AvgPool", ["x"],
                 {"ksize": [1, 3, 3, 1], "strides": [1, 2, 2, 1],
                  "padding": "SAME"})
    ref2 = _CPU_OPS["AvgPool"](node2, [Xg.float().cpu()], {})
    Y2 = torch.zeros(2, HO, WO, 64, dtype=torch.bfloat16, device="cuda")
    E.avgpool_bf16(Xg, Y2, 3, 3, 2, 2, pt, pl)
    torch.testing.assert_close(Y2.float().cpu(), ref2, rtol=1e-2, atol=1e-2)


def test_global_mean(E):
    X = torch.randn(4, 7, 7, 2048)
    Xg = _bf(X)
    Y = torch.zeros(4, 2048, dtype=torch.bfloat16, device="cuda")
    E.global_mean_bf16(Xg, Y)
    ref = Xg.float().mean(dim=(1, 2))
    torch.testing.assert_close(Y.float(), ref, rtol=1e-2, atol=1e-2)


def test_softmax(E):
    X = torch.randn(64, 1000) * 3
    Xg = _bf(X)
    Y = torch.zeros(64, 1000, dtype=torch.float32, device="cuda")
    E.softmax_bf16(Xg, Y)
    ref = torch.softmax(Xg.float(), dim=-1)
    torch.testing.assert_close(Y, ref, rtol=1e-3, atol=1e-4)
    torch.testing.assert_close(Y.sum(-1), torch.ones(64, device="cuda"),
                               rtol=1e-4, atol=1e-4)


def test_elementwise(E):
    X = torch.randn(3, 8, 8, 64)
    Xg = _bf(X)
    # cast roundtrip
    Xf = torch.zeros_like(Xg, dtype=torch.float32)
    E.cast_bf16_f32(Xg, Xf)
    torch.testing.assert_close(Xf, Xg.float())
    Xb2 = torch.zeros_like(Xg)
    E.cast_f32_bf16(Xf, Xb2)
    torch.testing.assert_close(Xb2.float(), Xg.float())
    # add + relu
    B = _bf(torch.randn(3, 8, 8, 64))
    Y = torch.zeros_like(Xg)
    E.add_relu_bf16(Xg, B, Y, True)
    torch.testing.assert_close(Y.float(), torch.relu(Xg.float() + B.float()),
                               rtol=1e-2, atol=1e-2)
    # bias + gelu
    bias = torch.randn(64, device="cuda")
    Y2 = torch.zeros_like(Xg)
    E.bias_act_bf16(Xg, bias, Y2, 5)
    import torch.nn.functional as F
    torch.testing.assert_close(Y2.float(), F.gelu(Xg.float() + bias),
                               rtol=2e-2, atol=2e-2)
    # channel pad
    Xs = _bf(torch.randn(4, 5, 5, 3))
    Yp = torch.zeros(4, 5, 5, 8, dtype=torch.bfloat16, device="cuda")
    E.channel_pad_bf16(Xs, Yp)
    torch.testing.assert_close(Yp[..., :3].float(), Xs.float())
    assert Yp[..., 3:].abs().sum().item() == 0
    # scale/shift (unfused BN)
    sc = torch.rand(64, device="cuda") + 0.5
    shf = torch.randn(64, device="cuda")
    Y3 = torch.zeros_like(Xg)
    E.scale_shift_bf16(Xg, sc, shf, Y3, False)
    torch.testing.assert_close(Y3.float(), Xg.float() * sc + shf,
                               rtol=1e-2, atol=1e-2)


def test_layernorm(E):
    import torch.nn.functional as F
    for rows, C in [(196 * 2, 768), (64, 2048), (33, 64)]:
        X = torch.randn(rows, C) * 2 + 0.5
        Xg = _bf(X)
        g = torch.rand(C, device="cuda") + 0.5
        b = torch.randn(C, device="cuda")
        Y = torch.zeros_like(Xg)
        E.layernorm_bf16(Xg, g, b, Y, 1e-6)
        ref = F.layer_norm(Xg.float(), (C,), g, b, 1e-6)
        torch.testing.assert_close(Y.float(), ref, rtol=2e-2, atol=2e-2)


def test_gemm_batched(E):
    torch.manual_seed(7)
    batch, M, N, K = 24, 196, 196, 64
    A = _bf(torch.randn(batch, M, K))
    B = _bf(torch.randn(batch, N, K))
    C = torch.zeros(batch, M, N, dtype=torch.bfloat16, device="cuda")
    E.gemm_batched_bf16(A, B, C)
    ref = torch.matmul(A.float(), B.float().transpose(1, 2))
    torch.testing.assert_close(C.float(), ref, rtol=2e-2, atol=0.2)
    # P@V shape with padded K (196 -> 200)
    batch, M, N, K = 24, 196, 64, 200
    A = _bf(torch.randn(batch, M, K))
    B = _bf(torch.randn(batch, N, K))
    C = torch.zeros(batch, M, N, dtype=torch.bfloat16, device="cuda")
    E.gemm_batched_bf16(A, B, C)
    ref = torch.matmul(A.float(), B.float().transpose(1, 2))
    torch.testing.assert_close(C.float(), ref, rtol=2e-2, atol=0.3)


def test_attn_fused(E):
    """Fused attention vs torch fp32 reference on strided head views."""
    torch.manual_seed(11)
    B, H, S, D = 3, 4, 196, 64
    # natural qkv layout [B, S, H, D] (no transposes, like the executor)
    q = _bf(torch.randn(B, S, H, D))
    k = _bf(torch.randn(B, S, H, D))
    v = _bf(torch.randn(B, S, H, D))
    out = torch.zeros(B, S, H, D, dtype=torch.bfloat16, device="cuda")
    scale = 1.0 / D ** 0.5
    E.attn_fused_bf16(q, k, v, out, B, H, S,
                      q.stride(0), q.stride(2), q.stride(1),
                      k.stride(0), k.stride(2), k.stride(1),
                      v.stride(0), v.stride(2), v.stride(1),
                      out.stride(0), out.stride(2), out.stride(1),
                      scale)
    qf = q.float().permute(0, 2, 1, 3)
    kf = k.float().permute(0, 2, 1, 3)
    vf = v.float().permute(0, 2, 1, 3)
    ref = torch.softmax(qf @ kf.transpose(-1, -2) * scale, -1) @ vf
    got = out.float().permute(0, 2, 1, 3)
    torch.testing.assert_close(got, ref, rtol=2e-2, atol=2e-2)
    # spiked input forces large rescales (guide §5.4 rule 26)
    q2 = q.clone()
    q2[:, 5] *= 30
    k2 = k.clone()
    k2[:, 100] *= 30
    out2 = torch.zeros_like(out)
    E.attn_fused_bf16(q2, k2, v, out2, B, H, S,
                      q2.stride(0), q2.stride(2), q2.stride(1),
                      k2.stride(0), k2.stride(2), k2.stride(1),
                      v.stride(0), v.stride(2), v.stride(1),
                      out2.stride(0), out2.stride(2), out2.stride(1),
                      scale)
    ref2 = torch.softmax(q2.float().permute(0, 2, 1, 3) @
                         k2.float().permute(0, 2, 1, 3).transpose(-1, -2)
                         * scale, -1) @ vf
    torch.testing.assert_close(out2.float().permute(0, 2, 1, 3), ref2,
                               rtol=3e-2, atol=3e-2)


@pytest.mark.parametrize("C,stride,act,res", [
    (32, 1, 2, False),    # MobileNet stem-depthwise shape class
    (144, 2, 2, False),   # stride-2 downsample
    (960, 1, 1, False),   # widest depthwise
    (12, 1, 0, True),     # channel tail (C%8!=0) + residual
])
def test_dwconv_bf16(E, C, stride, act, res):
    """Depthwise conv vs torch fp32 (TF SAME semantics, groups=C)."""
    torch.manual_seed(C + stride)
    N, H, W, R = 2, 17, 23, 3
    x = torch.randn(N, H, W, C)
    w = torch.randn(R, R, C, 1) * 0.5
    bias = torch.randn(C)
    # TF SAME pads
    HO = (H + stride - 1) // stride
    WO = (W + stride - 1) // stride
    ph = max((HO - 1) * stride + R - H, 0)
    pw = max((WO - 1) * stride + R - W, 0)
    pt, pl = ph // 2, pw // 2
    xc = x.permute(0, 3, 1, 2)
    xp = torch.nn.functional.pad(xc, (pl, pw - pl, pt, ph - pt))
    wt = w.permute(2, 3, 0, 1).reshape(C, 1, R, R)
    ref = torch.nn.functional.conv2d(xp, wt, bias=bias, stride=stride,
                                     groups=C).permute(0, 2, 3, 1)
    rr = torch.randn(N, HO, WO, C) if res else None
    if res:
        ref = ref + rr
    if act == 1:
        ref = torch.relu(ref)
    elif act == 2:
        ref = torch.clamp(ref, 0.0, 6.0)

    y = torch.empty(N, HO, WO, C, dtype=torch.bfloat16, device="cuda")
    E.dwconv_bf16(_bf(x), _bf(w.reshape(-1)), bias.float().cuda(),
                  _bf(rr) if res else None, y, R, R, stride, stride,
                  pt, pl, act)
    torch.testing.assert_close(y.float().cpu(), ref, atol=0.1, rtol=0.05)


@pytest.mark.parametrize("N,H,W,C,R,S,O,stride,same", [
    (2, 21, 21, 32, 3, 3, 32, 1, False),   # inception stem conv1 class
    (2, 21, 21, 32, 3, 3, 64, 1, True),    # stem conv2 class (SAME)
    (2, 17, 17, 160, 1, 7, 160, 1, True),  # 17x17 1x7
    (2, 17, 17, 160, 7, 1, 192, 1, True),  # 17x17 7x1
    (2, 35, 35, 64, 3, 3, 96, 1, True),    # 35x35 3x3
    (2, 23, 23, 32, 3, 3, 32, 2, True),    # s2 through the direct path
])
def test_conv_direct_shapes(E, N, H, W, C, R, S, O, stride, same):
    """conv_direct.hip routing (im2col-free): these shapes satisfy
    C%32==0 / Cout%16==0 / R*S>=3 and must match the fp32 reference
    through the same conv2d_bf16 binding."""
    import torch.nn.functional as F
    torch.manual_seed(4)
    X = torch.randn(N, H, W, C) * 0.5
    Wt = torch.randn(R, S, C, O) * (2.0 / (R * S * C)) ** 0.5
    xc = _bf(X).float().cpu().permute(0, 3, 1, 2)
    wc = Wt.permute(3, 2, 0, 1)
    if same:
        from tensorflow_web_deploy_amd.engine.executor import tf_same_pad
        xc = tf_same_pad(xc, R, S, stride, stride)
    ref = F.conv2d(xc, wc, stride=stride).permute(0, 2, 3, 1).contiguous()

    Xg = _bf(X)
    K = R * S * C
    Kp = (K + 63) // 64 * 64
    W2 = torch.zeros(O, Kp)
    W2[:, :K] = Wt.permute(3, 0, 1, 2).reshape(O, K)
    W2g = _bf(W2)
    if same:
        from tensorflow_web_deploy_amd.engine.gpu_executor import _same_pads
        HO, WO, pt, pl = _same_pads(H, W, R, S, stride, stride)
    else:
        HO, WO = (H - R) // stride + 1, (W - S) // stride + 1
        pt = pl = 0
    Y = torch.zeros(N, HO, WO, O, dtype=torch.bfloat16, device="cuda")
    E.conv2d_bf16(Xg, W2g, Y, None, None, R, S, stride, stride, pt, pl,
                  False)
    torch.testing.assert_close(Y.float().cpu(), ref, rtol=3e-2,
                               atol=2e-2 * K ** 0.5)
    # bias + relu epilogue
    bias = torch.randn(O, device="cuda")
    E.conv2d_bf16(Xg, W2g, Y, bias, None, R, S, stride, stride, pt, pl, 1)
    ref2 = torch.relu(ref + bias.cpu())
    torch.testing.assert_close(Y.float().cpu(), ref2, rtol=3e-2,
                               atol=2e-2 * K ** 0.5)


@pytest.mark.parametrize("maxp", [True, False])
@pytest.mark.parametrize("H,C", [(35, 288), (17, 768), (15, 64), (7, 8)])
def test_pool_s1_same(E, maxp, H, C):
    """Stride-1 SAME pools route to the separable vertical-strip kernel
    (pool_s1_kernel); must match the CPU reference incl border counts."""
    torch.manual_seed(5)
    X = torch.randn(2, H, H, C)
    Xg = _bf(X)
    from tensorflow_web_deploy_amd.engine.executor import _CPU_OPS
    from tensorflow_web_deploy_amd.engine.ir import Node
    op = "MaxPool" if maxp else "AvgPool"
    node = Node("p", op, ["x"],
                {"ksize": [1, 3, 3, 1], "strides": [1, 1, 1, 1],
                 "padding": "SAME"})
    ref = _CPU_OPS[op](node, [Xg.float().cpu()], {})
    Y = torch.zeros(2, H, H, C, dtype=torch.bfloat16, device="cuda")
    if maxp:
        E.maxpool_bf16(Xg, Y, 3, 3, 1, 1, 1, 1)
    else:
        E.avgpool_bf16(Xg, Y, 3, 3, 1, 1, 1, 1)
    torch.testing.assert_close(Y.float().cpu(), ref, rtol=2e-2, atol=2e-2)


def test_attn_defer_max_spiked(E):
    """T13 defer-max hazard coverage (guide §5.4 rule 26): a huge key
    arriving LATE must flip the rescale branch and still match the
    fp32 reference within the documented defer-path bound."""
    torch.manual_seed(1)
    B, H, S, D = 4, 2, 196, 64
    q = (torch.randn(B, S, H, D) * 0.5).bfloat16().cuda()
    k = (torch.randn(B, S, H, D) * 0.5).bfloat16().cuda()
    v = (torch.randn(B, S, H, D) * 0.5).bfloat16().cuda()
    # spike one key in the LAST tile and one matching query so the
    # running max jumps past THR mid-stream
    k[0, 180, 0] = 10.0
    q[0, 3, 0] = 10.0
    o = torch.zeros_like(q)
    E.attn_fused_bf16(q, k, v, o, B, H, S,
                      q.stride(0), q.stride(2), q.stride(1),
                      k.stride(0), k.stride(2), k.stride(1),
                      v.stride(0), v.stride(2), v.stride(1),
                      o.stride(0), o.stride(2), o.stride(1), 0.125)
    torch.cuda.synchronize()
    qf = q.float().permute(0, 2, 1, 3)
    kf = k.float().permute(0, 2, 1, 3)
    vf = v.float().permute(0, 2, 1, 3)
    ref = (torch.softmax(qf @ kf.transpose(-1, -2) * 0.125, -1)
           @ vf).permute(0, 2, 1, 3)
    err = (o.float() - ref).abs().max().item()
    assert err < 2e-2, err
