import sys, os
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
import torch
from tensorflow_web_deploy_amd.ops.native import ext
E = ext()
def t(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True); e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / iters
X = torch.randn(64, 17, 17, 160, device="cuda").bfloat16()
W2 = torch.randn(160, 1152, device="cuda").bfloat16()
Y = torch.zeros_like(X)
ms = t(lambda: E.conv2d_bf16(X, W2, Y, None, None, 1, 7, 1, 1, 0, 3, 0))
fl = 2*64*17*17*160*1120
print(f"inc1x7: {ms*1000:.1f} us  {fl/ms/1e9:.0f} TF")
ms = t(lambda: E.conv2d_bf16(X, W2, Y, None, None, 7, 1, 1, 1, 3, 0, 0))
print(f"inc7x1: {ms*1000:.1f} us  {fl/ms/1e9:.0f} TF")
X2 = torch.randn(64, 17, 17, 192, device="cuda").bfloat16()
W3 = torch.randn(192, 1344+64-((1344)%64 or 64), device="cuda").bfloat16()
Kp = (7*192+63)//64*64
W3 = torch.randn(192, Kp, device="cuda").bfloat16()
Y2 = torch.zeros_like(X2)
ms = t(lambda: E.conv2d_bf16(X2, W3, Y2, None, None, 1, 7, 1, 1, 0, 3, 0))
fl2 = 2*64*17*17*192*7*192
print(f"inc1x7_c192: {ms*1000:.1f} us  {fl2/ms/1e9:.0f} TF")
Xs = torch.randn(64, 149, 149, 32, device="cuda").bfloat16()
Ws = torch.randn(32, 320, device="cuda").bfloat16()
Ys = torch.zeros(64, 147, 147, 32, dtype=torch.bfloat16, device="cuda")
ms = t(lambda: E.conv2d_bf16(Xs, Ws, Ys, None, None, 3, 3, 1, 1, 0, 0, 0))
fls = 2*64*147*147*32*288
print(f"stem1 3x3C32: {ms*1000:.1f} us  {fls/ms/1e9:.0f} TF")
X4 = torch.randn(64, 35, 35, 64, device="cuda").bfloat16()
W4 = torch.randn(96, 576, device="cuda").bfloat16()
Y4 = torch.zeros(64, 35, 35, 96, dtype=torch.bfloat16, device="cuda")
ms = t(lambda: E.conv2d_bf16(X4, W4, Y4, None, None, 3, 3, 1, 1, 1, 1, 0))
fl4 = 2*64*35*35*96*576
print(f"b35 3x3 64->96: {ms*1000:.1f} us  {fl4/ms/1e9:.0f} TF")
# Inception pools (35x35 avg 3x3s1 SAME C=288; 17x17 C=768)
Xp = torch.randn(64, 35, 35, 288, device="cuda").bfloat16()
Yp = torch.zeros_like(Xp)
ms = t(lambda: E.avgpool_bf16(Xp, Yp, 3, 3, 1, 1, 1, 1))
mb = 2 * 64*35*35*288*2 / 1e6
print(f"avgpool35 C288: {ms*1000:.1f} us  {mb/ms/1000:.2f} TB/s")
Xp2 = torch.randn(64, 17, 17, 768, device="cuda").bfloat16()
Yp2 = torch.zeros_like(Xp2)
ms = t(lambda: E.avgpool_bf16(Xp2, Yp2, 3, 3, 1, 1, 1, 1))
mb = 2 * 64*17*17*768*2 / 1e6
print(f"avgpool17 C768: {ms*1000:.1f} us  {mb/ms/1000:.2f} TB/s")
Xp3 = torch.randn(64, 147, 147, 64, device="cuda").bfloat16()
Yp3 = torch.zeros(64, 73, 73, 64, dtype=torch.bfloat16, device="cuda")
ms = t(lambda: E.maxpool_bf16(Xp3, Yp3, 3, 3, 2, 2, 0, 0))
mb = (64*147*147*64 + 64*73*73*64) * 2 / 1e6
print(f"maxpool147s2: {ms*1000:.1f} us  {mb/ms/1000:.2f} TB/s")
