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
