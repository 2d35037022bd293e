"""A/B the stem (7x7 s2 Cin=8-padded) wgrad tn2 route on hardware:
default (1,4,32) vs AMDTRAIN_TN2_14 variants; used to pin the per-shape
config choice (see wgrad.hip dispatch comments)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from amdtrain import _C
torch.manual_seed(0)
B, H = 512, 224
Ho = (H + 6 - 7) // 2 + 1
M = B * Ho * Ho
gy = torch.randn(M, 64, device="cuda").bfloat16()
x8 = torch.randn(B * H * H, 8, device="cuda").bfloat16()
def t(fn, n=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n
d = t(lambda: _C.tn2_wgrad(gy, x8, 49, B, H, H, 2, 2, 7, 7, 3))
print(f"stem wgrad current route: {d*1e3:.3f} ms")
