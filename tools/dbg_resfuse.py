import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from amdtrain.models import build_model
from amdtrain.models.resnet import _residual_fuse_enabled, Bottleneck
from amdtrain.ops import conv as CV

print("fuse enabled:", _residual_fuse_enabled())
m = build_model("resnet50").cuda().to(memory_format=torch.channels_last).train()
x = torch.randn(4, 3, 64, 64, device="cuda").contiguous(memory_format=torch.channels_last)

# wrap forward to inspect cells
cells = []
orig_fwd = Bottleneck.forward
def fwd(self, x):
    out = orig_fwd(self, x)
    c = getattr(self.conv1, "_amdtrain_grad_cell", None)
    cells.append(("leftover" if c is not None else "consumed", None))
    return out
Bottleneck.forward = fwd

# instrument tap + conv backward
armed, stashed, fused = [0], [0], [0]
orig_tap_bwd = CV.ResidualGradTap.backward
def tap_bwd(ctx, g):
    if ctx.cell.armed:
        armed[0] += 1
    r = orig_tap_bwd(ctx, g)
    if ctx.cell.g is not None:
        stashed[0] += 1
    return r
CV.ResidualGradTap.backward = staticmethod(tap_bwd)

with torch.autocast("cuda", dtype=torch.bfloat16):
    y = m(x)
loss = y.float().square().mean()
loss.backward()
torch.cuda.synchronize()
print("cell states:", {s: sum(1 for t in cells if t[0]==s) for s in set(c[0] for c in cells)})
print("tap backward armed:", armed[0], "stashed:", stashed[0])
