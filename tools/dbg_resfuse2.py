import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from amdtrain.models import build_model

def run(fuse, ext):
    # ext=False: plain fp32 forward (convs fall back to MIOpen fp32, no
    # autocast) = ground truth
    os.environ["AMDTRAIN_RESFUSE"] = "1" if fuse else "0"
    torch.manual_seed(0)
    m = build_model("resnet50").cuda().to(memory_format=torch.channels_last).train()
    torch.manual_seed(1)
    x = torch.randn(4, 3, 64, 64, device="cuda").contiguous(memory_format=torch.channels_last)
    if ext:
        with torch.autocast("cuda", dtype=torch.bfloat16):
            y = m(x)
    else:
        y = m(x)  # fp32 torch fallback path = ground truth
    loss = y.float().square().mean()
    loss.backward()
    g = {n: p.grad.float().clone() for n, p in m.named_parameters()}
    return g

gt = run(False, False)   # fp32 torch
g0 = run(False, True)    # custom, no fusion
g1 = run(True, True)     # custom, fused
for n in ("conv1.weight", "layer1.1.conv1.weight", "fc.weight"):
    e0 = (g0[n] - gt[n]).abs().max().item()
    e1 = (g1[n] - gt[n]).abs().max().item()
    d = (g1[n] - g0[n]).abs().max().item()
    print(f"{n:28s} err_nofuse={e0:.4f} err_fused={e1:.4f} fused_vs_nofuse={d:.4f}")
