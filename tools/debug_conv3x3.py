"""Localize conv3x3 forward mismatches: error map by position, per-tap checks."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

from amdtrain import _C

DEV = "cuda:0"


def run(cin, cout, hw, stride, n=1):
    torch.manual_seed(0)
    x = torch.randn(n, cin, hw, hw, device=DEV) \
        .contiguous(memory_format=torch.channels_last).bfloat16()
    w = (torch.randn(cout, cin, 3, 3, device=DEV) * ((9 * cin) ** -0.5)) \
        .bfloat16()
    x2d = x.permute(0, 2, 3, 1).reshape(-1, cin)
    w2d = w.contiguous(memory_format=torch.channels_last) \
        .permute(0, 2, 3, 1).reshape(cout, 9 * cin)
    y2d = _C.conv3x3_fwd(x2d, n, hw, hw, stride, w2d)
    ho = (hw + 2 - 3) // stride + 1
    y = y2d.view(n, ho, ho, cout).permute(0, 3, 1, 2).float()
    yr = F.conv2d(x.float(), w.float(), stride=stride, padding=1)
    err = (y - yr).abs()
    print(f"cin={cin} cout={cout} hw={hw} s={stride}: max={err.max().item():.4f} "
          f"mean={err.mean().item():.5f}")
    # error by output position (max over n, c)
    emap = err.amax(dim=(0, 1))
    print("err rows (max per row):",
          [round(v, 2) for v in emap.amax(dim=1)[:8].tolist()],
          "...", [round(v, 2) for v in emap.amax(dim=1)[-4:].tolist()])
    print("err cols (max per col):",
          [round(v, 2) for v in emap.amax(dim=0)[:8].tolist()],
          "...", [round(v, 2) for v in emap.amax(dim=0)[-4:].tolist()])
    # interior-only error
    if ho > 4:
        print("interior max:", err[:, :, 2:-2, 2:-2].max().item())
    # channel pattern
    ec = err.amax(dim=(0, 2, 3))
    bad = (ec > 0.3).nonzero().flatten()
    print("bad channels:", bad[:16].tolist(), "count", bad.numel())

    # single-tap test: weight nonzero only at one tap
    for tap in range(9):
        wz = torch.zeros_like(w)
        kh, kw = tap // 3, tap % 3
        wz[:, :, kh, kw] = w[:, :, kh, kw]
        w2z = wz.contiguous(memory_format=torch.channels_last) \
            .permute(0, 2, 3, 1).reshape(cout, 9 * cin)
        yz = _C.conv3x3_fwd(x2d, n, hw, hw, stride, w2z) \
            .view(n, ho, ho, cout).permute(0, 3, 1, 2).float()
        yzr = F.conv2d(x.float(), wz.float(), stride=stride, padding=1)
        e = (yz - yzr).abs().max().item()
        print(f"  tap {tap} ({kh},{kw}): max err {e:.4f}")


if __name__ == "__main__":
    run(64, 64, 56, 1)
    run(64, 64, 12, 1)
    run(64, 128, 12, 2)
