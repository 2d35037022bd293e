"""A/B: conv3x3 128x128 kernels vs the 256x256 (8p3) structure on
layer3/layer4 shapes, refchecked vs F.conv2d."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402
from amdtrain import _C  # noqa: E402


def time_fn(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    import time
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def run(n, cin, cout, hw, stride=1):
    torch.manual_seed(0)
    x = torch.randn(n, cin, hw, hw, device="cuda") \
        .contiguous(memory_format=torch.channels_last).bfloat16()
    w = (torch.randn(cout, cin, 3, 3, device="cuda") * (9 * cin) ** -0.5) \
        .bfloat16()
    x2d = x.permute(0, 2, 3, 1).reshape(-1, cin)
    w2d = w.contiguous(memory_format=torch.channels_last) \
        .permute(0, 2, 3, 1).reshape(cout, 9 * cin)
    ho = (hw + 2 - 3) // stride + 1

    # fwd refcheck
    ref = F.conv2d(x[:1].float(), w.float(), stride=stride, padding=1)
    for name, fn in [("old", lambda: _C.conv3x3_fwd(x2d, n, hw, hw, stride, w2d)),
                     ("8p", lambda: _C.conv3x3_8p(x2d, n, hw, hw, stride, w2d, False))]:
        y = fn().view(n, ho, ho, cout).permute(0, 3, 1, 2)[:1].float()
        err = (y - ref).abs().max().item()
        ok = err < 0.5
        print(f"  fwd {name}: err {err:.4f} {'OK' if ok else 'FAIL'}")
        if not ok:
            return
    flops = 2.0 * n * ho * ho * cout * 9 * cin
    t_old = time_fn(lambda: _C.conv3x3_fwd(x2d, n, hw, hw, stride, w2d))
    t_8p = time_fn(lambda: _C.conv3x3_8p(x2d, n, hw, hw, stride, w2d, False))
    print(f"fwd n={n} c={cin}->{cout} hw={hw} s={stride}: "
          f"old {t_old*1e6:7.1f} us ({flops/t_old/1e12:6.1f} TF) | "
          f"8p {t_8p*1e6:7.1f} us ({flops/t_8p/1e12:6.1f} TF)", flush=True)

    if stride == 1 and cout % 64 == 0:
        gy = torch.randn(n, cout, ho, ho, device="cuda") \
            .contiguous(memory_format=torch.channels_last).bfloat16()
        gy2d = gy.permute(0, 2, 3, 1).reshape(-1, cout)
        # reference dgrad via old kernel (already parity-tested vs torch)
        dref = _C.conv3x3_dgrad(gy2d, n, hw, hw, stride, w2d)
        # 8p dgrad needs the permuted weight [Cin, 9*Cout] (no rotation)
        wrot = w2d.view(cout, 9, cin).permute(2, 1, 0).reshape(cin, 9 * cout) \
            .contiguous()
        d8 = _C.conv3x3_8p(gy2d, n, hw, hw, stride, wrot, True)
        err = (d8.float() - dref.float()).abs().max().item()
        print(f"  dgrad 8p vs old: err {err:.4f} {'OK' if err < 0.5 else 'FAIL'}")
        if err < 0.5:
            t_old = time_fn(lambda: _C.conv3x3_dgrad(gy2d, n, hw, hw, stride, w2d))
            t_8p = time_fn(lambda: _C.conv3x3_8p(gy2d, n, hw, hw, stride, wrot, True))
            print(f"dgrad: old {t_old*1e6:7.1f} us | 8p {t_8p*1e6:7.1f} us",
                  flush=True)


if __name__ == "__main__":
    run(512, 256, 256, 14)   # layer3 3x3 (K=2304)
    run(512, 512, 512, 7)    # layer4 3x3 (K=4608)
    run(512, 128, 128, 28)   # layer2 3x3 (K=1152)
    run(128, 256, 256, 14)
