"""Micro-benchmark: round-1 TN wgrad kernels (gemm_tn / conv3x3_wgrad,
per-tap + atomics) vs the v2 tr-read core (tn2_wgrad) on the ResNet-50
b512 wgrad shapes, refchecked vs fp32 matmul."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
from amdtrain import _C  # noqa: E402

B = int(os.environ.get("BENCH_B", "512"))


def time_fn(fn, iters=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def run_1x1(tag, M, N, K):
    torch.manual_seed(0)
    dY = torch.randn(M, N, device="cuda").bfloat16()
    X = torch.randn(M, K, device="cuda").bfloat16()
    new = _C.tn2_wgrad(dY, X)
    old = _C.gemm_tn(dY, X, 0)
    err = (new - old).abs().max().item()
    ok = err < 0.5
    t_old = time_fn(lambda: _C.gemm_tn(dY, X, 0))
    t_new = time_fn(lambda: _C.tn2_wgrad(dY, X))
    tf = 2.0 * M * N * K
    print(f"1x1 {tag:22s} M={M:8d} N={N:4d} K={K:4d}  "
          f"old {t_old*1e3:7.3f} ms ({tf/t_old/1e12:6.1f} TF)  "
          f"new {t_new*1e3:7.3f} ms ({tf/t_new/1e12:6.1f} TF)  "
          f"x{t_old/t_new:4.2f} {'OK' if ok else 'REFCHECK FAIL err=%.3f' % err}")
    return t_old, t_new


def run_3x3(tag, n, c_in, c_out, hw, s):
    torch.manual_seed(0)
    ho = (hw + 2 - 3) // s + 1
    M = n * ho * ho
    x2d = torch.randn(n * hw * hw, c_in, device="cuda").bfloat16()
    gy2d = torch.randn(M, c_out, device="cuda").bfloat16()
    new = _C.tn2_wgrad(gy2d, x2d, 9, n, hw, hw, s, 2)
    old = _C.conv3x3_wgrad(gy2d, x2d, n, hw, hw, s)
    err = (new - old).abs().max().item()
    ok = err < 0.5
    t_old = time_fn(lambda: _C.conv3x3_wgrad(gy2d, x2d, n, hw, hw, s))
    t_new = time_fn(lambda: _C.tn2_wgrad(gy2d, x2d, 9, n, hw, hw, s, 2))
    tf = 2.0 * M * c_out * 9 * c_in
    print(f"3x3 {tag:22s} M={M:8d} N={c_out:4d} K9={9*c_in:4d}  "
          f"old {t_old*1e3:7.3f} ms ({tf/t_old/1e12:6.1f} TF)  "
          f"new {t_new*1e3:7.3f} ms ({tf/t_new/1e12:6.1f} TF)  "
          f"x{t_old/t_new:4.2f} {'OK' if ok else 'REFCHECK FAIL err=%.3f' % err}")
    return t_old, t_new


def main():
    print(f"== ResNet-50 wgrad shapes at batch {B} ==")
    tot_old = tot_new = 0.0
    # conv3x3 sites (count x shape per fwd pass)
    for cnt, (cin, cout, hw, s) in [
            (3, (64, 64, 56, 1)),
            (1, (128, 128, 56, 2)), (3, (128, 128, 28, 1)),
            (1, (256, 256, 28, 2)), (5, (256, 256, 14, 1)),
            (1, (512, 512, 14, 2)), (2, (512, 512, 7, 1))]:
        o, nw = run_3x3(f"{cin}->{cout} {hw}x{hw}/{s}", B, cin, cout, hw, s)
        tot_old += cnt * o
        tot_new += cnt * nw
    # 1x1 sites
    for cnt, (M, N, K) in [
            (1, (B * 56 * 56, 64, 64)), (2, (B * 56 * 56, 64, 256)),
            (3, (B * 56 * 56, 256, 64)),
            (1, (B * 56 * 56, 128, 256)), (3, (B * 28 * 28, 128, 512)),
            (4, (B * 28 * 28, 512, 128)),
            (1, (B * 28 * 28, 256, 512)), (5, (B * 14 * 14, 256, 1024)),
            (6, (B * 14 * 14, 1024, 256)),
            (1, (B * 14 * 14, 512, 1024)), (2, (B * 7 * 7, 512, 2048)),
            (3, (B * 7 * 7, 2048, 512))]:
        o, nw = run_1x1(f"{N}x{K}", M, N, K)
        tot_old += cnt * o
        tot_new += cnt * nw
    print(f"\nTOTAL wgrad/step (weighted): old {tot_old*1e3:.2f} ms  "
          f"new {tot_new*1e3:.2f} ms  speedup x{tot_old/tot_new:.2f}")


if __name__ == "__main__":
    main()
