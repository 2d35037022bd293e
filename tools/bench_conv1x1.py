"""Per-shape analysis of the 1x1-conv GEMMs (gemm_bt fwd/dgrad): measured
time vs the HBM roofline (6.3 TB/s achievable) and MFMA roofline, to decide
whether further kernel work can pay.  A-traffic counts nbn re-reads (each
column-block of tiles re-reads the full A panel unless L2 catches it)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
from amdtrain import _C  # noqa: E402

B = int(os.environ.get("BENCH_B", "512"))
BW = 6.3e12
PEAK = 2.5e15


def t_fn(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def run(tag, M, N, K):
    torch.manual_seed(0)
    A = torch.randn(M, K, device="cuda").bfloat16()
    Bm = torch.randn(N, K, device="cuda").bfloat16()
    t = t_fn(lambda: _C.gemm_bt(A, Bm, False))
    fl = 2.0 * M * N * K
    nbn = (N + 127) // 128
    nbm = (M + 127) // 128
    # traffic model: A read per column-block (L2 may absorb), B read per
    # row-block (tiny), C written once
    bytes_min = 2.0 * (M * K + N * K + M * N)
    bytes_rr = 2.0 * (M * K * nbn + N * K * nbm + M * N)
    print(f"{tag:14s} M={M:8d} N={N:5d} K={K:5d}  {t*1e3:7.3f} ms "
          f"({fl/t/1e12:6.1f} TF, {bytes_min/t/1e12:5.2f} TB/s min-traffic) "
          f"bw-floor {bytes_min/BW*1e3:6.3f} ms  rr-floor {bytes_rr/BW*1e3:6.3f} ms "
          f"mfma-floor {fl/PEAK*1e3:6.3f} ms")


def main():
    print(f"== 1x1 shapes at b{B} (fwd + dgrad views) ==")
    for tag, M, N, K in [
            ("L1 c1 fwd",   B*56*56, 64, 64),    # also dgrad of same
            ("L1 c1' fwd",  B*56*56, 64, 256),
            ("L1 c3 fwd",   B*56*56, 256, 64),
            ("L1 ds fwd",   B*56*56, 256, 64),
            ("L2 c1 fwd",   B*28*28, 128, 512),
            ("L2 c3 fwd",   B*28*28, 512, 128),
            ("L3 c1 fwd",   B*14*14, 256, 1024),
            ("L3 c3 fwd",   B*14*14, 1024, 256),
            ("L4 c1 fwd",   B*7*7, 512, 2048),
            ("L4 c3 fwd",   B*7*7, 2048, 512)]:
        run(tag, M, N, K)


if __name__ == "__main__":
    main()
