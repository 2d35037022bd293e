"""Micro-benchmark: gemm_bt (128x128 m97 structure) vs gemm_bt_8p (256x256)
on square + ResNet conv shapes, refchecked vs torch.matmul."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
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


def run(M, N, K, check=True):
    torch.manual_seed(0)
    A = torch.randn(M, K, device="cuda").bfloat16()
    B = torch.randn(N, K, device="cuda").bfloat16()
    if check:
        ref = (A[:512].float() @ B.float().t())
        for name, fn in [("bt", lambda: _C.gemm_bt(A, B, False)),
                         ("8p", lambda: _C.gemm_bt_8p(A, B)),
                         ("8p3", lambda: _C.gemm_bt_8p3(A, B))]:
            out = fn()[:512].float()
            err = (out - ref).abs().max().item()
            scale = ref.abs().mean().item() + 1e-6
            ok = err < max(0.5, scale * 0.2)
            print(f"  {name} refcheck: max err {err:.4f} ({'OK' if ok else 'FAIL'})")
            if not ok:
                return
    flops = 2.0 * M * N * K
    t_bt = time_fn(lambda: _C.gemm_bt(A, B, False))
    t_8p = time_fn(lambda: _C.gemm_bt_8p(A, B))
    t_83 = time_fn(lambda: _C.gemm_bt_8p3(A, B))
    print(f"M={M} N={N} K={K}: bt {t_bt*1e6:8.1f} us ({flops/t_bt/1e12:7.1f} TF)"
          f" | 8p {t_8p*1e6:8.1f} us ({flops/t_8p/1e12:7.1f} TF)"
          f" | 8p3 {t_83*1e6:8.1f} us ({flops/t_83/1e12:7.1f} TF)", flush=True)


if __name__ == "__main__":
    run(4096, 4096, 4096)
    run(8192, 8192, 8192, check=False)
    # ResNet-50 1x1 shapes at b512
    run(512 * 49, 2048, 512)     # layer4 expand
    run(512 * 49, 512, 2048)     # layer4 reduce
    run(512 * 196, 1024, 256)    # layer3 expand
    run(512 * 784, 128, 512)     # layer2 reduce (N=128)
