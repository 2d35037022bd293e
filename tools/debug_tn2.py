"""Debug harness for the v2 TN wgrad core: verifies the tr16 probe mapping
first, then a minimal single-chunk tn2_wgrad, printing the error structure
(which 16x16 fragment region is wrong) to localize layout bugs."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
from amdtrain import _C  # noqa: E402


def probe():
    inp = torch.arange(256, dtype=torch.int16, device="cuda")
    out = _C.tr16_probe(inp).cpu()
    bad = 0
    for lane in range(64):
        for j in range(4):
            expect = (lane & 15) + j * 16 + (lane >> 4) * 64
            got = out[lane, j].item()
            if got != expect:
                bad += 1
                if bad <= 12:
                    print(f"  probe mismatch lane={lane} j={j} "
                          f"expect={expect} got={got}")
    print(f"tr16_probe: {bad} mismatches of 256")
    if bad:
        print("full map (lane-major rows of 4):")
        print(out.view(64, 4).numpy())
    return bad == 0


def tiny(M, N, K, tag):
    torch.manual_seed(0)
    dY = torch.randn(M, N, device="cuda").bfloat16()
    X = torch.randn(M, K, device="cuda").bfloat16()
    dw = _C.tn2_wgrad(dY, X)
    ref = dY.float().t() @ X.float()
    err = (dw - ref).abs()
    print(f"{tag}: M={M} N={N} K={K} maxerr={err.max().item():.4f} "
          f"meanerr={err.mean().item():.4f} "
          f"nan={torch.isnan(dw).sum().item()}")
    if err.max().item() > 0.5 * max(1, M ** 0.5 * 0.05):
        # per-16x16-fragment error map
        fr = err.view(N // 16, 16, K // 16, 16).amax(dim=(1, 3))
        print("per-frag max err (rows=n frags, cols=k frags):")
        torch.set_printoptions(precision=2, linewidth=200)
        print(fr.cpu())
        # inspect a single wrong element
        idx = torch.nonzero(err > err.max() * 0.5)[0]
        n_, k_ = idx[0].item(), idx[1].item()
        print(f"sample wrong ({n_},{k_}): got {dw[n_,k_].item():.4f} "
              f"want {ref[n_,k_].item():.4f}")
        # is it a transpose?
        terr = (dw - ref.t()).abs().max().item() if N == K else -1
        print(f"vs-transposed-ref maxerr: {terr:.4f}")
    return err.max().item()


if __name__ == "__main__":
    ok = probe()
    tiny(32, 64, 64, "single-chunk (2,2)")
    tiny(64, 64, 64, "one-full-chunk (2,2)")
    tiny(256, 64, 64, "multi-chunk (2,2)")
    tiny(64, 64, 256, "(1,4) config")
    tiny(64, 256, 64, "(4,1) config")
    tiny(4096, 128, 128, "msplit (2,2)")
