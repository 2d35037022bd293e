"""Isolate the NaN: constant-batch resnet18 training with per-step loss and
grad norms, toggling conv implementations."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402


def run(tag, steps=6, lr=0.01, arch="resnet18", bs=4, hw=64):
    from amdtrain.models import build_model
    from amdtrain.ops import CrossEntropyLoss, FusedSGD
    torch.manual_seed(1)
    m = build_model(arch, num_classes=10).to("cuda") \
        .to(memory_format=torch.channels_last)
    opt = FusedSGD(m.parameters(), lr=lr, momentum=0.9)
    crit = CrossEntropyLoss()
    x = torch.randn(bs, 3, hw, hw, device="cuda") \
        .contiguous(memory_format=torch.channels_last)
    t = torch.randint(0, 10, (bs,), device="cuda")
    print(f"--- {tag} (lr={lr})")
    for s in range(steps):
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = m(x)
        loss = crit(out, t)
        loss.backward()
        gn = sum(float(p.grad.float().norm() ** 2) for p in m.parameters()
                 if p.grad is not None) ** 0.5
        wn = sum(float(p.float().norm() ** 2) for p in m.parameters()) ** 0.5
        on = float(out.float().abs().max())
        print(f"step {s}: loss={loss.item():.4f} gradnorm={gn:.3e} "
              f"wnorm={wn:.3e} |out|max={on:.3e}", flush=True)
        opt.step()


if __name__ == "__main__":
    run("all-custom")
    os.environ["AMDTRAIN_CONV1X1"] = "miopen"
    os.environ["AMDTRAIN_CONV3X3"] = "miopen"
    os.environ["AMDTRAIN_CONVSTEM"] = "miopen"
    run("miopen-convs")
    os.environ.pop("AMDTRAIN_CONV3X3")
    run("custom3x3-only")
    os.environ["AMDTRAIN_CONV3X3"] = "miopen"
    os.environ.pop("AMDTRAIN_CONVSTEM")
    run("customstem-only")
    os.environ["AMDTRAIN_CONVSTEM"] = "miopen"
    os.environ.pop("AMDTRAIN_CONV1X1")
    run("custom1x1-only")
    # lower lr sanity
    os.environ.pop("AMDTRAIN_CONV3X3", None)
    os.environ.pop("AMDTRAIN_CONVSTEM", None)
    run("all-custom lr=1e-3", lr=1e-3)
