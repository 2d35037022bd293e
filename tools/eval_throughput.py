"""Forward-only (inference) throughput for a trained-architecture model.

Measures eval-mode ResNet forward passes (BN running stats, no autograd)
through the same hand-written gfx950 kernel path the training step uses:
on-GPU uint8 normalize -> bf16 autocast NHWC forward.

    python tools/eval_throughput.py [--arch resnet50] [--batch 1536]
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--arch", default="resnet50")
    p.add_argument("--batch", type=int, default=1536)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    args = p.parse_args()

    from amdtrain.models import build_model
    from amdtrain.ops import functional as OF

    m = build_model(args.arch).cuda() \
        .to(memory_format=torch.channels_last).eval()
    B = args.batch
    x = torch.randint(0, 256, (B, 3, 224, 224), dtype=torch.uint8,
                      device="cuda") \
        .contiguous(memory_format=torch.channels_last)

    def fwd():
        with torch.autocast("cuda", dtype=torch.bfloat16):
            return m(OF.normalize_u8(x, dtype=torch.bfloat16))

    with torch.no_grad():
        for _ in range(args.warmup):
            fwd()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            fwd()
        torch.cuda.synchronize()
        el = time.perf_counter() - t0

    print(json.dumps({
        "metric": f"inference images/sec ({args.arch} eval fwd, bf16)",
        "value": round(args.steps * B / el, 1),
        "ms_per_batch": round(el / args.steps * 1000, 3),
        "batch": B,
    }))


if __name__ == "__main__":
    main()
