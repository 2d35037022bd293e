"""Flagship benchmark: ResNet-50 ImageNet training step throughput on MI355X.

Measures BASELINE.json's headline metric — ImageNet images/sec (whole job)
for ResNet-50 on synthetic 3x224x224 data with random-init weights — through
the full launcher-style DDP training step: on-GPU uint8 normalize, bf16
autocast forward (NHWC), fused CE loss, backward with bucketed RCCL
all-reduce overlap, fused multi-tensor SGD step.

    python bench.py --gpus N --steps K --warmup W
    # N>1 is launched by the driver as:
    # python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
    #   --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Timing contract: W untimed warmup steps; barrier + torch.cuda.synchronize();
exactly K timed steps; barrier + synchronize; elapsed = MAX over ranks; one
JSON line from rank 0.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch-per-gpu", type=int, default=3072)  # swept: +2% over 2048, 136 GB/GPU
    p.add_argument("--arch", type=str, default="resnet50")
    p.add_argument("--dtype", type=str, default="bf16",
                   choices=["bf16", "fp32"])
    p.add_argument("--bucket-mb", type=float, default=50.0)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--no-ext", action="store_true",
                   help="A/B: plain-PyTorch ops instead of the HIP kernels")
    p.add_argument("--profile", type=str, default="",
                   help="write a torch.profiler kernel table here (3 steps)")
    p.add_argument("--style", type=str, default="ddp",
                   choices=["ddp", "apex", "horovod"],
                   help="launch-style variant to benchmark (BASELINE configs)")
    p.add_argument("--hipgraph", type=str, default="off",
                   choices=["auto", "on", "off"],
                   help="capture the steady-state step into a hipGraph and "
                        "replay it (measured NEUTRAL at the default batch — "
                        "the 313 ms step already hides launch overhead — so "
                        "off by default; capture verified correct on "
                        "hardware, world==1 only)")
    return p.parse_args()


def main():
    args = parse_args()
    # MIOpen exhaustive find can stall for minutes on new shapes (observed at
    # b512); FAST find keeps the library paths predictable for A/B runs
    os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")
    if args.no_ext:
        os.environ["AMDTRAIN_DISABLE_EXT"] = "1"
        os.environ["AMDTRAIN_ALLOW_EAGER"] = "1"

    import amdtrain.comm as comm
    from amdtrain.models import build_model
    from amdtrain.ops import CrossEntropyLoss, FusedSGD
    from amdtrain.ops import functional as OF
    from amdtrain.parallel import NativeDDP

    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    cpu_mode = os.environ.get("AMDTRAIN_BENCH_CPU") == "1"  # test-only
    if world > 1:
        comm.init_from_env(backend="gloo" if cpu_mode else None)

    if cpu_mode:
        device = torch.device("cpu")
    else:
        assert torch.cuda.is_available(), "bench.py requires an MI355X"
        torch.cuda.set_device(local_rank)
        device = torch.device(f"cuda:{local_rank}")
        torch.backends.cudnn.benchmark = True

    B = args.batch_per_gpu
    S = args.image_size
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32

    model = build_model(args.arch).to(device) \
        .to(memory_format=torch.channels_last)
    criterion = CrossEntropyLoss()
    optimizer = FusedSGD(model.parameters(), lr=0.1, momentum=0.9,
                         weight_decay=1e-4)
    amp_handle = None
    if args.style == "apex":
        from amdtrain.parallel import amp as amp_mod
        model, optimizer = amp_mod.initialize(model, optimizer,
                                              opt_level="O2", dtype=dtype)
        amp_handle = amp_mod
    if args.style == "horovod":
        import amdtrain.comm as C
        from amdtrain.parallel import Compression, DistributedOptimizer
        C.broadcast_module_state(model, src=0)
        optimizer = DistributedOptimizer(optimizer,
                                         model.named_parameters(),
                                         compression=Compression.fp16,
                                         fusion_mb=args.bucket_mb)
    elif world > 1:
        model = NativeDDP(model, bucket_cap_mb=args.bucket_mb)
    model.train()

    # synthetic data of the benchmark shape, resident on GPU (per-rank seed)
    g = torch.Generator(device="cpu").manual_seed(1234 + rank)
    images_u8 = torch.randint(0, 256, (B, 3, S, S), dtype=torch.uint8,
                              generator=g).to(device) \
        .contiguous(memory_format=torch.channels_last)
    targets = torch.randint(0, 1000, (B,), generator=g).to(device)

    def step():
        x = OF.normalize_u8(images_u8, dtype=dtype)
        if dtype == torch.bfloat16 and amp_handle is None:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                out = model(x)
        else:
            out = model(x)
        loss = criterion(out, targets)
        if hasattr(model, "reducer"):
            model.zero_grad()  # zero the bucket flats (grads are views)
        elif hasattr(optimizer, "reducer"):
            optimizer.zero_grad()
        else:
            # fresh grads each step: avoids 161 fill_ kernels + 161
            # AccumulateGrad add_ kernels per step
            optimizer.zero_grad(set_to_none=True)
        if amp_handle is not None:
            with amp_handle.scale_loss(loss, optimizer) as scaled:
                scaled.backward()
        else:
            loss.backward()
        optimizer.step()
        return loss

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize()

    # hipGraph capture (north-star: HIP streams and graphs): the whole
    # steady-state step — normalize, forward, backward, fused SGD — records
    # once and replays with zero per-kernel launch overhead.  world>1 keeps
    # eager launches (RCCL collectives inside capture are not exercised).
    use_graph = (args.hipgraph == "on"
                 or (args.hipgraph == "auto" and world == 1
                     and args.style == "ddp" and device.type == "cuda"
                     and not args.profile))
    graph = None
    if use_graph:
        for _ in range(3):  # allocator warmup before capture
            step()
        sync()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            graph_loss = step()

        def step_graph():
            graph.replay()
            return graph_loss
    for _ in range(args.warmup):
        (step_graph if graph is not None else step)()
    if args.profile and rank == 0:
        from amdtrain.utils.profiling import profile_steps
        with profile_steps(args.profile):
            for _ in range(3):
                step()
        sync()
    comm.barrier()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = (step_graph if graph is not None else step)()
    comm.barrier()
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if world > 1:
        t = torch.tensor([elapsed], device=device)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    if rank == 0:
        total_images = args.steps * B * world
        ips = total_images / elapsed
        baseline_ips = 1078.0  # reference DDP: 1.28M imgs / 1188.5 s on 4xV100
        result = {
            "metric": "ImageNet images/sec (ResNet-50 train step, whole job)",
            "value": round(ips, 1),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(ips / baseline_ips, 2),
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": args.arch,
                "global_batch": B * world,
                "image_size": S,
                "seq_len": None,
                "parallelism": f"dp{world}",
                "style": args.style,
                "sec_per_epoch_est": round(1_281_167 / ips, 1),
                "hip_ext": not args.no_ext,
                "hipgraph": bool(graph is not None),
                "peak_mem_gb": round(
                    torch.cuda.max_memory_allocated() / 2**30, 2)
                if device.type == "cuda" else None,
                "final_loss": round(float(loss.item()), 4),
            },
        }
        print(json.dumps(result), flush=True)

    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
