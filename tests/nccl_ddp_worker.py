"""torchrun worker for the 2-process RCCL (nccl-backend) NativeDDP GPU test.

Run by tests/test_gpu_multidev.py::test_nccl_ddp_2proc as
  torchrun --nproc-per-node 2 tests/nccl_ddp_worker.py
Each rank trains NativeDDP ResNet-18 (bf16, custom kernels) on its own GPU
over real RCCL, then checks:
  * parameters stay BITWISE identical across ranks after optimizer steps
    (the bucketed AVG all-reduce + identical init must keep replicas in
    lockstep);
  * the reducer enqueued its bucket all-reduces during backward (overlap);
  * DDP gradients equal the big-batch gradient (rank0 recompute).
Prints NCCL_DDP_OK on success (asserted by the pytest wrapper).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

from amdtrain.models import build_model  # noqa: E402
from amdtrain.ops import CrossEntropyLoss, FusedSGD  # noqa: E402
from amdtrain.parallel import NativeDDP  # noqa: E402


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local = int(os.environ["LOCAL_RANK"])
    torch.cuda.set_device(local)
    dist.init_process_group("nccl")

    torch.manual_seed(0)
    model = build_model("resnet18", num_classes=10).cuda() \
        .to(memory_format=torch.channels_last)
    ddp = NativeDDP(model, bucket_cap_mb=8.0)  # several buckets
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9,
                   weight_decay=1e-4)
    crit = CrossEntropyLoss()

    torch.manual_seed(100)  # same on all ranks: build the global batch
    for step in range(3):
        gx = torch.randn(8 * world, 3, 64, 64, device="cuda")
        gt = torch.randint(0, 10, (8 * world,), device="cuda")
        x = gx[rank * 8:(rank + 1) * 8] \
            .contiguous(memory_format=torch.channels_last)
        t = gt[rank * 8:(rank + 1) * 8]
        opt.zero_grad(set_to_none=False)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = ddp(x)
        loss = crit(out, t)
        loss.backward()
        assert ddp.reducer.last_overlap_launches >= \
            len(ddp.reducer.buckets) - 1, "all-reduces did not overlap"
        opt.step()
    torch.cuda.synchronize()

    # replicas must be in lockstep: compare a full flat param vector
    flat = torch.cat([p.detach().reshape(-1).float()
                      for p in model.parameters()])
    flats = [torch.empty_like(flat) for _ in range(world)]
    dist.all_gather(flats, flat)
    for r in range(1, world):
        diff = (flats[r] - flats[0]).abs().max().item()
        assert diff == 0.0, f"rank {r} diverged from rank 0 by {diff}"

    # one more step: DDP grads == big-batch grads (computed on every rank
    # from the full global batch with sync disabled)
    torch.manual_seed(200)
    gx = torch.randn(8 * world, 3, 64, 64, device="cuda")
    gt = torch.randint(0, 10, (8 * world,), device="cuda")
    x = gx[rank * 8:(rank + 1) * 8] \
        .contiguous(memory_format=torch.channels_last)
    opt.zero_grad(set_to_none=False)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = ddp(x)
    crit(out, gt[rank * 8:(rank + 1) * 8]).backward()
    ddp_grads = [p.grad.detach().float().clone()
                 for p in model.parameters()]

    opt.zero_grad(set_to_none=False)
    with ddp.no_sync():
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = ddp(gx.contiguous(memory_format=torch.channels_last))
        crit(out, gt).backward()
    for g_ddp, p in zip(ddp_grads, model.parameters()):
        g_big = p.grad.detach().float()
        # bf16-path noise only: both compute the same mean-over-global-batch
        lim = 0.05 * g_big.abs().max().item() + 2e-2
        assert (g_ddp - g_big).abs().max().item() <= lim, \
            (g_ddp - g_big).abs().max().item()

    dist.barrier()
    if rank == 0:
        print("NCCL_DDP_OK")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
