"""torchrun worker for the 2-process RCCL horovod-style engine GPU test.

Exercises DistributedOptimizer (hook-based fused all-reduce with fp16
gradient compression) + rank-0 parameter/optimizer-state broadcast over
REAL RCCL — reference horovod_distributed.py:149-164.  Prints HVD_OK.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

import amdtrain.comm as comm  # noqa: E402
from amdtrain.models import build_model  # noqa: E402
from amdtrain.ops import CrossEntropyLoss, FusedSGD  # noqa: E402
from amdtrain.parallel import Compression, DistributedOptimizer  # noqa: E402


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(int(os.environ["LOCAL_RANK"]))
    dist.init_process_group("nccl")

    torch.manual_seed(1000 + rank)  # DIFFERENT init per rank on purpose:
    model = build_model("resnet18", num_classes=10).cuda() \
        .to(memory_format=torch.channels_last)
    comm.broadcast_module_state(model, src=0)  # ...broadcast must fix it
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
    comm.broadcast_optimizer_state(opt, src=0,
                                   device=torch.device("cuda"))
    opt = DistributedOptimizer(opt, model.named_parameters(),
                               compression=Compression.fp16, fusion_mb=8.0)
    crit = CrossEntropyLoss()

    torch.manual_seed(77)  # same data stream on all ranks
    for step in range(3):
        gx = torch.randn(8 * world, 3, 64, 64, device="cuda")
        gt = torch.randint(0, 10, (8 * world,), device="cuda")
        x = gx[rank * 8:(rank + 1) * 8] \
            .contiguous(memory_format=torch.channels_last)
        opt.zero_grad(set_to_none=False)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(x)
        crit(out, gt[rank * 8:(rank + 1) * 8]).backward()
        opt.step()
    torch.cuda.synchronize()

    flat = torch.cat([p.detach().reshape(-1).float()
                      for p in model.parameters()])
    flats = [torch.empty_like(flat) for _ in range(world)]
    dist.all_gather(flats, flat)
    for r in range(1, world):
        diff = (flats[r] - flats[0]).abs().max().item()
        assert diff == 0.0, f"rank {r} diverged by {diff}"
    dist.barrier()
    if rank == 0:
        print("HVD_OK")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
