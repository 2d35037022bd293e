"""Multi-node Slurm DDP entrypoint (reference distributed_slurm_main.py).

Per-node process reads ``SLURM_PROCID`` / ``SLURM_NPROCS`` / ``SLURM_JOBID``
(reference :124-128), rendezvouses over a shared-filesystem file
``file://<realpath(dist_file)>.<jobid>`` (reference :129-130), and spawns
one worker per local GPU; the global rank is ``node_rank * ngpus + gpu``
(reference :136).  Reference quirks preserved: NO per-iteration metric
all-reduce (plain local ``loss.item()``, :273-275), unsharded validation
(:192-203), checkpoints written by EVERY rank (:237-243), per-epoch CSV
(:209,229-235).
"""

from __future__ import annotations

import os

import torch
import torch.multiprocessing as mp

from .. import comm
from ..config import base_parser
from .common import StyleConfig, run_worker

STYLE = StyleConfig(name="distributed_slurm_main", wrap="ddp",
                    shard_val=False, reduce_metrics=False,
                    ckpt_rank0_only=False)


def parse_args(argv=None):
    p = base_parser("MI355X Slurm multi-node DDP ImageNet training")
    p.add_argument("--dist-file", default="distfile", type=str,
                   help="shared-filesystem rendezvous file "
                        "(reference distributed_slurm_main.py:102-105)")
    return p.parse_args(argv)


def worker(local_gpu: int, ngpus: int, node_rank: int, world_size: int,
           args) -> None:
    rank = node_rank * ngpus + local_gpu  # reference :136
    comm.init_from_file(rank=rank, world_size=world_size,
                        file_path=args.dist_file,
                        job_id=os.environ.get("SLURM_JOBID"))
    if not args.epoch_csv and rank == 0:
        args.epoch_csv = "slurm_epochs.csv"
    # per-GPU batch = total / ngpus_per_node (NOT / world_size) — reference
    # distributed_slurm_main.py:155 divides by ngpus_per_node, so each node
    # processes the full --batch-size
    run_worker(local_gpu, ngpus, args, STYLE, global_rank=rank)


def main(argv=None) -> None:
    args = parse_args(argv)
    node_rank = int(os.environ.get("SLURM_PROCID", 0))
    nnodes = int(os.environ.get("SLURM_NPROCS", 1))
    ngpus = torch.cuda.device_count() or 1
    world_size = nnodes * ngpus
    if world_size == 1:
        run_worker(0, 1, args, STYLE)
        return
    mp.spawn(worker, nprocs=ngpus,
             args=(ngpus, node_rank, world_size, args))


if __name__ == "__main__":
    main()
