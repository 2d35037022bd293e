"""Horovod-style DistributedOptimizer entrypoint (reference horovod_distributed.py).

The reference launches with ``horovodrun -np 4 -H localhost:4`` (MPI) and
drives Horovod's C++ allreduce engine.  The MI355X-native equivalent needs
no MPI: workers spawn per GPU (or run under torchrun) and the
``DistributedOptimizer`` wrapper performs hook-based averaged gradient
all-reduce with fp16 compression over RCCL (reference :149-164 semantics:
rank-0 parameter + optimizer-state broadcast, fp16 grad compression,
avg-allreduce metric reduction).
"""

from __future__ import annotations

import os

import torch
import torch.multiprocessing as mp

from .. import comm
from ..config import base_parser
from .common import StyleConfig, run_worker

STYLE = StyleConfig(name="horovod_distributed", wrap="horovod",
                    compression="fp16")


def parse_args(argv=None):
    p = base_parser("MI355X Horovod-style DistributedOptimizer training")
    p.add_argument("--dist-addr", default="127.0.0.1", type=str)
    p.add_argument("--dist-port", default=23457, type=int)
    p.add_argument("--nprocs", default=0, type=int)
    p.add_argument("--compression", default="fp16",
                   choices=["none", "fp16", "bf16"],
                   help="gradient compression (hvd.Compression parity)")
    return p.parse_args(argv)


def worker(local_rank: int, nprocs: int, args) -> None:
    comm.init_from_tcp(rank=local_rank, world_size=nprocs,
                       addr=args.dist_addr, port=args.dist_port)
    style = StyleConfig(**{**STYLE.__dict__, "compression": args.compression})
    run_worker(local_rank, nprocs, args, style)


def main(argv=None) -> None:
    args = parse_args(argv)
    style = StyleConfig(**{**STYLE.__dict__, "compression": args.compression})
    if "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
        # torchrun-launched
        comm.init_from_env()
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
        run_worker(local_rank, comm.get_world_size(), args, style,
                   global_rank=comm.get_rank())
        return
    if "OMPI_COMM_WORLD_SIZE" in os.environ \
            and int(os.environ["OMPI_COMM_WORLD_SIZE"]) > 1:
        # MPI-style launch (`mpirun -np N python -m ...` — the reference's
        # `horovodrun -np 4 -H localhost:4` lineage, start.sh:4).  Open MPI
        # exports the rank/size env vars; rendezvous stays TCP on
        # --dist-addr/--dist-port (single node: the 127.0.0.1 default works;
        # multi-node: pass rank 0's address or use `-x MASTER_ADDR`).
        rank = int(os.environ["OMPI_COMM_WORLD_RANK"])
        world = int(os.environ["OMPI_COMM_WORLD_SIZE"])
        local_rank = int(os.environ.get("OMPI_COMM_WORLD_LOCAL_RANK", rank))
        addr = os.environ.get("MASTER_ADDR", args.dist_addr)
        comm.init_from_tcp(rank=rank, world_size=world, addr=addr,
                           port=args.dist_port)
        run_worker(local_rank, world, args, style, global_rank=rank)
        return
    nprocs = args.nprocs or torch.cuda.device_count() or 1
    if nprocs == 1:
        run_worker(0, 1, args, style)
        return
    mp.spawn(worker, nprocs=nprocs, args=(nprocs, args))


if __name__ == "__main__":
    main()
