"""Spawn-style DDP entrypoint (reference multiprocessing_distributed.py).

Self-launching: ``python -m amdtrain.cli.multiprocessing_distributed`` forks
one worker per GPU via ``torch.multiprocessing.spawn`` (reference :114) with
an explicit TCP rendezvous ``tcp://127.0.0.1:23456`` (reference :132-135).
Seeds are set inside the worker, as in the reference (:120-128).
"""

from __future__ import annotations

import torch
import torch.multiprocessing as mp

from .. import comm
from ..config import base_parser
from .common import StyleConfig, run_worker

STYLE = StyleConfig(name="multiprocessing_distributed", wrap="ddp")


def parse_args(argv=None):
    p = base_parser("MI355X spawn-style DDP ImageNet training")
    p.add_argument("--dist-addr", default="127.0.0.1", type=str)
    p.add_argument("--dist-port", default=23456, type=int)
    p.add_argument("--nprocs", default=0, type=int,
                   help="worker count (default: all visible GPUs)")
    return p.parse_args(argv)


def worker(local_rank: int, nprocs: int, args) -> None:
    comm.init_from_tcp(rank=local_rank, world_size=nprocs,
                       addr=args.dist_addr, port=args.dist_port)
    run_worker(local_rank, nprocs, args, STYLE)


def main(argv=None) -> None:
    args = parse_args(argv)
    nprocs = args.nprocs or torch.cuda.device_count() or 1
    if nprocs == 1:
        run_worker(0, 1, args, STYLE)
        return
    mp.spawn(worker, nprocs=nprocs, args=(nprocs, args))


if __name__ == "__main__":
    main()
