"""Launcher-style DDP entrypoint (reference distributed.py).

One process per GPU, started externally::

    python -m torch.distributed.run --nproc-per-node=8 \
        --master-addr 127.0.0.1 -m amdtrain.cli.distributed -a resnet50 --synthetic

Accepts the legacy ``--local_rank`` flag (reference distributed.py:73-76) as
well as torchrun's LOCAL_RANK env var; rendezvous is env:// over RCCL
(reference distributed.py:132).
"""

from __future__ import annotations

import os

import torch

from .. import comm
from ..config import base_parser
from .common import StyleConfig, run_worker

STYLE = StyleConfig(name="distributed", wrap="ddp")


def parse_args(argv=None):
    p = base_parser("MI355X launcher-style DDP ImageNet training")
    p.add_argument("--local_rank", "--local-rank", default=None, type=int,
                   help="node-local rank (injected by the launcher)")
    return p.parse_args(argv)


def main(argv=None) -> float:
    args = parse_args(argv)
    local_rank = args.local_rank
    if local_rank is None:
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
        comm.init_from_env()
        nprocs = comm.get_world_size()
    else:
        nprocs = 1
    return run_worker(local_rank, nprocs, args, STYLE,
                      global_rank=comm.get_rank())


if __name__ == "__main__":
    main()
