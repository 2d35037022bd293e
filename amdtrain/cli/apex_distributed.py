"""Apex-style mixed-precision DDP entrypoint (reference apex_distributed.py).

Launcher-style process-per-GPU with:
  * ``amp.initialize(model, optimizer)`` mixed precision
    (reference :216; our O2 = half model + fp32 master weights + fused
    unscale/inf-check HIP kernels, O1 = autocast),
  * ``scale_loss`` backward (reference :328-329),
  * the GPU-side side-stream prefetcher with on-GPU normalize
    (reference data_prefetcher :115-169),
  * the reference's quirk of NOT sharding the validation set — every rank
    evaluates the full val set (reference :246-257).
"""

from __future__ import annotations

import os

from .. import comm
from ..config import base_parser
from .common import StyleConfig, run_worker

STYLE = StyleConfig(name="apex_distributed", wrap="ddp", amp_level="O2",
                    shard_val=False, use_prefetcher=True)


def parse_args(argv=None):
    p = base_parser("MI355X Apex-style AMP DDP ImageNet training")
    p.add_argument("--local_rank", "--local-rank", default=None, type=int)
    p.add_argument("--opt-level", default="O2", choices=["O1", "O2"],
                   help="AMP opt level (apex parity; default O2)")
    return p.parse_args(argv)


def main(argv=None) -> float:
    args = parse_args(argv)
    style = StyleConfig(**{**STYLE.__dict__, "amp_level": args.opt_level})
    local_rank = args.local_rank
    if local_rank is None:
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
        comm.init_from_env()
        nprocs = comm.get_world_size()
    else:
        nprocs = 1
    return run_worker(local_rank, nprocs, args, style,
                      global_rank=comm.get_rank())


if __name__ == "__main__":
    main()
