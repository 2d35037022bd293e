"""Process-group bootstrap — all three reference rendezvous styles over RCCL.

On ROCm the ``"nccl"`` backend string IS RCCL (torch-ROCm's ProcessGroupNCCL
compiles against it); collectives ride xGMI between the 8 MI355X GPUs.
The three init styles mirror the reference exactly:

  * env://  — launcher-style, MASTER_ADDR/PORT + RANK/WORLD_SIZE from the
              environment (reference distributed.py:132)
  * tcp://  — spawn-style explicit host:port + world_size + rank
              (reference multiprocessing_distributed.py:132-135)
  * file:// — Slurm-style shared-filesystem rendezvous
              (reference distributed_slurm_main.py:129-140)

CPU tests use the same call paths with backend="gloo".
"""

from __future__ import annotations

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


def backend_for_device(device: Optional[str] = None) -> str:
    """'nccl' (=RCCL on ROCm) when GPUs drive the job, else 'gloo'."""
    if device is not None:
        return "nccl" if str(device).startswith("cuda") else "gloo"
    return "nccl" if torch.cuda.is_available() else "gloo"


_DEFAULT_TIMEOUT = datetime.timedelta(minutes=10)


def init_from_env(backend: Optional[str] = None,
                  timeout: datetime.timedelta = _DEFAULT_TIMEOUT) -> int:
    """env:// rendezvous. Returns the global rank."""
    backend = backend or backend_for_device()
    dist.init_process_group(backend=backend, init_method="env://",
                            timeout=timeout)
    return dist.get_rank()


def init_from_tcp(rank: int, world_size: int,
                  addr: str = "127.0.0.1", port: int = 23456,
                  backend: Optional[str] = None,
                  timeout: datetime.timedelta = _DEFAULT_TIMEOUT) -> int:
    """tcp://addr:port rendezvous (reference's tcp://127.0.0.1:23456)."""
    backend = backend or backend_for_device()
    dist.init_process_group(backend=backend,
                            init_method=f"tcp://{addr}:{port}",
                            world_size=world_size, rank=rank, timeout=timeout)
    return dist.get_rank()


def init_from_file(rank: int, world_size: int, file_path: str,
                   job_id: Optional[str] = None,
                   backend: Optional[str] = None,
                   timeout: datetime.timedelta = _DEFAULT_TIMEOUT) -> int:
    """file:// rendezvous; ``job_id`` is appended like the reference's
    ``file://<realpath(dist_file)>.<SLURM_JOBID>`` (distributed_slurm_main.py:129-130)."""
    backend = backend or backend_for_device()
    path = os.path.realpath(file_path)
    if job_id is not None:
        path = f"{path}.{job_id}"
    dist.init_process_group(backend=backend, init_method=f"file://{path}",
                            world_size=world_size, rank=rank, timeout=timeout)
    return dist.get_rank()


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1
