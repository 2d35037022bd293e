"""Kernel-level profiling helpers (SURVEY §5 tracing parity, MI355X-native).

Wraps torch.profiler (roctracer-backed on ROCm) to capture a per-kernel time
table for a few training steps — the in-framework complement to out-of-band
``rocprofv3 --stats`` captures (see profiles/).
"""

from __future__ import annotations

import contextlib
from typing import Optional

import torch


@contextlib.contextmanager
def profile_steps(out_path: Optional[str] = None, row_limit: int = 40):
    """Profile the enclosed steps; writes a kernel time table to
    ``out_path`` (or stdout)."""
    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=False) as prof:
        yield prof
    table = prof.key_averages().table(
        sort_by="self_cuda_time_total", row_limit=row_limit)
    if out_path:
        with open(out_path, "w") as fh:
            fh.write(table)
    else:
        print(table, flush=True)
