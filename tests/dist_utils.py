"""Helpers for multi-process CPU (gloo) tests: spawn N ranks running a
module-level function with a file:// rendezvous."""

import io
import os
import tempfile

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _entry(rank, world, file_path, fn, args, q):
    try:
        dist.init_process_group("gloo", init_method=f"file://{file_path}",
                                world_size=world, rank=rank)
        out = fn(rank, world, *args)
        # serialize through bytes: fd-based tensor sharing breaks once the
        # producer process exits
        buf = io.BytesIO()
        torch.save(out, buf)
        q.put((rank, "ok", buf.getvalue()))
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, "err", traceback.format_exc()))
        raise
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_distributed(fn, world: int = 2, args: tuple = ()):
    """Run ``fn(rank, world, *args)`` on ``world`` gloo ranks; returns
    {rank: result}."""
    with tempfile.TemporaryDirectory() as td:
        file_path = os.path.join(td, "rdv")
        ctx = mp.get_context("spawn")
        q = ctx.SimpleQueue()
        procs = [ctx.Process(target=_entry,
                             args=(r, world, file_path, fn, args, q))
                 for r in range(world)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(world):
            rank, status, out = q.get()
            if status == "err":
                for p in procs:
                    p.terminate()
                raise RuntimeError(f"rank {rank} failed:\n{out}")
            results[rank] = torch.load(io.BytesIO(out), weights_only=False)
        for p in procs:
            p.join(timeout=60)
        return results
