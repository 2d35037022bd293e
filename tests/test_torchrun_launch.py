"""Launcher-style end-to-end test: run the real entrypoints under
``torch.distributed.run`` with 2 processes on CPU/gloo — the exact launch
pattern the reference's start.sh uses (and the benchmark driver)."""

import os
import subprocess
import sys

import pytest
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

COMMON = ["-a", "resnet18", "--synthetic",
          "--synthetic-train-size", "8", "--synthetic-val-size", "8",
          "--image-size", "32", "-b", "8", "--epochs", "1",
          "-j", "0", "--max-steps", "2", "-p", "1", "--dtype", "fp32"]


def _torchrun(module, extra, tmp_path, port):
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node=2",
           "--master-addr", "127.0.0.1", "--master-port", str(port),
           "-m", module] + COMMON + extra
    env = dict(os.environ, PYTHONPATH=ROOT + os.pathsep +
               os.environ.get("PYTHONPATH", ""))
    return subprocess.run(cmd, cwd=str(tmp_path), env=env,
                          capture_output=True, text=True, timeout=420)


def test_torchrun_ddp_entrypoint(tmp_path):
    r = _torchrun("amdtrain.cli.distributed", [], tmp_path, 29611)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    # rank-0-only checkpoint (reference distributed.py:218-225)
    assert (tmp_path / "checkpoint.pth.tar").exists()
    ck = torch.load(str(tmp_path / "checkpoint.pth.tar"), weights_only=False)
    assert ck["arch"] == "resnet18"
    assert "Acc@1" in r.stdout  # validate summary printed


def test_torchrun_apex_entrypoint(tmp_path):
    r = _torchrun("amdtrain.cli.apex_distributed", ["--opt-level", "O1"],
                  tmp_path, 29613)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert (tmp_path / "checkpoint.pth.tar").exists()


def test_torchrun_horovod_style_entrypoint(tmp_path):
    r = _torchrun("amdtrain.cli.horovod_distributed",
                  ["--compression", "none"], tmp_path, 29615)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert (tmp_path / "checkpoint.pth.tar").exists()


def test_torchrun_bench_contract_cpu(tmp_path):
    """Run bench.py exactly as the scaling driver does (torchrun, 2 procs) —
    CPU mode via AMDTRAIN_BENCH_CPU=1 — and validate the JSON contract."""
    import json
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node=2",
           "--master-addr", "127.0.0.1", "--master-port", "29617",
           os.path.join(ROOT, "bench.py"), "--gpus", "2", "--steps", "2",
           "--warmup", "1", "--batch-per-gpu", "2", "--arch", "resnet18",
           "--image-size", "32", "--dtype", "fp32"]
    env = dict(os.environ, AMDTRAIN_BENCH_CPU="1",
               AMDTRAIN_DISABLE_EXT="1", AMDTRAIN_ALLOW_EAGER="1",
               PYTHONPATH=ROOT + os.pathsep + os.environ.get("PYTHONPATH", ""))
    r = subprocess.run(cmd, cwd=str(tmp_path), env=env, capture_output=True,
                       text=True, timeout=420)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2 and d["steps"] == 2
    assert d["unit"] == "images/sec" and d["higher_is_better"] is True
    assert d["config"]["global_batch"] == 4
    assert d["scaling"] == "weak" and d["vs_baseline"] is not None


def test_torchrun_bench_contract_styles_cpu(tmp_path):
    """The bench's --style apex/horovod variants must satisfy the same
    JSON contract as ddp (the styles share the timing harness; this
    guards the style-flag plumbing the driver never exercises)."""
    import json
    for port, style in ((29618, "horovod"), (29619, "apex")):
        cmd = [sys.executable, "-m", "torch.distributed.run",
               "--nnodes=1", "--nproc-per-node=2",
               "--master-addr", "127.0.0.1", "--master-port", str(port),
               os.path.join(ROOT, "bench.py"), "--gpus", "2", "--steps",
               "2", "--warmup", "1", "--batch-per-gpu", "2", "--arch",
               "resnet18", "--image-size", "32", "--dtype", "fp32",
               "--style", style]
        env = dict(os.environ, AMDTRAIN_BENCH_CPU="1",
                   AMDTRAIN_DISABLE_EXT="1", AMDTRAIN_ALLOW_EAGER="1",
                   PYTHONPATH=ROOT + os.pathsep +
                   os.environ.get("PYTHONPATH", ""))
        r = subprocess.run(cmd, cwd=str(tmp_path), env=env,
                           capture_output=True, text=True, timeout=420)
        assert r.returncode == 0, style + ": " + r.stdout[-1500:] \
            + r.stderr[-1500:]
        line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
        d = json.loads(line)
        assert d["config"]["style"] == style
        assert d["n_gpus"] == 2 and d["config"]["global_batch"] == 4
