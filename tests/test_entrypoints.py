"""End-to-end entrypoint tests on CPU: each launch style runs a tiny
synthetic epoch through its real ``main()`` (world_size=1 here; multi-rank
engine behavior is covered by test_parallel_training.py)."""

import os

import pytest
import torch

COMMON = ["-a", "resnet18", "--synthetic",
          "--synthetic-train-size", "8", "--synthetic-val-size", "8",
          "--image-size", "64", "-b", "4", "--epochs", "1",
          "-j", "0", "--max-steps", "2", "-p", "1", "--dtype", "fp32"]


def _run_in(tmp_path, fn, argv):
    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        return fn(argv)
    finally:
        os.chdir(cwd)


def test_distributed_entrypoint(tmp_path):
    from amdtrain.cli.distributed import main
    _run_in(tmp_path, main, COMMON)
    assert (tmp_path / "checkpoint.pth.tar").exists()
    ck = torch.load(str(tmp_path / "checkpoint.pth.tar"),
                    weights_only=False)
    assert set(ck.keys()) == {"epoch", "arch", "state_dict", "best_acc1"}
    assert ck["arch"] == "resnet18"


def test_multiprocessing_entrypoint(tmp_path):
    from amdtrain.cli.multiprocessing_distributed import main
    _run_in(tmp_path, main, COMMON + ["--nprocs", "1"])
    assert (tmp_path / "checkpoint.pth.tar").exists()


def test_apex_entrypoint(tmp_path):
    from amdtrain.cli.apex_distributed import main
    # O1 on CPU (O2 halving is exercised in test_amp)
    _run_in(tmp_path, main, COMMON + ["--opt-level", "O1"])
    assert (tmp_path / "checkpoint.pth.tar").exists()


def test_horovod_entrypoint(tmp_path):
    from amdtrain.cli.horovod_distributed import main
    _run_in(tmp_path, main, COMMON + ["--nprocs", "1"])
    assert (tmp_path / "checkpoint.pth.tar").exists()


def test_dataparallel_entrypoint(tmp_path):
    from amdtrain.cli.dataparallel import main
    _run_in(tmp_path, main, COMMON)
    assert (tmp_path / "checkpoint.pth.tar").exists()
    assert (tmp_path / "dataparallel_epochs.csv").exists()


def test_slurm_entrypoint(tmp_path):
    from amdtrain.cli.distributed_slurm_main import main
    _run_in(tmp_path, main, COMMON)
    assert (tmp_path / "checkpoint.pth.tar").exists()


def test_evaluate_flag(tmp_path):
    from amdtrain.cli.distributed import main
    acc = _run_in(tmp_path, main, COMMON + ["-e"])
    assert isinstance(acc, float)


def test_resume(tmp_path):
    from amdtrain.cli.distributed import main
    _run_in(tmp_path, main, COMMON)
    # resume from the checkpoint written above (start_epoch advances to 1)
    _run_in(tmp_path, main, COMMON + ["--resume", "checkpoint.pth.tar"])
