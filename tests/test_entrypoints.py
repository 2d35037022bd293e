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


def test_resume_restores_best_acc1(tmp_path):
    """Resume must restore best_acc1 from the checkpoint so the first
    post-resume epoch can't silently overwrite model_best.pth.tar with a
    worse model (advisor finding r1)."""
    from amdtrain.cli.distributed import main
    _run_in(tmp_path, main, COMMON)
    ck_path = tmp_path / "checkpoint.pth.tar"
    ck = torch.load(str(ck_path), weights_only=False)
    ck["best_acc1"] = 99.9  # a 2-step random-init run can't beat this
    torch.save(ck, str(ck_path))
    # sentinel "best" file: a correct resume must NOT overwrite it
    (tmp_path / "model_best.pth.tar").write_bytes(b"sentinel-best")
    best_bytes = (tmp_path / "model_best.pth.tar").read_bytes()
    # later --epochs wins in argparse: run one more epoch (start_epoch=1)
    _run_in(tmp_path, main,
            COMMON + ["--epochs", "2", "--resume", "checkpoint.pth.tar"])
    new_ck = torch.load(str(ck_path), weights_only=False)
    assert new_ck["best_acc1"] == 99.9  # carried through, not reset to 0
    # model_best untouched (no false is_best)
    assert (tmp_path / "model_best.pth.tar").read_bytes() == best_bytes


def test_slurm_rank_math_and_batch_division(monkeypatch):
    """Global rank = node_rank*ngpus + gpu (reference
    distributed_slurm_main.py:136) and per-GPU batch divides by
    ngpus_per_node, NOT world_size (reference :155)."""
    from amdtrain.cli import distributed_slurm_main as M

    calls = {}
    monkeypatch.setattr(M.comm, "init_from_file",
                        lambda **kw: calls.update(init=kw))

    def fake_run_worker(local_gpu, nprocs, args, style, global_rank=None):
        calls.update(local_gpu=local_gpu, nprocs=nprocs,
                     global_rank=global_rank)
    monkeypatch.setattr(M, "run_worker", fake_run_worker)

    args = M.parse_args(COMMON)
    # node 1 of 2, gpu 3 of 4 -> global rank 7; batch divisor must be 4
    M.worker(local_gpu=3, ngpus=4, node_rank=1, world_size=8, args=args)
    assert calls["init"]["rank"] == 7
    assert calls["init"]["world_size"] == 8
    assert calls["global_rank"] == 7
    assert calls["nprocs"] == 4  # run_worker divides batch by this


def test_horovod_mpi_style_launch(tmp_path):
    """MPI-style launch of the horovod entrypoint: two processes with
    OMPI_COMM_WORLD_* env vars (the `horovodrun -np 2` lineage,
    reference start.sh:4) rendezvous over TCP and train in lockstep."""
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    procs = []
    for rank in range(2):
        env = dict(os.environ,
                   OMPI_COMM_WORLD_SIZE="2",
                   OMPI_COMM_WORLD_RANK=str(rank),
                   OMPI_COMM_WORLD_LOCAL_RANK=str(rank),
                   PYTHONPATH=root + os.pathsep +
                   os.environ.get("PYTHONPATH", ""))
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "amdtrain.cli.horovod_distributed"]
            + COMMON + ["--dist-port", "29659"],
            cwd=str(tmp_path), env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    outs = []
    for p in procs:
        out, err = p.communicate(timeout=420)
        outs.append((p.returncode, out, err))
    for rc, out, err in outs:
        assert rc == 0, out[-2000:] + err[-2000:]
    assert (tmp_path / "checkpoint.pth.tar").exists()
