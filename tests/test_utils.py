"""Unit tests for meters, accuracy, LR schedule, checkpoint schema
(reference behaviors per SURVEY §2a)."""

import os

import pytest
import torch

from amdtrain.utils import (AverageMeter, ProgressMeter, accuracy,
                            adjust_learning_rate, load_checkpoint,
                            save_checkpoint)
from amdtrain.utils.checkpoint import make_checkpoint_state
from amdtrain.utils.csvlog import EpochTimer


def test_average_meter():
    m = AverageMeter("Loss", ":.4e")
    m.update(2.0)
    m.update(4.0, n=3)
    assert m.val == 4.0
    assert m.count == 4
    assert m.sum == 14.0
    assert abs(m.avg - 3.5) < 1e-9
    assert "Loss" in str(m) and "(" in str(m)


def test_progress_meter_format(capsys):
    m = AverageMeter("Acc@1", ":6.2f")
    m.update(50.0)
    p = ProgressMeter(100, [m], prefix="Epoch: [3]")
    line = p.display(7)
    assert line.startswith("Epoch: [3][  7/100]")
    assert "Acc@1" in line


def test_accuracy_topk():
    # logits where sample i's true class ranks exactly i-th (distinct values:
    # torch.topk tie order is implementation-defined)
    logits = torch.tensor([
        [9.0, 5.0, 4.0, 3.0, 2.0],   # target 0 -> rank 0
        [9.0, 5.0, 4.0, 3.0, 2.0],   # target 1 -> rank 1
        [9.0, 5.0, 4.0, 3.0, 2.0],   # target 4 -> rank 4
        [9.0, 5.0, 4.0, 3.0, 2.0],   # target 2 -> rank 2
    ])
    target = torch.tensor([0, 1, 4, 2])
    acc1, acc5 = accuracy(logits, target, topk=(1, 5))
    assert acc1.shape == (1,)
    assert abs(acc1.item() - 25.0) < 1e-6
    assert abs(acc5.item() - 100.0) < 1e-6
    # cross-check against torch.topk semantics on random data
    g = torch.Generator().manual_seed(0)
    logits = torch.randn(64, 100, generator=g)
    target = torch.randint(0, 100, (64,), generator=g)
    a1, a3 = accuracy(logits, target, topk=(1, 3))
    _, pred = logits.topk(3, 1, True, True)
    correct = pred.eq(target.view(-1, 1))
    ref1 = correct[:, :1].any(1).float().sum() * 100 / 64
    ref3 = correct.any(1).float().sum() * 100 / 64
    assert torch.allclose(a1, ref1.reshape(1))
    assert torch.allclose(a3, ref3.reshape(1))


def test_adjust_learning_rate():
    opt = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=0.1)
    # reference: lr = lr0 * 0.1 ** (epoch // 30)
    for epoch, want in [(0, 0.1), (29, 0.1), (30, 0.01), (59, 0.01),
                        (60, 0.001), (89, 0.001)]:
        lr = adjust_learning_rate(opt, epoch, 0.1)
        assert abs(lr - want) < 1e-12
        assert abs(opt.param_groups[0]["lr"] - want) < 1e-12


def test_checkpoint_roundtrip(tmp_path):
    model = torch.nn.Linear(4, 2)
    state = make_checkpoint_state(epoch=3, arch="resnet18", model=model,
                                  best_acc1=55.5)
    # reference dict schema (distributed.py:219-225)
    assert set(state.keys()) == {"epoch", "arch", "state_dict", "best_acc1"}
    assert state["epoch"] == 4  # epoch + 1
    ck = tmp_path / "checkpoint.pth.tar"
    best = tmp_path / "model_best.pth.tar"
    save_checkpoint(state, is_best=True, filename=str(ck),
                    best_filename=str(best))
    assert ck.exists() and best.exists()
    model2 = torch.nn.Linear(4, 2)
    loaded = load_checkpoint(str(ck), model2)
    assert loaded["arch"] == "resnet18"
    for a, b in zip(model.parameters(), model2.parameters()):
        assert torch.equal(a, b)


def test_checkpoint_module_prefix(tmp_path):
    model = torch.nn.Linear(4, 2)
    sd = {"module." + k: v for k, v in model.state_dict().items()}
    torch.save({"epoch": 1, "arch": "x", "state_dict": sd, "best_acc1": 0.0},
               tmp_path / "c.pth.tar")
    model2 = torch.nn.Linear(4, 2)
    load_checkpoint(str(tmp_path / "c.pth.tar"), model2)
    for a, b in zip(model.parameters(), model2.parameters()):
        assert torch.equal(a, b)


def test_epoch_timer_csv(tmp_path):
    path = tmp_path / "epochs.csv"
    t = EpochTimer(str(path))
    t.start()
    t.stop(0)
    t.start()
    t.stop(1)
    t.close()
    lines = path.read_text().strip().splitlines()
    assert lines[0] == "epoch,seconds"
    assert len(lines) == 3


def test_profile_steps_cpu(tmp_path):
    """profile_steps wraps torch.profiler and writes a kernel table
    (CPU activities here; CUDA rows appear on hardware)."""
    import torch
    from amdtrain.utils.profiling import profile_steps
    out = tmp_path / "prof.txt"
    with profile_steps(str(out)):
        torch.randn(64, 64) @ torch.randn(64, 64)
    text = out.read_text()
    assert "Name" in text and "CPU" in text
