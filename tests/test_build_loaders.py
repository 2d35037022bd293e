"""Loader construction: synthetic fallback, sharding wiring, quirk flags."""

import argparse

import torch

from amdtrain.data.build import build_datasets, build_loaders


def _args(**kw):
    d = dict(data="", synthetic=True, synthetic_train_size=16,
             synthetic_val_size=8, image_size=32, batch_size=4, workers=0)
    d.update(kw)
    return argparse.Namespace(**d)


def test_synthetic_fallback_when_no_data_dir():
    train, val = build_datasets(_args(synthetic=False, data="/nonexistent"))
    assert len(train) == 16 and len(val) == 8


def test_loaders_sharded_train_and_val():
    tl, vl, ts, vs = build_loaders(_args(), world_size=2, rank=1,
                                   distributed=True, distributed_val=True)
    assert ts is not None and ts.rank == 1 and ts.num_replicas == 2
    assert vs is not None  # distributed evaluation shard
    assert len(ts) == 8  # 16/2
    xb, yb = next(iter(tl))
    assert xb.shape == (4, 3, 32, 32)


def test_apex_quirk_unsharded_val():
    tl, vl, ts, vs = build_loaders(_args(), world_size=2, rank=0,
                                   distributed=True, distributed_val=False)
    assert ts is not None
    assert vs is None  # every rank evaluates the full val set
    assert len(vl.dataset) == 8


def test_single_process_loaders_shuffle():
    tl, vl, ts, vs = build_loaders(_args(), world_size=1, rank=0,
                                   distributed=False)
    assert ts is None and vs is None


def test_engine_max_steps():
    from torch.utils.data import DataLoader
    from amdtrain.data import SyntheticImageNet
    from amdtrain.engine.loops import TrainState, train
    from amdtrain.ops import CrossEntropyLoss, FusedSGD
    torch.manual_seed(0)
    m = torch.nn.Sequential(torch.nn.Flatten(), torch.nn.Linear(3 * 16 * 16, 5))
    opt = FusedSGD(m.parameters(), lr=0.01)
    ds = SyntheticImageNet(length=40, image_size=16, num_classes=5)
    loader = DataLoader(ds, batch_size=4)
    state = TrainState(device=torch.device("cpu"), channels_last=False,
                       max_steps=3, print_freq=100)
    steps_before = [p.detach().clone() for p in m.parameters()]
    train(loader, m, CrossEntropyLoss(), opt, 0, state)
    # only 3 of 10 batches consumed; params moved
    assert any(not torch.equal(a, b)
               for a, b in zip(steps_before, m.parameters()))
