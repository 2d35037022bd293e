"""Distributed evaluation parity (reference distributed.py:279-324 +
README.md:586-662): sharded validation with fused metric all-reduce must
agree with a single-process evaluation of the full set."""

import torch

from dist_utils import run_distributed


def _make(seed=0):
    torch.manual_seed(seed)
    model = torch.nn.Sequential(torch.nn.Flatten(), torch.nn.Linear(12, 5))
    return model


def _dataset(n=32):
    torch.manual_seed(42)
    xs = torch.randn(n, 3, 2, 2)
    ts = torch.randint(0, 5, (n,))
    return list(zip(xs, ts))


def _eval_fn(rank, world):
    from torch.utils.data import DataLoader
    from amdtrain.data import DistributedSampler
    from amdtrain.engine.loops import TrainState, validate
    model = _make()
    data = _dataset()
    sampler = DistributedSampler(data, world, rank, shuffle=False)
    loader = DataLoader(data, batch_size=4, sampler=sampler)
    state = TrainState(device=torch.device("cpu"), world_size=world,
                       rank=rank, reduce_metrics=True, channels_last=False,
                       print_freq=100)
    crit = torch.nn.CrossEntropyLoss()
    return validate(loader, model, crit, state)


def test_distributed_eval_matches_serial():
    res = run_distributed(_eval_fn, world=2)
    # serial reference: full set, one process
    from torch.utils.data import DataLoader
    from amdtrain.engine.loops import TrainState, validate
    model = _make()
    loader = DataLoader(_dataset(), batch_size=4)
    state = TrainState(device=torch.device("cpu"), world_size=1, rank=0,
                       channels_last=False, print_freq=100)
    serial = validate(loader, model, torch.nn.CrossEntropyLoss(), state)
    # 32 samples / 2 ranks / batch 4: equal shards, no padding -> the
    # mean-of-batch-means equals the serial value exactly
    assert abs(res[0] - serial) < 1e-4
    assert abs(res[0] - res[1]) < 1e-9  # all ranks agree


def _o2_ddp_fn(rank, world, steps):
    """AMP O2 (mixed bf16/fp32 param dtypes) + NativeDDP on gloo."""
    from amdtrain.parallel import NativeDDP, amp
    from amdtrain.ops import FusedSGD
    torch.manual_seed(rank)
    model = torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.LayerNorm(16), torch.nn.ReLU(),
        torch.nn.Linear(16, 4))
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
    model, opt = amp.initialize(model, opt, opt_level="O2",
                                dtype=torch.bfloat16)
    ddp = NativeDDP(model, bucket_cap_mb=0.0001)
    for step in range(steps):
        torch.manual_seed(4000 + step * world + rank)
        x = torch.randn(6, 8).bfloat16()
        t = torch.randint(0, 4, (6,))
        ddp.zero_grad()
        loss = torch.nn.functional.cross_entropy(ddp(x).float(), t)
        with amp.scale_loss(loss, opt) as scaled:
            scaled.backward()
        opt.step()
    return [p.detach().float().clone() for p in model.parameters()]


def test_o2_ddp_mixed_dtype_lockstep():
    res = run_distributed(_o2_ddp_fn, world=2, args=(3,))
    for a, b in zip(res[0], res[1]):
        assert torch.allclose(a, b, atol=1e-5)
