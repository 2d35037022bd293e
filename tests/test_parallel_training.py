"""End-to-end parallel-engine tests on CPU/gloo: NativeDDP training keeps
replicas in lock-step and matches the equivalent single-process big-batch
run; Horovod-style DistributedOptimizer averages gradients through hooks."""

import torch

from dist_utils import run_distributed


def _tiny_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(8, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))


def _ddp_train_fn(rank, world, steps):
    from amdtrain.parallel import NativeDDP
    from amdtrain.ops import FusedSGD
    torch.manual_seed(rank * 7 + 1)  # different init; broadcast will fix it
    model = _tiny_model(seed=rank)
    ddp = NativeDDP(model, bucket_cap_mb=0.0001)
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9,
                   weight_decay=1e-4)
    losses = []
    for step in range(steps):
        torch.manual_seed(1000 + step * world + rank)
        x = torch.randn(6, 8)
        t = torch.randint(0, 4, (6,))
        ddp.zero_grad()
        loss = torch.nn.functional.cross_entropy(ddp(x), t)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return ([p.detach().clone() for p in model.parameters()], losses)


def _single_train(steps, world=2):
    """Equivalent big-batch single-process run: concatenate both ranks'
    batches; grads average since CE-mean over equal-size shards == mean of
    shard means."""
    model = _tiny_model(seed=0)  # rank 0's init is broadcast in DDP
    opt_params = list(model.parameters())
    from amdtrain.ops import FusedSGD
    opt = FusedSGD(opt_params, lr=0.05, momentum=0.9, weight_decay=1e-4)
    for step in range(steps):
        xs, ts = [], []
        for rank in range(world):
            torch.manual_seed(1000 + step * world + rank)
            xs.append(torch.randn(6, 8))
            ts.append(torch.randint(0, 4, (6,)))
        x = torch.cat(xs)
        t = torch.cat(ts)
        opt.zero_grad(set_to_none=False)
        loss = torch.nn.functional.cross_entropy(model(x), t)
        loss.backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def test_ddp_matches_big_batch():
    res = run_distributed(_ddp_train_fn, world=2, args=(4,))
    params0, _ = res[0]
    params1, _ = res[1]
    # ranks stay in lock-step
    for a, b in zip(params0, params1):
        assert torch.allclose(a, b, atol=1e-6)
    # and match the single-process big-batch equivalent
    ref = _single_train(4)
    for a, r in zip(params0, ref):
        assert torch.allclose(a, r, atol=1e-5), (a - r).abs().max()


def _hvd_train_fn(rank, world, steps):
    from amdtrain.parallel import DistributedOptimizer, Compression
    from amdtrain.parallel.horovod_style import broadcast_parameters
    from amdtrain.comm import broadcast_optimizer_state
    from amdtrain.ops import FusedSGD
    model = _tiny_model(seed=rank)
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
    broadcast_parameters(model, src=0)
    broadcast_optimizer_state(opt, src=0)
    opt = DistributedOptimizer(opt, model.named_parameters(),
                               compression=Compression.none,
                               fusion_mb=0.0001)
    for step in range(steps):
        torch.manual_seed(2000 + step * world + rank)
        x = torch.randn(6, 8)
        t = torch.randint(0, 4, (6,))
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), t)
        loss.backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def test_horovod_style_lockstep():
    res = run_distributed(_hvd_train_fn, world=2, args=(3,))
    for a, b in zip(res[0], res[1]):
        assert torch.allclose(a, b, atol=1e-6)


def _slurm_rank_math():
    # reference distributed_slurm_main.py:136: rank = node_rank*ngpus + gpu
    ranks = []
    for node_rank in range(2):
        for gpu in range(4):
            ranks.append(node_rank * 4 + gpu)
    return ranks


def test_slurm_global_rank_math():
    assert _slurm_rank_math() == list(range(8))


def _apex_style_fn(rank, world, steps):
    """AMP O1 + NativeDDP + scale_loss on gloo (the apex entrypoint's path)."""
    import torch.nn.functional as TF
    from amdtrain.parallel import NativeDDP, amp
    from amdtrain.ops import FusedSGD
    model = _tiny_model(seed=rank)
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
    model, opt = amp.initialize(model, opt, opt_level="O1",
                                dtype=torch.bfloat16)
    ddp = NativeDDP(model, bucket_cap_mb=0.0001)
    for step in range(steps):
        torch.manual_seed(3000 + step * world + rank)
        x = torch.randn(6, 8)
        t = torch.randint(0, 4, (6,))
        ddp.zero_grad()
        loss = TF.cross_entropy(ddp(x), t)
        with amp.scale_loss(loss, opt) as scaled:
            scaled.backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def test_apex_style_ddp_lockstep():
    res = run_distributed(_apex_style_fn, world=2, args=(3,))
    for a, b in zip(res[0], res[1]):
        assert torch.allclose(a, b, atol=1e-6)
