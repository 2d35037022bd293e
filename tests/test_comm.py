"""Multi-rank (gloo, CPU) tests of the comm layer: reduce_mean, fused
MetricReducer, module/optimizer state broadcast — the rank math and
collective semantics that run over RCCL on the GPU node."""

import torch

from dist_utils import run_distributed


def _reduce_mean_fn(rank, world):
    from amdtrain.comm import reduce_mean
    t = torch.tensor([float(rank + 1)])
    out = reduce_mean(t, world)
    return out.item()


def test_reduce_mean_two_ranks():
    res = run_distributed(_reduce_mean_fn, world=2)
    # mean of 1.0 and 2.0
    assert abs(res[0] - 1.5) < 1e-6 and abs(res[1] - 1.5) < 1e-6


def _metric_reducer_fn(rank, world):
    from amdtrain.comm import MetricReducer
    r = MetricReducer(3, torch.device("cpu"))
    vals = [torch.tensor(float(rank)), torch.tensor(10.0 * rank),
            torch.tensor(1.0)]
    r.reduce(vals)
    return r.items()


def test_metric_reducer_fused():
    res = run_distributed(_metric_reducer_fn, world=2)
    for rank in (0, 1):
        loss, a, b = res[rank]
        assert abs(loss - 0.5) < 1e-6
        assert abs(a - 5.0) < 1e-6
        assert abs(b - 1.0) < 1e-6


def _broadcast_fn(rank, world):
    from amdtrain.comm import broadcast_module_state
    torch.manual_seed(rank)  # ranks start with DIFFERENT weights
    m = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.BatchNorm1d(8))
    broadcast_module_state(m, src=0)
    return [t.clone() for t in m.state_dict().values()]


def test_broadcast_module_state():
    res = run_distributed(_broadcast_fn, world=2)
    for a, b in zip(res[0], res[1]):
        assert torch.equal(a, b)


def _bcast_opt_fn(rank, world):
    from amdtrain.comm import broadcast_optimizer_state
    torch.manual_seed(rank)
    p = torch.nn.Parameter(torch.randn(4))
    opt = torch.optim.SGD([p], lr=0.1, momentum=0.9)
    p.grad = torch.randn(4)
    opt.step()  # creates momentum state, different per rank
    broadcast_optimizer_state(opt, src=0)
    buf = opt.state[p]["momentum_buffer"]
    return buf.clone()


def test_broadcast_optimizer_state():
    res = run_distributed(_bcast_opt_fn, world=2)
    assert torch.equal(res[0], res[1])


def _barrier_fn(rank, world):
    from amdtrain.comm import barrier
    barrier()
    return True


def test_barrier():
    res = run_distributed(_barrier_fn, world=2)
    assert res[0] and res[1]
