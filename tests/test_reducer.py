"""Bucketed-reducer correctness on CPU/gloo: averaged gradients equal the
mean of per-rank gradients; grads stay as bucket views; no_sync works;
compression path converges to the same values (within half precision)."""

import torch

from dist_utils import run_distributed


def _make_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.ReLU(), torch.nn.Linear(16, 4))


def _reducer_fn(rank, world, compression):
    from amdtrain.parallel import BucketedReducer
    model = _make_model()  # same init on both ranks (same seed)
    reducer = BucketedReducer(list(model.parameters()), bucket_cap_mb=0.0001,
                              compression=compression)
    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randn(4, 8)
    y = model(x).sum()
    y.backward()
    return [p.grad.clone() for p in model.parameters()]


def _local_grads(seed):
    model = _make_model()
    torch.manual_seed(seed)
    x = torch.randn(4, 8)
    model(x).sum().backward()
    return [p.grad.clone() for p in model.parameters()]


def test_reducer_averages_gradients():
    res = run_distributed(_reducer_fn, world=2, args=("none",))
    g0 = _local_grads(100)
    g1 = _local_grads(101)
    expected = [(a + b) / 2 for a, b in zip(g0, g1)]
    for rank in (0, 1):
        for got, want in zip(res[rank], expected):
            assert torch.allclose(got, want, atol=1e-6), (got, want)


def test_reducer_compression_bf16():
    res = run_distributed(_reducer_fn, world=2, args=("bf16",))
    g0 = _local_grads(100)
    g1 = _local_grads(101)
    expected = [(a + b) / 2 for a, b in zip(g0, g1)]
    for got, want in zip(res[0], expected):
        assert torch.allclose(got, want, atol=0.05, rtol=0.05)
    # both ranks identical
    for a, b in zip(res[0], res[1]):
        assert torch.equal(a, b)


def _no_sync_fn(rank, world):
    from amdtrain.parallel import BucketedReducer
    model = _make_model()
    reducer = BucketedReducer(list(model.parameters()))
    torch.manual_seed(200 + rank)
    x = torch.randn(4, 8)
    with reducer.no_sync():
        model(x).sum().backward()
    return [p.grad.clone() for p in model.parameters()]


def test_reducer_no_sync_keeps_local():
    res = run_distributed(_no_sync_fn, world=2)
    l0 = _local_grads(200)
    for got, want in zip(res[0], l0):
        assert torch.allclose(got, want, atol=1e-6)
    # ranks differ (no sync happened)
    assert any(not torch.allclose(a, b)
               for a, b in zip(res[0], res[1]))


def test_reducer_single_process_grad_views():
    from amdtrain.parallel import BucketedReducer
    model = _make_model()
    reducer = BucketedReducer(list(model.parameters()))
    x = torch.randn(4, 8)
    model(x).sum().backward()
    ref = _make_model()
    ref_x = x.clone()
    ref(ref_x).sum().backward()
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.grad, q.grad, atol=1e-6)
    # grads are views into bucket flats
    flats = reducer.grad_buffers()
    total = sum(f.numel() for f in flats)
    assert total == sum(p.numel() for p in model.parameters())
    reducer.zero_grad()
    for p in model.parameters():
        assert torch.all(p.grad == 0)


def test_reducer_accumulation_across_backwards():
    from amdtrain.parallel import BucketedReducer
    model = _make_model()
    BucketedReducer(list(model.parameters()))
    x = torch.randn(4, 8)
    model(x).sum().backward()
    g1 = [p.grad.clone() for p in model.parameters()]
    model(x).sum().backward()  # accumulates into the same views
    for p, g in zip(model.parameters(), g1):
        assert torch.allclose(p.grad, 2 * g, atol=1e-5)


def test_reducer_four_ranks():
    """4-rank gloo: averaged gradients with multiple buckets (rank math at
    higher world sizes, closer to the 8-GPU node)."""
    res = run_distributed(_reducer_fn, world=4, args=("none",))
    grads = [_local_grads(100 + r) for r in range(4)]
    expected = [sum(gs) / 4 for gs in zip(*grads)]
    for rank in range(4):
        for got, want in zip(res[rank], expected):
            assert torch.allclose(got, want, atol=1e-6)


def test_allreduce_overlaps_backward():
    """The reducer must ENQUEUE bucket all-reduces while backward is still
    running (not flush everything in the final callback) — the overlap is
    the reducer's whole point (SURVEY §5 comm plan item ii)."""
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = [ctx.Process(target=_overlap_worker, args=(r, 2, q))
          for r in range(2)]
    for p in ps:
        p.start()
    for p in ps:
        p.join(180)
    results = [q.get() for _ in range(2)]
    for nbuckets, overlapped in results:
        assert nbuckets >= 3  # tiny cap must split the model
        # every bucket except possibly the last (flushed by the callback)
        # launched during backward
        assert overlapped >= nbuckets - 1, (nbuckets, overlapped)


def _overlap_worker(rank, world, q):
    import os
    import torch.distributed as dist
    from amdtrain.parallel import NativeDDP

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29655")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)
    m = torch.nn.Sequential(*[torch.nn.Linear(256, 256) for _ in range(8)])
    ddp = NativeDDP(m, bucket_cap_mb=0.5)  # ~0.26 MB per layer -> ~4 buckets
    x = torch.randn(4, 256)
    ddp(x).sum().backward()
    q.put((len(ddp.reducer.buckets), ddp.reducer.last_overlap_launches))
    dist.destroy_process_group()
