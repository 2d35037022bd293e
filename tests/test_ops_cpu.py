"""CPU reference-path tests for the op layer (the same references the GPU
kernels are verified against in tests/test_gpu_kernels.py)."""

import torch
import torch.nn.functional as F

from amdtrain.ops import CrossEntropyLoss, FusedSGD
from amdtrain.ops import functional as OF


def test_cross_entropy_matches_torch():
    torch.manual_seed(0)
    logits = torch.randn(16, 10, requires_grad=True)
    target = torch.randint(0, 10, (16,))
    crit = CrossEntropyLoss()
    loss = crit(logits, target)
    ref = F.cross_entropy(logits, target)
    assert torch.allclose(loss, ref, atol=1e-6)
    loss.backward()
    g1 = logits.grad.clone()
    logits.grad = None
    ref.backward()
    assert torch.allclose(g1, logits.grad, atol=1e-6)


def test_topk_ranks_matches_topk():
    torch.manual_seed(1)
    logits = torch.randn(32, 50)
    target = torch.randint(0, 50, (32,))
    ranks = OF.topk_ranks(logits, target)
    _, pred = logits.topk(50, 1, True, True)
    for i in range(32):
        want = (pred[i] == target[i]).nonzero()[0, 0].item()
        assert ranks[i].item() == want


def test_normalize_u8():
    x = torch.randint(0, 256, (2, 3, 8, 8), dtype=torch.uint8)
    y = OF.normalize_u8(x, dtype=torch.float32)
    m = torch.tensor(OF.IMAGENET_MEAN_255).reshape(1, 3, 1, 1)
    s = torch.tensor(OF.IMAGENET_STD_255).reshape(1, 3, 1, 1)
    ref = (x.float() - m) / s
    assert torch.allclose(y, ref, atol=1e-5)


def test_multi_tensor_scale_check():
    ts = [torch.ones(4), torch.full((3,), 2.0)]
    found = torch.zeros(1)
    OF.multi_tensor_scale_check(ts, 0.5, found)
    assert torch.allclose(ts[0], torch.full((4,), 0.5))
    assert torch.allclose(ts[1], torch.ones(3))
    assert found.item() == 0.0
    ts = [torch.tensor([1.0, float("inf")])]
    OF.multi_tensor_scale_check(ts, 1.0, found)
    assert found.item() == 1.0


def test_multi_tensor_cast_roundtrip():
    src = [torch.randn(8), torch.randn(3)]
    dst = [torch.empty(8, dtype=torch.bfloat16),
           torch.empty(3, dtype=torch.bfloat16)]
    OF.multi_tensor_cast(src, dst)
    for s, d in zip(src, dst):
        assert torch.allclose(s, d.float(), atol=0.01, rtol=0.01)


def test_fused_sgd_matches_torch_sgd():
    torch.manual_seed(0)
    p1 = torch.nn.Parameter(torch.randn(10))
    p2 = torch.nn.Parameter(torch.randn(5, 5))
    q1 = torch.nn.Parameter(p1.detach().clone())
    q2 = torch.nn.Parameter(p2.detach().clone())
    ours = FusedSGD([p1, p2], lr=0.1, momentum=0.9, weight_decay=1e-4)
    ref = torch.optim.SGD([q1, q2], lr=0.1, momentum=0.9, weight_decay=1e-4)
    for step in range(5):
        torch.manual_seed(10 + step)
        g1, g2 = torch.randn(10), torch.randn(5, 5)
        p1.grad, p2.grad = g1.clone(), g2.clone()
        q1.grad, q2.grad = g1.clone(), g2.clone()
        ours.step()
        ref.step()
        assert torch.allclose(p1, q1, atol=1e-7), step
        assert torch.allclose(p2, q2, atol=1e-7), step


def test_fused_sgd_nesterov():
    p = torch.nn.Parameter(torch.randn(6))
    q = torch.nn.Parameter(p.detach().clone())
    ours = FusedSGD([p], lr=0.05, momentum=0.9, nesterov=True)
    ref = torch.optim.SGD([q], lr=0.05, momentum=0.9, nesterov=True)
    for step in range(4):
        g = torch.randn(6)
        p.grad = g.clone()
        q.grad = g.clone()
        ours.step()
        ref.step()
        assert torch.allclose(p, q, atol=1e-7)
