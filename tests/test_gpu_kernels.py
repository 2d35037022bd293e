"""GPU numerics tests: every hand-written gfx950 HIP kernel is compared
against a plain PyTorch fp32 reference of the same op (SURVEY §4 item b)."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _require_ext():
    from amdtrain.ops import functional as OF
    assert OF.ext_available(), (
        "amdtrain._C must be built in-tree on the GPU box")
    return OF


def _nhwc(x):
    return x.contiguous(memory_format=torch.channels_last)


# ---------- normalize_u8 ----------

def test_normalize_u8():
    OF = _require_ext()
    x = torch.randint(0, 256, (4, 3, 33, 35), dtype=torch.uint8, device=DEV)
    y = OF.normalize_u8(x, dtype=torch.float32)
    m = torch.tensor(OF.IMAGENET_MEAN_255, device=DEV).reshape(1, 3, 1, 1)
    s = torch.tensor(OF.IMAGENET_STD_255, device=DEV).reshape(1, 3, 1, 1)
    ref = (x.float() - m) / s
    assert torch.allclose(y, ref, atol=1e-4)
    yb = OF.normalize_u8(x, dtype=torch.bfloat16)
    assert yb.dtype == torch.bfloat16
    assert torch.allclose(yb.float(), ref, atol=0.05, rtol=0.02)
    yh = OF.normalize_u8(x, dtype=torch.float16)  # O2-fp16 prefetcher path
    assert yh.dtype == torch.float16
    assert torch.allclose(yh.float(), ref, atol=0.01, rtol=0.005)


# ---------- cross entropy ----------

@pytest.mark.parametrize("C", [10, 100, 1000])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_cross_entropy(dtype, C):
    # C=10 regression: lanes beyond C must not poison the logsumexp
    # reduction with exp(-inf - (-inf)) = NaN
    OF = _require_ext()
    torch.manual_seed(0)
    logits = torch.randn(64, C, device=DEV, dtype=dtype,
                         requires_grad=True)
    target = torch.randint(0, C, (64,), device=DEV)
    loss = OF.cross_entropy(logits, target)
    ref_logits = logits.detach().float().requires_grad_(True)
    ref = F.cross_entropy(ref_logits, target)
    atol = 1e-5 if dtype == torch.float32 else 3e-3
    assert torch.allclose(loss.float(), ref, atol=atol)
    loss.backward()
    ref.backward()
    assert torch.allclose(logits.grad.float(), ref_logits.grad,
                          atol=(1e-6 if dtype == torch.float32 else 1e-4))


# ---------- topk ranks / accuracy ----------

def test_topk_ranks_gpu():
    OF = _require_ext()
    torch.manual_seed(1)
    logits = torch.randn(128, 1000, device=DEV)
    target = torch.randint(0, 1000, (128,), device=DEV)
    ranks = OF.topk_ranks(logits, target)
    _, pred = logits.topk(1000, 1, True, True)
    for i in range(0, 128, 7):
        want = (pred[i] == target[i]).nonzero()[0, 0].item()
        assert ranks[i].item() == want


def test_accuracy_gpu_matches_cpu():
    from amdtrain.utils import accuracy
    _require_ext()
    torch.manual_seed(2)
    logits = torch.randn(256, 1000, device=DEV)
    target = torch.randint(0, 1000, (256,), device=DEV)
    a1, a5 = accuracy(logits, target, topk=(1, 5))
    c1, c5 = accuracy(logits.cpu(), target.cpu(), topk=(1, 5))
    assert torch.allclose(a1.cpu(), c1, atol=1e-4)
    assert torch.allclose(a5.cpu(), c5, atol=1e-4)


# ---------- fused SGD ----------

def test_multi_tensor_sgd():
    from amdtrain.ops import FusedSGD
    _require_ext()
    torch.manual_seed(0)
    shapes = [(64,), (128, 64), (3, 3, 64, 64), (1000,), (7,)]
    ps = [torch.nn.Parameter(torch.randn(s, device=DEV)) for s in shapes]
    qs = [torch.nn.Parameter(p.detach().clone()) for p in ps]
    ours = FusedSGD(ps, lr=0.1, momentum=0.9, weight_decay=1e-4)
    ref = torch.optim.SGD(qs, lr=0.1, momentum=0.9, weight_decay=1e-4)
    for step in range(4):
        torch.manual_seed(10 + step)
        for p, q in zip(ps, qs):
            g = torch.randn_like(p)
            p.grad = g.clone()
            q.grad = g.clone()
        ours.step()
        ref.step()
    for p, q in zip(ps, qs):
        assert torch.allclose(p, q, atol=1e-6), (p - q).abs().max()


def test_multi_tensor_sgd_zero_grad_fold():
    from amdtrain.ops import FusedSGD
    _require_ext()
    p = torch.nn.Parameter(torch.randn(1000, device=DEV))
    opt = FusedSGD([p], lr=0.1, momentum=0.9)
    p.grad = torch.randn_like(p)
    opt.step(zero_grad=True)
    assert torch.all(p.grad == 0)


# ---------- scale/check + cast ----------

def test_multi_tensor_scale_check_gpu():
    OF = _require_ext()
    ts = [torch.randn(1000, device=DEV), torch.randn(37, device=DEV)]
    refs = [t.clone() for t in ts]
    found = torch.zeros(1, device=DEV)
    OF.multi_tensor_scale_check(ts, 0.25, found)
    for t, r in zip(ts, refs):
        assert torch.allclose(t, r * 0.25, atol=1e-7)
    assert found.item() == 0.0
    bad = [torch.tensor([1.0, float("inf")], device=DEV)]
    OF.multi_tensor_scale_check(bad, 1.0, found)
    torch.cuda.synchronize()
    assert found.item() == 1.0


def test_multi_tensor_cast_gpu():
    OF = _require_ext()
    src = [torch.randn(999, device=DEV), torch.randn(64, device=DEV)]
    dst = [torch.empty(999, device=DEV, dtype=torch.bfloat16),
           torch.empty(64, device=DEV, dtype=torch.bfloat16)]
    OF.multi_tensor_cast(src, dst)
    torch.cuda.synchronize()
    for s, d in zip(src, dst):
        assert torch.allclose(s.bfloat16().float(), d.float())
    # mixed-direction group (bf16 -> fp32)
    back = [torch.empty(999, device=DEV), torch.empty(64, device=DEV)]
    OF.multi_tensor_cast(dst, back)
    for d, b in zip(dst, back):
        assert torch.allclose(d.float(), b)


# ---------- batch norm ----------

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("C", [64, 256, 2048])
def test_bn_train_forward(dtype, C):
    from amdtrain.models.resnet import FusedBatchNorm2d
    _require_ext()
    torch.manual_seed(0)
    bn = FusedBatchNorm2d(C).to(DEV)
    ref = torch.nn.BatchNorm2d(C).to(DEV)
    ref.load_state_dict(bn.state_dict())
    x = torch.randn(4, C, 14, 14, device=DEV, dtype=dtype) * 2 + 0.5
    y = bn.forward_relu(_nhwc(x))
    yr = F.relu(ref(x.float()))
    tol = 1e-4 if dtype == torch.float32 else 0.05
    assert torch.allclose(y.float(), yr, atol=tol, rtol=0.02)
    # running stats updated identically (fp32 both)
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-4)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-3)
    assert bn.num_batches_tracked.item() == ref.num_batches_tracked.item()


def test_bn_eval_forward():
    from amdtrain.models.resnet import FusedBatchNorm2d
    _require_ext()
    torch.manual_seed(0)
    C = 128
    bn = FusedBatchNorm2d(C).to(DEV)
    ref = torch.nn.BatchNorm2d(C).to(DEV)
    # random running stats
    bn.running_mean.normal_()
    bn.running_var.uniform_(0.5, 2.0)
    ref.load_state_dict(bn.state_dict())
    bn.eval()
    ref.eval()
    x = torch.randn(2, C, 8, 8, device=DEV)
    with torch.no_grad():
        y = bn(_nhwc(x))
        yr = ref(x)
    assert torch.allclose(y.float(), yr, atol=1e-4, rtol=1e-4)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_bn_add_relu_backward(dtype):
    from amdtrain.models.resnet import FusedBatchNorm2d
    from amdtrain.ops import fused as OFU
    _require_ext()
    torch.manual_seed(3)
    C = 64
    bn = FusedBatchNorm2d(C).to(DEV)
    ref = torch.nn.BatchNorm2d(C).to(DEV)
    ref.load_state_dict(bn.state_dict())

    x = torch.randn(4, C, 7, 7, device=DEV, dtype=dtype)
    z = torch.randn_like(x)
    xg = _nhwc(x).requires_grad_(True)
    zg = _nhwc(z).requires_grad_(True)
    y = OFU.bn_add_relu(xg, bn, zg)
    loss = (y.float() ** 2).sum()
    loss.backward()

    xr = x.detach().float().requires_grad_(True)
    zr = z.detach().float().requires_grad_(True)
    yr = F.relu(ref(xr) + zr)
    (yr ** 2).sum().backward()

    tol = 2e-3 if dtype == torch.float32 else 0.2
    rtol = 1e-3 if dtype == torch.float32 else 0.05
    assert torch.allclose(y.float(), yr, atol=tol, rtol=rtol)
    assert torch.allclose(xg.grad.float(), xr.grad, atol=tol * 5, rtol=rtol)
    assert torch.allclose(zg.grad.float(), zr.grad, atol=tol * 5, rtol=rtol)
    assert torch.allclose(bn.weight.grad, ref.weight.grad, atol=tol * 10,
                          rtol=rtol)
    assert torch.allclose(bn.bias.grad, ref.bias.grad, atol=tol * 10,
                          rtol=rtol)


# ---------- pooling ----------

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_maxpool(dtype):
    from amdtrain.ops import fused as OFU
    _require_ext()
    torch.manual_seed(4)
    x = torch.randn(2, 64, 112, 112, device=DEV, dtype=dtype)
    xg = _nhwc(x).requires_grad_(True)
    y = OFU.max_pool_3x3_s2(xg)
    g = torch.randn_like(y)
    y.backward(g)

    # reference in the SAME dtype: argmax tie-breaks differ between a bf16
    # and an fp32 view of the data, which reroutes single-element gradients
    xr = x.detach().clone().requires_grad_(True)
    yr = F.max_pool2d(xr, 3, 2, 1)
    yr.backward(g)
    assert y.shape == yr.shape
    assert torch.allclose(y.float(), yr.float(), atol=1e-6)
    assert torch.allclose(xg.grad.float(), xr.grad.float(), atol=1e-6)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_global_avg_pool(dtype):
    from amdtrain.ops import fused as OFU
    _require_ext()
    torch.manual_seed(5)
    x = torch.randn(3, 2048, 7, 7, device=DEV, dtype=dtype)
    xg = _nhwc(x).requires_grad_(True)
    y = OFU.global_avg_pool(xg)
    (y.float() ** 2).sum().backward()

    xr = x.detach().float().requires_grad_(True)
    yr = F.adaptive_avg_pool2d(xr, (1, 1))
    (yr ** 2).sum().backward()
    assert torch.allclose(y.float(), yr, atol=0.01 if dtype != torch.float32 else 1e-5)
    assert torch.allclose(xg.grad.float(), xr.grad,
                          atol=0.05 if dtype != torch.float32 else 1e-4)


# ---------- whole model ----------

def test_resnet18_forward_matches_cpu():
    from amdtrain.models import build_model
    _require_ext()
    torch.manual_seed(6)
    m = build_model("resnet18", num_classes=10)
    m_gpu = build_model("resnet18", num_classes=10)
    m_gpu.load_state_dict(m.state_dict())
    m_gpu = m_gpu.to(DEV).to(memory_format=torch.channels_last)
    m.eval()
    m_gpu.eval()
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        y_cpu = m(x)
        y_gpu = m_gpu(_nhwc(x.to(DEV)))
    assert torch.allclose(y_gpu.cpu(), y_cpu, atol=5e-3, rtol=1e-3), \
        (y_gpu.cpu() - y_cpu).abs().max()


def test_train_step_resnet50_bf16():
    """One full training step of the flagship config on GPU: custom BN/pool/
    CE/SGD kernels + autocast bf16 conv."""
    from amdtrain.models import build_model
    from amdtrain.ops import CrossEntropyLoss, FusedSGD
    from amdtrain.ops import functional as OF
    _require_ext()
    torch.manual_seed(7)
    m = build_model("resnet50").to(DEV).to(memory_format=torch.channels_last)
    crit = CrossEntropyLoss()
    opt = FusedSGD(m.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-4)
    imgs = torch.randint(0, 256, (8, 3, 224, 224), dtype=torch.uint8,
                         device=DEV)
    x = OF.normalize_u8(imgs, dtype=torch.bfloat16)
    t = torch.randint(0, 1000, (8,), device=DEV)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = m(x)
    loss = crit(out, t)
    opt.zero_grad(set_to_none=False)
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()
    assert all(p.grad is not None for p in m.parameters())


@pytest.mark.parametrize("path", ["conv1x1", "conv3x3"])
def test_conv_bn_fused_stats_parity(path):
    """conv-epilogue BN statistics == BN's own reduce pass (running stats,
    mean, output all agree between the fused and unfused paths)."""
    import os
    from amdtrain.models.resnet import FusedBatchNorm2d
    from amdtrain.ops.conv import conv1x1_mfma, conv3x3_mfma
    _require_ext()
    torch.manual_seed(0)
    cin, cout, hw = 64, 128, 28
    x = torch.randn(4, cin, hw, hw, device=DEV) \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(cout, cin, 1, 1, device=DEV) * cin ** -0.5) \
        if path == "conv1x1" else \
        (torch.randn(cout, cin, 3, 3, device=DEV) * (9 * cin) ** -0.5)
    fn = conv1x1_mfma if path == "conv1x1" else conv3x3_mfma

    bn_a = FusedBatchNorm2d(cout).to(DEV)
    bn_b = FusedBatchNorm2d(cout).to(DEV)
    bn_b.load_state_dict(bn_a.state_dict())

    with torch.autocast("cuda", dtype=torch.bfloat16):
        y_fused_conv = fn(x, w, 1)
    assert getattr(y_fused_conv, "_amdtrain_bn_stats", None) is not None
    y_fused = bn_a.forward_relu(y_fused_conv)

    os.environ["AMDTRAIN_FUSE_BNSTATS"] = "0"
    try:
        with torch.autocast("cuda", dtype=torch.bfloat16):
            y_plain_conv = fn(x, w, 1)
        assert getattr(y_plain_conv, "_amdtrain_bn_stats", None) is None
        y_plain = bn_b.forward_relu(y_plain_conv)
    finally:
        os.environ.pop("AMDTRAIN_FUSE_BNSTATS")

    # stats from the fp32 accumulator vs from the rounded bf16 output:
    # small tolerance, but running stats and outputs must agree closely
    assert torch.allclose(bn_a.running_mean, bn_b.running_mean, atol=1e-2,
                          rtol=1e-2)
    assert torch.allclose(bn_a.running_var, bn_b.running_var, atol=1e-2,
                          rtol=1e-2)
    assert torch.allclose(y_fused.float(), y_plain.float(), atol=0.05,
                          rtol=0.05), (y_fused - y_plain).abs().max()


def test_deterministic_wgrad(monkeypatch):
    """AMDTRAIN_DETERMINISTIC=1 (set by --seed) makes conv wgrads bitwise
    reproducible (single-accumulator split-M)."""
    monkeypatch.setenv("AMDTRAIN_DETERMINISTIC", "1")
    from amdtrain import _C
    _require_ext()
    torch.manual_seed(0)
    dY = torch.randn(4096, 64, device=DEV).bfloat16()
    X = torch.randn(4096, 128, device=DEV).bfloat16()
    a = _C.gemm_tn(dY, X, 0)
    b = _C.gemm_tn(dY, X, 0)
    assert torch.equal(a, b)
    x2d = torch.randn(2 * 14 * 14, 64, device=DEV).bfloat16()
    gy2d = torch.randn(2 * 14 * 14, 64, device=DEV).bfloat16()
    a = _C.conv3x3_wgrad(gy2d, x2d, 2, 14, 14, 1)
    b = _C.conv3x3_wgrad(gy2d, x2d, 2, 14, 14, 1)
    assert torch.equal(a, b)


@pytest.mark.parametrize("cfg", [
    (3, 16, 7, 2, 3, 22),    # stem-like (Cin=3, 7x7 s2 p3)
    (32, 48, 5, 1, 2, 17),   # odd 5x5
    (16, 32, 7, 3, 2, 21),   # stride 3
])
def test_conv_generic_dgrad(cfg):
    """Generic-conv input gradient vs fp32 torch."""
    cin, cout, k, s, p, hw = cfg
    from amdtrain.ops.conv import conv_stem_mfma
    torch.manual_seed(0)
    x = torch.randn(2, cin, hw, hw, device=DEV) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    w = torch.randn(cout, cin, k, k, device=DEV) \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = conv_stem_mfma(x, w, s, p)
    gy = torch.randn_like(y)
    y.backward(gy)
    xr = x.detach().clone().float().requires_grad_(True)
    wr = w.detach().clone().float().requires_grad_(True)
    yr = F.conv2d(xr, wr, stride=s, padding=p)
    yr.backward(gy.float())
    assert torch.allclose(x.grad.float(), xr.grad, atol=0.5, rtol=0.05), \
        (x.grad.float() - xr.grad).abs().max().item()
    assert torch.allclose(w.grad.float(), wr.grad, atol=1.0, rtol=0.05), \
        (w.grad.float() - wr.grad).abs().max().item()
