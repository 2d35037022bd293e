"""GPU tests for the implicit-GEMM 3x3 conv kernels vs F.conv2d (fp32 ref,
random asymmetric data — transpose/rotation-detecting)."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _ext():
    from amdtrain.ops import functional as OF
    assert OF.ext_available()
    from amdtrain import _C
    return _C


@pytest.mark.parametrize("stride", [1, 2])
@pytest.mark.parametrize("cin,cout,hw", [
    (64, 64, 56),     # layer1 middle conv
    (128, 128, 29),   # odd spatial size
    (256, 256, 14),
])
def test_conv3x3_fwd_bwd(stride, cin, cout, hw):
    from amdtrain.ops.conv import conv3x3_mfma
    _ext()
    torch.manual_seed(0)
    x = torch.randn(2, cin, hw, hw, device=DEV) \
        .contiguous(memory_format=torch.channels_last)
    w = torch.randn(cout, cin, 3, 3, device=DEV) * ((9 * cin) ** -0.5)

    xg = x.clone().requires_grad_(True)
    wg = w.clone().requires_grad_(True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = conv3x3_mfma(xg, wg, stride)
    g = torch.randn_like(y.float())
    y.float().backward(g)

    xr = x.clone().float().requires_grad_(True)
    wr = w.clone().float().requires_grad_(True)
    yr = F.conv2d(xr, wr, stride=stride, padding=1)
    yr.backward(g)

    assert y.shape == yr.shape, (y.shape, yr.shape)
    assert torch.allclose(y.float(), yr, atol=0.3, rtol=0.05), \
        (y.float() - yr).abs().max()
    assert torch.allclose(xg.grad.float(), xr.grad, atol=0.5, rtol=0.1), \
        (xg.grad.float() - xr.grad).abs().max()
    assert torch.allclose(wg.grad.float(), wr.grad, atol=2.0, rtol=0.1), \
        (wg.grad.float() - wr.grad).abs().max()


def test_resnet50_all_custom_convs_step():
    """Flagship step with BOTH custom conv paths on (stem is the
    conv_generic implicit-GEMM kernel, conv_stem.hip)."""
    from amdtrain.models import build_model
    from amdtrain.ops import CrossEntropyLoss, FusedSGD
    _ext()
    torch.manual_seed(1)
    m = build_model("resnet50").to(DEV).to(memory_format=torch.channels_last)
    opt = FusedSGD(m.parameters(), lr=0.01, momentum=0.9)
    x = torch.randn(4, 3, 224, 224, device=DEV) \
        .contiguous(memory_format=torch.channels_last)
    t = torch.randint(0, 1000, (4,), device=DEV)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = m(x)
    loss = CrossEntropyLoss()(out, t)
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()


def test_resnet18_output_parity_custom_vs_miopen(monkeypatch):
    """Same weights, same input: custom conv stack vs the torch-ROCm
    fallback stack agree."""
    from amdtrain.models import build_model
    _ext()
    torch.manual_seed(2)
    m = build_model("resnet18", num_classes=100).to(DEV) \
        .to(memory_format=torch.channels_last)
    m.eval()
    x = torch.randn(2, 3, 96, 96, device=DEV) \
        .contiguous(memory_format=torch.channels_last)
    with torch.no_grad(), torch.autocast("cuda", dtype=torch.bfloat16):
        y_custom = m(x)
    monkeypatch.setenv("AMDTRAIN_CONV1X1", "miopen")
    monkeypatch.setenv("AMDTRAIN_CONV3X3", "miopen")
    with torch.no_grad(), torch.autocast("cuda", dtype=torch.bfloat16):
        y_miopen = m(x)
    assert torch.allclose(y_custom.float(), y_miopen.float(), atol=0.5,
                          rtol=0.05), (y_custom - y_miopen).abs().max()


@pytest.mark.parametrize("cfg", [
    (3, 64, 7, 2, 3, 112),   # the ResNet stem
    (32, 64, 5, 1, 2, 17),   # odd generic config
])
def test_conv_generic_stem(cfg):
    from amdtrain.ops.conv import conv_stem_mfma
    _ext()
    cin, cout, k, stride, pad, hw = cfg
    torch.manual_seed(5)
    x = torch.randn(2, cin, hw, hw, device=DEV) \
        .contiguous(memory_format=torch.channels_last)
    w = torch.randn(cout, cin, k, k, device=DEV) * ((k * k * cin) ** -0.5)
    wg = w.clone().requires_grad_(True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = conv_stem_mfma(x, wg, stride, pad)
    g = torch.randn_like(y.float())
    y.float().backward(g)

    wr = w.clone().float().requires_grad_(True)
    yr = F.conv2d(x.float(), wr, stride=stride, padding=pad)
    yr.backward(g)
    assert y.shape == yr.shape
    assert torch.allclose(y.float(), yr, atol=0.3, rtol=0.05), \
        (y.float() - yr).abs().max()
    assert torch.allclose(wg.grad.float(), wr.grad, atol=2.0, rtol=0.1), \
        (wg.grad.float() - wr.grad).abs().max()


@pytest.mark.parametrize("cfg", [(128, 32, 14, 1), (64, 8, 14, 2),
                                 (256, 32, 7, 1)])
def test_conv3x3_grouped_vs_torch(cfg):
    """Grouped 3x3 (ResNeXt) through the block-diagonalized dense path vs
    fp32 torch grouped conv: fwd, input grad, weight grad."""
    width, groups, hw, s = cfg
    from amdtrain.ops.conv import conv3x3_grouped_mfma
    torch.manual_seed(0)
    sg = width // groups
    x = torch.randn(4, width, hw, hw, device="cuda") \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    w = torch.randn(width, sg, 3, 3, device="cuda") \
        .contiguous(memory_format=torch.channels_last).requires_grad_(True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = conv3x3_grouped_mfma(x, w, s, groups)
    gy = torch.randn_like(y)
    y.backward(gy)
    xr = x.detach().clone().float().requires_grad_(True)
    wr = w.detach().clone().float().requires_grad_(True)
    yr = F.conv2d(xr, wr, stride=s, padding=1, groups=groups)
    yr.backward(gy.float())
    assert torch.allclose(y.float(), yr, atol=0.5, rtol=0.05), \
        (y.float() - yr).abs().max().item()
    assert torch.allclose(x.grad.float(), xr.grad, atol=0.5, rtol=0.05), \
        (x.grad.float() - xr.grad).abs().max().item()
    assert torch.allclose(w.grad.float(), wr.grad, atol=1.0, rtol=0.05), \
        (w.grad.float() - wr.grad).abs().max().item()
