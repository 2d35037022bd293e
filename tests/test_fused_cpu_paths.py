"""CPU fallback branches of the fused op wrappers (the same semantics the
GPU kernels implement)."""

import torch
import torch.nn.functional as F

from amdtrain.models.resnet import FusedBatchNorm2d
from amdtrain.ops import fused as OF
from amdtrain.ops.conv import AmdConv2d


def test_bn_add_relu_cpu_matches_manual():
    torch.manual_seed(0)
    bn = FusedBatchNorm2d(8)
    ref = torch.nn.BatchNorm2d(8)
    ref.load_state_dict(bn.state_dict())
    x = torch.randn(3, 8, 5, 5)
    z = torch.randn(3, 8, 5, 5)
    y = OF.bn_add_relu(x, bn, z)
    yr = F.relu(ref(x) + z)
    assert torch.allclose(y, yr, atol=1e-6)
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-7)
    assert bn.num_batches_tracked == ref.num_batches_tracked


def test_bn_eval_cpu():
    bn = FusedBatchNorm2d(4)
    bn.running_mean.normal_()
    bn.running_var.uniform_(0.5, 2.0)
    ref = torch.nn.BatchNorm2d(4)
    ref.load_state_dict(bn.state_dict())
    bn.eval()
    ref.eval()
    x = torch.randn(2, 4, 3, 3)
    with torch.no_grad():
        assert torch.allclose(bn(x), ref(x), atol=1e-6)


def test_amdconv2d_cpu_falls_through_to_torch():
    torch.manual_seed(1)
    c = AmdConv2d(8, 16, kernel_size=3, padding=1, bias=False)
    r = torch.nn.Conv2d(8, 16, kernel_size=3, padding=1, bias=False)
    r.load_state_dict(c.state_dict())
    x = torch.randn(2, 8, 6, 6)
    assert torch.allclose(c(x), r(x), atol=1e-6)


def test_amdconv2d_grouped_guard():
    # grouped convs must use the library path (custom kernels assume groups=1)
    c = AmdConv2d(8, 8, kernel_size=3, padding=1, groups=4, bias=False)
    x = torch.randn(1, 8, 5, 5)
    y = c(x)
    assert y.shape == (1, 8, 5, 5)


def test_pool_cpu_paths():
    x = torch.randn(2, 4, 9, 9)
    assert torch.allclose(OF.max_pool_3x3_s2(x),
                          F.max_pool2d(x, 3, 2, 1))
    assert torch.allclose(OF.global_avg_pool(x),
                          F.adaptive_avg_pool2d(x, (1, 1)))


def test_residual_tap_unarmed_passthrough():
    """ResidualGradTap must behave as identity for autograd unless the
    producing BN armed the mailbox (the safety latch that keeps every
    fallback path correct)."""
    import torch
    from amdtrain.ops.conv import GradCell, ResidualGradTap

    cell = GradCell()
    x = torch.randn(4, requires_grad=True)
    (ResidualGradTap.apply(x, cell) * 2).sum().backward()
    assert torch.allclose(x.grad, torch.full((4,), 2.0))
    assert cell.g is None

    cell2 = GradCell()
    cell2.armed = True
    x2 = torch.randn(4, requires_grad=True)
    ResidualGradTap.apply(x2, cell2).sum().backward()
    assert x2.grad is None  # rerouted, not accumulated
    assert torch.allclose(cell2.g, torch.ones(4))


def test_gradcell_survives_amp_cast():
    """GradCell instances must pass through torch.amp.custom_fwd's
    cast_inputs traversal UNTOUCHED (a dict would be deep-copied and
    silently disconnect the mailbox — the round-2 bug this guards)."""
    import torch
    from amdtrain.ops.conv import GradCell

    cell = GradCell()
    out = torch.amp.autocast_mode._cast((cell,), "cuda", torch.bfloat16)
    assert out[0] is cell
