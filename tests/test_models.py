"""Model-family tests: shapes, torchvision-matching parameter counts and
state_dict naming (checkpoint interop with the reference)."""

import pytest
import torch

from amdtrain.models import build_model, model_names

# torchvision reference parameter counts for ImageNet-1k heads
PARAM_COUNTS = {
    "resnet18": 11_689_512,
    "resnet34": 21_797_672,
    "resnet50": 25_557_032,
    "resnet101": 44_549_160,
    "resnet152": 60_192_808,
    "resnext50_32x4d": 25_028_904,
    "wide_resnet50_2": 68_883_240,
}


def test_registry():
    assert "resnet18" in model_names()
    assert "resnet50" in model_names()
    with pytest.raises(ValueError):
        build_model("nope")


@pytest.mark.parametrize("arch", sorted(PARAM_COUNTS))
def test_param_counts(arch):
    m = build_model(arch)
    assert sum(p.numel() for p in m.parameters()) == PARAM_COUNTS[arch]


@pytest.mark.parametrize("arch", ["resnet18", "resnet50"])
def test_forward_backward(arch):
    m = build_model(arch, num_classes=10)
    x = torch.randn(2, 3, 224, 224)
    y = m(x)
    assert y.shape == (2, 10)
    y.sum().backward()
    assert all(p.grad is not None for p in m.parameters())


def test_state_dict_keys_torchvision_compatible():
    m = build_model("resnet50")
    keys = set(m.state_dict().keys())
    # spot-check the torchvision naming contract
    for k in ["conv1.weight", "bn1.weight", "bn1.running_mean",
              "layer1.0.conv1.weight", "layer1.0.downsample.0.weight",
              "layer1.0.downsample.1.running_var", "layer4.2.bn3.bias",
              "fc.weight", "fc.bias"]:
        assert k in keys, k


def test_eval_mode_stats_frozen():
    m = build_model("resnet18", num_classes=10)
    m.eval()
    rm = m.bn1.running_mean.clone()
    with torch.no_grad():
        m(torch.randn(2, 3, 64, 64))
    assert torch.equal(rm, m.bn1.running_mean)


def test_train_mode_updates_stats():
    m = build_model("resnet18", num_classes=10)
    m.train()
    rm = m.bn1.running_mean.clone()
    nbt = m.bn1.num_batches_tracked.clone()
    m(torch.randn(2, 3, 64, 64))
    assert not torch.equal(rm, m.bn1.running_mean)
    assert m.bn1.num_batches_tracked == nbt + 1


def test_all_convs_are_amd_dispatch():
    from amdtrain.ops.conv import AmdConv2d
    m = build_model("resnet50")
    convs = [mod for mod in m.modules() if isinstance(mod, torch.nn.Conv2d)]
    assert len(convs) == 53
    assert all(isinstance(c, AmdConv2d) for c in convs), \
        "every conv (incl. the stem) must go through the dispatching Conv2d"
