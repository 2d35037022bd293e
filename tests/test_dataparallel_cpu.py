"""CPU-testable pieces of the scatter/gather DataParallel engine."""

import torch

from amdtrain.parallel.dataparallel import _Gather


def test_gather_forward_backward_cpu():
    a = torch.randn(3, 4, requires_grad=True)
    b = torch.randn(5, 4, requires_grad=True)
    out = _Gather.apply(torch.device("cpu"), a, b)
    assert out.shape == (8, 4)
    assert torch.equal(out[:3], a)
    g = torch.randn(8, 4)
    out.backward(g)
    assert torch.allclose(a.grad, g[:3])
    assert torch.allclose(b.grad, g[3:])


def test_sgdp_single_device_passthrough_cpu(monkeypatch):
    """With one device the wrapper must behave as a plain module call."""
    import amdtrain.parallel.dataparallel as dp

    class FakeDP(dp.ScatterGatherDataParallel):
        def __init__(self, module):  # bypass cuda device setup
            torch.nn.Module.__init__(self)
            self.module = module
            self.devices = [torch.device("cpu")]
            self.output_device = torch.device("cpu")
            self._replicas = [module]
            self._master_params = list(module.parameters())

    m = torch.nn.Linear(4, 2)
    w = FakeDP(m)
    x = torch.randn(3, 4)
    y = w(x)
    assert torch.allclose(y, m(x))
    assert set(w.state_dict().keys()) == set(m.state_dict().keys())
