"""GPU numerics for the v2 TN wgrad core (ops/csrc/wgrad.hip): the
ds_read_b64_tr_b16 layout probe, then tn2_wgrad vs plain fp32 PyTorch
references for all three gather modes and every wave-grid config
(SURVEY §4 item b; reference wgrad sites distributed.py:268)."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _ext():
    from amdtrain.ops import functional as OF
    assert OF.ext_available()
    return OF.require_ext()


def test_tr16_probe_layout():
    """Hardware transpose read semantics: lane l elem j must read
    lds[(l&15) + j*16 + (l>>4)*64] (guide m156 mapping) — the whole v2
    wgrad LDS layout is derived from this."""
    e = _ext()
    inp = torch.arange(256, dtype=torch.int16, device=DEV)
    out = e.tr16_probe(inp).cpu()  # [64 lanes, 4 elems]
    for lane in range(64):
        for j in range(4):
            expect = (lane & 15) + j * 16 + (lane >> 4) * 64
            assert out[lane, j].item() == expect, (lane, j, out[lane, j])


@pytest.mark.parametrize("shape", [
    (4099, 64, 64),     # (2,2) config, ragged M tail
    (4096, 64, 256),    # (1,4) skinny-N config
    (4096, 256, 64),    # (4,1) skinny-K config
    (2048, 256, 512),   # (2,2) multi-tile
    (1000, 1000, 2048), # ragged N (FC-shaped)
])
def test_tn2_plain_vs_ref(shape):
    M, N, K = shape
    e = _ext()
    torch.manual_seed(0)
    dY = torch.randn(M, N, device=DEV).bfloat16()
    X = torch.randn(M, K, device=DEV).bfloat16()
    dw = e.tn2_wgrad(dY, X)
    ref = dY.float().t() @ X.float()
    assert dw.shape == (N, K) and dw.dtype == torch.float32
    tol = 0.05 * (M ** 0.5)
    assert torch.allclose(dw, ref, atol=tol, rtol=0.02), \
        (dw - ref).abs().max().item()


def test_tn2_deterministic():
    """No atomics + fixed split -> bitwise identical across runs."""
    e = _ext()
    torch.manual_seed(1)
    dY = torch.randn(3000, 128, device=DEV).bfloat16()
    X = torch.randn(3000, 256, device=DEV).bfloat16()
    a = e.tn2_wgrad(dY, X)
    b = e.tn2_wgrad(dY, X)
    assert torch.equal(a, b)


@pytest.mark.parametrize("chans", [(128, 256), (512, 1024)])
def test_tn2_strided_vs_ref(chans):
    """gmode 1: strided-1x1 wgrad (downsample convs)."""
    cout, cin = chans
    e = _ext()
    torch.manual_seed(0)
    n, h, w, s = 4, 14, 14, 2
    ho, wo = h // s, w // s
    x = torch.randn(n, cin, h, w, device=DEV).bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    gy = torch.randn(n, cout, ho, wo, device=DEV).bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    x2d = x.permute(0, 2, 3, 1).reshape(-1, cin)
    gy2d = gy.permute(0, 2, 3, 1).reshape(-1, cout)
    dw = e.tn2_wgrad(gy2d, x2d, 1, n, h, w, s, 1)
    wref = torch.nn.grad.conv2d_weight(
        x.float(), (cout, cin, 1, 1), gy.float(), stride=s)
    assert torch.allclose(dw.view(cout, cin),
                          wref.view(cout, cin), atol=0.5, rtol=0.02), \
        (dw.view(cout, cin) - wref.view(cout, cin)).abs().max().item()


@pytest.mark.parametrize("cfg", [
    (64, 64, 28, 1),    # (1,4) config (N=64, K9=576)
    (128, 128, 14, 1),  # (2,2)
    (128, 128, 15, 2),  # (2,2) stride 2, odd input
    (256, 256, 7, 1),   # (2,2) wide K9
])
def test_tn2_conv3x3_vs_ref(cfg):
    """gmode 2: all 9 taps in one launch vs torch conv2d_weight."""
    cout, cin, hw, s = cfg
    e = _ext()
    torch.manual_seed(0)
    n = 4
    x = torch.randn(n, cin, hw, hw, device=DEV).bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    ho = (hw + 2 - 3) // s + 1
    gy = torch.randn(n, cout, ho, ho, device=DEV).bfloat16() \
        .contiguous(memory_format=torch.channels_last)
    x2d = x.permute(0, 2, 3, 1).reshape(-1, cin)
    gy2d = gy.permute(0, 2, 3, 1).reshape(-1, cout)
    dw = e.tn2_wgrad(gy2d, x2d, 9, n, hw, hw, s, 2)  # [Cout, 9*Cin]
    wref = torch.nn.grad.conv2d_weight(
        x.float(), (cout, cin, 3, 3), gy.float(), stride=s, padding=1)
    # [Cout, 9*Cin] -> [Cout, Cin, 3, 3] (tap-major columns)
    dw4 = dw.view(cout, 3, 3, cin).permute(0, 3, 1, 2)
    assert torch.allclose(dw4, wref, atol=0.5, rtol=0.02), \
        (dw4 - wref).abs().max().item()


def test_tn2_matches_old_kernels():
    """v2 against the round-1 kernels on a ResNet-50 shape of each family."""
    e = _ext()
    torch.manual_seed(2)
    # plain 1x1
    dY = torch.randn(6272, 256, device=DEV).bfloat16()
    X = torch.randn(6272, 64, device=DEV).bfloat16()
    new = e.tn2_wgrad(dY, X)
    old = e.gemm_tn(dY, X, 0)
    assert torch.allclose(new, old, atol=0.5, rtol=0.01)
    # conv3x3
    n, cin, cout, hw = 4, 128, 128, 28
    x2d = torch.randn(n * hw * hw, cin, device=DEV).bfloat16()
    gy2d = torch.randn(n * hw * hw, cout, device=DEV).bfloat16()
    new = e.tn2_wgrad(gy2d, x2d, 9, n, hw, hw, 1, 2)
    old = e.conv3x3_wgrad(gy2d, x2d, n, hw, hw, 1)
    assert torch.allclose(new, old, atol=0.5, rtol=0.01), \
        (new - old).abs().max().item()
