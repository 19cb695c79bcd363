"""Conv2dTF TF-SAME padding semantics + coords_grid."""
import torch
import torch.nn.functional as F

from raft_amd.models.layers import Conv2dTF, coords_grid


def _ref_same_conv(x, w, b, stride, k):
    """Explicit TF SAME: pad total = (ceil(in/s)-1)*s + k - in, beg=total//2."""
    import math
    ih, iw = x.shape[-2:]
    th = max((math.ceil(ih / stride) - 1) * stride + k - ih, 0)
    tw = max((math.ceil(iw / stride) - 1) * stride + k - iw, 0)
    x = F.pad(x, (tw // 2, tw - tw // 2, th // 2, th - th // 2))
    return F.conv2d(x, w, b, stride)


def test_stride2_k7_asymmetric():
    conv = Conv2dTF(3, 8, 7, stride=2)
    x = torch.randn(2, 3, 64, 96)
    out = conv(x)
    ref = _ref_same_conv(x, conv.weight, conv.bias, 2, 7)
    assert out.shape == (2, 8, 32, 48)
    assert torch.allclose(out, ref, atol=1e-6)


def test_stride2_k3():
    conv = Conv2dTF(4, 4, 3, stride=2)
    x = torch.randn(1, 4, 54, 128)
    out = conv(x)
    ref = _ref_same_conv(x, conv.weight, conv.bias, 2, 3)
    assert out.shape == (1, 4, 27, 64)
    assert torch.allclose(out, ref, atol=1e-6)


def test_stride1_matches_plain_same_padding():
    conv = Conv2dTF(4, 4, 3)
    x = torch.randn(1, 4, 17, 23)
    ref = F.conv2d(x, conv.weight, conv.bias, 1, 1)
    assert torch.allclose(conv(x), ref, atol=1e-6)


def test_rect_kernel_1x5():
    conv = Conv2dTF(2, 2, (1, 5))
    x = torch.randn(1, 2, 6, 10)
    out = conv(x)
    assert out.shape == (1, 2, 6, 10)
    ref = F.conv2d(x, conv.weight, conv.bias, 1, (0, 2))
    assert torch.allclose(out, ref, atol=1e-6)


def test_coords_grid():
    g = coords_grid(2, 3, 4)
    assert g.shape == (2, 2, 3, 4)
    # channel 0 = x varies along W; channel 1 = y along H
    assert torch.equal(g[0, 0, 0], torch.tensor([0.0, 1.0, 2.0, 3.0]))
    assert torch.equal(g[0, 1, :, 0], torch.tensor([0.0, 1.0, 2.0]))
    assert torch.equal(g[0], g[1])
