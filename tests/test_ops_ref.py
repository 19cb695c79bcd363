"""Golden-path semantics of the pure-PyTorch reference ops.

These pin the numerics the HIP kernels must reproduce: edge-clamp bilinear
with trunc-toward-zero corners and clamped-corner weights
(networks/utils.py:39-99), the [::-1] window order (model_utils.py:237),
TF-VALID pooling, the x0.25-scaled mask softmax axis and patch order of the
convex upsample (networks/RAFT.py:119-134).
"""
import math

import torch

from raft_amd.ops import torch_ref as R


def test_corr_volume_matches_naive():
    B, C, H, W = 2, 16, 6, 8
    f1 = torch.randn(B, C, H, W)
    f2 = torch.randn(B, C, H, W)
    corr = R.corr_volume(f1, f2)
    assert corr.shape == (B, H * W, H, W)
    # naive double loop on one batch element
    b, q, y2, x2 = 1, 13, 3, 5
    expect = (f1[b, :, q // W, q % W] * f2[b, :, y2, x2]).sum() / math.sqrt(C)
    assert torch.allclose(corr[b, q, y2, x2], expect, atol=1e-5)


def test_pyramid_pool_floor_division():
    corr = torch.randn(1, 4, 54, 128)
    pyr = R.corr_pyramid_pool(corr, 4)
    shapes = [tuple(p.shape[-2:]) for p in pyr]
    # TF VALID: 54x128 -> 27x64 -> 13x32 -> 6x16 (floor)
    assert shapes == [(54, 128), (27, 64), (13, 32), (6, 16)]
    assert torch.allclose(pyr[1][0, 0, 0, 0], corr[0, 0, :2, :2].mean())


def test_bilinear_sample_interior():
    corr = torch.arange(12, dtype=torch.float32).reshape(1, 3, 4)
    # value at (x,y) = y*4 + x => bilinear at (1.5, 0.5) = 0.5*... = 3.5
    out = R.bilinear_sample_volume(corr, torch.tensor([[1.5]]),
                                   torch.tensor([[0.5]]))
    assert torch.allclose(out, torch.tensor([[3.5]]))


def test_bilinear_sample_trunc_negative():
    """Negative coords use trunc (toward zero), giving extrapolation weights
    — reference behavior (tf.cast), not floor."""
    corr = torch.tensor([[[1.0, 2.0], [3.0, 4.0]]])  # [1,2,2]
    # x=-0.5: x0=trunc(-0.5)=0, x1=1, qx = 1-(-0.5) = 1.5
    out = R.bilinear_sample_volume(corr, torch.tensor([[-0.5]]),
                                   torch.tensor([[0.0]]))
    # wa=1.5, wc=-0.5 -> 1.5*1 + (-0.5)*2 = 0.5
    assert torch.allclose(out, torch.tensor([[0.5]]))


def test_bilinear_sample_edge_clamp():
    corr = torch.tensor([[[1.0, 2.0], [3.0, 4.0]]])
    # far out of range -> clamped corners equal -> plain corner value
    out = R.bilinear_sample_volume(corr, torch.tensor([[10.0]]),
                                   torch.tensor([[10.0]]))
    assert torch.allclose(out, torch.tensor([[4.0]]))


def test_corr_lookup_window_order():
    """Tap k of level 0 must be offset (dx = k//(2r+1)-r, dy = k%(2r+1)-r)."""
    B, H, W, r = 1, 3, 4, 1
    K = 2 * r + 1
    HW = H * W
    # volume value = y2*W + x2 for every query pixel
    base = torch.arange(W).float().repeat(H, 1) + \
        torch.arange(H).float().unsqueeze(1) * W
    corr = base.reshape(1, 1, H, W).expand(B, HW, H, W).contiguous()
    coords = torch.stack(torch.meshgrid(torch.arange(H, dtype=torch.float32),
                                        torch.arange(W, dtype=torch.float32),
                                        indexing="ij")[::-1], dim=-1)
    coords = coords.unsqueeze(0)  # [1,H,W,2] (x,y)
    out = R.corr_lookup([corr], coords, r)
    assert out.shape == (B, K * K, H, W)
    y, x = 1, 2
    for k in range(K * K):
        dx = k // K - r
        dy = k % K - r
        ex = min(max(x + dx, 0), W - 1)
        ey = min(max(y + dy, 0), H - 1)
        # interior taps: exact integer coords -> exact values
        assert abs(float(out[0, k, y, x]) - (ey * W + ex)) < 1e-5, k


def test_corr_lookup_level_scaling():
    """Level i centroid is coords / 2^i."""
    B, H, W = 1, 2, 4
    HW = H * W
    l0 = torch.zeros(B, HW, H, W)
    l1 = torch.arange(B * HW * 1 * 2, dtype=torch.float32).reshape(B, HW, 1, 2)
    coords = torch.full((B, H, W, 2), 2.0)
    out = R.corr_lookup([l0, l1], coords, radius=0)
    # level-1 centroid = (1,1) clamped into [0..1]x[0..0] -> tap at (1,0)
    q = 0
    assert float(out[0, 1, 0, 0]) == float(l1[0, q, 0, 1])


def test_gru_gates():
    h = torch.randn(2, 8, 4, 4)
    z = torch.randn_like(h)
    q = torch.randn_like(h)
    out = R.gru_gates(h, z, q)
    zs = torch.sigmoid(z)
    expect = (1 - zs) * h + zs * torch.tanh(q)
    assert torch.allclose(out, expect, atol=1e-6)


def test_convex_upsample_delta_mask():
    """A mask concentrated on tap k makes output pixel = 8*flow of the
    k-th 3x3 neighbor (zero-padded SAME)."""
    B, H, W = 1, 3, 4
    flow = torch.randn(B, 2, H, W)
    k = 5  # k=5 -> ky=1, kx=2 (row-major 3x3) -> neighbor (y, x+1)
    mask = torch.full((B, 9, 8, 8, H, W), -50.0)
    mask[:, k] = 50.0
    out = R.convex_upsample(flow, mask.reshape(B, 576, H, W))
    assert out.shape == (B, 2, 8 * H, 8 * W)
    y, x = 1, 1
    nbr = flow[0, :, y, x + 1] * 8
    got = out[0, :, 8 * y + 3, 8 * x + 6]
    assert torch.allclose(got, nbr, atol=1e-4)
    # zero-pad: neighbor outside image contributes 0
    y, x = 0, W - 1
    got_edge = out[0, :, 8 * y, 8 * x]
    mask2 = torch.full((B, 9, 8, 8, H, W), -50.0)
    mask2[:, 8] = 50.0  # (dy=1, dx=1) neighbor of (0, W-1): x+1 out of range
    out2 = R.convex_upsample(flow, mask2.reshape(B, 576, H, W))
    assert torch.allclose(out2[0, :, 0, 8 * x], torch.zeros(2), atol=1e-4)


def test_convex_upsample_mask_channel_factorization():
    """Channel c = k*64 + dy*8 + dx (TF reshape NHWC 576->(9,1,8,8))."""
    B, H, W = 1, 2, 2
    flow = torch.ones(B, 2, H, W)
    mask = torch.full((B, 576, H, W), -50.0)
    k, dy, dx = 4, 3, 7   # k=4 = center tap
    mask[:, k * 64 + dy * 8 + dx] = 50.0
    out = R.convex_upsample(flow, mask)
    # the (dy,dx) sub-pixel of cell (0,0) sees tap 4 (identity neighbor) = 8
    assert abs(float(out[0, 0, dy, dx]) - 8.0) < 1e-4


def test_upflow8_align_corners():
    flow = torch.randn(1, 2, 3, 5)
    up = R.upflow8(flow)
    assert up.shape == (1, 2, 24, 40)
    # align_corners=True preserves the corners exactly (no x8 value scale)
    assert torch.allclose(up[..., 0, 0], flow[..., 0, 0], atol=1e-6)
    assert torch.allclose(up[..., -1, -1], flow[..., -1, -1], atol=1e-6)
