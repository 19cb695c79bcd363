"""Pure-PyTorch golden implementations of the RAFT hot ops.

These mirror the reference TF graph *exactly* (file:line citations below) and
are the numerics oracle for the HIP kernels (tests compare HIP output against
these in fp32).  They also serve as the CPU execution path.

Conventions: NCHW activations; coords are ``[B, H, W, 2]`` float with last
dim (x, y); correlation volume level i is ``[B, H1*W1, H2/2^i, W2/2^i]``.
"""
from __future__ import annotations

import math
from typing import List

import torch
import torch.nn.functional as F


def corr_volume(fmap1: torch.Tensor, fmap2: torch.Tensor) -> torch.Tensor:
    """All-pairs correlation ``C = F1·F2ᵀ / sqrt(c)``.

    Mirrors networks/model_utils.py:199-215 (reshape + matmul + 1/sqrt(c)).
    fmap*: [B, C, H, W] -> returns [B, H*W, H, W] in fp32 (accumulation is
    always fp32 regardless of input dtype — SURVEY.md §7 hard part (e)).
    """
    B, C, H, W = fmap1.shape
    f1 = fmap1.reshape(B, C, H * W).transpose(1, 2).float()  # [B, HW, C]
    f2 = fmap2.reshape(B, C, H * W).float()                  # [B, C, HW]
    corr = torch.matmul(f1, f2) / math.sqrt(C)
    return corr.reshape(B, H * W, H, W)


def corr_pyramid_pool(corr: torch.Tensor, num_levels: int = 4) -> List[torch.Tensor]:
    """2x2/2 average-pool pyramid over the *target* dims, TF VALID semantics
    (floor division) — model_utils.py:215-219.

    corr: [B, H1*W1, H2, W2]; returns [level0, ..., level{num_levels-1}].
    """
    pyramid = [corr]
    for _ in range(num_levels - 1):
        if corr.shape[-2] < 2 or corr.shape[-1] < 2:
            # degenerate level (tiny input): keep the last level so the
            # channel count contract (L*KK) holds; lookup clamps anyway
            pyramid.append(corr)
            continue
        corr = F.avg_pool2d(corr, 2, stride=2)  # default floor, matches VALID
        pyramid.append(corr)
    return pyramid


def bilinear_sample_volume(corr: torch.Tensor, x: torch.Tensor,
                           y: torch.Tensor) -> torch.Tensor:
    """Edge-clamp bilinear gather from [N, H2, W2] at per-sample (x, y).

    Replicates networks/utils.py:39-99 (tf_grid_sample) exactly:
      * corner ints by *truncation toward zero* (tf.cast, utils.py:54-57),
        not floor — differs for negative coords;
      * both corners clamped to the image, weights computed from the
        CLAMPED x1/y1 via qx = x1 - x (utils.py:60-89) — the edge-clamp
        boundary (vs zeros-pad in official PyTorch grid_sample);
      * weights wa=qx*qy, wb=qx*(1-qy), wc=(1-qx)*qy, wd=(1-qx)(1-qy).

    x, y: [N, K] sample coordinates into each of the N slices.
    Returns [N, K].
    """
    N, H2, W2 = corr.shape
    xt = torch.trunc(x)
    yt = torch.trunc(y)
    x0 = torch.clamp(xt.long(), 0, W2 - 1)
    x1 = torch.clamp(xt.long() + 1, 0, W2 - 1)
    y0 = torch.clamp(yt.long(), 0, H2 - 1)
    y1 = torch.clamp(yt.long() + 1, 0, H2 - 1)

    n = torch.arange(N, device=corr.device).unsqueeze(1)  # [N,1]
    Ia = corr[n, y0, x0]
    Ib = corr[n, y1, x0]
    Ic = corr[n, y0, x1]
    Id = corr[n, y1, x1]

    qx = x1.to(x.dtype) - x
    qy = y1.to(y.dtype) - y
    wa = qx * qy
    wb = qx * (1.0 - qy)
    wc = (1.0 - qx) * qy
    wd = (1.0 - qx) * (1.0 - qy)
    return wa * Ia + wb * Ib + wc * Ic + wd * Id


def corr_lookup(pyramid: List[torch.Tensor], coords: torch.Tensor,
                radius: int) -> torch.Tensor:
    """Multi-scale (2r+1)^2-tap window lookup — model_utils.py:224-249.

    Window order: reference builds delta = stack(meshgrid(dy, dx)[::-1]) so
    tap index k ↔ offset (dx = k // (2r+1) - r, dy = k % (2r+1) - r)
    (model_utils.py:235-237): the x offset varies along the *slow* window
    axis. Channel layout of the output: [lvl0 k0..k_{K-1}, lvl1 ..., ...].

    pyramid: list of [B, H1*W1, H2_i, W2_i]; coords: [B, H, W, 2] (x, y).
    Returns [B, L*(2r+1)^2, H, W] (NCHW, ready for the motion encoder).
    """
    B, H, W, _ = coords.shape
    K = 2 * radius + 1
    dev = coords.device
    dt = coords.dtype
    r = float(radius)
    # offsets along tap index k: off_x slow, off_y fast (see docstring)
    off = torch.linspace(-r, r, K, device=dev, dtype=dt)
    off_x = off.repeat_interleave(K)  # [K*K]
    off_y = off.repeat(K)             # [K*K]

    cf = coords.reshape(B * H * W, 2)
    out_levels = []
    for i, corr in enumerate(pyramid):
        Bc, HW, H2, W2 = corr.shape
        c = corr.reshape(B * H * W, H2, W2)
        cx = cf[:, 0:1] / (2 ** i) + off_x.unsqueeze(0)  # [N, K*K]
        cy = cf[:, 1:2] / (2 ** i) + off_y.unsqueeze(0)
        sampled = bilinear_sample_volume(c, cx, cy)      # [N, K*K]
        out_levels.append(sampled.reshape(B, H, W, K * K))
    out = torch.cat(out_levels, dim=-1)                  # [B, H, W, L*K*K]
    return out.permute(0, 3, 1, 2).contiguous()


def gru_gates(h: torch.Tensor, z_act: torch.Tensor,
              q_act: torch.Tensor) -> torch.Tensor:
    """Pointwise GRU state update ``h' = (1-σ(z))·h + σ(z)·tanh(q)``.

    The gate math of model_utils.py:146,154,168 with the conv activations
    already applied by the caller's convs (z_act/q_act are pre-activation).
    """
    z = torch.sigmoid(z_act)
    q = torch.tanh(q_act)
    return (1.0 - z) * h + z * q


def convex_upsample(flow: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
    """8x convex upsample with a learned 9-way mask — networks/RAFT.py:119-134.

    flow: [B, 2, H, W]; mask: [B, 576, H, W] with channel c = k*64 + dy*8 + dx
    (reference reshapes NHWC 576 -> (9, 1, 8, 8), RAFT.py:125).  Softmax over
    the 9 taps; taps are the 3x3 zero-padded neighborhood of 8*flow
    (tf.extract_image_patches SAME, RAFT.py:128).  Output [B, 2, 8H, 8W];
    out(8y+dy, 8x+dx) = Σ_k m[k,dy,dx,y,x] · 8·flow[nbr_k(y,x)].
    """
    B, _, H, W = flow.shape
    m = mask.reshape(B, 9, 8, 8, H, W)
    m = torch.softmax(m, dim=1)
    # F.unfold channel order is (c, ky, kx) with c slowest; view separates it.
    patches = F.unfold(8.0 * flow, 3, padding=1).reshape(B, 2, 9, 1, 1, H, W)
    up = (m.unsqueeze(1) * patches).sum(dim=2)           # [B, 2, 8, 8, H, W]
    up = up.permute(0, 1, 4, 2, 5, 3)                    # [B, 2, H, 8, W, 8]
    return up.reshape(B, 2, 8 * H, 8 * W)


def upflow8(flow: torch.Tensor) -> torch.Tensor:
    """Bilinear x8 upsample, align_corners=True — networks/utils.py:105-111.

    NOTE (deliberate reference quirk): the reference does *not* multiply the
    flow values by 8 on this path (RAFT.py:104-105), unlike official RAFT.
    Callers that want physically-scaled flow pass the result through
    ``8 * upflow8(flow)`` themselves (see RaftConfig.scale_small_upflow).
    """
    B, C, H, W = flow.shape
    return F.interpolate(flow, size=(8 * H, 8 * W), mode="bilinear",
                         align_corners=True)
