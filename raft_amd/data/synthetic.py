"""Synthetic frame-pair generator (benchmarks + training without datasets).

The environment has no network access for Sintel/FlyingThings, so training
(config 3) and the EPE harness run on procedurally generated pairs: a random
textured image warped by a known smooth flow field, giving (im1, im2,
flow_gt) triplets with exact ground truth.
"""
from __future__ import annotations

from typing import Tuple

import torch
import torch.nn.functional as F


def _random_texture(b: int, h: int, w: int, g: torch.Generator) -> torch.Tensor:
    """Multi-octave random texture in [0,1], BGR-like 3 channels."""
    img = torch.zeros(b, 3, h, w)
    for scale in (4, 8, 16, 32):
        noise = torch.rand(b, 3, max(h // scale, 1), max(w // scale, 1),
                           generator=g)
        img += F.interpolate(noise, size=(h, w), mode="bilinear",
                             align_corners=False) / 4.0
    return img.clamp(0, 1)


def random_flow(b: int, h: int, w: int, g: torch.Generator,
                max_mag: float = 16.0) -> torch.Tensor:
    """Smooth random flow field [B,2,H,W] with |f| <~ max_mag.

    Kept very smooth (3x4 control grid) so the backward-warp construction's
    ground truth is consistent to ~0.1*max_mag EPE: the warp identity
    im2(x+f) = im1(x + f - f(x+f)) deviates by |∇f|·|f|, and the coarse grid
    bounds |∇f| ~ max_mag / (min(H,W)/3)."""
    coarse = (torch.rand(b, 2, 3, 4, generator=g) * 2 - 1) * max_mag
    return F.interpolate(coarse, size=(h, w), mode="bilinear",
                         align_corners=True)


def warp(img: torch.Tensor, flow: torch.Tensor) -> torch.Tensor:
    """Backward-warp img by flow (im2(x) = im1(x + flow(x)) inverse map)."""
    b, _, h, w = img.shape
    ys, xs = torch.meshgrid(torch.arange(h, dtype=torch.float32),
                            torch.arange(w, dtype=torch.float32),
                            indexing="ij")
    grid = torch.stack([xs, ys], dim=0)[None] + flow
    gx = 2.0 * grid[:, 0] / max(w - 1, 1) - 1.0
    gy = 2.0 * grid[:, 1] / max(h - 1, 1) - 1.0
    return F.grid_sample(img, torch.stack([gx, gy], dim=-1),
                         mode="bilinear", padding_mode="border",
                         align_corners=True)


def synthetic_pair(batch: int, height: int, width: int, seed: int = 0,
                   max_mag: float = 12.0
                   ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (image1, image2, flow_gt): image2 is image1 warped so that
    pixels of image1 move by flow_gt from frame 1 to frame 2."""
    g = torch.Generator().manual_seed(seed)
    im1 = _random_texture(batch, height, width, g)
    flow = random_flow(batch, height, width, g, max_mag)
    # im2(x) = im1(x - flow(x)) approximately realizes forward flow `flow`
    im2 = warp(im1, -flow)
    return im1, im2, flow
