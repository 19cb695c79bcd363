"""Test-time dataflow: file pairs -> model-ready tensors.

Re-design of dataflow/test_dataflow.py: decode -> BGR -> (test mode) resize
to input_size -> float32 [0,1] (:56-61, :85-87, :96-97), batched.  The
reference used tensorpack DataFlow + cv2; here it is a plain iterator over
torch tensors with the same numerics (bilinear resize).

Training augmentations (the reference defined but never used them,
test_dataflow.py:19-41) are implemented for the training path: shared-param
photometric jitter across the pair, random horizontal flip with frame-order
swap, random crop.
"""
from __future__ import annotations

from typing import Iterator, List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.nn.functional as F

from raft_amd.data.imageio import read_png


def load_image(path: str) -> torch.Tensor:
    """PNG -> [3,H,W] float32 BGR in [0,1]."""
    img = read_png(path)  # HxWx3 uint8 BGR
    return torch.from_numpy(img.astype(np.float32) / 255.0).permute(2, 0, 1)


def resize_to(img: torch.Tensor, size: Tuple[int, int]) -> torch.Tensor:
    """Bilinear resize to (H, W) — the reference's cv2.resize test path
    (test_dataflow.py:85-87)."""
    if img.shape[-2:] == tuple(size):
        return img
    return F.interpolate(img[None], size=size, mode="bilinear",
                         align_corners=False)[0]


class PairDataflow:
    """Iterates (im1, im2) batches from a list of file pairs."""

    def __init__(self, filelist: Sequence[Tuple[str, str]],
                 input_size: Optional[Tuple[int, int]] = (432, 1024),
                 batch: int = 1):
        self.filelist = list(filelist)
        self.input_size = input_size
        self.batch = batch

    def __len__(self) -> int:
        return (len(self.filelist) + self.batch - 1) // self.batch

    def size(self) -> int:   # reference dataflow API (test_dataflow.py:118)
        return len(self)

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        cur1: List[torch.Tensor] = []
        cur2: List[torch.Tensor] = []
        for p1, p2 in self.filelist:
            im1 = load_image(p1)
            im2 = load_image(p2)
            if self.input_size is not None:
                im1 = resize_to(im1, self.input_size)
                im2 = resize_to(im2, self.input_size)
            cur1.append(im1)
            cur2.append(im2)
            if len(cur1) == self.batch:
                yield torch.stack(cur1), torch.stack(cur2)
                cur1, cur2 = [], []
        if cur1:
            yield torch.stack(cur1), torch.stack(cur2)


# ---------------------------------------------------------------- training aug
def augment_pair(im1: torch.Tensor, im2: torch.Tensor, flow: torch.Tensor,
                 g: torch.Generator,
                 crop: Optional[Tuple[int, int]] = None):
    """Shared-parameter photometric + geometric augmentation of a pair
    (reference intent: test_dataflow.py:20-41 — contrast/gamma shared across
    the pair, h-flip with frame swap, random crop)."""
    # photometric: shared contrast & gamma
    c = 0.8 + 0.4 * torch.rand((), generator=g).item()
    gamma = 0.8 + 0.4 * torch.rand((), generator=g).item()
    im1 = (im1 * c).clamp(0, 1) ** gamma
    im2 = (im2 * c).clamp(0, 1) ** gamma
    # horizontal flip (x component negates; the reference's frame-order swap
    # variant only applies without ground truth — it would invalidate flow)
    if torch.rand((), generator=g).item() < 0.5:
        im1 = torch.flip(im1, [-1])
        im2 = torch.flip(im2, [-1])
        flow = torch.flip(flow, [-1]) * torch.tensor(
            [-1.0, 1.0]).view(1, 2, 1, 1)
    if crop is not None:
        ch, cw = crop
        H, W = im1.shape[-2:]
        y0 = int(torch.randint(0, max(H - ch, 1) + 1, (1,), generator=g))
        x0 = int(torch.randint(0, max(W - cw, 1) + 1, (1,), generator=g))
        im1 = im1[..., y0:y0 + ch, x0:x0 + cw]
        im2 = im2[..., y0:y0 + ch, x0:x0 + cw]
        flow = flow[..., y0:y0 + ch, x0:x0 + cw]
    return im1, im2, flow
