"""Flow post-processing extras — parity with the reference's flow_utils.py
beyond the RAFT path: USM/contrast augmentation (:123-135), static-region
masking (:155-159), flow reversal with conflict averaging and
nearest-neighbor hole filling (:166-274), and a guided filter (the
reference used cv2.ximgproc; reimplemented with box filters).  The
reference's ``calc_flow`` wrapped cv2 DIS optical flow; here the framework
itself is the flow engine (``calc_flow`` runs RAFT).

All NumPy; the reversal is vectorized (np.add.at splatting) instead of the
reference's per-pixel Python loops.
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np


# ------------------------------------------------------------- augmentation
def _gaussian_kernel1d(sigma: float) -> np.ndarray:
    radius = max(int(3.0 * sigma + 0.5), 1)
    x = np.arange(-radius, radius + 1, dtype=np.float64)
    k = np.exp(-0.5 * (x / sigma) ** 2)
    return k / k.sum()


def gaussian_blur(img: np.ndarray, sigma: float) -> np.ndarray:
    """Separable Gaussian blur with edge replication (cv2.GaussianBlur
    equivalent for the USM use)."""
    k = _gaussian_kernel1d(sigma)
    r = len(k) // 2
    out = img.astype(np.float64)
    pad = np.pad(out, [(r, r), (0, 0)] + [(0, 0)] * (img.ndim - 2),
                 mode="edge")
    out = np.apply_along_axis(lambda m: np.convolve(m, k, "valid"), 0, pad)
    pad = np.pad(out, [(0, 0), (r, r)] + [(0, 0)] * (img.ndim - 2),
                 mode="edge")
    out = np.apply_along_axis(lambda m: np.convolve(m, k, "valid"), 1, pad)
    return out


def aug_img(im: np.ndarray, contrast: float = 1.5, bias: float = 0.0,
            usm_amount: float = 0.5, usm_sigma: float = 5.0) -> np.ndarray:
    """Contrast stretch + unsharp mask (reference flow_utils.py:123-135:
    addWeighted(im, 1.5, blur, -0.5))."""
    im = np.clip(contrast * im.astype(np.float64) + bias, 0, 255)
    blurred = gaussian_blur(im, usm_sigma)
    usm = np.clip((1.0 + usm_amount) * im - usm_amount * blurred, 0, 255)
    return usm.astype(np.uint8)


# ---------------------------------------------------------------- masking
def set_static_flow(flow01: np.ndarray, im0: np.ndarray, bg: np.ndarray,
                    thresh: float = 5.0) -> np.ndarray:
    """Zero the flow where im0 matches the background plate
    (flow_utils.py:155-159)."""
    diff = np.abs(bg.astype(np.int32) - im0.astype(np.int32))
    static = np.all(diff < thresh, axis=-1, keepdims=True)
    return np.where(static, 0.0, flow01)


# ------------------------------------------------------------ flow reversal
def reverse_flow(flow01: np.ndarray, static_mask: Optional[np.ndarray] = None,
                 time_step: float = 1.0
                 ) -> Tuple[np.ndarray, np.ndarray]:
    """Reverse a forward flow field by splatting (-flow) at the rounded
    target positions, averaging conflicts, then filling holes with the
    mean of the nearest valid pixels in the four axis directions —
    the reference's reverse_flow_avg_skip_static (flow_utils.py:166-274)
    with its FLOW_PROJECTION_ROUND branch, vectorized.

    static_mask: optional [H,W] bool — pixels excluded from projection
    (the reference marked them inf and skipped).
    Returns (flow10, hole_mask_before_fill).
    """
    h, w = flow01.shape[:2]
    f = flow01.astype(np.float64) * time_step
    ys, xs = np.mgrid[0:h, 0:w]
    tx = np.clip(np.rint(f[:, :, 0] + xs), 0, w - 1).astype(np.int64)
    ty = np.clip(np.rint(f[:, :, 1] + ys), 0, h - 1).astype(np.int64)

    valid = np.ones((h, w), bool)
    if static_mask is not None:
        valid &= ~static_mask.astype(bool)

    flat = (ty * w + tx)[valid]
    flow10 = np.zeros((h * w, 2), np.float64)
    count = np.zeros(h * w, np.float64)
    np.add.at(flow10, flat, -f[valid])
    np.add.at(count, flat, 1.0)

    filled = count > 1e-7
    flow10[filled] /= count[filled, None]
    flow10 = flow10.reshape(h, w, 2)
    holes = ~filled.reshape(h, w)

    if holes.any():
        flow10 = _fill_holes_nearest4(flow10, holes)
    return flow10.astype(np.float32), holes


def _fill_holes_nearest4(flow: np.ndarray, holes: np.ndarray) -> np.ndarray:
    """For each hole, average the nearest valid pixel up/down/left/right
    (reference fiil_ind, flow_utils.py:229-262), computed by directional
    propagation instead of per-pixel scans."""
    h, w = holes.shape
    out = flow.copy()

    def directional(valid_val, axis, reverse):
        """Nearest valid value scanning along axis (propagate fill)."""
        v = valid_val.copy()
        ok = ~holes
        idx_range = range(h if axis == 0 else w)
        if reverse:
            idx_range = reversed(list(idx_range))
        last = None
        lastok = None
        res = np.full_like(flow, np.nan)
        has = np.zeros((h, w), bool)
        for i in idx_range:
            sl = (i, slice(None)) if axis == 0 else (slice(None), i)
            cur_ok = ok[sl]
            if last is None:
                last = np.where(cur_ok[:, None], v[sl], np.nan)
                lastok = cur_ok.copy()
            else:
                last = np.where(cur_ok[:, None], v[sl], last)
                lastok = lastok | cur_ok
            res[sl] = last
            has[sl] = lastok
        return res, has

    sums = np.zeros_like(flow)
    cnt = np.zeros((h, w), np.float64)
    for axis in (0, 1):
        for rev in (False, True):
            val, has = directional(flow, axis, rev)
            use = holes & has
            sums[use] += val[use]
            cnt[use] += 1.0
    fillable = holes & (cnt > 0)
    out[fillable] = sums[fillable] / cnt[fillable, None]
    out[holes & (cnt == 0)] = 0.0
    return out


# ------------------------------------------------------------ guided filter
def box_filter(img: np.ndarray, r: int) -> np.ndarray:
    """Mean filter with window (2r+1)^2, edge-replicated."""
    pad = np.pad(img, [(r, r), (r, r)] + [(0, 0)] * (img.ndim - 2),
                 mode="edge")
    c = np.cumsum(np.cumsum(pad, axis=0), axis=1)
    c = np.pad(c, [(1, 0), (1, 0)] + [(0, 0)] * (img.ndim - 2))
    k = 2 * r + 1
    out = (c[k:, k:] - c[:-k, k:] - c[k:, :-k] + c[:-k, :-k]) / (k * k)
    return out


def guided_filter(guide: np.ndarray, src: np.ndarray, radius: int = 9,
                  eps: float = 2.0) -> np.ndarray:
    """Edge-preserving smoothing of src guided by guide (He et al. 2010) —
    the post-processing the reference applied to DIS flow via
    cv2.ximgproc.guidedFilter (flow_utils.py:151)."""
    g = guide.astype(np.float64)
    if g.ndim == 3:
        g = g.mean(axis=2)
    s = src.astype(np.float64)
    single = s.ndim == 2
    if single:
        s = s[:, :, None]
    mg = box_filter(g, radius)
    ms = box_filter(s, radius)
    mgs = box_filter(g[:, :, None] * s, radius)
    var_g = box_filter(g * g, radius) - mg * mg
    cov = mgs - mg[:, :, None] * ms
    a = cov / (var_g[:, :, None] + eps)
    b = ms - a * mg[:, :, None]
    out = box_filter(a, radius) * g[:, :, None] + box_filter(b, radius)
    return out[:, :, 0] if single else out


# -------------------------------------------------------------- calc_flow
def calc_flow(im0: np.ndarray, im1: np.ndarray, model=None, iters: int = 12,
              post_filter: bool = True) -> np.ndarray:
    """Dense flow im0 -> im1. The reference wrapped cv2 DIS optical flow
    (flow_utils.py:137-153); here the framework itself is the engine: runs
    the (given or default raft-small) model, optionally guided-filters the
    result like the reference did."""
    import torch

    from raft_amd import RAFT, RaftConfig
    from raft_amd.engine.inference import InferenceEngine

    if model is None:
        model = RAFT(RaftConfig(small=True)).eval()
    engine = InferenceEngine(model, iters=iters)
    to_t = lambda im: torch.from_numpy(
        im.astype(np.float32) / 255.0).permute(2, 0, 1)[None]
    flow = engine(to_t(im0), to_t(im1))[0].float().permute(1, 2, 0)
    flow = flow.cpu().numpy()
    if post_filter:
        flow = guided_filter(im0, flow, radius=9, eps=2.0).astype(np.float32)
    return flow
