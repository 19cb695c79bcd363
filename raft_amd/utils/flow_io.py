""".flo file IO (Middlebury 'PIEH' format) and flow resize — functional
parity with flow_utils.py:277-318 of the reference."""
from __future__ import annotations

import numpy as np
import torch
import torch.nn.functional as F

_TAG = b"PIEH"


def read_flo(path: str) -> np.ndarray:
    with open(path, "rb") as f:
        if f.read(4) != _TAG:
            raise ValueError(f"{path}: missing PIEH header")
        w = int(np.fromfile(f, np.int32, 1)[0])
        h = int(np.fromfile(f, np.int32, 1)[0])
        flow = np.fromfile(f, np.float32, w * h * 2).reshape(h, w, 2)
    return flow


def write_flo(path: str, flow: np.ndarray) -> None:
    flow = np.asarray(flow, np.float32)
    assert flow.ndim == 3 and flow.shape[2] == 2
    with open(path, "wb") as f:
        f.write(_TAG)
        np.array([flow.shape[1], flow.shape[0]], np.int32).tofile(f)
        flow.tofile(f)


def resize_flow(flow: np.ndarray, dw: int, dh: int) -> np.ndarray:
    """Bilinear-resize a [H,W,2] flow field and rescale the vector
    magnitudes by the size ratio (flow_utils.py:277-284)."""
    h, w = flow.shape[:2]
    t = torch.from_numpy(np.ascontiguousarray(flow)).permute(2, 0, 1)[None]
    r = F.interpolate(t, size=(dh, dw), mode="bilinear", align_corners=False)
    r = r[0].permute(1, 2, 0).numpy().copy()
    r[:, :, 0] *= dw / float(w)
    r[:, :, 1] *= dh / float(h)
    return r
