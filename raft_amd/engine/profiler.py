"""FLOP / parameter reporting (working replacement for the reference's
crashing flops mode, infer_raft.py:80-95) and simple step timers."""
from __future__ import annotations

import time
from typing import Optional

import torch
import torch.nn as nn


def count_model_flops(model, height: int, width: int,
                      iters: Optional[int] = None, batch: int = 1) -> dict:
    """Count conv FLOPs via forward hooks (2*MACs, tf.profiler convention)
    plus the correlation GEMM and pyramid lookup, on a real CPU forward at
    the given shape."""
    iters = iters or model.cfg.iters
    counts = {"conv_flops": 0}

    hooks = []

    def conv_hook(mod, inp, out):
        kh, kw = mod.kernel_size
        counts["conv_flops"] += (
            2 * mod.in_channels * mod.out_channels * kh * kw
            * out.shape[-2] * out.shape[-1] * out.shape[0])

    for m in model.modules():
        if isinstance(m, nn.Conv2d):
            hooks.append(m.register_forward_hook(conv_hook))
    model = model.eval()
    with torch.no_grad():
        model(torch.rand(batch, 3, height, width),
              torch.rand(batch, 3, height, width), iters=iters)
    for h in hooks:
        h.remove()

    h8, w8 = height // 8, width // 8
    hw = h8 * w8
    c = model.cfg.fnet_dim
    corr_gemm = 2 * hw * hw * c * batch
    K = 2 * model.cfg.corr_radius + 1
    lookup = batch * hw * model.cfg.corr_levels * K * K * 8 * iters
    total = counts["conv_flops"] + corr_gemm + lookup
    return {
        "model": "raft-small" if model.cfg.small else "raft-things",
        "input": [batch, 3, height, width],
        "iters": iters,
        "params": sum(p.numel() for p in model.parameters()),
        "conv_flops": counts["conv_flops"],
        "corr_gemm_flops": corr_gemm,
        "corr_lookup_flops": lookup,
        "total_flops": total,
        "total_gflops": total / 1e9,
    }


class StepTimer:
    """CUDA-event (HIP-event) step timer with warmup discard."""

    def __init__(self):
        self.times = []

    def timeit(self, fn, steps: int, warmup: int = 3) -> float:
        dev = torch.cuda.is_available()
        for _ in range(warmup):
            fn()
        if dev:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(steps):
            fn()
        if dev:
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / steps
        self.times.append(dt)
        return dt
