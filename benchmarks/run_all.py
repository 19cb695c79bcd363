#!/usr/bin/env python3
"""Run every BASELINE.json config on the current device and emit a JSON
report (benchmarks/results_<host>.json + markdown table).

  1. raft-small, 2x128x256, 12 iters, CPU-only plumbing path
  2. raft-things, 2x436x1024, bf16, 32 iters, 1 GPU          (headline)
  3. raft-things training, batch 16 @ 2x368x768, 12 iters    (per-GPU shard
     batch 2 when run 1-GPU; DP=8 is the driver's scaling run)
  4. raft-things, 2x1080x1920, full-res volume, 32 iters
  5. raft-small mixed-batch dynamic H x W inference

Usage: python benchmarks/run_all.py [--quick]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, steps, warmup, sync):
    for _ in range(warmup):
        fn()
    sync()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    sync()
    return (time.perf_counter() - t0) / steps


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--out", default=None)
    args = ap.parse_args()

    from raft_amd import RAFT, RaftConfig
    from raft_amd.engine.inference import InferenceEngine
    from raft_amd.engine.trainer import Trainer, TrainConfig

    have_gpu = torch.cuda.is_available()
    dev = torch.device("cuda" if have_gpu else "cpu")
    sync = (lambda: torch.cuda.synchronize()) if have_gpu else (lambda: None)
    steps = 4 if args.quick else 10
    results = []

    def record(name, ms, unit_frames=1, extra=None):
        r = {"config": name, "ms_per_step": ms * 1e3,
             "fps": unit_frames / ms, **(extra or {})}
        results.append(r)
        print(json.dumps(r))

    # config 1 — CPU plumbing
    m = RAFT(RaftConfig(small=True)).eval()
    x1 = torch.rand(1, 3, 128, 256)
    x2 = torch.rand(1, 3, 128, 256)
    with torch.no_grad():
        ms = timeit(lambda: m(x1, x2, iters=12), 3, 1, lambda: None)
    record("1: raft-small 2x128x256 12it cpu", ms)

    if not have_gpu:
        _write(results, args.out)
        return

    # config 2 — headline inference
    m2 = RAFT(RaftConfig(small=False)).to(dev).eval()
    eng = InferenceEngine(m2, iters=32, dtype=torch.bfloat16)
    a = torch.rand(1, 3, 436, 1024)
    b = torch.rand(1, 3, 436, 1024)
    ms = timeit(lambda: eng(a, b), steps, 3, sync)
    record("2: raft-things 2x436x1024 bf16 32it", ms)

    # config 3 — training step (per-GPU shard of the DP=8 batch-16 job)
    m3 = RAFT(RaftConfig(small=False))
    tr = Trainer(m3, TrainConfig(num_steps=steps + 4, iters=12, batch=2,
                                 height=368, width=768), device=dev)
    xt1 = torch.rand(2, 3, 368, 768, device=dev)
    xt2 = torch.rand(2, 3, 368, 768, device=dev)
    gt = torch.randn(2, 2, 368, 768, device=dev)
    ms = timeit(lambda: tr.step(xt1, xt2, gt), max(steps // 2, 2), 2, sync)
    record("3: raft-things train b2@368x768 12it (per-GPU shard)", ms,
           unit_frames=2)

    # config 4 — full-res volume resident
    eng4 = InferenceEngine(RAFT(RaftConfig(small=False)).to(dev).eval(),
                           iters=32, dtype=torch.bfloat16)
    c = torch.rand(1, 3, 1080, 1920)
    d = torch.rand(1, 3, 1080, 1920)
    ms = timeit(lambda: eng4(c, d), max(steps // 2, 2), 2, sync)
    peak = torch.cuda.memory_stats()["allocated_bytes.all.peak"] / 2**30
    record("4: raft-things 2x1080x1920 bf16 32it full-res volume", ms,
           extra={"peak_gib": round(peak, 2)})

    # config 5 — mixed-batch dynamic shapes
    eng5 = InferenceEngine(RAFT(RaftConfig(small=True)).to(dev).eval(),
                           iters=12, dtype=torch.bfloat16)
    shapes = [(368, 768), (436, 1024), (288, 512), (436, 1024)]
    batches = [(torch.rand(1, 3, h, w), torch.rand(1, 3, h, w))
               for h, w in shapes]

    def step5():
        for p, q in batches:
            eng5(p, q)

    ms = timeit(step5, max(steps // 2, 2), 2, sync)
    record("5: raft-small dynamic HxW (4 shapes/step)", ms, unit_frames=4)

    _write(results, args.out)


def _write(results, out):
    out = out or os.path.join(os.path.dirname(os.path.abspath(__file__)),
                              "results.json")
    with open(out, "w") as f:
        json.dump({"device": "mi355x" if torch.cuda.is_available() else "cpu",
                   "results": results}, f, indent=2)
    print(f"wrote {out}")


if __name__ == "__main__":
    main()
