"""Production soak: many frames through the engine across mixed shapes;
asserts memory stays bounded (no allocator growth from the fused path's
packed-weight caches or per-iteration tensors)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from raft_amd import RAFT, RaftConfig
from raft_amd.engine.inference import InferenceEngine


def main():
    dev = torch.device("cuda")
    eng = InferenceEngine(RAFT(RaftConfig(small=False)).to(dev).eval(),
                          iters=16, dtype=torch.bfloat16)
    shapes = [(436, 1024), (368, 768), (288, 512), (436, 1024)]
    batches = {s: (torch.rand(1, 3, *s), torch.rand(1, 3, *s))
               for s in shapes}
    # warm all shapes
    for s in shapes:
        eng(*batches[s])
    torch.cuda.synchronize()
    base = torch.cuda.memory_allocated()
    t0 = time.perf_counter()
    n = 240
    for i in range(n):
        s = shapes[i % len(shapes)]
        out = eng(*batches[s])
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    cur = torch.cuda.memory_allocated()
    peak = torch.cuda.max_memory_allocated()
    print(f"{n} frames, {1e3 * dt / n:.2f} ms avg (mixed shapes), "
          f"alloc drift {(cur - base) / 2**20:.1f} MiB, "
          f"peak {peak / 2**30:.2f} GiB")
    assert cur - base < 64 * 2**20, "allocator growth: possible leak"
    assert torch.isfinite(out.float()).all()
    print("soak OK")


if __name__ == "__main__":
    main()
