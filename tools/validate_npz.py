"""Validate a .npz checkpoint against the reference key schema.

  python tools/validate_npz.py weights.npz [--small]

Reports missing/extra/mis-shaped keys vs the model's expected tree
 — useful before loading converted reference weights.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("npz")
    ap.add_argument("--small", action="store_true")
    args = ap.parse_args(argv)

    from raft_amd import RAFT, RaftConfig
    from raft_amd.utils import checkpoint as ckpt

    model = RAFT(RaftConfig(small=args.small))
    expected = set(ckpt.expected_npz_keys(model))
    sd = model.state_dict()
    archive = np.load(args.npz)
    present = {k[:-2] if k.endswith(":0") else k for k in archive.files}

    missing = sorted(expected - present)
    extra = sorted(present - expected)
    shape_errors = []
    for k in sorted(expected & present):
        pt_key, transpose = ckpt.tf_key_to_torch(k)
        arr = archive[k if k in archive.files else k + ":0"]
        want = tuple(sd[pt_key].shape)
        got = tuple(arr.shape)
        if transpose:
            got = (got[3], got[2], got[0], got[1])
        if got != want:
            shape_errors.append((k, got, want))

    print(f"expected {len(expected)} keys; present {len(present)}")
    for name, items in (("MISSING", missing), ("EXTRA", extra)):
        for k in items[:20]:
            print(f"{name}: {k}")
        if len(items) > 20:
            print(f"... and {len(items) - 20} more {name.lower()}")
    for k, got, want in shape_errors[:20]:
        print(f"SHAPE: {k}: {got} != {want}")
    ok = not missing and not shape_errors
    print("OK" if ok else "INVALID")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
