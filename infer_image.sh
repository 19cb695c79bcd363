#!/bin/bash
# Demo invocations — parity with the reference's infer_image.sh:
# raft-things and raft-small on the canonical Sintel frame pair.
python infer_raft.py --mode test --im1 frame_0016.png --im2 frame_0017.png --load release_weight/raft-things.npz --out .
python infer_raft.py --mode test --im1 frame_0016.png --im2 frame_0017.png --load release_weight/raft-small.npz --small --out .
