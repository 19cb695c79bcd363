#!/bin/bash
# Demo invocations — parity with the reference's infer_image.sh:
# raft-things and raft-small on the canonical frame pair. This image has
# no network access, so if the Sintel frames / released weights are not
# present we synthesize a warped demo pair (and run with random-init
# weights) instead of failing.
set -e
cd "$(dirname "$0")"
if [ ! -f frame_0016.png ] || [ ! -f frame_0017.png ]; then
    python - << 'PY'
import numpy as np
from raft_amd.data.synthetic import synthetic_pair
from raft_amd.data.imageio import write_png
im1, im2, _ = synthetic_pair(1, 384, 768, seed=16)
for name, t in (("frame_0016.png", im1), ("frame_0017.png", im2)):
    write_png(name, (t[0].permute(1, 2, 0).numpy() * 255).astype("uint8"))
    print("synthesized", name)
PY
fi
LOAD_THINGS=""; LOAD_SMALL=""
[ -f release_weight/raft-things.npz ] && LOAD_THINGS="--load release_weight/raft-things.npz"
[ -f release_weight/raft-small.npz ] && LOAD_SMALL="--load release_weight/raft-small.npz"
python infer_raft.py --mode test --im1 frame_0016.png --im2 frame_0017.png $LOAD_THINGS --out .
python infer_raft.py --mode test --im1 frame_0016.png --im2 frame_0017.png $LOAD_SMALL --small --out .
