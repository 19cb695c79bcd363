// Common device-side helpers for the raft_amd gfx950 kernels.
// Target: MI355X (CDNA4, wave64, 256 CUs / 8 XCDs). No CUDA compat paths.
#pragma once

#include <hip/hip_runtime.h>

#define RAFT_DEV __device__ __forceinline__

typedef float floatx4 __attribute__((ext_vector_type(4)));

// ceil-div
constexpr int cdiv(int a, int b) { return (a + b - 1) / b; }

// Edge-clamp bilinear corner/weight math replicating the reference's
// tf_grid_sample (networks/utils.py:39-99): corner ints by trunc-toward-zero
// (tf.cast), both corners clamped, weights from the CLAMPED far corner
// (qx = x1c - x) — can exceed [0,1] for out-of-range coords; weights always
// sum to 1.
struct BilinearTap {
    int x0, x1, y0, y1;
    float wa, wb, wc, wd;  // (x0,y0),(x0,y1),(x1,y0),(x1,y1)
};

RAFT_DEV BilinearTap make_tap(float x, float y, int W, int H) {
    BilinearTap t;
    int xt = (int)x;  // C cast truncates toward zero == tf.cast
    int yt = (int)y;
    t.x0 = min(max(xt, 0), W - 1);
    t.x1 = min(max(xt + 1, 0), W - 1);
    t.y0 = min(max(yt, 0), H - 1);
    t.y1 = min(max(yt + 1, 0), H - 1);
    float qx = (float)t.x1 - x;
    float qy = (float)t.y1 - y;
    t.wa = qx * qy;
    t.wb = qx * (1.0f - qy);
    t.wc = (1.0f - qx) * qy;
    t.wd = (1.0f - qx) * (1.0f - qy);
    return t;
}
