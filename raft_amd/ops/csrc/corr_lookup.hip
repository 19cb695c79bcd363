// K3: multi-scale correlation pyramid lookup (the #1 hot kernel) and its
// backward scatter.  Replaces the reference's per-level meshgrid window +
// tf_grid_sample gather_nd chain (model_utils.py:224-249,
// networks/utils.py:39-99).
//
// Forward: out[b, lvl*KK + k, y, x] = bilinear(corr_lvl[b, y*W+x, :, :],
//          coords(b,y,x)/2^lvl + (dx, dy)) with tap order
//          dx = k / (2r+1) - r, dy = k % (2r+1) - r  (the reference's [::-1]
//          window order, model_utils.py:237) and edge-clamp trunc bilinear
//          (common.h make_tap).
//
// Thread-per-output-element, x fastest -> fully coalesced stores; the 4
// corner gathers of neighboring threads overlap heavily and ride L1/L2
// (total tap traffic per call is ~tens of MB — latency-, not BW-bound).
//
// Backward: same indexing, atomicAdd of w*grad into the 4 corners of the
// per-query slice. Different queries own disjoint slices, so contention is
// only among a query's own 324 taps.

#include "common.h"

struct Levels {
    const float* ptr[4];
    float* gptr[4];
    int H[4];
    int W[4];
};

extern "C" __global__ void corr_lookup_fwd_f32(
    Levels lv, const float* __restrict__ coords,  // [B, H, W, 2]
    float* __restrict__ out,                      // [B, L*KK, H, W]
    int Bq, int H, int W, int num_levels, int radius, long long total) {
    const int K = 2 * radius + 1;
    const int KK = K * K;
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const int x = (int)(idx % W);
        const int y = (int)((idx / W) % H);
        const int c = (int)((idx / ((long long)W * H)) % (num_levels * KK));
        const int b = (int)(idx / ((long long)W * H * num_levels * KK));
        const int lvl = c / KK;
        const int k = c - lvl * KK;

        const long long ci = (((long long)b * H + y) * W + x) * 2;
        const float inv = 1.0f / (float)(1 << lvl);
        const float cx = coords[ci] * inv + (float)(k / K - radius);
        const float cy = coords[ci + 1] * inv + (float)(k % K - radius);

        const int H2 = lv.H[lvl], W2 = lv.W[lvl];
        const float* slice = lv.ptr[lvl]
            + ((size_t)b * H * W + (size_t)y * W + x) * (size_t)H2 * W2;
        BilinearTap t = make_tap(cx, cy, W2, H2);
        const float Ia = slice[t.y0 * W2 + t.x0];
        const float Ib = slice[t.y1 * W2 + t.x0];
        const float Ic = slice[t.y0 * W2 + t.x1];
        const float Id = slice[t.y1 * W2 + t.x1];
        out[idx] = t.wa * Ia + t.wb * Ib + t.wc * Ic + t.wd * Id;
    }
}

extern "C" __global__ void corr_lookup_bwd_f32(
    Levels lv, const float* __restrict__ coords,
    const float* __restrict__ grad_out,           // [B, L*KK, H, W]
    int Bq, int H, int W, int num_levels, int radius, long long total) {
    const int K = 2 * radius + 1;
    const int KK = K * K;
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const float g = grad_out[idx];
        const int x = (int)(idx % W);
        const int y = (int)((idx / W) % H);
        const int c = (int)((idx / ((long long)W * H)) % (num_levels * KK));
        const int b = (int)(idx / ((long long)W * H * num_levels * KK));
        const int lvl = c / KK;
        const int k = c - lvl * KK;

        const long long ci = (((long long)b * H + y) * W + x) * 2;
        const float inv = 1.0f / (float)(1 << lvl);
        const float cx = coords[ci] * inv + (float)(k / K - radius);
        const float cy = coords[ci + 1] * inv + (float)(k % K - radius);

        const int H2 = lv.H[lvl], W2 = lv.W[lvl];
        float* slice = lv.gptr[lvl]
            + ((size_t)b * H * W + (size_t)y * W + x) * (size_t)H2 * W2;
        BilinearTap t = make_tap(cx, cy, W2, H2);
        atomicAdd(&slice[t.y0 * W2 + t.x0], t.wa * g);
        atomicAdd(&slice[t.y1 * W2 + t.x0], t.wb * g);
        atomicAdd(&slice[t.y0 * W2 + t.x1], t.wc * g);
        atomicAdd(&slice[t.y1 * W2 + t.x1], t.wd * g);
    }
}

// ----------------------------------------------------------- host launchers
extern "C" void launch_corr_lookup_fwd_f32(
    const float* const* level_ptrs, const int* level_h, const int* level_w,
    const float* coords, float* out, int B, int H, int W, int num_levels,
    int radius, hipStream_t s) {
    Levels lv{};
    for (int i = 0; i < num_levels; ++i) {
        lv.ptr[i] = level_ptrs[i];
        lv.H[i] = level_h[i];
        lv.W[i] = level_w[i];
    }
    const int K = 2 * radius + 1;
    const long long total = (long long)B * H * W * num_levels * K * K;
    int blocks = (int)min((total + 255) / 256, (long long)8192);
    hipLaunchKernelGGL(corr_lookup_fwd_f32, dim3(blocks), dim3(256), 0, s,
                       lv, coords, out, B, H, W, num_levels, radius, total);
}

extern "C" void launch_corr_lookup_bwd_f32(
    float* const* grad_level_ptrs, const int* level_h, const int* level_w,
    const float* coords, const float* grad_out, int B, int H, int W,
    int num_levels, int radius, hipStream_t s) {
    Levels lv{};
    for (int i = 0; i < num_levels; ++i) {
        lv.gptr[i] = grad_level_ptrs[i];
        lv.H[i] = level_h[i];
        lv.W[i] = level_w[i];
    }
    const int K = 2 * radius + 1;
    const long long total = (long long)B * H * W * num_levels * K * K;
    int blocks = (int)min((total + 255) / 256, (long long)8192);
    hipLaunchKernelGGL(corr_lookup_bwd_f32, dim3(blocks), dim3(256), 0, s,
                       lv, coords, grad_out, B, H, W, num_levels, radius,
                       total);
}
