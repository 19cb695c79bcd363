// Instance-norm kernels for the fused NHWC encoders (fnet uses
// InstanceNorm(center=False, scale=False) — model_utils.py:13: no affine,
// eps 1e-5). Two passes: per-(b,c) mean/rstd reduction over H*W, then a
// pointwise apply with optional relu and optional residual
// (relu(res + relu(xhat))) — the relu-before-add residual quirk
// (model_utils.py:28-31) is honored by the caller choosing apply modes.

#include "common.h"
#include <hip/hip_bf16.h>

// stats: grid (B, ceil(C/64)); block 256 = 64 channels x 4 row-groups
extern "C" __global__ __launch_bounds__(256) void inorm_stats_k(
    const __hip_bfloat16* __restrict__ in,   // [B, H*W, C]
    float* __restrict__ mean, float* __restrict__ rstd,  // [B, C]
    int HW, int C, float eps) {
    __shared__ float red[2][4][64];
    const int b = blockIdx.x;
    const int c0 = blockIdx.y * 64;
    const int c = c0 + (threadIdx.x & 63);
    const int g = threadIdx.x >> 6;          // row-group 0..3
    float s = 0.f, s2 = 0.f;
    if (c < C) {
        const __hip_bfloat16* base = in + (size_t)b * HW * C + c;
        for (int p = g; p < HW; p += 4) {
            const float v = (float)base[(size_t)p * C];
            s += v;
            s2 += v * v;
        }
    }
    red[0][g][threadIdx.x & 63] = s;
    red[1][g][threadIdx.x & 63] = s2;
    __syncthreads();
    if (g == 0 && c < C) {
        float ts = 0.f, ts2 = 0.f;
        for (int i = 0; i < 4; ++i) {
            ts += red[0][i][threadIdx.x & 63];
            ts2 += red[1][i][threadIdx.x & 63];
        }
        const float m = ts / HW;
        const float var = fmaxf(ts2 / HW - m * m, 0.0f);
        mean[(size_t)b * C + c] = m;
        rstd[(size_t)b * C + c] = rsqrtf(var + eps);
    }
}

// apply: y = act((x - mean) * rstd) [+ residual, outer relu]
// mode 0: xhat            (downsample.1 norm: no relu)
// mode 1: relu(xhat)      (conv1/norm1 path)
// mode 2: relu(res + relu(xhat))   (block output, relu-before-add quirk)
extern "C" __global__ void inorm_apply_k(
    const __hip_bfloat16* __restrict__ in,   // [B, HW, C]
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const __hip_bfloat16* __restrict__ res,  // mode 2 only
    __hip_bfloat16* __restrict__ out,
    int HW, int C, int mode, long long total) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < total; i += stride) {
        const int c = (int)(i % C);
        const long long bc = (i / ((long long)HW * C)) * C + c;
        float v = ((float)in[i] - mean[bc]) * rstd[bc];
        if (mode >= 1) v = fmaxf(v, 0.0f);
        if (mode == 2) v = fmaxf((float)res[i] + v, 0.0f);
        out[i] = (__hip_bfloat16)v;
    }
}

extern "C" void launch_inorm_stats(const void* in, float* mean, float* rstd,
                                   int B, int HW, int C, float eps,
                                   hipStream_t s) {
    dim3 grid(B, cdiv(C, 64));
    hipLaunchKernelGGL(inorm_stats_k, grid, dim3(256), 0, s,
                       (const __hip_bfloat16*)in, mean, rstd, HW, C, eps);
}

extern "C" void launch_inorm_apply(const void* in, const float* mean,
                                   const float* rstd, const void* res,
                                   void* out, int B, int HW, int C,
                                   int mode, hipStream_t s) {
    const long long total = (long long)B * HW * C;
    int blocks = (int)min((total + 255) / 256, (long long)2048);
    hipLaunchKernelGGL(inorm_apply_k, dim3(blocks), dim3(256), 0, s,
                       (const __hip_bfloat16*)in, mean, rstd,
                       (const __hip_bfloat16*)res, (__hip_bfloat16*)out, HW,
                       C, mode, total);
}
