// Instance-norm kernels for the fused NHWC encoders (fnet uses
// InstanceNorm(center=False, scale=False) — model_utils.py:13: no affine,
// eps 1e-5). Two passes: per-(b,c) mean/rstd reduction over H*W, then a
// pointwise apply with optional relu and optional residual
// (relu(res + relu(xhat))) — the relu-before-add residual quirk
// (model_utils.py:28-31) is honored by the caller choosing apply modes.

#include "common.h"
#include <hip/hip_bf16.h>

// stats, pass 1: split reduction — grid (B, ceil(C/64), S position
// partitions); each block writes its partition's moments to its OWN slab
// slot (no atomics: float atomicAdd order made the whole fused path
// nondeterministic — caught by the bit-determinism race screen).
// (A (B, C/64) grid was 2 blocks on a 256-CU chip: 3.2 ms/call.)
extern "C" __global__ __launch_bounds__(256) void inorm_stats_part_k(
    const __hip_bfloat16* __restrict__ in,   // [B, H*W, C]
    float* __restrict__ acc,                 // [B, C, S, 2] slabs
    int HW, int C) {
    __shared__ float red[2][4][64];
    const int b = blockIdx.x;
    const int c0 = blockIdx.y * 64;
    const int c = c0 + (threadIdx.x & 63);
    const int g = threadIdx.x >> 6;          // row-group 0..3
    const int S = gridDim.z;
    float s = 0.f, s2 = 0.f;
    if (c < C) {
        const __hip_bfloat16* base = in + (size_t)b * HW * C + c;
        for (int p = blockIdx.z * 4 + g; p < HW; p += 4 * S) {
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
        const size_t slot = (((size_t)b * C + c) * S + blockIdx.z) * 2;
        acc[slot] = ts;
        acc[slot + 1] = ts2;
    }
}

// stats, pass 2: finalize mean/rstd — fixed-order sum over the S slabs
// (deterministic by construction)
extern "C" __global__ void inorm_stats_fin_k(
    const float* __restrict__ acc, float* __restrict__ mean,
    float* __restrict__ rstd, int HW, int S, long long BC, float eps) {
    const long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= BC) return;
    float ts = 0.f, ts2 = 0.f;
    const float* slab = acc + (size_t)i * S * 2;
    for (int z = 0; z < S; ++z) {
        ts += slab[z * 2];
        ts2 += slab[z * 2 + 1];
    }
    const float m = ts / HW;
    const float var = fmaxf(ts2 / HW - m * m, 0.0f);
    mean[i] = m;
    rstd[i] = rsqrtf(var + eps);
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

extern "C" int inorm_stats_partitions(int HW) {
    return (int)min((long long)cdiv(HW, 1024), (long long)128);
}

extern "C" void launch_inorm_stats(const void* in, float* acc, float* mean,
                                   float* rstd, int B, int HW, int C,
                                   float eps, hipStream_t s) {
    const int S = inorm_stats_partitions(HW);
    dim3 grid(B, cdiv(C, 64), S);
    hipLaunchKernelGGL(inorm_stats_part_k, grid, dim3(256), 0, s,
                       (const __hip_bfloat16*)in, acc, HW, C);
    const long long BC = (long long)B * C;
    hipLaunchKernelGGL(inorm_stats_fin_k,
                       dim3((unsigned)((BC + 255) / 256)), dim3(256), 0, s,
                       acc, mean, rstd, HW, S, BC, eps);
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
