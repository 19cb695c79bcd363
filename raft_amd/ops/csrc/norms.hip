// Instance-norm kernels for the fused NHWC encoders (fnet uses
// InstanceNorm(center=False, scale=False) — model_utils.py:13: no affine,
// eps 1e-5). Two passes: per-(b,c) mean/rstd reduction over H*W, then a
// pointwise apply with optional relu and optional residual
// (relu(res + relu(xhat))) — the relu-before-add residual quirk
// (model_utils.py:28-31) is honored by the caller choosing apply modes.

#include "common.h"
#include <hip/hip_bf16.h>

__device__ inline float bf_bits_to_f(unsigned int bits16) {
    union { unsigned int i; float f; } u;
    u.i = bits16 << 16;
    return u.f;
}

// stats, pass 1: split reduction — grid (B, ceil(C/64), S position
// partitions); each block writes its partition's moments to its OWN slab
// slot (no atomics: float atomicAdd order made the whole fused path
// nondeterministic — caught by the bit-determinism race screen).
// (A (B, C/64) grid was 2 blocks on a 256-CU chip: 3.2 ms/call.)
//
// Vectorized: each lane loads 8 consecutive channels per position as one
// 16-byte uint4 (a 2-byte-per-lane version measured 66 us avg — pure
// memory latency at ~3 waves/CU occupancy; 8 accumulator chains + 8x the
// bytes in flight fix that). Lane layout: lane&7 -> channel octet within
// the 64-channel group, lane>>3 -> position row (32 rows/iteration).
// Requires C % 8 == 0 (guaranteed by the launcher; scalar fallback below).
extern "C" __global__ __launch_bounds__(256) void inorm_stats_part_k(
    const __hip_bfloat16* __restrict__ in,   // [B, H*W, C]
    float* __restrict__ acc,                 // [B, C, S, 2] slabs
    int HW, int C) {
    __shared__ float red[2][32][64];
    const int b = blockIdx.x;
    const int c0 = blockIdx.y * 64 + (threadIdx.x & 7) * 8;  // octet base
    const int g = threadIdx.x >> 3;          // position row-group 0..31
    const int S = gridDim.z;
    float s[8], s2[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) { s[j] = 0.f; s2[j] = 0.f; }
    if (c0 + 8 <= C) {
        const __hip_bfloat16* base = in + (size_t)b * HW * C + c0;
        for (int p = blockIdx.z * 32 + g; p < HW; p += 32 * S) {
            const uint4 v =
                *reinterpret_cast<const uint4*>(base + (size_t)p * C);
            const unsigned int w[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
            for (int j = 0; j < 4; ++j) {
                const float lo = bf_bits_to_f(w[j] & 0xffffu);
                const float hi = bf_bits_to_f(w[j] >> 16);
                s[2 * j] += lo;     s2[2 * j] += lo * lo;
                s[2 * j + 1] += hi; s2[2 * j + 1] += hi * hi;
            }
        }
    } else if (c0 < C) {             // ragged tail of a non-/64 C
        const __hip_bfloat16* base = in + (size_t)b * HW * C;
        for (int p = blockIdx.z * 32 + g; p < HW; p += 32 * S)
            for (int j = 0; j < 8 && c0 + j < C; ++j) {
                const float v = (float)base[(size_t)p * C + c0 + j];
                s[j] += v;
                s2[j] += v * v;
            }
    }
    const int cl = (threadIdx.x & 7) * 8;    // channel base within group
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        red[0][g][cl + j] = s[j];
        red[1][g][cl + j] = s2[j];
    }
    __syncthreads();
    const int c = blockIdx.y * 64 + (threadIdx.x & 63);
    if (threadIdx.x < 64 && c < C) {
        float ts = 0.f, ts2 = 0.f;
        for (int i = 0; i < 32; ++i) {       // fixed order: deterministic
            ts += red[0][i][threadIdx.x];
            ts2 += red[1][i][threadIdx.x];
        }
        const size_t slot = (((size_t)b * C + c) * S + blockIdx.z) * 2;
        acc[slot] = ts;
        acc[slot + 1] = ts2;
    }
}

// stats, pass 2: finalize mean/rstd — one wavefront per (b,c); the slab
// sum is a lane-strided partial + fixed shuffle tree (deterministic: the
// reduction order is a pure function of lane indices). A thread-per-(b,c)
// version walked S~110 slots serially: ~10 us of dependent adds.
extern "C" __global__ __launch_bounds__(256) void inorm_stats_fin_k(
    const float* __restrict__ acc, float* __restrict__ mean,
    float* __restrict__ rstd, int HW, int S, long long BC, float eps) {
    const long long i = (long long)blockIdx.x * 4 + (threadIdx.x >> 6);
    if (i >= BC) return;
    const int lane = threadIdx.x & 63;
    float ts = 0.f, ts2 = 0.f;
    const float* slab = acc + (size_t)i * S * 2;
    for (int z = lane; z < S; z += 64) {
        ts += slab[z * 2];
        ts2 += slab[z * 2 + 1];
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        ts += __shfl_down(ts, off, 64);
        ts2 += __shfl_down(ts2, off, 64);
    }
    if (lane == 0) {
        const float m = ts / HW;
        const float var = fmaxf(ts2 / HW - m * m, 0.0f);
        mean[i] = m;
        rstd[i] = rsqrtf(var + eps);
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

// vectorized apply (C % 8 == 0): 16-byte loads/stores, 8 channels/lane;
// the octet never straddles a channel boundary so c is octet-uniform-free
// but mean/rstd indexing stays per-channel (L2-resident: BC floats).
extern "C" __global__ void inorm_apply_v8_k(
    const __hip_bfloat16* __restrict__ in,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const __hip_bfloat16* __restrict__ res,
    __hip_bfloat16* __restrict__ out,
    int HW, int C, int mode, long long total8) {   // total/8 octets
    const long long stride = (long long)gridDim.x * blockDim.x;
    const int C8 = C >> 3;
    for (long long o = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         o < total8; o += stride) {
        const int c0 = (int)(o % C8) * 8;
        const long long bc0 = (o / ((long long)HW * C8)) * C + c0;
        const uint4 v = reinterpret_cast<const uint4*>(in)[o];
        uint4 rv = make_uint4(0u, 0u, 0u, 0u);
        if (mode == 2) rv = reinterpret_cast<const uint4*>(res)[o];
        const unsigned int w[4] = {v.x, v.y, v.z, v.w};
        const unsigned int rw[4] = {rv.x, rv.y, rv.z, rv.w};
        unsigned int ow[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            float lo = (bf_bits_to_f(w[j] & 0xffffu) - mean[bc0 + 2 * j]) *
                       rstd[bc0 + 2 * j];
            float hi = (bf_bits_to_f(w[j] >> 16) - mean[bc0 + 2 * j + 1]) *
                       rstd[bc0 + 2 * j + 1];
            if (mode >= 1) { lo = fmaxf(lo, 0.f); hi = fmaxf(hi, 0.f); }
            if (mode == 2) {
                lo = fmaxf(bf_bits_to_f(rw[j] & 0xffffu) + lo, 0.f);
                hi = fmaxf(bf_bits_to_f(rw[j] >> 16) + hi, 0.f);
            }
            const unsigned short lb =
                __hip_bfloat16_raw(__float2bfloat16(lo)).x;
            const unsigned short hb =
                __hip_bfloat16_raw(__float2bfloat16(hi)).x;
            ow[j] = (unsigned int)lb | ((unsigned int)hb << 16);
        }
        reinterpret_cast<uint4*>(out)[o] =
            make_uint4(ow[0], ow[1], ow[2], ow[3]);
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
                       dim3((unsigned)((BC + 3) / 4)), dim3(256), 0, s,
                       acc, mean, rstd, HW, S, BC, eps);
}

extern "C" void launch_inorm_apply(const void* in, const float* mean,
                                   const float* rstd, const void* res,
                                   void* out, int B, int HW, int C,
                                   int mode, hipStream_t s) {
    const long long total = (long long)B * HW * C;
    if ((C & 7) == 0) {
        const long long total8 = total >> 3;
        int blocks = (int)min((total8 + 255) / 256, (long long)2048);
        hipLaunchKernelGGL(inorm_apply_v8_k, dim3(blocks), dim3(256), 0, s,
                           (const __hip_bfloat16*)in, mean, rstd,
                           (const __hip_bfloat16*)res, (__hip_bfloat16*)out,
                           HW, C, mode, total8);
        return;
    }
    int blocks = (int)min((total + 255) / 256, (long long)2048);
    hipLaunchKernelGGL(inorm_apply_k, dim3(blocks), dim3(256), 0, s,
                       (const __hip_bfloat16*)in, mean, rstd,
                       (const __hip_bfloat16*)res, (__hip_bfloat16*)out, HW,
                       C, mode, total);
}
