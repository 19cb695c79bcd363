// K1: all-pairs correlation volume  C[b, m, n] = sum_k F1[b,k,m] * F2[b,k,n] / sqrt(K)
// (replaces the reference's reshape+matmul+divide, model_utils.py:199-215).
//
// MFMA f32 GEMM (v_mfma_f32_16x16x4_f32, exact fp32 at the 157 TF f32 rate).
// Both operands arrive K-major ([B, C, H*W] — NCHW fmaps flattened), which is
// exactly the LDS staging layout the MFMA A/B fragments want:
//   lane l reads A_lds[k = l>>4][m0 + (l&15)] — 16 consecutive floats per
//   lane group, rows padded +1 float to kill the 2-way bank conflict between
//   lane groups (row stride 128 ≡ 0 mod 32 banks otherwise).
// Tile: 128x128 per 256-thread block (4 waves, 2x2, each wave 64x64 = 4x4
// fragments), BK=16 staged per iteration, bounds-checked staging so any
// H*W works (config 4's 135*240=32400 is not tile-divisible).
// bf16 inputs are handled by the host casting to fp32 for v1 (exact);
// a bf16 MFMA path is the planned upgrade once profiled.

#include "common.h"

#define BM 128
#define BN 128
#define BK 16
#define LDA (BM + 1)   // +1 float pad: see header comment

extern "C" __global__ __launch_bounds__(256)
void corr_volume_f32(const float* __restrict__ f1,   // [B, K, M]
                     const float* __restrict__ f2,   // [B, K, N]
                     float* __restrict__ out,        // [B, M, N]
                     int M, int N, int K, float scale) {
    __shared__ float sA[BK][LDA];
    __shared__ float sB[BK][LDA];

    const int b = blockIdx.z;
    const int m0 = blockIdx.y * BM;
    const int n0 = blockIdx.x * BN;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;          // 0..3
    const int wm = (wave >> 1) * 64;    // wave row offset in tile
    const int wn = (wave & 1) * 64;

    const float* A = f1 + (size_t)b * K * M;
    const float* Bp = f2 + (size_t)b * K * N;

    floatx4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    // staging: 256 threads x 8 floats = 2048 = BK*BM. Thread t loads
    // row kr = t/16, cols m0 + (t%16)*8 .. +7 (float4 x2, bounds-checked).
    const int skr = tid >> 4;
    const int scol = (tid & 15) * 8;

    for (int k0 = 0; k0 < K; k0 += BK) {
#pragma unroll
        for (int u = 0; u < 8; ++u) {
            int ma = m0 + scol + u;
            int na = n0 + scol + u;
            sA[skr][scol + u] = (ma < M) ? A[(size_t)(k0 + skr) * M + ma] : 0.f;
            sB[skr][scol + u] = (na < N) ? Bp[(size_t)(k0 + skr) * N + na] : 0.f;
        }
        __syncthreads();

#pragma unroll
        for (int kk = 0; kk < BK / 4; ++kk) {
            const int krow = kk * 4 + (lane >> 4);
            float a_frag[4], b_frag[4];
#pragma unroll
            for (int i = 0; i < 4; ++i)
                a_frag[i] = sA[krow][wm + i * 16 + (lane & 15)];
#pragma unroll
            for (int j = 0; j < 4; ++j)
                b_frag[j] = sB[krow][wn + j * 16 + (lane & 15)];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 4; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                        a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
    }

    // epilogue: C/D mapping col = lane&15, row = (lane>>4)*4 + r
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int m = m0 + wm + i * 16 + (lane >> 4) * 4 + r;
                int n = n0 + wn + j * 16 + (lane & 15);
                if (m < M && n < N)
                    out[((size_t)b * M + m) * N + n] = acc[i][j][r] * scale;
            }
        }
    }
}

// K2: 2x2/2 average pool over the target dims (TF VALID / floor), one level.
// corr [B, Q, H, W] -> [B, Q, H/2, W/2].  Bandwidth-bound; grid-stride,
// coalesced on x.
extern "C" __global__ void corr_pool2x_f32(const float* __restrict__ in,
                                           float* __restrict__ out,
                                           int H, int W, int Ho, int Wo,
                                           long long total) {
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const int xo = (int)(idx % Wo);
        const int yo = (int)((idx / Wo) % Ho);
        const long long q = idx / ((long long)Wo * Ho);   // fused b*Q index
        const float* src = in + (q * H + 2 * yo) * W + 2 * xo;
        out[idx] = 0.25f * (src[0] + src[1] + src[W] + src[W + 1]);
    }
}

// ----------------------------------------------------------- host launchers
extern "C" void launch_corr_volume_f32(const float* f1, const float* f2,
                                       float* out, int Bsz, int M, int N,
                                       int K, float scale, hipStream_t s) {
    dim3 grid(cdiv(N, BN), cdiv(M, BM), Bsz);
    hipLaunchKernelGGL(corr_volume_f32, grid, dim3(256), 0, s,
                       f1, f2, out, M, N, K, scale);
}

extern "C" void launch_corr_pool2x_f32(const float* in, float* out, int H,
                                       int W, int Ho, int Wo, long long total,
                                       hipStream_t s) {
    int blocks = (int)min((total + 255) / 256, (long long)2048);
    hipLaunchKernelGGL(corr_pool2x_f32, dim3(blocks), dim3(256), 0, s,
                       in, out, H, W, Ho, Wo, total);
}
