// Fused fp32 MFMA GEMM with elementwise epilogues — the MI355X hot path for
// CosineRandomFeatures (out = cos(X W + b) * scale) and friends.
//
// Replaces the reference's Breeze/OpenBLAS GEMM + per-element cos pass
// (ref: nodes.stats.CosineRandomFeatures): the
// D-wide intermediate X*W never round-trips through HBM.
//
// Design (CDNA4 / gfx950, cdna_hip_programming.md §3/§5):
//   * v_mfma_f32_32x32x2_f32 — exact f32 at the 157 TF f32 vector rate
//     (no xf32 on gfx950; guide §3 "FP32-input MFMA").
//   * 128x128 macro tile, 4 waves, each wave a 2x2 grid of 32x32 MFMA
//     tiles; BK=32 (16 MFMA k-steps of K=2 per tile pair).
//   * LDS: A tile [128][33] f32 (pad 1 breaks the 32-bank row stride:
//     bank (row*33+k)%32 = (row+k)%32, conflict-free per 32-lane group),
//     B tile [32][128] (column-consecutive reads are conflict-free).
//   * XCD-aware block swizzle for L2 locality on the 8-XCD chip.
// Correctness-first v1 (guide's untuned shape measured ~122 TF of the
// 155 TF ceiling); pipelining upgrades come with rocprof evidence.

#include "common.h"
#include <math.h>

#define BM 128
#define BN 128
#define BK 32
#define APAD 1

enum Epilogue { EP_NONE = 0, EP_BIAS = 1, EP_COS = 2, EP_RELU = 3 };

template <int EPI>
__global__ __launch_bounds__(256)
void gemm_fused_f32_kernel(const float* __restrict__ A,  // [M, K] row-major
                           const float* __restrict__ B,  // [K, N] row-major
                           const float* __restrict__ bias,  // [N] or null
                           float* __restrict__ C,        // [M, N]
                           int M, int K, int N, float scale, float alpha) {
    // double-buffered LDS, 2-phase pipeline (guide T3 minimum form +
    // T14 reg-staging): per K-tile, issue next tile's global loads into
    // registers FIRST, run the MFMA phase on the current buffer (hides
    // the ~900-cycle HBM latency), then ds_write the landed registers
    // into the other buffer; ONE barrier per tile.
    __shared__ float As[2][BM][BK + APAD];
    __shared__ float Bs[2][BK][BN];

    const unsigned nbn = (unsigned)ks_ceil_div(N, BN);
    const unsigned nwg = nbn * (unsigned)ks_ceil_div(M, BM);
    unsigned wg = ks_xcd_swizzle(blockIdx.x, nwg);
    const int brow = (int)(wg / nbn) * BM;
    const int bcol = (int)(wg % nbn) * BN;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = wave >> 1, wc = wave & 1;   // wave's 64x64 quadrant
    const int l31 = lane & 31;
    const int lk = lane >> 5;                  // 0/1: k offset inside MFMA

    // per-thread staging geometry (same for every tile)
    const int a_kv = (tid & 7) * 4;            // A: k offset 0..28
    const int a_r0 = tid >> 3;                 // A: row 0..31 (x4 passes)
    const int b_nv = (tid & 31) * 4;           // B: col offset 0..124
    const int b_k0 = tid >> 5;                 // B: k row 0..7 (x4 passes)

    float4 ra[4], rb[4];

    auto stage_load = [&](int k0) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int gr = brow + a_r0 + p * 32;
            const int gk = k0 + a_kv;
            float4 v = {0, 0, 0, 0};
            if (gr < M) {
                if (gk + 3 < K) {
                    v = *reinterpret_cast<const float4*>(&A[(size_t)gr * K + gk]);
                } else {
#pragma unroll
                    for (int q = 0; q < 4; ++q)
                        if (gk + q < K) ((float*)&v)[q] = A[(size_t)gr * K + gk + q];
                }
            }
            ra[p] = v;
        }
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int gk = k0 + b_k0 + p * 8;
            const int gn = bcol + b_nv;
            float4 v = {0, 0, 0, 0};
            if (gk < K) {
                if (gn + 3 < N) {
                    v = *reinterpret_cast<const float4*>(&B[(size_t)gk * N + gn]);
                } else {
#pragma unroll
                    for (int q = 0; q < 4; ++q)
                        if (gn + q < N) ((float*)&v)[q] = B[(size_t)gk * N + gn + q];
                }
            }
            rb[p] = v;
        }
    };

    auto stage_write = [&](int buf) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int row = a_r0 + p * 32;
            As[buf][row][a_kv + 0] = ra[p].x;
            As[buf][row][a_kv + 1] = ra[p].y;
            As[buf][row][a_kv + 2] = ra[p].z;
            As[buf][row][a_kv + 3] = ra[p].w;
            *reinterpret_cast<float4*>(&Bs[buf][b_k0 + p * 8][b_nv]) = rb[p];
        }
    };

    f32x16 acc[2][2];
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.0f);

    const int ntiles = ks_ceil_div(K, BK);
    stage_load(0);
    stage_write(0);
    __syncthreads();

    int cur = 0;
    for (int t = 0; t < ntiles; ++t) {
        if (t + 1 < ntiles) stage_load((t + 1) * BK);  // async under MFMA
        // ---- MFMA inner loop: 16 k-steps of K=2
#pragma unroll
        for (int kk = 0; kk < BK; kk += 2) {
            float a0 = As[cur][wr * 64 + 0 + l31][kk + lk];
            float a1 = As[cur][wr * 64 + 32 + l31][kk + lk];
            float b0 = Bs[cur][kk + lk][wc * 64 + 0 + l31];
            float b1 = Bs[cur][kk + lk][wc * 64 + 32 + l31];
            acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
            acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
            acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
            acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
        }
        if (t + 1 < ntiles) stage_write(cur ^ 1);      // vmcnt at first use
        __syncthreads();
        cur ^= 1;
    }

    // ---- epilogue: C/D layout col = lane&31, row = (r&3) + 8*(r>>2) + 4*(lane>>5)
#pragma unroll
    for (int ti = 0; ti < 2; ++ti) {
#pragma unroll
        for (int tj = 0; tj < 2; ++tj) {
            const int col = bcol + wc * 64 + tj * 32 + l31;
            if (col >= N) continue;
            const float bv = (EPI != EP_NONE && bias != nullptr) ? bias[col] : 0.0f;
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = brow + wr * 64 + ti * 32 + (r & 3) + 8 * (r >> 2) + 4 * lk;
                if (row >= M) continue;
                float v = acc[ti][tj][r] + bv;
                if (EPI == EP_COS) v = __cosf(v) * scale;
                else if (EPI == EP_RELU) v = fmaxf(v - alpha, 0.0f) * scale;
                C[(size_t)row * N + col] = v;
            }
        }
    }
}

// --------------------------------------------------------------------------
// 256x256 tile, 8 waves (2M x 4N), BK=32 — the big-shape variant.
// f32 MFMA GEMM at a 128^2 tile is HBM-bound (32 flops/byte -> needs
// 4.8 TB/s of the ~6.3 achievable); 256^2 doubles arithmetic intensity.
// 1 block/CU (130 KB LDS), 2 waves/SIMD; same 2-phase reg-staged pipeline.
// --------------------------------------------------------------------------
#define BM2 256
#define BN2 256

template <int EPI>
__global__ __launch_bounds__(512, 2)
void gemm_fused_f32_256_kernel(const float* __restrict__ A,
                               const float* __restrict__ B,
                               const float* __restrict__ bias,
                               float* __restrict__ C,
                               int M, int K, int N, float scale, float alpha) {
    __shared__ float As[2][BM2][BK + APAD];
    __shared__ float Bs[2][BK][BN2];

    const unsigned nbn = (unsigned)ks_ceil_div(N, BN2);
    const unsigned nwg = nbn * (unsigned)ks_ceil_div(M, BM2);
    unsigned wg = ks_xcd_swizzle(blockIdx.x, nwg);
    const int brow = (int)(wg / nbn) * BM2;
    const int bcol = (int)(wg % nbn) * BN2;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = wave >> 2;                 // 0..1: 128-row half
    const int wc = wave & 3;                  // 0..3: 64-col quarter
    const int l31 = lane & 31;
    const int lk = lane >> 5;

    const int a_kv = (tid & 7) * 4;           // k offset
    const int a_r0 = tid >> 3;                // row 0..63 (x4 passes)
    const int b_nv = (tid & 63) * 4;          // col offset
    const int b_k0 = tid >> 6;                // k row 0..7 (x4 passes)

    float4 ra[4], rb[4];

    auto load_a_piece = [&](int k0, int p) -> float4 {
        const int gr = brow + a_r0 + p * 64;
        const int gk = k0 + a_kv;
        float4 v = {0, 0, 0, 0};
        if (gr < M) {
            if (gk + 3 < K) {
                v = *reinterpret_cast<const float4*>(&A[(size_t)gr * K + gk]);
            } else {
#pragma unroll
                for (int q = 0; q < 4; ++q)
                    if (gk + q < K) ((float*)&v)[q] = A[(size_t)gr * K + gk + q];
            }
        }
        return v;
    };
    auto load_b_piece = [&](int k0, int p) -> float4 {
        const int gk = k0 + b_k0 + p * 8;
        const int gn = bcol + b_nv;
        float4 v = {0, 0, 0, 0};
        if (gk < K) {
            if (gn + 3 < N) {
                v = *reinterpret_cast<const float4*>(&B[(size_t)gk * N + gn]);
            } else {
#pragma unroll
                for (int q = 0; q < 4; ++q)
                    if (gn + q < N) ((float*)&v)[q] = B[(size_t)gk * N + gn + q];
            }
        }
        return v;
    };
    auto write_a_piece = [&](int buf, int p) {
        const int row = a_r0 + p * 64;
        As[buf][row][a_kv + 0] = ra[p].x;
        As[buf][row][a_kv + 1] = ra[p].y;
        As[buf][row][a_kv + 2] = ra[p].z;
        As[buf][row][a_kv + 3] = ra[p].w;
    };
    auto write_b_piece = [&](int buf, int p) {
        *reinterpret_cast<float4*>(&Bs[buf][b_k0 + p * 8][b_nv]) = rb[p];
    };
    auto stage_load = [&](int k0) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            ra[p] = load_a_piece(k0, p);
            rb[p] = load_b_piece(k0, p);
        }
    };
    auto stage_write = [&](int buf) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            write_a_piece(buf, p);
            write_b_piece(buf, p);
        }
    };

    f32x16 acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.0f);

    const int ntiles = ks_ceil_div(K, BK);
    stage_load(0);
    stage_write(0);
    __syncthreads();

    // phase-interleaved k-loop: the ablation probe (gemm_ablate.hip)
    // measures the MFMA-only structure at 150 TF and the block staging at
    // ~7x real-time, so the 120-TF full kernel loses ~20% purely to the
    // staging segment draining the matrix pipe.  Spreading the 8 load
    // pieces over phases 0-3 and the 8 write pieces over phases 4-7 (each
    // beside 16 MFMAs) lets the co-resident wave's MFMAs cover them, and
    // the load->write distance (~8 k-steps) covers the HBM latency.
    int cur = 0;
    for (int t = 0; t < ntiles; ++t) {
        const bool more = t + 1 < ntiles;
#pragma unroll
        for (int ph = 0; ph < 8; ++ph) {
            if (more && ph < 4) {
                ra[ph] = load_a_piece((t + 1) * BK, ph);
                rb[ph] = load_b_piece((t + 1) * BK, ph);
            }
#pragma unroll
            for (int kk = 4 * ph; kk < 4 * ph + 4; kk += 2) {
                float a0 = As[cur][wr * 128 + 0 + l31][kk + lk];
                float a1 = As[cur][wr * 128 + 32 + l31][kk + lk];
                float a2 = As[cur][wr * 128 + 64 + l31][kk + lk];
                float a3 = As[cur][wr * 128 + 96 + l31][kk + lk];
                float b0 = Bs[cur][kk + lk][wc * 64 + 0 + l31];
                float b1 = Bs[cur][kk + lk][wc * 64 + 32 + l31];
                acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
                acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
                acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
                acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
                acc[2][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a2, b0, acc[2][0], 0, 0, 0);
                acc[2][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a2, b1, acc[2][1], 0, 0, 0);
                acc[3][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a3, b0, acc[3][0], 0, 0, 0);
                acc[3][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a3, b1, acc[3][1], 0, 0, 0);
            }
            if (more && ph >= 4) {
                write_a_piece(cur ^ 1, ph - 4);
                write_b_piece(cur ^ 1, ph - 4);
            }
        }
        __syncthreads();
        cur ^= 1;
    }

#pragma unroll
    for (int ti = 0; ti < 4; ++ti) {
#pragma unroll
        for (int tj = 0; tj < 2; ++tj) {
            const int col = bcol + wc * 64 + tj * 32 + l31;
            if (col >= N) continue;
            const float bv = (EPI != EP_NONE && bias != nullptr) ? bias[col] : 0.0f;
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = brow + wr * 128 + ti * 32 + (r & 3) + 8 * (r >> 2) + 4 * lk;
                if (row >= M) continue;
                float v = acc[ti][tj][r] + bv;
                if (EPI == EP_COS) v = __cosf(v) * scale;
                else if (EPI == EP_RELU) v = fmaxf(v - alpha, 0.0f) * scale;
                C[(size_t)row * N + col] = v;
            }
        }
    }
}

static int launch_gemm_fused(const float* A, const float* B, const float* bias,
                             float* C, int M, int K, int N, float scale,
                             float alpha, int epilogue, hipStream_t stream) {
    // big shapes go to the 256^2 tile (2x arithmetic intensity; the 128^2
    // tile is HBM-bound) — but only when the 256-tile grid still fills the
    // 256-CU chip (a launch needs >= #CUs workgroups).
    const int nwg256 = ks_ceil_div(M, BM2) * ks_ceil_div(N, BN2);
    const bool big = (M >= 512 && N >= 512 && nwg256 >= 256);
    const int nwg = big ? nwg256 : ks_ceil_div(M, BM) * ks_ceil_div(N, BN);
    dim3 grid(nwg), block(big ? 512 : 256);
#define KS_LAUNCH(EPI_)                                                     \
    do {                                                                    \
        if (big)                                                            \
            hipLaunchKernelGGL(gemm_fused_f32_256_kernel<EPI_>, grid, block, \
                               0, stream, A, B, bias, C, M, K, N, scale,    \
                               alpha);                                      \
        else                                                                \
            hipLaunchKernelGGL(gemm_fused_f32_kernel<EPI_>, grid, block, 0, \
                               stream, A, B, bias, C, M, K, N, scale,       \
                               alpha);                                      \
    } while (0)
    switch (epilogue) {
        case EP_NONE: KS_LAUNCH(EP_NONE); break;
        case EP_BIAS: KS_LAUNCH(EP_BIAS); break;
        case EP_COS:  KS_LAUNCH(EP_COS); break;
        case EP_RELU: KS_LAUNCH(EP_RELU); break;
        default: return (int)hipErrorInvalidValue;
    }
#undef KS_LAUNCH
    return (int)hipGetLastError();
}

KS_EXPORT int ks_gemm_fused_f32(const float* A, const float* B,
                                const float* bias, float* C, int M, int K,
                                int N, float scale, float alpha, int epilogue,
                                hipStream_t stream) {
    return launch_gemm_fused(A, B, bias, C, M, K, N, scale, alpha, epilogue,
                             stream);
}
