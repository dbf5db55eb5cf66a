// Ablation probe for the 256x256 f32 MFMA GEMM structure (not shipped in
// the hot path): VARIANT 0 = full kernel, 1 = MFMA-only (tile staged once;
// k-loop runs MFMAs on stale LDS — numerically wrong, perf ceiling of the
// compute phase), 2 = staging-only (no MFMA; asm keep-alive prevents DCE).
#include "common.h"

#define ABM 256
#define ABN 256
#define ABK 32
#define ABPAD 1

template <int VARIANT>
__global__ __launch_bounds__(512, 2)
void gemm_ablate_kernel(const float* __restrict__ A,
                        const float* __restrict__ B,
                        float* __restrict__ C, int M, int K, int N) {
    __shared__ float As[2][ABM][ABK + ABPAD];
    __shared__ float Bs[2][ABK][ABN];

    const unsigned nbn = (unsigned)ks_ceil_div(N, ABN);
    unsigned wg = ks_xcd_swizzle(blockIdx.x,
                                 nbn * (unsigned)ks_ceil_div(M, ABM));
    const int brow = (int)(wg / nbn) * ABM;
    const int bcol = (int)(wg % nbn) * ABN;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = wave >> 2, wc = wave & 3;
    const int l31 = lane & 31;
    const int lk = lane >> 5;

    const int a_kv = (tid & 7) * 4;
    const int a_r0 = tid >> 3;
    const int b_nv = (tid & 63) * 4;
    const int b_k0 = tid >> 6;

    float4 ra[4], rb[4];
    auto stage_load = [&](int k0) {
#pragma unroll
        for (int p = 0; p < 4; ++p)
            ra[p] = *reinterpret_cast<const float4*>(
                &A[(size_t)(brow + a_r0 + p * 64) * K + k0 + a_kv]);
#pragma unroll
        for (int p = 0; p < 4; ++p)
            rb[p] = *reinterpret_cast<const float4*>(
                &B[(size_t)(k0 + b_k0 + p * 8) * N + bcol + b_nv]);
    };
    auto stage_write = [&](int buf) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            const int row = a_r0 + p * 64;
            As[buf][row][a_kv + 0] = ra[p].x;
            As[buf][row][a_kv + 1] = ra[p].y;
            As[buf][row][a_kv + 2] = ra[p].z;
            As[buf][row][a_kv + 3] = ra[p].w;
            *reinterpret_cast<float4*>(&Bs[buf][b_k0 + p * 8][b_nv]) = rb[p];
        }
    };

    f32x16 acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j) acc[i][j] = (f32x16)(0.0f);

    const int ntiles = K / ABK;
    stage_load(0);
    stage_write(0);
    __syncthreads();

    int cur = 0;
    for (int t = 0; t < ntiles; ++t) {
        if (VARIANT != 1 && t + 1 < ntiles) stage_load((t + 1) * ABK);
        if (VARIANT != 2) {
#pragma unroll
            for (int kk = 0; kk < ABK; kk += 2) {
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
        } else {
            // keep staged registers live (rule 17: skip-DCE)
            asm volatile("" :: "v"(ra[0].x), "v"(rb[0].x));
        }
        if (VARIANT != 1 && t + 1 < ntiles) stage_write(cur ^ 1);
        __syncthreads();
        cur ^= 1;
    }

#pragma unroll
    for (int ti = 0; ti < 4; ++ti)
#pragma unroll
        for (int tj = 0; tj < 2; ++tj) {
            const int col = bcol + wc * 64 + tj * 32 + l31;
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = brow + wr * 128 + ti * 32
                                + (r & 3) + 8 * (r >> 2) + 4 * lk;
                C[(size_t)row * N + col] = acc[ti][tj][r];
            }
        }
}

KS_EXPORT int ks_gemm_ablate_f32(const float* A, const float* B, float* C,
                                 int M, int K, int N, int variant,
                                 hipStream_t stream) {
    dim3 grid(ks_ceil_div(M, ABM) * ks_ceil_div(N, ABN)), block(512);
    switch (variant) {
        case 0: hipLaunchKernelGGL(gemm_ablate_kernel<0>, grid, block, 0, stream, A, B, C, M, K, N); break;
        case 1: hipLaunchKernelGGL(gemm_ablate_kernel<1>, grid, block, 0, stream, A, B, C, M, K, N); break;
        case 2: hipLaunchKernelGGL(gemm_ablate_kernel<2>, grid, block, 0, stream, A, B, C, M, K, N); break;
        default: return (int)hipErrorInvalidValue;
    }
    return (int)hipGetLastError();
}
