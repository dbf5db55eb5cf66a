// Banded triangular pooling kernels (gfx950) — the SIFT/LCS gcd-lattice
// spatial pooling without the 36x-waste dense GEMM.
//
// The pooling matrix M [W, Q] has only 2*bin-1 nonzeros per column (a
// triangle window at center bin/2 + q*g).  hipBLASLt runs the dense
// [B*8*H, W] @ [W, Q] form at its flop rate (~10 ms at bench scale-2),
// but the useful flops are 15/256 of that: these kernels evaluate the
// band directly and run at the HBM floor instead.
//
//   ks_pool_lastdim_f32: in [R, W] -> out [R, Q]   (pool the last dim)
//   ks_pool_middim_f32:  in [B, H, C] -> out [B, C, Qy]
//     (pool the middle dim AND transpose so qy lands contiguous — the
//      layout the descriptor-assembly gather wants)
//
// Window: out = sum_t tap[t] * in[c_q - (L-1)/2 + t], zero boundary,
// centers c_q = c0 + q*g, tap length L = 2*bin - 1.
//
// ref: VLFeat dsift spatial binning — redesigned as
// lattice pooling kernels.

#include "common.h"

#define PMAXL 97   // max tap length (bin <= 49)

__global__ __launch_bounds__(256)
void pool_lastdim_kernel(const float* __restrict__ in,   // [R, W]
                         float* __restrict__ out,        // [R, Q]
                         const float* __restrict__ tap,  // [L]
                         int L, int c0, int g, int W, int Q, long R) {
    // 4 waves per block, one row per wave; row staged in LDS
    const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
    const int lane = threadIdx.x & 63;
    const int wv = threadIdx.x >> 6;
    extern __shared__ float sm[];
    float* taps = sm;                 // [PMAXL]
    float* rows = sm + PMAXL;         // [4][W]
    for (int t = threadIdx.x; t < L; t += 256) taps[t] = tap[t];
    if (row < R) {
        const float* ir = in + row * W;
        for (int c = lane; c < W; c += 64) rows[wv * W + c] = ir[c];
    }
    __syncthreads();
    if (row >= R) return;
    const float* r = rows + wv * W;
    const int half = (L - 1) / 2;
    for (int q = lane; q < Q; q += 64) {
        const int base = c0 + q * g - half;
        float s = 0.0f;
        const int t0 = max(0, -base);
        const int t1 = min(L, W - base);
        for (int t = t0; t < t1; ++t) s += taps[t] * r[base + t];
        out[row * Q + q] = s;
    }
}

KS_EXPORT int ks_pool_lastdim_f32(const float* in, float* out,
                                  const float* tap, int L, int c0, int g,
                                  int W, int Q, long R, hipStream_t stream) {
    if (L > PMAXL) return (int)hipErrorInvalidValue;
    long blocks = (R + 3) / 4;
    size_t lds = (size_t)(PMAXL + 4 * W) * 4;
    if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
    hipLaunchKernelGGL(pool_lastdim_kernel, dim3((unsigned)blocks), dim3(256),
                       lds, stream, in, out, tap, L, c0, g, W, Q, R);
    return (int)hipGetLastError();
}

// pool the middle dim of [B, H, C] and TRANSPOSE: out [B, C, Qy].
// One block per (b, 32-column tile): the [H, 32] tile is staged in LDS,
// threads map (2 columns x 128 qy-lanes) so the out writes coalesce.
#define MCT 32

__global__ __launch_bounds__(256)
void pool_middim_kernel(const float* __restrict__ in,   // [B, H, C]
                        float* __restrict__ out,        // [B, C, Qy]
                        const float* __restrict__ tap,  // [L]
                        int L, int c0, int g, int H, int C, int Qy) {
    const int b = blockIdx.x;
    const int ct0 = blockIdx.y * MCT;
    const int cw = min(MCT, C - ct0);
    extern __shared__ float sm[];
    float* taps = sm;                  // [PMAXL]
    float* tile = sm + PMAXL;          // [H][MCT + 1] (pad: without it
    // every lane of the q-sweep reads column c of the tile -> all 64
    // lanes land on ONE bank; measured 25.5 ms vs 2.2 with the pad)
    for (int t = threadIdx.x; t < L; t += 256) taps[t] = tap[t];
    const float* ib = in + ((size_t)b * H) * C;
    for (int i = threadIdx.x; i < H * MCT; i += 256) {
        const int h = i / MCT, c = i % MCT;
        tile[h * (MCT + 1) + c] = (c < cw)
            ? ib[(size_t)h * C + ct0 + c] : 0.0f;
    }
    __syncthreads();
    const int half = (L - 1) / 2;
    // thread -> (column pair, qy lane): lanes sweep qy so writes coalesce
    const int cl = threadIdx.x >> 7;          // 0/1
    const int ql = threadIdx.x & 127;
    for (int c = cl; c < cw; c += 2) {
        for (int q = ql; q < Qy; q += 128) {
            const int base = c0 + q * g - half;
            float s = 0.0f;
            const int t0 = max(0, -base);
            const int t1 = min(L, H - base);
            for (int t = t0; t < t1; ++t)
                s += taps[t] * tile[(base + t) * (MCT + 1) + c];
            out[((size_t)b * C + ct0 + c) * Qy + q] = s;
        }
    }
}

KS_EXPORT int ks_pool_middim_f32(const float* in, float* out,
                                 const float* tap, int L, int c0, int g,
                                 int H, int C, int Qy, int B,
                                 hipStream_t stream) {
    if (L > PMAXL) return (int)hipErrorInvalidValue;
    size_t lds = (size_t)(PMAXL + H * (MCT + 1)) * 4;
    if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
    dim3 grid(B, (C + MCT - 1) / MCT);
    hipLaunchKernelGGL(pool_middim_kernel, grid, dim3(256), lds, stream,
                       in, out, tap, L, c0, g, H, C, Qy);
    return (int)hipGetLastError();
}
