// K2/K3: fused GMM posterior + Fisher-Vector kernels (gfx950).
//
// ref: EncEval GMM (K2 — E-step posteriors) and EncEval
// Fisher Vector (K3 — fused posterior + moment accumulation, one workgroup
// per image, LDS-resident GMM params).
//
// Both kernels use the expanded diagonal-Gaussian form with
// host-precomputed coefficients:
//   logit_k(x) = -sum_d x_d^2 a_kd + sum_d x_d b_kd - c_k
//   a = 1/(2 var), b = mu/var,
//   c_k = sum_d mu^2/(2var) + 0.5 sum_d log var + D/2 log 2pi - log w_k
// Posteriors are a row softmax over K.  The per-row logits are computed
// TWICE (max pass + exp pass) instead of stored — at 2KD flops per row the
// recompute is cheaper than spilling K registers (guide §6 G6, rule 20).
//
// Shape support (covers the FV pipeline configs): D <= 64, K <= 128,
// K % 16 == 0 for the FV kernel.  Larger shapes fall back to the batched
// GEMM formulation in Python (ops/images/fisher_vector.py).

#include "common.h"
#include <math.h>

#define ROWS_PER_TILE 128

// --------------------------------------------------------------------------
// K2: gamma[n, K] = softmax_k(logit_k(x_row)), plus sum of per-row
// logsumexp into ll_out[0] (block-reduced atomics).
// One thread per row; X tile staged in LDS; params in LDS.
// --------------------------------------------------------------------------
__global__ __launch_bounds__(256)
void gmm_posteriors_kernel(const float* __restrict__ X,   // [n, D]
                           const float* __restrict__ a,   // [K, D]
                           const float* __restrict__ b,   // [K, D]
                           const float* __restrict__ c,   // [K]
                           float* __restrict__ gamma,     // [n, K]
                           float* __restrict__ ll_out,    // [1] or null
                           int n, int D, int K) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* As = (float*)smem;              // [K, D]
    float* Bs = As + K * D;                // [K, D]
    float* Cs = Bs + K * D;                // [K]
    float* Xs = Cs + K;                    // [ROWS_PER_TILE, D]

    for (int i = threadIdx.x; i < K * D; i += 256) {
        As[i] = a[i];
        Bs[i] = b[i];
    }
    for (int i = threadIdx.x; i < K; i += 256) Cs[i] = c[i];
    __syncthreads();

    float ll_local = 0.0f;
    const int tiles = (n + ROWS_PER_TILE - 1) / ROWS_PER_TILE;
    for (int tile = blockIdx.x; tile < tiles; tile += gridDim.x) {
        const int row0 = tile * ROWS_PER_TILE;
        const int rows = min(ROWS_PER_TILE, n - row0);
        // stage X rows (coalesced float copy)
        __syncthreads();
        for (int i = threadIdx.x; i < rows * D; i += 256)
            Xs[i] = X[(size_t)row0 * D + i];
        __syncthreads();
        // threads 0..rows-1 each own a row; others idle this tile
        if ((int)threadIdx.x < rows) {
            const float* xr = &Xs[threadIdx.x * D];
            // pass 1: max logit
            float m = -1e30f;
            for (int k = 0; k < K; ++k) {
                const float* ak = &As[k * D];
                const float* bk = &Bs[k * D];
                float t = -Cs[k];
                for (int d = 0; d < D; ++d) {
                    float x = xr[d];
                    t += x * (bk[d] - x * ak[d]);
                }
                m = fmaxf(m, t);
            }
            // pass 2: exp-sum, write normalized
            float s = 0.0f;
            float* grow = &gamma[(size_t)(row0 + threadIdx.x) * K];
            for (int k = 0; k < K; ++k) {
                const float* ak = &As[k * D];
                const float* bk = &Bs[k * D];
                float t = -Cs[k];
                for (int d = 0; d < D; ++d) {
                    float x = xr[d];
                    t += x * (bk[d] - x * ak[d]);
                }
                float e = __expf(t - m);
                grow[k] = e;
                s += e;
            }
            float inv = 1.0f / s;
            for (int k = 0; k < K; ++k) grow[k] *= inv;
            ll_local += m + __logf(s);
        }
    }
    if (ll_out != nullptr) {
        // block-reduce ll and one atomic per block
        __shared__ float red[4];
        float v = ll_local;
#pragma unroll
        for (int off = 32; off; off >>= 1) v += __shfl_down(v, off, 64);
        if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = v;
        __syncthreads();
        if (threadIdx.x == 0)
            atomicAdd(ll_out, red[0] + red[1] + red[2] + red[3]);
    }
}

KS_EXPORT int ks_gmm_posteriors_f32(const float* X, const float* a,
                                    const float* b, const float* c,
                                    float* gamma, float* ll_out, int n,
                                    int D, int K, hipStream_t stream) {
    if (D > 64 || K > 128) return (int)hipErrorInvalidValue;
    int tiles = ks_ceil_div(n, ROWS_PER_TILE);
    int blocks = tiles < 2048 ? tiles : 2048;
    size_t lds = (size_t)(2 * K * D + K + ROWS_PER_TILE * D) * 4;
    hipLaunchKernelGGL(gmm_posteriors_kernel, dim3(blocks), dim3(256), lds,
                       stream, X, a, b, c, gamma, ll_out, n, D, K);
    return (int)hipGetLastError();
}

// --------------------------------------------------------------------------
// K3: Fisher Vector, one workgroup per image.
//   out[b] = [ G_mu (K*D) ; G_sigma (K*D) ]
// Phase A (per 128-descriptor chunk): thread t computes descriptor t's
// posteriors into LDS gamma[128][K] (two-pass softmax as above).
// Phase B: 4 waves split K; wave w, lane d accumulates its k-slice of
//   S0[k] = sum gamma,  S1[k][d] = sum gamma x_d,  S2[k][d] = sum gamma x_d^2
// in REGISTERS over the chunk (xd LDS reads are lane-consecutive ->
// conflict-free; gamma reads are wave-uniform -> broadcast).
// Epilogue converts (S0, S1, S2) to the improved-FV form and stores.
// --------------------------------------------------------------------------
template <int KPW>  // K per wave = K/4, compile-time for register indexing
__global__ __launch_bounds__(256)
void fisher_vector_kernel(const float* __restrict__ X,   // [B, n, D]
                          const float* __restrict__ a,   // [K, D]
                          const float* __restrict__ bco, // [K, D]
                          const float* __restrict__ c,   // [K]
                          const float* __restrict__ mu,  // [K, D]
                          const float* __restrict__ var, // [K, D]
                          const float* __restrict__ w,   // [K]
                          float* __restrict__ out,       // [B, 2*K*D]
                          int n, int D) {
    const int K = KPW * 4;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* As = (float*)smem;             // [K, D]
    float* Bs = As + K * D;               // [K, D]
    float* Cs = Bs + K * D;               // [K]
    float* Xs = Cs + K;                   // [128, D]
    float* Gs = Xs + 128 * D;             // [128, K] gamma chunk

    const int img = blockIdx.x;
    const float* Xi = X + (size_t)img * n * D;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;

    for (int i = tid; i < K * D; i += 256) {
        As[i] = a[i];
        Bs[i] = bco[i];
    }
    for (int i = tid; i < K; i += 256) Cs[i] = c[i];

    float S0[KPW];
    float S1[KPW];
    float S2[KPW];
#pragma unroll
    for (int k = 0; k < KPW; ++k) S0[k] = S1[k] = S2[k] = 0.0f;

    for (int chunk0 = 0; chunk0 < n; chunk0 += 128) {
        const int rows = min(128, n - chunk0);
        __syncthreads();
        for (int i = tid; i < rows * D; i += 256)
            Xs[i] = Xi[(size_t)chunk0 * D + i];
        __syncthreads();
        // ---- phase A: posteriors for this chunk (threads 0..rows-1)
        if (tid < rows) {
            const float* xr = &Xs[tid * D];
            float m = -1e30f;
            for (int k = 0; k < K; ++k) {
                const float* ak = &As[k * D];
                const float* bk = &Bs[k * D];
                float t = -Cs[k];
                for (int d = 0; d < D; ++d) {
                    float x = xr[d];
                    t += x * (bk[d] - x * ak[d]);
                }
                m = fmaxf(m, t);
            }
            float s = 0.0f;
            for (int k = 0; k < K; ++k) {
                const float* ak = &As[k * D];
                const float* bk = &Bs[k * D];
                float t = -Cs[k];
                for (int d = 0; d < D; ++d) {
                    float x = xr[d];
                    t += x * (bk[d] - x * ak[d]);
                }
                float e = __expf(t - m);
                Gs[tid * K + k] = e;
                s += e;
            }
            float inv = 1.0f / s;
            for (int k = 0; k < K; ++k) Gs[tid * K + k] *= inv;
        }
        __syncthreads();
        // ---- phase B: accumulate this wave's k-slice; lane = d column
        const int k0 = wave * KPW;
        if (lane < D) {
            for (int src = 0; src < rows; ++src) {
                const float xd = Xs[src * D + lane];
                const float* gs = &Gs[src * K + k0];
#pragma unroll
                for (int k = 0; k < KPW; ++k) {
                    const float g = gs[k];          // wave-uniform broadcast
                    S0[k] += g;
                    S1[k] += g * xd;
                    S2[k] += g * xd * xd;
                }
            }
        }
    }
    __syncthreads();
    // ---- epilogue: improved-FV scaling, write G_mu then G_sigma
    //   G_mu[k,d]  = (S1 - mu S0) / (sig * n * sqrt(w))
    //   G_sig[k,d] = ((S2 - 2 mu S1 + mu^2 S0)/var - S0) / (n * sqrt(2w))
    if (lane < D) {
        const int k0 = wave * KPW;
        float* omu = out + (size_t)img * (2 * K * D);
        float* osig = omu + K * D;
#pragma unroll
        for (int k = 0; k < KPW; ++k) {
            const int kk = k0 + k;
            const float mkd = mu[kk * D + lane];
            const float vkd = var[kk * D + lane];
            const float wk = w[kk];
            const float sig = sqrtf(vkd);
            const float gm = (S1[k] - mkd * S0[k]) /
                             (sig * n * sqrtf(wk));
            const float gs = ((S2[k] - 2.0f * mkd * S1[k] + mkd * mkd * S0[k])
                              / vkd - S0[k]) / (n * sqrtf(2.0f * wk));
            omu[kk * D + lane] = gm;
            osig[kk * D + lane] = gs;
        }
    }
}

KS_EXPORT int ks_fisher_vector_f32(const float* X, const float* a,
                                   const float* b, const float* c,
                                   const float* mu, const float* var,
                                   const float* w, float* out, int B, int n,
                                   int D, int K, hipStream_t stream) {
    if (D > 64 || K > 128 || (K % 16) != 0) return (int)hipErrorInvalidValue;
    size_t lds = (size_t)(2 * K * D + K + 128 * D + 128 * K) * 4;
    if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
    dim3 grid(B), block(256);
    switch (K / 4) {
        case 4:  hipLaunchKernelGGL(fisher_vector_kernel<4>, grid, block, lds, stream, X, a, b, c, mu, var, w, out, n, D); break;
        case 8:  hipLaunchKernelGGL(fisher_vector_kernel<8>, grid, block, lds, stream, X, a, b, c, mu, var, w, out, n, D); break;
        case 12: hipLaunchKernelGGL(fisher_vector_kernel<12>, grid, block, lds, stream, X, a, b, c, mu, var, w, out, n, D); break;
        case 16: hipLaunchKernelGGL(fisher_vector_kernel<16>, grid, block, lds, stream, X, a, b, c, mu, var, w, out, n, D); break;
        case 24: hipLaunchKernelGGL(fisher_vector_kernel<24>, grid, block, lds, stream, X, a, b, c, mu, var, w, out, n, D); break;
        case 32: hipLaunchKernelGGL(fisher_vector_kernel<32>, grid, block, lds, stream, X, a, b, c, mu, var, w, out, n, D); break;
        default: return (int)hipErrorInvalidValue;
    }
    return (int)hipGetLastError();
}
