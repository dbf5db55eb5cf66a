// GMM EM M-step + next-iteration coefficient build, fused into ONE
// launch (gfx950).
//
// The fused E-step (fv_mfma.hip STATS mode) leaves per-iteration partial
// sufficient statistics out[b] = [S0 (K) | S1 (K*D) | S2 (K*D) | ll].
// Round 1 summed the partials and rebuilt (mu, var, w) and the E-step
// operand P with ~18 small torch launches per iteration (~0.7 ms of
// launch overhead at K=256, D=64 — comparable to the E-step kernel
// itself).  This kernel does all of it in one launch:
//
//   per component k (one 64-wide wavefront):
//     Nk  = sum_b S0[b,k];  Sx[d] = sum_b S1[b,k,d];  Sxx likewise
//     mu = Sx/Nk, var = clamp(Sxx/Nk - mu^2, min_var), w = Nk/n_tot
//     P[:,k] = [-1/(2var) ; mu/var ; -(c) ; 0] with
//     c = sum_d mu_d^2/(2 var_d) + 0.5 sum_d log var_d
//         + D/2 log(2 pi) - log w            (the fv_mfma operand)
//   block 0 additionally sums the ll partials into ll_out[0].
//
// Distributed mode (pre_reduced != nullptr): the host has already summed
// partials and all-reduced them across ranks in fp64; the kernel consumes
// that [K + 2KD + 1] fp64 vector instead of the [nb, *] fp32 partials, so
// multi-GPU EM takes exactly one extra all-reduce per iteration and the
// same two launches.
//
// ref: EncEval computeGMM — redesigned as a
// two-launch-per-iteration distributed EM rather than a port.

#include "common.h"
#include <math.h>

#define MSD 64   // max D

__global__ __launch_bounds__(64)
void gmm_mstep_kernel(const float* __restrict__ partials,  // [nb, W]
                      const double* __restrict__ pre,      // [W] or null
                      int nb, int K, int D, float n_tot, float min_var,
                      float* __restrict__ mu,    // [K, D]
                      float* __restrict__ var,   // [K, D]
                      float* __restrict__ w,     // [K]
                      float* __restrict__ P,     // [160, K] fv_mfma operand
                      float* __restrict__ ll) {  // [1]
    const int k = blockIdx.x;
    const int lane = threadIdx.x;
    const int W = K + 2 * K * D + 1;

    double Nk = 0.0, Sx = 0.0, Sxx = 0.0;
    if (pre != nullptr) {
        if (lane == 0) Nk = pre[k];
        if (lane < D) {
            Sx = pre[K + k * D + lane];
            Sxx = pre[K + K * D + k * D + lane];
        }
    } else {
        if (lane == 0)
            for (int b = 0; b < nb; ++b) Nk += (double)partials[(size_t)b * W + k];
        if (lane < D)
            for (int b = 0; b < nb; ++b) {
                const float* p = partials + (size_t)b * W;
                Sx += (double)p[K + k * D + lane];
                Sxx += (double)p[K + K * D + k * D + lane];
            }
    }
    Nk = __shfl(Nk, 0, 64);
    const double nk = fmax(Nk, 1e-10);
    const float wk = (float)(Nk / (double)n_tot);
    float m = 0.0f, v = 1.0f;
    if (lane < D) {
        m = (float)(Sx / nk);
        v = fmaxf((float)(Sxx / nk) - m * m, min_var);
        mu[(size_t)k * D + lane] = m;
        var[(size_t)k * D + lane] = v;
        const float inv2v = 0.5f / v;
        P[(size_t)lane * K + k] = -inv2v;                  // -1/(2v)
        P[(size_t)(MSD + lane) * K + k] = m / v;           // mu/v
    }
    if (lane == 0) w[k] = wk;
    // c-row reduction over d (lanes >= D contribute 0)
    float part = 0.0f;
    if (lane < D) part = m * m * (0.5f / v) + 0.5f * __logf(v);
#pragma unroll
    for (int off = 32; off; off >>= 1) part += __shfl_down(part, off, 64);
    if (lane == 0) {
        const float c = part + 0.5f * (float)D * logf(2.0f * (float)M_PI)
                        - __logf(fmaxf(wk, 1e-30f));
        P[(size_t)(2 * MSD) * K + k] = -c;
    }
    // zero the pad rows of this component's P column once per call
    for (int r = 2 * MSD + 1 + lane; r < 160; r += 64)
        P[(size_t)r * K + k] = 0.0f;
    if (D < MSD) {
        for (int r = D + lane; r < MSD; r += 64) {
            P[(size_t)r * K + k] = 0.0f;
            P[(size_t)(MSD + r) * K + k] = 0.0f;
        }
    }
    if (k == 0 && lane == 0) {
        double s = 0.0;
        if (pre != nullptr) s = pre[W - 1];
        else
            for (int b = 0; b < nb; ++b) s += (double)partials[(size_t)b * W + W - 1];
        ll[0] = (float)s;
    }
}

KS_EXPORT int ks_gmm_mstep_f32(const float* partials, const double* pre,
                               int nb, int K, int D, float n_tot,
                               float min_var, float* mu, float* var,
                               float* w, float* P, float* ll,
                               hipStream_t stream) {
    if (D > MSD) return (int)hipErrorInvalidValue;
    hipLaunchKernelGGL(gmm_mstep_kernel, dim3(K), dim3(64), 0, stream,
                       partials, pre, nb, K, D, n_tot, min_var, mu, var, w,
                       P, ll);
    return (int)hipGetLastError();
}
