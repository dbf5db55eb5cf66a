// Empirical layout probe for v_mfma_f32_32x32x16_f16 (gfx950).
//
// Assumed mapping (to verify before the fv_mfma f16-stats rewrite):
//   A[m=32][k=16]: lane l supplies A[l&31][8*(l>>5) + j], j = 0..7
//   B[k=16][n=32]: lane l supplies B[8*(l>>5) + j][l&31]
//   D[m][n]:       col = l&31, row = (reg&3) + 8*(reg>>2) + 4*(l>>5)
// One wavefront computes D = A @ B and writes it row-major; the host
// compares against a CPU reference with asymmetric A and B.

#include "common.h"

typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));

__global__ __launch_bounds__(64)
void probe_mfma16_kernel(const _Float16* __restrict__ A,  // [32][16] row-major
                         const _Float16* __restrict__ B,  // [16][32] row-major
                         float* __restrict__ D) {         // [32][32] row-major
    const int l = threadIdx.x;
    const int l31 = l & 31;
    const int lk = l >> 5;
    f16x8 a, b;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        a[j] = A[l31 * 16 + 8 * lk + j];
        b[j] = B[(8 * lk + j) * 32 + l31];
    }
    f32x16 c = (f32x16)(0.0f);
    c = __builtin_amdgcn_mfma_f32_32x32x16_f16(a, b, c, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
        const int row = (r & 3) + 8 * (r >> 2) + 4 * lk;
        D[row * 32 + l31] = c[r];
    }
}

KS_EXPORT int ks_probe_mfma16(const void* A, const void* B, float* D,
                              hipStream_t stream) {
    hipLaunchKernelGGL(probe_mfma16_kernel, dim3(1), dim3(64), 0, stream,
                       (const _Float16*)A, (const _Float16*)B, D);
    return (int)hipGetLastError();
}
