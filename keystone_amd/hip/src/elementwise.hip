// Fused memory-bound row ops for the FV/feature pipelines.
//
// ref targets: SignedHellingerMapper -> NormalizeRows
// (the FV power + L2 normalization pair, applied to 32k–256k-dim rows) and
// SymmetricRectifier.  Eager torch runs these as 5–6 HBM passes; the fused
// kernels do 2 (hellinger+norm) or 1 (rectifier).
//
// CDNA4 notes: float4 loads (16 B/lane, G13); one workgroup per row with a
// grid-stride inner loop; wave shuffle + LDS reduction for the row norm.

#include "common.h"

// out[i] = sign(x)*sqrt(|x|) / ||sign(x)*sqrt(|x|)||_2  per row, one kernel.
__global__ __launch_bounds__(256)
void hellinger_l2_kernel(const float* __restrict__ X, float* __restrict__ out,
                         int rows, int cols, float eps) {
    __shared__ float red[4];
    const int row = blockIdx.x;
    if (row >= rows) return;
    const float* xr = X + (size_t)row * cols;
    float* orow = out + (size_t)row * cols;
    const int tid = threadIdx.x;

    // pass 1: y = sign*sqrt(|x|), write y, accumulate sum(y^2) (= sum|x|)
    float ss = 0.0f;
    int c = tid * 4;
    for (; c + 3 < cols; c += 256 * 4) {
        float4 v = *reinterpret_cast<const float4*>(&xr[c]);
        float4 y;
        y.x = copysignf(sqrtf(fabsf(v.x)), v.x);
        y.y = copysignf(sqrtf(fabsf(v.y)), v.y);
        y.z = copysignf(sqrtf(fabsf(v.z)), v.z);
        y.w = copysignf(sqrtf(fabsf(v.w)), v.w);
        ss += fabsf(v.x) + fabsf(v.y) + fabsf(v.z) + fabsf(v.w);
        *reinterpret_cast<float4*>(&orow[c]) = y;
    }
    for (; c < cols; ++c) {
        float v = xr[c];
        orow[c] = copysignf(sqrtf(fabsf(v)), v);
        ss += fabsf(v);
    }
    // wave + LDS reduce
#pragma unroll
    for (int off = 32; off; off >>= 1) ss += __shfl_down(ss, off, 64);
    if ((tid & 63) == 0) red[tid >> 6] = ss;
    __syncthreads();
    float total = red[0] + red[1] + red[2] + red[3];
    float inv = rsqrtf(fmaxf(total, eps));

    // pass 2: scale
    for (c = tid * 4; c + 3 < cols; c += 256 * 4) {
        float4 y = *reinterpret_cast<float4*>(&orow[c]);
        y.x *= inv; y.y *= inv; y.z *= inv; y.w *= inv;
        *reinterpret_cast<float4*>(&orow[c]) = y;
    }
    for (; c < cols; ++c) orow[c] *= inv;
}

KS_EXPORT int ks_hellinger_l2_f32(const float* X, float* out, int rows,
                                  int cols, float eps, hipStream_t stream) {
    if (rows == 0 || cols == 0) return 0;
    hipLaunchKernelGGL(hellinger_l2_kernel, dim3(rows), dim3(256), 0, stream,
                       X, out, rows, cols, eps);
    return (int)hipGetLastError();
}

// SymmetricRectifier: out = [max(0, x - alpha); max(0, -x - alpha)]
// (feature doubling, ref: nodes.images.SymmetricRectifier).
__global__ __launch_bounds__(256)
void sym_rectify_kernel(const float* __restrict__ X, float* __restrict__ out,
                        int rows, int cols, float alpha) {
    const size_t n = (size_t)rows * cols;
    for (size_t i = (size_t)blockIdx.x * 256 + threadIdx.x; i < n;
         i += (size_t)gridDim.x * 256) {
        const int r = (int)(i / cols), c = (int)(i % cols);
        float v = X[i];
        float* orow = out + (size_t)r * (2 * cols);
        orow[c] = fmaxf(v - alpha, 0.0f);
        orow[cols + c] = fmaxf(-v - alpha, 0.0f);
    }
}

KS_EXPORT int ks_sym_rectify_f32(const float* X, float* out, int rows,
                                 int cols, float alpha, hipStream_t stream) {
    if (rows == 0 || cols == 0) return 0;
    const size_t n = (size_t)rows * cols;
    int blocks = (int)((n + 255) / 256);
    if (blocks > 2048) blocks = 2048;   // grid-stride (G11)
    hipLaunchKernelGGL(sym_rectify_kernel, dim3(blocks), dim3(256), 0, stream,
                       X, out, rows, cols, alpha);
    return (int)hipGetLastError();
}

// In-place fused bias + cos epilogue: X[r, c] = cos(X[r, c] + b[c]) * scale.
// Pairs with a library GEMM (hipBLASLt ~150 TF f32) to form the fastest
// random-cosine-features path: GEMM + ONE elementwise pass instead of
// eager's 3-4 passes.
__global__ __launch_bounds__(256)
void bias_cos_kernel(float* __restrict__ X, const float* __restrict__ b,
                     int rows, int cols, float scale) {
    const size_t n = (size_t)rows * cols;
    size_t i = ((size_t)blockIdx.x * 256 + threadIdx.x) * 4;
    const size_t stride = (size_t)gridDim.x * 256 * 4;
    for (; i + 3 < n; i += stride) {
        float4 v = *reinterpret_cast<float4*>(&X[i]);
        const int c = (int)(i % cols);
        // cols is a multiple of 4 in practice; handle wrap generically
        if (c + 3 < cols) {
            v.x = __cosf(v.x + b[c]) * scale;
            v.y = __cosf(v.y + b[c + 1]) * scale;
            v.z = __cosf(v.z + b[c + 2]) * scale;
            v.w = __cosf(v.w + b[c + 3]) * scale;
            *reinterpret_cast<float4*>(&X[i]) = v;
        } else {
            for (int q = 0; q < 4; ++q)
                X[i + q] = __cosf(X[i + q] + b[(i + q) % cols]) * scale;
        }
    }
    if (blockIdx.x == 0 && threadIdx.x < 4) {
        for (size_t j = n - (n & 3) + threadIdx.x; j < n; j += 4)
            X[j] = __cosf(X[j] + b[j % cols]) * scale;
    }
}

KS_EXPORT int ks_bias_cos_f32(float* X, const float* b, int rows, int cols,
                              float scale, hipStream_t stream) {
    if (rows == 0 || cols == 0) return 0;
    const size_t n = (size_t)rows * cols;
    int blocks = (int)((n / 4 + 255) / 256);
    if (blocks > 2048) blocks = 2048;
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(bias_cos_kernel, dim3(blocks), dim3(256), 0, stream,
                       X, b, rows, cols, scale);
    return (int)hipGetLastError();
}
