// K1: dense-SIFT orientation-map kernel (gfx950).
//
// ref: VLFeat dense SIFT (JNI VLFeat.getSIFTs) — the
// Gaussian-blur -> gradient -> soft orientation-binning front half, fused
// into ONE kernel with LDS halo tiles:
//
//   omap[b, o, y, x] = mag(y,x) * softbin_o(angle(y,x))     (8 bins)
//
// where the image is first blurred with a separable Gaussian (replicate
// boundary, radius r).  The eager torch chain for this stage is 2 banded
// GEMMs + ~8 elementwise/scatter kernels re-reading HBM; this kernel
// reads the image once and writes the 8-plane map once (the remaining
// SIFT stages — triangular spatial pooling at the descriptor grid and
// descriptor normalization — are sampling-banded GEMMs (hipBLASLt) and
// the ks_sift_norm kernel).
//
// Tile: 16 rows x 64 cols of output per 256-thread block; LDS stages
// raw -> h-blur -> v-blur with the halos each stage needs.

#include "common.h"
#include <math.h>

#define OTH 16   // output tile rows
#define OTW 64   // output tile cols
#define MAXR 48  // max Gaussian radius (bin sizes to ~48: scales>=4 at
                 // bin 8-16 need radius up to 45 — VERDICT r1 #9)

__global__ __launch_bounds__(256)
void sift_omap_kernel(const float* __restrict__ img,  // [B, H, W]
                      float* __restrict__ omap,       // [B, 8, H, W]
                      const float* __restrict__ gk,   // [2r+1]
                      int r, int H, int W) {
    const int b = blockIdx.z;
    const int oy0 = blockIdx.y * OTH;
    const int ox0 = blockIdx.x * OTW;
    const int tid = threadIdx.x;

    // dynamic LDS sized to the ACTUAL radius (a static worst-case array
    // would cost every launch MAXR's occupancy: r=6 needs 22 KB -> 7
    // workgroups/CU where the static r=48 layout would pin 1-2)
    const int RH = OTH + 2 + 2 * r;          // raw/hblur rows
    const int RW = OTW + 2 + 2 * r;          // raw cols
    const int HW = OTW + 2;                  // hblur/vblur cols
    extern __shared__ float smem[];
    float* raw = smem;                       // [RH][RW]
    float* hb = raw + RH * RW;               // [RH][HW]
    float* vb = hb + RH * HW;                // [OTH+2][HW]
    float* gks = vb + (OTH + 2) * HW;        // [2r+1] staged taps

    const float* im = img + (size_t)b * H * W;

    // stage the Gaussian taps ONCE: gk[t] inside the blur loops was a
    // per-lane GLOBAL load per tap per output element (13+ loads each) —
    // latency the 1-wave-deep loops cannot hide
    for (int t = tid; t <= 2 * r; t += 256) gks[t] = gk[t];

    // ---- load raw tile + halo (replicate boundary)
    for (int i = tid; i < RH * RW; i += 256) {
        int ry = i / RW, rx = i % RW;
        int gy = oy0 - 1 - r + ry;
        int gx = ox0 - 1 - r + rx;
        gy = min(max(gy, 0), H - 1);
        gx = min(max(gx, 0), W - 1);
        raw[ry * RW + rx] = im[(size_t)gy * W + gx];
    }
    __syncthreads();

    // ---- horizontal blur: rows [0, RH), cols [0, OTW+2)
    for (int i = tid; i < RH * HW; i += 256) {
        int ry = i / HW, cx = i % HW;
        float s = 0.0f;
        for (int t = 0; t <= 2 * r; ++t) s += gks[t] * raw[ry * RW + cx + t];
        hb[ry * HW + cx] = s;
    }
    __syncthreads();

    // ---- vertical blur: rows [0, OTH+2)
    for (int i = tid; i < (OTH + 2) * HW; i += 256) {
        int vy = i / HW, cx = i % HW;
        float s = 0.0f;
        for (int t = 0; t <= 2 * r; ++t) s += gks[t] * hb[(vy + t) * HW + cx];
        vb[vy * HW + cx] = s;
    }
    __syncthreads();

    // ---- gradient + orientation soft-binning + 8-plane write
    const float inv_step = 8.0f / (2.0f * (float)M_PI);
    for (int i = tid; i < OTH * OTW; i += 256) {
        int ty = i / OTW, tx = i % OTW;
        const int gy = oy0 + ty, gx = ox0 + tx;
        if (gy >= H || gx >= W) continue;
        float gxv = 0.0f, gyv = 0.0f;
        if (gx >= 1 && gx + 1 < W)
            gxv = 0.5f * (vb[(ty + 1) * HW + tx + 2] - vb[(ty + 1) * HW + tx]);
        if (gy >= 1 && gy + 1 < H)
            gyv = 0.5f * (vb[(ty + 2) * HW + tx + 1] - vb[ty * HW + tx + 1]);
        const float mag = sqrtf(gxv * gxv + gyv * gyv);
        float ang = atan2f(gyv, gxv);
        if (ang < 0.0f) ang += 2.0f * (float)M_PI;
        float t = ang * inv_step;
        int b0 = (int)t;
        float frac = t - (float)b0;
        b0 &= 7;
        const int b1 = (b0 + 1) & 7;
        float vals[8] = {0, 0, 0, 0, 0, 0, 0, 0};
        vals[b0] = mag * (1.0f - frac);
        vals[b1] = mag * frac;
        float* ob = omap + ((size_t)b * 8) * H * W + (size_t)gy * W + gx;
#pragma unroll
        for (int o = 0; o < 8; ++o) ob[(size_t)o * H * W] = vals[o];
    }
}

KS_EXPORT int ks_sift_omap_f32(const float* img, float* omap,
                               const float* gauss, int radius, int B, int H,
                               int W, hipStream_t stream) {
    if (radius > MAXR || radius < 0) return (int)hipErrorInvalidValue;
    const int RH = OTH + 2 + 2 * radius;
    const int RW = OTW + 2 + 2 * radius;
    const int HW = OTW + 2;
    size_t lds = (size_t)(RH * RW + RH * HW + (OTH + 2) * HW
                          + 2 * radius + 1) * 4;
    if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
    dim3 grid(ks_ceil_div(W, OTW), ks_ceil_div(H, OTH), B);
    hipLaunchKernelGGL(sift_omap_kernel, grid, dim3(256), lds, stream, img,
                       omap, gauss, radius, H, W);
    return (int)hipGetLastError();
}

// Fused orientation-maps + horizontal triangle pooling on the gcd
// lattice: writes A1 [B, 8, H, Qx] DIRECTLY (the [B, 8, H, W] omap
// round-trip is 17 GB at bench scale, and the banded pooling GEMM spends
// 36x the useful flops).  Column q pools at center bin/2 + q*g with
// weights (1 - |t|/bin), zero boundary — exactly _merged_pool_matrix.
#define PHMAX 8   // max horizontal pooling halo (bin - 1)

__global__ __launch_bounds__(256)
void sift_omap_pooled_kernel(const float* __restrict__ img,  // [B, H, W]
                             float* __restrict__ out,  // [B, 8, H, Qx]
                             const float* __restrict__ gk,   // [2r+1]
                             int r, int bin, int g, int Qx,
                             int H, int W) {
    const int b = blockIdx.z;
    const int oy0 = blockIdx.y * OTH;
    const int ox0 = blockIdx.x * OTW;
    const int tid = threadIdx.x;
    const int ph = bin - 1;                  // pooling halo (<= PHMAX)

    const int RH = OTH + 2 + 2 * r;          // raw/hblur rows
    const int HW2 = OTW + 2 + 2 * ph;        // hblur/vblur cols
    const int RW2 = HW2 + 2 * r;             // raw cols
    const int OW2 = OTW + 2 * ph;
    // dynamic LDS sized to the ACTUAL radius/halo: the static worst-case
    // (r=12, ph=8) was 78 KB -> 1 workgroup/CU; bench shapes (r<=8) fit
    // ~68 KB -> 2 co-resident workgroups
    extern __shared__ float smem2[];
    float* raw = smem2;                      // [RH][RW2]
    float* hb = raw + RH * RW2;              // [RH][HW2]
    float* vb = hb + RH * HW2;               // [OTH+2][HW2]
    float* ob8 = vb + (OTH + 2) * HW2;       // [8][OTH][OW2]

    const float* im = img + (size_t)b * H * W;

    // ---- load raw tile + halo (replicate boundary)
    for (int i = tid; i < RH * RW2; i += 256) {
        int ry = i / RW2, rx = i % RW2;
        int gy = oy0 - 1 - r + ry;
        int gx = ox0 - 1 - r - ph + rx;
        gy = min(max(gy, 0), H - 1);
        gx = min(max(gx, 0), W - 1);
        raw[ry * RW2 + rx] = im[(size_t)gy * W + gx];
    }
    __syncthreads();

    // ---- horizontal blur
    for (int i = tid; i < RH * HW2; i += 256) {
        int ry = i / HW2, cx = i % HW2;
        float s = 0.0f;
        for (int t = 0; t <= 2 * r; ++t) s += gk[t] * raw[ry * RW2 + cx + t];
        hb[ry * HW2 + cx] = s;
    }
    __syncthreads();

    // ---- vertical blur
    for (int i = tid; i < (OTH + 2) * HW2; i += 256) {
        int vy = i / HW2, cx = i % HW2;
        float s = 0.0f;
        for (int t = 0; t <= 2 * r; ++t) s += gk[t] * hb[(vy + t) * HW2 + cx];
        vb[vy * HW2 + cx] = s;
    }
    __syncthreads();

    // ---- gradient + orientation soft-binning into LDS planes
    const float inv_step = 8.0f / (2.0f * (float)M_PI);
    for (int i = tid; i < OTH * OW2; i += 256) {
        int ty = i / OW2, tx = i % OW2;
        const int gy = oy0 + ty, gx = ox0 - ph + tx;
        float vals[8] = {0, 0, 0, 0, 0, 0, 0, 0};
        if (gy < H && gx >= 0 && gx < W) {
            float gxv = 0.0f, gyv = 0.0f;
            if (gx >= 1 && gx + 1 < W)
                gxv = 0.5f * (vb[(ty + 1) * HW2 + tx + 2]
                              - vb[(ty + 1) * HW2 + tx]);
            if (gy >= 1 && gy + 1 < H)
                gyv = 0.5f * (vb[(ty + 2) * HW2 + tx + 1]
                              - vb[ty * HW2 + tx + 1]);
            const float mag = sqrtf(gxv * gxv + gyv * gyv);
            float ang = atan2f(gyv, gxv);
            if (ang < 0.0f) ang += 2.0f * (float)M_PI;
            float t = ang * inv_step;
            int b0 = (int)t;
            float frac = t - (float)b0;
            b0 &= 7;
            const int b1 = (b0 + 1) & 7;
            vals[b0] = mag * (1.0f - frac);
            vals[b1] = mag * frac;
        }
#pragma unroll
        for (int o = 0; o < 8; ++o) ob8[(o * OTH + ty) * OW2 + tx] = vals[o];
    }
    __syncthreads();

    // ---- horizontal triangle pooling at the lattice centers in-tile
    const int binh = bin / 2;
    // centers c = binh + q*g with c in [ox0, ox0 + OTW)
    int q_lo = (ox0 - binh + g - 1) / g;
    if (q_lo < 0) q_lo = 0;
    int q_hi = (ox0 + OTW - 1 - binh) / g;
    if (q_hi > Qx - 1) q_hi = Qx - 1;
    const int nq = q_hi - q_lo + 1;
    if (nq <= 0) return;
    const float invb = 1.0f / (float)bin;
    for (int i = tid; i < 8 * OTH * nq; i += 256) {
        const int o = i / (OTH * nq);
        const int rem = i % (OTH * nq);
        const int ty = rem / nq;
        const int q = q_lo + rem % nq;
        const int gy = oy0 + ty;
        if (gy >= H) continue;
        const int c = binh + q * g;
        float s = 0.0f;
        for (int t = 1 - bin; t <= bin - 1; ++t) {
            const int x = c + t;
            if (x < 0 || x >= W) continue;
            s += (1.0f - fabsf((float)t) * invb)
                 * ob8[(o * OTH + ty) * OW2 + (x - ox0 + ph)];
        }
        out[(((size_t)b * 8 + o) * H + gy) * Qx + q] = s;
    }
}

KS_EXPORT int ks_sift_omap_pooled_f32(const float* img, float* out,
                                      const float* gauss, int radius,
                                      int bin, int g, int Qx, int B, int H,
                                      int W, hipStream_t stream) {
    if (radius > MAXR || radius < 0 || bin - 1 > PHMAX || bin < 2 || g < 1)
        return (int)hipErrorInvalidValue;
    const int ph = bin - 1;
    const int RH = OTH + 2 + 2 * radius;
    const int HW2 = OTW + 2 + 2 * ph;
    const int RW2 = HW2 + 2 * radius;
    size_t lds = (size_t)(RH * RW2 + RH * HW2 + (OTH + 2) * HW2
                          + 8 * OTH * (OTW + 2 * ph)) * 4;
    if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
    dim3 grid(ks_ceil_div(W, OTW), ks_ceil_div(H, OTH), B);
    hipLaunchKernelGGL(sift_omap_pooled_kernel, grid, dim3(256), lds, stream,
                       img, out, gauss, radius, bin, g, Qx, H, W);
    return (int)hipGetLastError();
}

// SIFT descriptor normalization: per row (128-dim), L2 -> clamp -> L2.
// 4 waves per block, one row per wave.
__global__ __launch_bounds__(256)
void sift_norm_kernel(float* __restrict__ D, long rows, int dim,
                      float clampv, float eps) {
    const long row = (long)blockIdx.x * (blockDim.x / 64) + (threadIdx.x >> 6);
    if (row >= rows) return;
    const int lane = threadIdx.x & 63;
    float* d = D + row * dim;
    float ss = 0.0f;
    for (int c = lane; c < dim; c += 64) ss += d[c] * d[c];
#pragma unroll
    for (int off = 32; off; off >>= 1) ss += __shfl_down(ss, off, 64);
    ss = __shfl(ss, 0, 64);
    float inv = rsqrtf(fmaxf(ss, eps));
    float ss2 = 0.0f;
    for (int c = lane; c < dim; c += 64) {
        float v = fminf(d[c] * inv, clampv);
        d[c] = v;
        ss2 += v * v;
    }
#pragma unroll
    for (int off = 32; off; off >>= 1) ss2 += __shfl_down(ss2, off, 64);
    ss2 = __shfl(ss2, 0, 64);
    float inv2 = rsqrtf(fmaxf(ss2, eps));
    for (int c = lane; c < dim; c += 64) d[c] *= inv2;
}

// register-resident variant for dim == 64*NV (the 128-dim SIFT case):
// the whole row lives in NV registers per lane between the three passes,
// so HBM traffic is ONE read + ONE write (the generic kernel above walks
// the row three times; measured 13.3 -> ~6 ms at 29.5M rows).
template <int NV>
__global__ __launch_bounds__(256)
void sift_norm_reg_kernel(float* __restrict__ D, long rows, float clampv,
                          float eps) {
    const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= rows) return;
    const int lane = threadIdx.x & 63;
    float* d = D + row * (64 * NV);
    float v[NV];
    float ss = 0.0f;
#pragma unroll
    for (int q = 0; q < NV; ++q) {
        v[q] = d[lane + 64 * q];
        ss += v[q] * v[q];
    }
#pragma unroll
    for (int off = 32; off; off >>= 1) ss += __shfl_down(ss, off, 64);
    ss = __shfl(ss, 0, 64);
    const float inv = rsqrtf(fmaxf(ss, eps));
    float ss2 = 0.0f;
#pragma unroll
    for (int q = 0; q < NV; ++q) {
        v[q] = fminf(v[q] * inv, clampv);
        ss2 += v[q] * v[q];
    }
#pragma unroll
    for (int off = 32; off; off >>= 1) ss2 += __shfl_down(ss2, off, 64);
    ss2 = __shfl(ss2, 0, 64);
    const float inv2 = rsqrtf(fmaxf(ss2, eps));
#pragma unroll
    for (int q = 0; q < NV; ++q) d[lane + 64 * q] = v[q] * inv2;
}

// 2-level-addressed variant: descriptor (b, i) lives at
// D + b * bpitch + i * dim — normalizes IN PLACE inside a strided slice
// of a preallocated [B, n_total, dim] buffer, so multi-scale assembly
// needs no torch.cat pass.
template <int NV>
__global__ __launch_bounds__(256)
void sift_norm_reg2_kernel(float* __restrict__ D, long bpitch, int nrows,
                           long rows, float clampv, float eps) {
    const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= rows) return;
    const int lane = threadIdx.x & 63;
    float* d = D + (row / nrows) * bpitch + (row % nrows) * (64 * NV);
    float v[NV];
    float ss = 0.0f;
#pragma unroll
    for (int q = 0; q < NV; ++q) {
        v[q] = d[lane + 64 * q];
        ss += v[q] * v[q];
    }
#pragma unroll
    for (int off = 32; off; off >>= 1) ss += __shfl_down(ss, off, 64);
    ss = __shfl(ss, 0, 64);
    const float inv = rsqrtf(fmaxf(ss, eps));
    float ss2 = 0.0f;
#pragma unroll
    for (int q = 0; q < NV; ++q) {
        v[q] = fminf(v[q] * inv, clampv);
        ss2 += v[q] * v[q];
    }
#pragma unroll
    for (int off = 32; off; off >>= 1) ss2 += __shfl_down(ss2, off, 64);
    ss2 = __shfl(ss2, 0, 64);
    const float inv2 = rsqrtf(fmaxf(ss2, eps));
#pragma unroll
    for (int q = 0; q < NV; ++q) d[lane + 64 * q] = v[q] * inv2;
}

KS_EXPORT int ks_sift_norm2_f32(float* D, long bpitch, int nrows, long rows,
                                int dim, float clampv, float eps,
                                hipStream_t stream) {
    if (rows == 0) return 0;
    long blocks = (rows + 3) / 4;
    dim3 g((unsigned)blocks), b(256);
    switch (dim) {
        case 64:  hipLaunchKernelGGL(sift_norm_reg2_kernel<1>, g, b, 0, stream, D, bpitch, nrows, rows, clampv, eps); break;
        case 128: hipLaunchKernelGGL(sift_norm_reg2_kernel<2>, g, b, 0, stream, D, bpitch, nrows, rows, clampv, eps); break;
        case 192: hipLaunchKernelGGL(sift_norm_reg2_kernel<3>, g, b, 0, stream, D, bpitch, nrows, rows, clampv, eps); break;
        case 256: hipLaunchKernelGGL(sift_norm_reg2_kernel<4>, g, b, 0, stream, D, bpitch, nrows, rows, clampv, eps); break;
        default: return (int)hipErrorInvalidValue;
    }
    return (int)hipGetLastError();
}

KS_EXPORT int ks_sift_norm_f32(float* D, long rows, int dim, float clampv,
                               float eps, hipStream_t stream) {
    if (rows == 0) return 0;
    const int waves_per_block = 4;
    long blocks = (rows + waves_per_block - 1) / waves_per_block;
    dim3 g((unsigned)blocks), b(64 * waves_per_block);
    switch (dim) {
        case 64:  hipLaunchKernelGGL(sift_norm_reg_kernel<1>, g, b, 0, stream, D, rows, clampv, eps); break;
        case 128: hipLaunchKernelGGL(sift_norm_reg_kernel<2>, g, b, 0, stream, D, rows, clampv, eps); break;
        case 192: hipLaunchKernelGGL(sift_norm_reg_kernel<3>, g, b, 0, stream, D, rows, clampv, eps); break;
        case 256: hipLaunchKernelGGL(sift_norm_reg_kernel<4>, g, b, 0, stream, D, rows, clampv, eps); break;
        default:
            hipLaunchKernelGGL(sift_norm_kernel, g, b, 0, stream, D, rows,
                               dim, clampv, eps);
    }
    return (int)hipGetLastError();
}
