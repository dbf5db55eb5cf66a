// K3 v2: Fisher Vector as a fused MFMA kernel (gfx950).
//
// The batched-GEMM FV path materializes gamma [B, n, K] (tens of GB at
// benchmark scale) and re-reads it for the moment GEMMs; this kernel
// keeps everything on-chip: per image (one workgroup), per 32-descriptor
// chunk:
//
//   phase L: logits[32, K] = Xext @ P           (f32 MFMA, acc in AGPRs)
//            with Xext = [x^2 | x | 1 | 0-pad] (LDS, built while staging)
//            and P = [-1/(2var) ; mu/var ; -c ; 0] (global, L2-resident)
//   phase S: softmax rows in LDS (8 lanes per row) -> gamma
//   phase A: S[K, XC] += gamma^T @ Xext         (MFMA, accumulators held
//            across ALL chunks of the image: S2 | S1 | S0 columns)
//
// The ones-column trick makes S0 just another GEMM column, and zero-padded
// tail descriptors contribute nothing (their Xext rows are zero), so no
// masking is needed anywhere.  Epilogue converts (S0, S1, S2) to the
// improved-FV form and writes [2*KT*D] per image.
//
// f16 stats pass (default, round 2): gamma and Xext are ALSO written to
// LDS as TRANSPOSED f16 panels ([comp][chunk-row] / [xcol][chunk-row],
// row stride 40 f16 = 80 B = five 16-B groups -> b128 fragment loads walk
// all 8 quad-bank groups), and phase A runs on v_mfma_f32_32x32x16_f16:
// k=32 takes 2 instructions x 32 cyc/SIMD instead of 16 x 64 — a 16x
// issue-throughput step that removes the stats pass from the critical
// path (f32 accumulate; only the gamma/x inputs round to f16, ~5e-4
// relative).  KEYSTONE_FV_F32STATS=1 selects the exact-f32 stats pass.
//
// Supported: D <= 64 (DT=64 padded), K in {64, 128, 256}.  4 waves.
//
// ref: EncEval gmm-fisher/fisher.cxx (calcAndGetFVs) —
// redesigned as a single fused kernel rather than a port.

#include "common.h"
#include <math.h>

#define DT 64
#define XC (2 * DT + 2)      // 130 live columns of Xext (col 129 = pad0)
#define XCP 160              // padded to 5 MFMA col-tiles
#define XSTRIDE (XCP + 1)    // LDS row stride: (161*r + c) % 32 distinct
#define CHUNK 32
#define SH (CHUNK + 8)       // f16 panel row stride (80 B = 16 B x 5, odd)

typedef _Float16 f16x8 __attribute__((ext_vector_type(8)));

// STATS mode: X is ONE flat [n, D] matrix processed grid-strided; block b
// writes its PARTIAL sufficient statistics to out[b] = [S0 (KT) | S1
// (KT*D) | S2 (KT*D) | ll (1)] — the GMM EM E-step in one launch (sum the
// partials host-side).  FV mode (STATS=false): one image per block.
// SPLIT (f16-only, OPT-IN via KEYSTONE_FV_SPLIT=1): TWO workgroups per
// image (gridDim.y == 2) — each computes the full logits+softmax but
// accumulates stats for only HALF the components.  Halved sacc (80
// AGPRs) + trimmed LDS (78.5 KB at K=256) let 2 workgroups co-reside
// per CU, aiming to fill the 69% non-MFMA time seen at occupancy 1.
// MEASURED (two generations): Gs-based split 48.6 ms; ONLINE-softmax
// split (no Gs panel, row reductions straight from the accumulator
// fragments, 4 barriers) 45.2 ms — vs 33.9 for the joint-span schedule
// at K=256/B=8192.  The duplicated logits MFMA (+93% issue work) always
// outweighs the co-residency win.  Kept opt-in as the documented A/B;
// the joint-span f16 kernel is the production path.
template <int KT, bool STATS, bool F16S, bool SPLIT>
__global__ __launch_bounds__(256, SPLIT ? 2 : 1)
void fv_mfma_kernel(const float* __restrict__ X,    // [B, n, D]
                    const float* __restrict__ P,    // [XCP, KT] padded
                    const float* __restrict__ mu,   // [KT, D]
                    const float* __restrict__ var,  // [KT, D]
                    const float* __restrict__ w,    // [KT]
                    float* __restrict__ out,        // [B, 2*KT*D]
                    int n, int D) {
    constexpr int KT32 = KT / 32;               // logits col-tiles
    constexpr int LT = (KT32 + 3) / 4;          // logits tiles per wave (max)
    constexpr int KH = SPLIT ? KT / 2 : KT;     // stats component span
    constexpr int KH32 = KH / 32;
    constexpr int ST = 5 * ((KH32 + 3) / 4);    // stats tiles per wave (max)
    // when KT32 is a multiple of 4 every wave's tiles are valid: the tile
    // guards below become compile-time (a RUNTIME guard around each load
    // makes hipcc branch + wait vmcnt(0) per element — guide §5 trap 4c)
    constexpr bool FULL = (KT32 % 4 == 0);
    constexpr bool FULLS = (KH32 % 4 == 0);
    // gamma row stride: (KT+8) % 32 == 8 puts softmax lane (row, sub)
    // (4 rows x 8 sublanes per 32-lane phase) on bank row*8 + sub -> all
    // 32 banks distinct.  The previous KT+1 stride gave bank row+sub:
    // ~3-way conflicts, 2.7e8 counted (profiles/pmc_summary.csv).
    constexpr int GST = KT + 8;
    // f32 words before the f16 region (16-B aligned for b128 loads).
    // SPLIT uses ONLINE softmax straight from the accumulator fragments:
    // no Gs buffer at all, just two tiny [32 rows][5] reduction panels.
    constexpr int NBUF = F16S ? 1 : 2;          // Xs f32 buffers
    constexpr int GSW = SPLIT ? 0 : CHUNK * GST;
    constexpr int RSW = SPLIT ? 2 * 32 * 5 : 0;
    constexpr int F32W =
        ((NBUF * CHUNK * XSTRIDE + GSW + KT + 1 + RSW) + 3) & ~3;

    extern __shared__ __attribute__((aligned(16))) char smem[];
    // gamma needs NO double buffer: stats(c-1) reads it during the joint
    // MFMA span, and logits(c) only overwrite it in the write phase after
    // the intervening barrier.
    float* Xs0 = (float*)smem;                  // [NBUF][CHUNK][XSTRIDE]
    float* Gs0 = Xs0 + NBUF * CHUNK * XSTRIDE;  // [CHUNK][GST] (!SPLIT)
    float* S0s = Gs0 + GSW;                     // [KT] broadcast (+ll)
    float* smM = S0s + KT + 1;                  // [32][5] row-max panel
    float* smS = smM + 32 * 5;                  // [32][5] row-sum panel
    // f16 region (F16S only): transposed panels for the f16 stats MFMA
    constexpr int NXH = SPLIT ? 1 : 2;          // XhT buffers
    _Float16* XhT0 = (_Float16*)(smem + (size_t)F32W * 4);  // [NXH][XCP][SH]
    _Float16* GhT = XhT0 + NXH * XCP * SH;                  // [KH][SH]

    const int img = blockIdx.x;
    const int kh0 = SPLIT ? (int)blockIdx.y * KH : 0;   // component base
    const float* Xi = STATS ? X : X + (size_t)img * n * D;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int l31 = lane & 31;
    const int lk = lane >> 5;                   // 0/1

    // stats accumulators: tile index t -> (ti, tj) = (t / 5, t % 5) with
    // ti strided over waves: ti = wave + 4 * (t / 5) ... flattened below.
    f32x16 sacc[ST];
#pragma unroll
    for (int t = 0; t < ST; ++t) sacc[t] = (f32x16)(0.0f);

    // Pipelined 2-deep schedule: chunk c's logits-MFMA and chunk c-1's
    // stats-MFMA form ONE joint MFMA span per iteration (the co-resident
    // wave covers the non-MFMA gaps), chunk c+1's global X loads are
    // issued ahead of the span (T14), and only 3 barriers/iteration
    // remain (MFMA span -> LDS writes -> softmax).
    // chunk index -> starting row (grid-strided in STATS mode)
    const int nc_all = (n + CHUNK - 1) / CHUNK;
    const int nc = STATS
        ? (nc_all - blockIdx.x + (int)gridDim.x - 1) / (int)gridDim.x
        : nc_all;
    auto chunk_start = [&](int i) {
        return STATS ? (blockIdx.x + (size_t)i * gridDim.x) * CHUNK
                     : (size_t)i * CHUNK;
    };
    if (STATS && tid == 0) S0s[KT] = 0.0f;      // ll accumulator
    if (nc == 0) {
        if (STATS) {
            __syncthreads();
            float* ob = out + (size_t)img * (KT + 2 * KT * D + 1);
            for (int i = tid; i < KT + 2 * KT * D + 1; i += 256) ob[i] = 0.0f;
        }
        return;
    }

    // zero the f32 pad columns (129..XCP-1) once; never touched again
    for (int buf = 0; buf < NBUF; ++buf)
        for (int i = tid; i < CHUNK * (XCP - XC + 1); i += 256) {
            const int r = i / (XCP - XC + 1), c = XC - 1 + i % (XCP - XC + 1);
            Xs0[(buf * CHUNK + r) * XSTRIDE + c] = 0.0f;
        }
    if (F16S) {
        // zero the f16 pad rows (cols 129..159 of Xext) in every buffer
        for (int i = tid; i < NXH * (XCP - XC + 1) * SH; i += 256) {
            const int buf = i / ((XCP - XC + 1) * SH);
            const int j = i % ((XCP - XC + 1) * SH);
            XhT0[(buf * XCP + XC - 1) * SH + j] = (_Float16)0.0f;
        }
    }

    // per-thread staging slots.
    //   F16S: thread owns column d = tid & 63 and 8 CONSECUTIVE rows
    //         r0 = (tid >> 6) * 8 — consecutive rows pack into ONE b128
    //         f16 write per panel row (conflict-free: 80-B row stride).
    //   f32:  thread owns 8 strided (row, d) slots as in round 1.
    float xreg[8];
    const int sd = tid & 63;                    // F16S staging column
    const int sr0 = (tid >> 6) * 8;             // F16S staging row base
    auto stage_load = [&](int c0, int rows) {
#pragma unroll
        for (int q = 0; q < 8; ++q) {
            const int r = F16S ? (sr0 + q) : (tid + 256 * q) / DT;
            const int d = F16S ? sd : (tid + 256 * q) % DT;
            xreg[q] = (r < rows && d < D)
                ? Xi[(size_t)(c0 + r) * D + d] : 0.0f;
        }
    };
    auto stage_write = [&](float* Xs, _Float16* Xh, int rows) {
        if (F16S) {
            f16x8 hx, hx2;
#pragma unroll
            for (int q = 0; q < 8; ++q) {
                const float x2 = xreg[q] * xreg[q];
                Xs[(sr0 + q) * XSTRIDE + sd] = x2;
                Xs[(sr0 + q) * XSTRIDE + DT + sd] = xreg[q];
                hx[q] = (_Float16)xreg[q];
                hx2[q] = (_Float16)x2;
            }
            *(f16x8*)&Xh[(size_t)sd * SH + sr0] = hx2;        // x^2 rows
            *(f16x8*)&Xh[(size_t)(DT + sd) * SH + sr0] = hx;  // x rows
            if (sd == 0) {
                f16x8 ones;
#pragma unroll
                for (int q = 0; q < 8; ++q)
                    ones[q] = (_Float16)((sr0 + q < rows) ? 1.0f : 0.0f);
                *(f16x8*)&Xh[(size_t)(2 * DT) * SH + sr0] = ones;
            }
        } else {
#pragma unroll
            for (int q = 0; q < 8; ++q) {
                const int i = tid + 256 * q;
                const int r = i / DT, d = i % DT;
                Xs[r * XSTRIDE + d] = xreg[q] * xreg[q];
                Xs[r * XSTRIDE + DT + d] = xreg[q];
            }
        }
        if (tid < CHUNK)
            Xs[tid * XSTRIDE + 2 * DT] = (tid < rows) ? 1.0f : 0.0f;
    };

    stage_load((int)chunk_start(0), min(CHUNK, n - (int)chunk_start(0)));
    stage_write(Xs0, XhT0, min(CHUNK, n - (int)chunk_start(0)));
    __syncthreads();

    // f16 stats-MFMA span for the chunk staged in XhT slot (ci & 1),
    // reading the gamma panel GhT (single-buffered; see barrier notes)
    auto stats_f16 = [&](int ci) {
        const _Float16* Xh = XhT0 + (SPLIT ? 0 : (size_t)(ci & 1) * XCP * SH);
#pragma unroll
        for (int kk = 0; kk < CHUNK; kk += 16) {
#pragma unroll
            for (int t = 0; t < ST; ++t) {
                const int ti = wave + 4 * (t / 5);
                const int tj = t % 5;
                if (FULLS || ti < KH32) {
                    const f16x8 a = *(const f16x8*)
                        &GhT[(size_t)(ti * 32 + l31) * SH + kk + 8 * lk];
                    const f16x8 b = *(const f16x8*)
                        &Xh[(size_t)(tj * 32 + l31) * SH + kk + 8 * lk];
                    sacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_f16(
                        a, b, sacc[t], 0, 0, 0);
                }
            }
        }
    };
    auto stats_f32 = [&](float* Xp, float* Gp) {
#pragma unroll 1
        for (int kk = 0; kk < CHUNK; kk += 2) {
#pragma unroll
            for (int t = 0; t < ST; ++t) {
                const int ti = wave + 4 * (t / 5);
                const int tj = t % 5;
                if (FULL || ti < KT32) {
                    const float a = Gp[(kk + lk) * GST + ti * 32 + l31];
                    const float b = Xp[(kk + lk) * XSTRIDE + tj * 32 + l31];
                    sacc[t] = __builtin_amdgcn_mfma_f32_32x32x2f32(
                        a, b, sacc[t], 0, 0, 0);
                }
            }
        }
    };

    // logits(c) MFMA span over Xs (shared by both schedules)
    auto logits_span = [&](f32x16* lacc) {
#pragma unroll
        for (int t = 0; t < LT; ++t) lacc[t] = (f32x16)(0.0f);
#pragma unroll 5
        for (int kk = 0; kk < XC; kk += 2) {
            const float a = Xs0[l31 * XSTRIDE + kk + lk];
#pragma unroll
            for (int t = 0; t < LT; ++t) {
                const int ct = wave + 4 * t;
                if (FULL || ct < KT32) {
                    const float b = P[(size_t)(kk + lk) * KT + ct * 32 + l31];
                    lacc[t] = __builtin_amdgcn_mfma_f32_32x32x2f32(
                        a, b, lacc[t], 0, 0, 0);
                }
            }
        }
    };
    if (SPLIT) {
        // ONLINE-softmax schedule (4 barriers, no Gs round-trip):
        //   [logits span] -> row-max shuffle/LDS reduce -> B1 ->
        //   [exp + row-sum reduce; Xs(c+1) f32 write] -> B2 ->
        //   [gamma -> own GhT tile] -> B3 -> [stats(c)] -> B4 ->
        //   [XhT(c+1) f16 write]  (next chunk's B3 covers visibility).
        // Per (lane, reg) the C-fragment row is unique:
        // row = (r & 3) + 8 * (r >> 2) + 4 * lk; lanes sharing lk are a
        // contiguous 32-lane group, so __shfl_xor(..., 32) reduces rows
        // across the in-wave columns.  At K=256 / LT=2 tile t == half
        // owns the WG's gamma half exactly.
        float ll_acc = 0.0f;
        for (int c = 0; c < nc; ++c) {
            const bool more = c + 1 < nc;
            const int rows_c = min(CHUNK, n - (int)chunk_start(c));
            const int rows1 = more
                ? min(CHUNK, n - (int)chunk_start(c + 1)) : 0;
            if (more) stage_load((int)chunk_start(c + 1), rows1);
            f32x16 lacc[LT];
            logits_span(lacc);
            float red[16];
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                red[r] = fmaxf(lacc[0][r], lacc[1][r]);
#pragma unroll
                for (int off = 16; off; off >>= 1)
                    red[r] = fmaxf(red[r], __shfl_xor(red[r], off, 32));
            }
            if (l31 == 0) {
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row = (r & 3) + 8 * (r >> 2) + 4 * lk;
                    smM[row * 5 + wave] = red[r];
                }
            }
            __syncthreads();   // B1
            float mrow[16];
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = (r & 3) + 8 * (r >> 2) + 4 * lk;
                mrow[r] = fmaxf(fmaxf(smM[row * 5 + 0], smM[row * 5 + 1]),
                                fmaxf(smM[row * 5 + 2], smM[row * 5 + 3]));
            }
#pragma unroll
            for (int t = 0; t < LT; ++t)
#pragma unroll
                for (int r = 0; r < 16; ++r)
                    lacc[t][r] = __expf(lacc[t][r] - mrow[r]);
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                red[r] = lacc[0][r] + lacc[1][r];
#pragma unroll
                for (int off = 16; off; off >>= 1)
                    red[r] += __shfl_xor(red[r], off, 32);
            }
            if (l31 == 0) {
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row = (r & 3) + 8 * (r >> 2) + 4 * lk;
                    smS[row * 5 + wave] = red[r];
                }
            }
            if (more) {
                // Xs(c+1): logits(c) consumed Xs before B1
#pragma unroll
                for (int q = 0; q < 8; ++q) {
                    Xs0[(sr0 + q) * XSTRIDE + sd] = xreg[q] * xreg[q];
                    Xs0[(sr0 + q) * XSTRIDE + DT + sd] = xreg[q];
                }
                if (tid < CHUNK)
                    Xs0[tid * XSTRIDE + 2 * DT] = (tid < rows1) ? 1.0f : 0.0f;
            }
            __syncthreads();   // B2
            // gamma for the OWN half tile (t == blockIdx.y) -> GhT
            {
                const int t = (int)blockIdx.y;
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row = (r & 3) + 8 * (r >> 2) + 4 * lk;
                    const float srow = smS[row * 5 + 0] + smS[row * 5 + 1]
                                     + smS[row * 5 + 2] + smS[row * 5 + 3];
                    const int col_local = wave * 32 + l31;
                    GhT[(size_t)col_local * SH + row] =
                        (_Float16)(lacc[t][r] / srow);
                    if (STATS && wave == 0 && l31 == 0 && t == 0
                            && row < rows_c)
                        ll_acc += mrow[r] + __logf(srow);
                }
            }
            __syncthreads();   // B3: GhT visible
            stats_f16(c);
            __syncthreads();   // B4: XhT(c) consumed
            if (more) {
                f16x8 hx, hx2;
#pragma unroll
                for (int q = 0; q < 8; ++q) {
                    hx[q] = (_Float16)xreg[q];
                    hx2[q] = (_Float16)(xreg[q] * xreg[q]);
                }
                *(f16x8*)&XhT0[(size_t)sd * SH + sr0] = hx2;
                *(f16x8*)&XhT0[(size_t)(DT + sd) * SH + sr0] = hx;
                if (sd == 0) {
                    f16x8 ones;
#pragma unroll
                    for (int q = 0; q < 8; ++q)
                        ones[q] = (_Float16)((sr0 + q < rows1) ? 1.0f : 0.0f);
                    *(f16x8*)&XhT0[(size_t)(2 * DT) * SH + sr0] = ones;
                }
            }
        }
        if (STATS && wave == 0 && l31 == 0 && blockIdx.y == 0 && kh0 == 0)
            atomicAdd(&S0s[KT], ll_acc);
    } else
    for (int c = 0; c < nc; ++c) {
        const int cur = c & 1;
        // F16S: ONE f32 staging buffer (logits(c) consumes it inside this
        // span; chunk c+1 overwrites it in the write phase).  f32 stats
        // keep the double buffer (stats(c-1) reads the other slot).
        float* Xs = F16S ? Xs0 : Xs0 + cur * CHUNK * XSTRIDE;
        float* Gs = Gs0;
        float* Xp = F16S ? Xs0 : Xs0 + (cur ^ 1) * CHUNK * XSTRIDE;
        const bool more = c + 1 < nc;
        if (more)
            stage_load((int)chunk_start(c + 1),
                       min(CHUNK, n - (int)chunk_start(c + 1)));

        // ---- joint MFMA span: logits(c) + stats(c-1)
        f32x16 lacc[LT];
#pragma unroll
        for (int t = 0; t < LT; ++t) lacc[t] = (f32x16)(0.0f);
#pragma unroll 5
        for (int kk = 0; kk < XC; kk += 2) {
            const float a = Xs[l31 * XSTRIDE + kk + lk];
#pragma unroll
            for (int t = 0; t < LT; ++t) {
                const int ct = wave + 4 * t;
                if (FULL || ct < KT32) {
                    const float b = P[(size_t)(kk + lk) * KT + ct * 32 + l31];
                    lacc[t] = __builtin_amdgcn_mfma_f32_32x32x2f32(
                        a, b, lacc[t], 0, 0, 0);
                }
            }
        }
        if (c > 0) {
            if (F16S) stats_f16(c - 1);
            else stats_f32(Xp, Gs0);
        }
        __syncthreads();   // stats done reading; logits accs private

        // ---- LDS writes: logits(c) -> Gs, staged chunk c+1 -> its slot
#pragma unroll
        for (int t = 0; t < LT; ++t) {
            const int ct = wave + 4 * t;
            if (FULL || ct < KT32) {
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row = (r & 3) + 8 * (r >> 2) + 4 * lk;
                    Gs[row * GST + ct * 32 + l31] = lacc[t][r];
                }
            }
        }
        if (more)
            stage_write(Xp, XhT0 + (size_t)((c + 1) & 1) * XCP * SH,
                        min(CHUNK, n - (int)chunk_start(c + 1)));
        __syncthreads();

        // ---- softmax(c) on Gs, 8 lanes per row; F16S writes the
        // normalized gamma TRANSPOSED into the f16 panel (the two halves
        // of a dword land in adjacent rows -> 2-way u16 write, measured
        // cheaper than a separate packing pass)
        {
            const int row = tid >> 3;
            const int sub = tid & 7;
            float m = -1e30f;
            for (int k = sub; k < KT; k += 8)
                m = fmaxf(m, Gs[row * GST + k]);
#pragma unroll
            for (int off = 4; off; off >>= 1)
                m = fmaxf(m, __shfl_xor(m, off, 8));
            float s = 0.0f;
            for (int k = sub; k < KT; k += 8) {
                float e = __expf(Gs[row * GST + k] - m);
                Gs[row * GST + k] = e;
                s += e;
            }
#pragma unroll
            for (int off = 4; off; off >>= 1)
                s += __shfl_xor(s, off, 8);
            const float inv = 1.0f / s;
            for (int k = sub; k < KT; k += 8) {
                if (F16S)
                    GhT[(size_t)k * SH + row] =
                        (_Float16)(Gs[row * GST + k] * inv);
                else
                    Gs[row * GST + k] *= inv;
            }
            if (STATS && sub == 0
                    && row < n - (int)chunk_start(c))
                atomicAdd(&S0s[KT], m + __logf(s));   // logsumexp sum
        }
        __syncthreads();
    }

    // ---- trailing stats for the last chunk (SPLIT runs stats in-loop)
    if (SPLIT) {
    } else if (F16S) {
        stats_f16(nc - 1);
    } else {
        const int cur = (nc - 1) & 1;
        stats_f32(Xs0 + cur * CHUNK * XSTRIDE, Gs0);
    }
    __syncthreads();

    if (STATS) {
        // ---- epilogue (STATS): write raw partials
        // ob = [S0 (KT) | S1 (KT*D) | S2 (KT*D) | ll]
        float* ob = out + (size_t)img * (KT + 2 * KT * D + 1);
#pragma unroll
        for (int t = 0; t < ST; ++t) {
            const int ti = wave + 4 * (t / 5);
            const int tj = t % 5;
            if (!(FULLS || ti < KH32)) continue;
            if (tj == 4) {
                if (l31 == 0) {
#pragma unroll
                    for (int r = 0; r < 16; ++r) {
                        const int kcomp = kh0 + ti * 32 + (r & 3) + 8 * (r >> 2) + 4 * lk;
                        ob[kcomp] = sacc[t][r];
                    }
                }
            } else {
                const int d = (tj & 1) * 32 + l31;
                if (d < D) {
                    const bool is_s2 = tj < 2;   // x^2 columns
#pragma unroll
                    for (int r = 0; r < 16; ++r) {
                        const int kcomp = kh0 + ti * 32 + (r & 3) + 8 * (r >> 2) + 4 * lk;
                        ob[KT + (is_s2 ? KT * D : 0) + kcomp * D + d] =
                            sacc[t][r];
                    }
                }
            }
        }
        if (tid == 0 && kh0 == 0) ob[KT + 2 * KT * D] = S0s[KT];
        return;
    }

    // ---- epilogue: S0 broadcast, then FV from (S0, S1, S2)
    // tile (ti, tj=4) col 128 = S0 column (j = l31 + 128 -> l31 == 0)
#pragma unroll
    for (int t = 0; t < ST; ++t) {
        const int ti = wave + 4 * (t / 5);
        const int tj = t % 5;
        if (ti < KH32 && tj == 4 && l31 == 0) {
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int kcomp = kh0 + ti * 32 + (r & 3) + 8 * (r >> 2) + 4 * lk;
                S0s[kcomp] = sacc[t][r];
            }
        }
    }
    __syncthreads();

    // S2 tiles: tj in {0, 1} (cols 0..63 = x^2) ; S1 tiles: tj in {2, 3}
    // For output element (kcomp, d): S2 at tile tj = d/32, S1 at
    // tj = 2 + d/32 — same lane (l31 = d%32), same r. Pair them up.
    const float invn = 1.0f / (float)n;
#pragma unroll
    for (int t = 0; t < ST; ++t) {
        const int ti = wave + 4 * (t / 5);
        const int tj = t % 5;
        if (ti >= KH32 || tj >= 2) continue;      // S2 tiles drive the pair
        const int d = tj * 32 + l31;
        if (d >= D) continue;
        // find the matching S1 tile (tj + 2) in this wave's tile list:
        // flattened index t + 2 (same ti, tj+2) — valid because tiles are
        // laid out tj-major within each ti.
        const f32x16 s2 = sacc[t];
        const f32x16 s1 = sacc[t + 2];
#pragma unroll
        for (int r = 0; r < 16; ++r) {
            const int kcomp = kh0 + ti * 32 + (r & 3) + 8 * (r >> 2) + 4 * lk;
            const float s0 = S0s[kcomp];
            const float m = mu[kcomp * D + d];
            const float v = var[kcomp * D + d];
            const float wk = w[kcomp];
            const float sig = sqrtf(v);
            const float gm = (s1[r] - m * s0) / (sig * (float)n * sqrtf(wk));
            const float gs = ((s2[r] - 2.0f * m * s1[r] + m * m * s0) / v
                              - s0) * invn / sqrtf(2.0f * wk);
            float* ob = out + (size_t)img * (2 * KT * D);
            ob[kcomp * D + d] = gm;
            ob[KT * D + kcomp * D + d] = gs;
        }
    }
}

// mode: 0 = f32 stats, 1 = f16 stats, 2 = f16 stats + K-split (2 WG/img,
// online softmax: no Gs panel, two [32][5] reduction panels instead)
static size_t fv_lds_bytes(int K, int mode) {
    const int nbuf = mode ? 1 : 2;
    const int gsw = (mode == 2) ? 0 : CHUNK * (K + 8);
    const int rsw = (mode == 2) ? 2 * 32 * 5 : 0;
    size_t f32w = ((size_t)(nbuf * CHUNK * XSTRIDE + gsw + K + 1 + rsw)
                   + 3) & ~(size_t)3;
    size_t b = f32w * 4;
    const int nxh = (mode == 2) ? 1 : 2;
    const int kh = (mode == 2) ? K / 2 : K;
    if (mode) b += (size_t)(nxh * XCP * SH + kh * SH) * 2;
    return b;
}

KS_EXPORT int ks_fv_mfma_f32(const float* X, const float* P, const float* mu,
                             const float* var, const float* w, float* out,
                             int B, int n, int D, int K, int f16_stats,
                             hipStream_t stream) {
    if (D > DT || (K != 64 && K != 128 && K != 256))
        return (int)hipErrorInvalidValue;
    int mode = f16_stats;
    if (mode == 2 && K != 256) mode = 1;   // split only tuned for K=256
    size_t lds = fv_lds_bytes(K, mode);
    if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
    dim3 grid(B, mode == 2 ? 2 : 1), block(256);
    switch (K * 4 + mode) {
        case 257: hipLaunchKernelGGL((fv_mfma_kernel<64, false, true, false>), grid, block, lds, stream, X, P, mu, var, w, out, n, D); break;
        case 256: hipLaunchKernelGGL((fv_mfma_kernel<64, false, false, false>), grid, block, lds, stream, X, P, mu, var, w, out, n, D); break;
        case 513: hipLaunchKernelGGL((fv_mfma_kernel<128, false, true, false>), grid, block, lds, stream, X, P, mu, var, w, out, n, D); break;
        case 512: hipLaunchKernelGGL((fv_mfma_kernel<128, false, false, false>), grid, block, lds, stream, X, P, mu, var, w, out, n, D); break;
        case 1025: hipLaunchKernelGGL((fv_mfma_kernel<256, false, true, false>), grid, block, lds, stream, X, P, mu, var, w, out, n, D); break;
        case 1024: hipLaunchKernelGGL((fv_mfma_kernel<256, false, false, false>), grid, block, lds, stream, X, P, mu, var, w, out, n, D); break;
        case 1026: hipLaunchKernelGGL((fv_mfma_kernel<256, false, true, true>), grid, block, lds, stream, X, P, mu, var, w, out, n, D); break;
        default: return (int)hipErrorInvalidValue;
    }
    return (int)hipGetLastError();
}

// GMM E-step sufficient statistics over ONE flat [n, D] matrix:
// out [n_blocks, K + 2*K*D + 1] partials (sum over dim 0 host-side).
KS_EXPORT int ks_gmm_stats_f32(const float* X, const float* P, float* out,
                               int n_blocks, int n, int D, int K,
                               int f16_stats, hipStream_t stream) {
    if (D > DT || (K != 64 && K != 128 && K != 256))
        return (int)hipErrorInvalidValue;
    int mode = f16_stats;
    if (mode == 2 && K != 256) mode = 1;
    size_t lds = fv_lds_bytes(K, mode);
    if (lds > 160 * 1024) return (int)hipErrorInvalidValue;
    dim3 grid(n_blocks, mode == 2 ? 2 : 1), block(256);
    switch (K * 4 + mode) {
        case 257: hipLaunchKernelGGL((fv_mfma_kernel<64, true, true, false>), grid, block, lds, stream, X, P, nullptr, nullptr, nullptr, out, n, D); break;
        case 256: hipLaunchKernelGGL((fv_mfma_kernel<64, true, false, false>), grid, block, lds, stream, X, P, nullptr, nullptr, nullptr, out, n, D); break;
        case 513: hipLaunchKernelGGL((fv_mfma_kernel<128, true, true, false>), grid, block, lds, stream, X, P, nullptr, nullptr, nullptr, out, n, D); break;
        case 512: hipLaunchKernelGGL((fv_mfma_kernel<128, true, false, false>), grid, block, lds, stream, X, P, nullptr, nullptr, nullptr, out, n, D); break;
        case 1025: hipLaunchKernelGGL((fv_mfma_kernel<256, true, true, false>), grid, block, lds, stream, X, P, nullptr, nullptr, nullptr, out, n, D); break;
        case 1024: hipLaunchKernelGGL((fv_mfma_kernel<256, true, false, false>), grid, block, lds, stream, X, P, nullptr, nullptr, nullptr, out, n, D); break;
        case 1026: hipLaunchKernelGGL((fv_mfma_kernel<256, true, true, true>), grid, block, lds, stream, X, P, nullptr, nullptr, nullptr, out, n, D); break;
        default: return (int)hipErrorInvalidValue;
    }
    return (int)hipGetLastError();
}
