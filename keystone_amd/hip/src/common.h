// Common helpers for keystone_amd CDNA4 (gfx950) kernels.
// Built standalone with hipcc (no torch ABI); launchers are extern "C" and
// take raw device pointers + the caller's hipStream_t (ctypes from Python).
#pragma once
#include <hip/hip_runtime.h>

#define KS_EXPORT extern "C" __attribute__((visibility("default")))

// fp32 MFMA accumulator: v_mfma_f32_32x32x2_f32 uses 16 f32 regs per lane.
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__host__ __device__ static inline int ks_ceil_div(int a, int b) {
    return (a + b - 1) / b;
}

// Bijective XCD-aware workgroup remap (MI355X: 8 XCDs, block b lands on
// XCD b%8): give each XCD a contiguous chunk of the grid so neighbouring
// tiles share an L2 (cdna_hip_programming.md T1; bijective form).
__device__ __forceinline__ unsigned ks_xcd_swizzle(unsigned wgid, unsigned nwg) {
    const unsigned NXCD = 8;
    if (nwg < NXCD) return wgid;
    unsigned q = nwg / NXCD, r = nwg % NXCD;
    unsigned xcd = wgid % NXCD, idx = wgid / NXCD;
    unsigned base = (xcd < r) ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q;
    return base + idx;
}
