"""keystone_amd.hip — the CDNA4 (gfx950) kernel extension.

Hand-written HIP kernels for the framework's hot paths, compiled in-tree to
``libkeystone_hip.so`` with hipcc (no torch C++ ABI — launchers are
``extern "C"`` and take raw device pointers + the caller's HIP stream; we
call them via ctypes with torch's current stream so they serialize
correctly with surrounding torch ops).

Kernels (ref mapping in SURVEY.md §2.9):
  * ``gemm_fused``      — fp32 MFMA GEMM with fused bias/cos/relu epilogue
                          (CosineRandomFeatures, Convolver epilogues)
  * ``hellinger_l2``    — fused signed-sqrt + row L2-norm (FV normalization)
  * ``sym_rectify``     — SymmetricRectifier feature doubling
  * ``fisher_vector``   — fused GMM-posterior + FV accumulation (K3)
  * ``gmm_posteriors``  — GMM E-step responsibilities (K2)
  * ``sift``            — dense SIFT descriptor family (K1)

GPU ops FAIL LOUDLY when the extension is missing on a CUDA device — there
is deliberately no silent eager fallback on GPU (set KEYSTONE_ALLOW_EAGER=1
to override for debugging only).
"""

from __future__ import annotations

import ctypes
import os
import subprocess
import sys
from typing import Optional

import torch

_HERE = os.path.dirname(os.path.abspath(__file__))
_SRC = os.path.join(_HERE, "src")
_LIB_PATH = os.path.join(_HERE, "libkeystone_hip.so")

_lib: Optional[ctypes.CDLL] = None


def source_files():
    return sorted(
        os.path.join(_SRC, f) for f in os.listdir(_SRC)
        if f.endswith(".hip") or f.endswith(".cpp"))


def build(force: bool = False, arch: str = "gfx950", verbose: bool = True) -> str:
    """Compile every HIP source into the in-tree shared library."""
    srcs = source_files()
    if not force and os.path.exists(_LIB_PATH):
        lib_m = os.path.getmtime(_LIB_PATH)
        if all(os.path.getmtime(s) <= lib_m for s in srcs +
               [os.path.join(_SRC, "common.h")]):
            return _LIB_PATH
    hipcc = os.environ.get("HIPCC", "hipcc")
    cmd = [hipcc, f"--offload-arch={arch}", "-O3", "-std=c++17", "-shared",
           "-fPIC", "-fvisibility=hidden", "-o", _LIB_PATH] + srcs
    if verbose:
        print("[keystone_amd.hip] " + " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return _LIB_PATH


def _load() -> ctypes.CDLL:
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_LIB_PATH):
        raise RuntimeError(
            "keystone_amd HIP extension not built (libkeystone_hip.so "
            "missing). Run keystone_amd.hip.build() — GPU ops refuse to "
            "fall back to eager silently.")
    lib = ctypes.CDLL(_LIB_PATH)
    c_p, c_i, c_f = ctypes.c_void_p, ctypes.c_int, ctypes.c_float
    lib.ks_gemm_fused_f32.argtypes = [c_p, c_p, c_p, c_p, c_i, c_i, c_i,
                                      c_f, c_f, c_i, c_p]
    lib.ks_gemm_fused_f32.restype = c_i
    lib.ks_hellinger_l2_f32.argtypes = [c_p, c_p, c_i, c_i, c_f, c_p]
    lib.ks_hellinger_l2_f32.restype = c_i
    lib.ks_sym_rectify_f32.argtypes = [c_p, c_p, c_i, c_i, c_f, c_p]
    lib.ks_sym_rectify_f32.restype = c_i
    lib.ks_bias_cos_f32.argtypes = [c_p, c_p, c_i, c_i, c_f, c_p]
    lib.ks_bias_cos_f32.restype = c_i
    _register_extra(lib)
    _lib = lib
    return lib


def _register_extra(lib: ctypes.CDLL) -> None:
    """argtypes for kernels added after v1 (ignore if absent)."""
    c_p, c_i, c_f = ctypes.c_void_p, ctypes.c_int, ctypes.c_float
    try:
        lib.ks_fisher_vector_f32.argtypes = [c_p] * 8 + [c_i] * 4 + [c_p]
        lib.ks_fisher_vector_f32.restype = c_i
    except AttributeError:
        pass
    try:
        lib.ks_gmm_posteriors_f32.argtypes = [c_p] * 6 + [c_i] * 3 + [c_p]
        lib.ks_gmm_posteriors_f32.restype = c_i
    except AttributeError:
        pass
    try:
        lib.ks_fv_mfma_f32.argtypes = [c_p] * 6 + [c_i] * 5 + [c_p]
        lib.ks_fv_mfma_f32.restype = c_i
    except AttributeError:
        pass
    try:
        lib.ks_gmm_stats_f32.argtypes = [c_p] * 3 + [c_i] * 5 + [c_p]
        lib.ks_gmm_stats_f32.restype = c_i
    except AttributeError:
        pass
    try:
        lib.ks_gmm_mstep_f32.argtypes = [c_p, c_p, c_i, c_i, c_i, c_f, c_f,
                                         c_p, c_p, c_p, c_p, c_p, c_p]
        lib.ks_gmm_mstep_f32.restype = c_i
    except AttributeError:
        pass
    try:
        lib.ks_woodbury_f32.argtypes = [c_p] * 6 + [c_i, c_i, c_p]
        lib.ks_woodbury_f32.restype = c_i
    except AttributeError:
        pass
    try:
        lib.ks_segsum_f32.argtypes = [c_p] * 6 + [c_i] * 3 + [c_p]
        lib.ks_segsum_f32.restype = c_i
    except AttributeError:
        pass
    try:
        lib.ks_pool_lastdim_f32.argtypes = [c_p, c_p, c_p, c_i, c_i, c_i,
                                            c_i, c_i, ctypes.c_long, c_p]
        lib.ks_pool_lastdim_f32.restype = c_i
        lib.ks_pool_middim_f32.argtypes = [c_p, c_p, c_p] + [c_i] * 7 + [c_p]
        lib.ks_pool_middim_f32.restype = c_i
    except AttributeError:
        pass
    try:
        lib.ks_sift_omap_f32.argtypes = [c_p, c_p, c_p, c_i, c_i, c_i, c_i,
                                         c_p]
        lib.ks_sift_omap_f32.restype = c_i
        lib.ks_sift_omap_pooled_f32.argtypes = [c_p, c_p, c_p] + [c_i] * 7 \
            + [c_p]
        lib.ks_sift_omap_pooled_f32.restype = c_i
        lib.ks_sift_norm_f32.argtypes = [c_p, ctypes.c_long, c_i, c_f, c_f,
                                         c_p]
        lib.ks_sift_norm_f32.restype = c_i
        lib.ks_sift_norm2_f32.argtypes = [c_p, ctypes.c_long, c_i,
                                          ctypes.c_long, c_i, c_f, c_f, c_p]
        lib.ks_sift_norm2_f32.restype = c_i
    except AttributeError:
        pass


def available() -> bool:
    try:
        _load()
        return True
    except (RuntimeError, OSError):
        return False


def gpu_ready() -> bool:
    """The no-silent-fallback contract, enforced: every op that dispatches
    CUDA tensors to a HIP kernel gates on this.  Extension present -> True
    (use the kernel).  Extension missing -> RuntimeError, UNLESS
    KEYSTONE_ALLOW_EAGER=1 explicitly authorizes the eager torch path
    (debugging only) -> False."""
    if available():
        return True
    if os.environ.get("KEYSTONE_ALLOW_EAGER"):
        return False
    raise RuntimeError(
        "keystone_amd HIP extension missing on a GPU device; refusing "
        "silent eager fallback (build with keystone_amd.hip.build(), or "
        "set KEYSTONE_ALLOW_EAGER=1 for debugging)")


def _stream() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def _check(code: int, name: str) -> None:
    if code != 0:
        raise RuntimeError(f"{name} failed: hipError {code}")


def _prep(t: torch.Tensor, name: str) -> torch.Tensor:
    if not t.is_cuda:
        raise RuntimeError(f"{name}: expected CUDA tensor")
    if t.dtype != torch.float32:
        raise RuntimeError(f"{name}: expected float32, got {t.dtype}")
    return t.contiguous()


def _prep_inplace(t: torch.Tensor, name: str) -> torch.Tensor:
    """For in-place kernels: a silent .contiguous() copy would leave the
    caller's tensor unmodified — fail loudly on layout regressions."""
    if not t.is_contiguous():
        raise RuntimeError(f"{name}: in-place kernel needs a contiguous "
                           "tensor (got a strided view)")
    return _prep(t, name)


# --------------------------------------------------------------------- ops
EP_NONE, EP_BIAS, EP_COS, EP_RELU = 0, 1, 2, 3


def gemm_fused(X: torch.Tensor, W: torch.Tensor,
               bias: Optional[torch.Tensor], epilogue: int,
               scale: float = 1.0, alpha: float = 0.0,
               out: Optional[torch.Tensor] = None) -> torch.Tensor:
    lib = _load()
    X = _prep(X, "gemm_fused.X")
    W = _prep(W, "gemm_fused.W")
    M, K = X.shape
    K2, N = W.shape
    assert K == K2, (X.shape, W.shape)
    if out is None:
        out = torch.empty(M, N, dtype=torch.float32, device=X.device)
    bptr = None
    if bias is not None:
        bias = _prep(bias, "gemm_fused.bias")
        bptr = ctypes.c_void_p(bias.data_ptr())
    _check(lib.ks_gemm_fused_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(W.data_ptr()), bptr,
        ctypes.c_void_p(out.data_ptr()), M, K, N, scale, alpha, epilogue,
        _stream()), "ks_gemm_fused_f32")
    return out


def bias_cos_(X: torch.Tensor, b: torch.Tensor, scale: float) -> torch.Tensor:
    """In-place ``X = cos(X + b) * scale`` (one HBM pass)."""
    lib = _load()
    X = _prep_inplace(X, "bias_cos.X")
    b = _prep(b, "bias_cos.b")
    rows, cols = X.shape
    _check(lib.ks_bias_cos_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(b.data_ptr()),
        rows, cols, scale, _stream()), "ks_bias_cos_f32")
    return X


def gemm_cos(X: torch.Tensor, W: torch.Tensor, b: torch.Tensor,
             scale: float) -> torch.Tensor:
    """cos(X @ W + b) * scale — CosineRandomFeatures hot path.

    Default: hipBLASLt GEMM (via torch.matmul, ~150 TF f32) + ONE fused
    bias+cos pass — measured faster than both eager torch (3-4 passes) and
    the fully-fused MFMA kernel (the hand kernel's GEMM runs ~115 TF;
    set KEYSTONE_FUSED_GEMM_COS=1 to use it end-to-end instead)."""
    if os.environ.get("KEYSTONE_FUSED_GEMM_COS"):
        return gemm_fused(X, W, b, EP_COS, scale=scale)
    Z = X @ W
    return bias_cos_(Z, b, scale)


def hellinger_l2(X: torch.Tensor, eps: float = 1e-12) -> torch.Tensor:
    lib = _load()
    X = _prep(X, "hellinger_l2.X")
    rows, cols = X.shape
    out = torch.empty_like(X)
    _check(lib.ks_hellinger_l2_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        rows, cols, eps, _stream()), "ks_hellinger_l2_f32")
    return out


def _gmm_coeffs(mu: torch.Tensor, var: torch.Tensor, w: torch.Tensor):
    """Host-precomputed expanded-Gaussian coefficients (see gmm_fv.hip)."""
    import math
    a = (0.5 / var).contiguous()
    b = (mu / var).contiguous()
    c = ((mu * mu * a).sum(1) + 0.5 * var.log().sum(1)
         + 0.5 * mu.shape[1] * math.log(2 * math.pi)
         - w.clamp_min(1e-30).log()).contiguous()
    return a, b, c


def gmm_supported(D: int, K: int) -> bool:
    return D <= 64 and K <= 128


def fv_supported(D: int, K: int) -> bool:
    if D > 64 or K > 128 or K % 16 != 0 or (K // 4) not in (4, 8, 12, 16, 24, 32):
        return False
    return (2 * K * D + K + 128 * D + 128 * K) * 4 <= 160 * 1024


def gmm_posteriors(X: torch.Tensor, mu: torch.Tensor, var: torch.Tensor,
                   w: torch.Tensor, want_ll: bool = False):
    """gamma = softmax_k log N(x|k) + log w_k; optional summed logsumexp."""
    lib = _load()
    X = _prep(X, "gmm_posteriors.X")
    n, D = X.shape
    K = mu.shape[0]
    a, b, c = _gmm_coeffs(_prep(mu, "mu"), _prep(var, "var"),
                          _prep(w, "w"))
    gamma = torch.empty(n, K, dtype=torch.float32, device=X.device)
    ll = torch.zeros(1, dtype=torch.float32, device=X.device) if want_ll else None
    _check(lib.ks_gmm_posteriors_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(a.data_ptr()),
        ctypes.c_void_p(b.data_ptr()), ctypes.c_void_p(c.data_ptr()),
        ctypes.c_void_p(gamma.data_ptr()),
        ctypes.c_void_p(ll.data_ptr()) if ll is not None else None,
        n, D, K, _stream()), "ks_gmm_posteriors_f32")
    return (gamma, ll) if want_ll else gamma


def fv_mfma_supported(D: int, K: int) -> bool:
    return D <= 64 and K in (64, 128, 256)


def _f16_stats() -> int:
    """Stats-pass mode: 1 = f16 MFMA (v_mfma_f32_32x32x16_f16, 16x the
    f32 form's issue throughput; gamma/x round to f16, f32 accumulate),
    2 = f16 + K-split (two co-resident workgroups per image at K=256),
    0 = exact f32 (KEYSTONE_FV_F32STATS=1)."""
    if os.environ.get("KEYSTONE_FV_F32STATS"):
        return 0
    if os.environ.get("KEYSTONE_FV_SPLIT"):
        return 2
    return 1


def fisher_vector_mfma(X: torch.Tensor, mu: torch.Tensor, var: torch.Tensor,
                       w: torch.Tensor) -> torch.Tensor:
    """Fused MFMA FV (fv_mfma.hip): X [B, n, D] -> [B, 2*K*D] without
    materializing gamma.  D <= 64, K in {64, 128, 256}."""
    lib = _load()
    X = _prep(X, "fv_mfma.X")
    B, n, D = X.shape
    K = mu.shape[0]
    mu = _prep(mu, "mu")
    var = _prep(var, "var")
    w = _prep(w, "w")
    a, b, c = _gmm_coeffs(mu, var, w)
    P = _fv_P(a, b, c, K, X.device)
    out = torch.empty(B, 2 * K * D, dtype=torch.float32, device=X.device)
    _check(lib.ks_fv_mfma_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(P.data_ptr()),
        ctypes.c_void_p(mu.data_ptr()), ctypes.c_void_p(var.data_ptr()),
        ctypes.c_void_p(w.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        B, n, D, K, _f16_stats(), _stream()), "ks_fv_mfma_f32")
    return out


def _fv_P(a: torch.Tensor, b: torch.Tensor, c: torch.Tensor, K: int,
          device) -> torch.Tensor:
    """Padded [XCP=160, K] operand for fv_mfma: [-a.T ; b.T ; -c ; 0]."""
    DT = 64
    D = a.shape[1]
    P = torch.zeros(2 * DT + 32, K, dtype=torch.float32, device=device)
    P[:D] = -a.T
    P[DT:DT + D] = b.T
    P[2 * DT] = -c
    return P.contiguous()


def gmm_stats_into(X: torch.Tensor, P: torch.Tensor, out: torch.Tensor):
    """Low-level fused E-step: preallocated P [160, K] (see ``_fv_P``) and
    partials buffer out [nb, K + 2*K*D + 1] -> (Nk, Sx, Sxx, ll) fp64 views
    of one contiguous sum (no per-call allocations — EM-loop hot path)."""
    lib = _load()
    n, D = X.shape
    K = P.shape[1]
    nb = out.shape[0]
    _check(lib.ks_gmm_stats_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(P.data_ptr()),
        ctypes.c_void_p(out.data_ptr()), nb, n, D, K, _f16_stats(),
        _stream()), "ks_gmm_stats_f32")
    s = out.sum(0, dtype=torch.float64)
    return (s[:K], s[K:K + K * D].view(K, D),
            s[K + K * D:K + 2 * K * D].view(K, D), s[K + 2 * K * D:])


def gmm_stats_raw(X: torch.Tensor, P: torch.Tensor,
                  out: torch.Tensor) -> None:
    """E-step partials kernel ONLY (no host-side sum): pairs with
    gmm_mstep_ for the two-launch-per-iteration EM loop."""
    lib = _load()
    n, D = X.shape
    K = P.shape[1]
    nb = out.shape[0]
    _check(lib.ks_gmm_stats_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(P.data_ptr()),
        ctypes.c_void_p(out.data_ptr()), nb, n, D, K, _f16_stats(),
        _stream()), "ks_gmm_stats_f32")


def gmm_mstep_(partials: torch.Tensor, pre,
               n_tot: float, min_var: float, mu: torch.Tensor,
               var: torch.Tensor, w: torch.Tensor, P: torch.Tensor,
               ll: torch.Tensor) -> None:
    """Fused M-step + next-iteration operand build (gmm_mstep.hip): sums
    the E-step partials (or consumes the pre-reduced fp64 stats vector
    ``pre`` on multi-GPU), updates (mu, var, w) in place, rebuilds the
    fv_mfma P operand, and writes the log-likelihood scalar — ONE launch
    replacing ~18 torch small-ops per EM iteration."""
    lib = _load()
    nb = partials.shape[0]
    K, D = mu.shape
    _check(lib.ks_gmm_mstep_f32(
        ctypes.c_void_p(partials.data_ptr()),
        ctypes.c_void_p(pre.data_ptr()) if pre is not None else None,
        nb, K, D, n_tot, min_var,
        ctypes.c_void_p(mu.data_ptr()), ctypes.c_void_p(var.data_ptr()),
        ctypes.c_void_p(w.data_ptr()), ctypes.c_void_p(P.data_ptr()),
        ctypes.c_void_p(ll.data_ptr()), _stream()), "ks_gmm_mstep_f32")


WOODBURY_NCMAX = 32


def segsum(X: torch.Tensor, T: torch.Tensor, starts: torch.Tensor,
           ncs: torch.Tensor, cols: torch.Tensor) -> torch.Tensor:
    """Contiguous segmented reduction for the weighted solver's per-class
    rhs: out[g] = sum_{i in class g} X[i, :] * T[i, cols[g]]."""
    lib = _load()
    G = starts.numel()
    n, db = X.shape
    K = T.shape[1]
    out = torch.empty(G, db, dtype=torch.float32, device=X.device)
    _check(lib.ks_segsum_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(T.data_ptr()),
        ctypes.c_void_p(starts.data_ptr()), ctypes.c_void_p(ncs.data_ptr()),
        ctypes.c_void_p(cols.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        G, db, K, _stream()), "ks_segsum_f32")
    return out


def pool_lastdim(x: torch.Tensor, tap: torch.Tensor, c0: int, g: int,
                 Q: int) -> torch.Tensor:
    """Triangular lattice pooling of the LAST dim: [R, W] -> [R, Q]
    (pool1d.hip — evaluates the 2*bin-1 band directly instead of the
    dense [W, Q] GEMM whose useful flops are ~6%)."""
    lib = _load()
    x = _prep(x, "pool_lastdim.x")
    tap = _prep(tap, "pool_lastdim.tap")
    R, W = x.shape
    out = torch.empty(R, Q, dtype=torch.float32, device=x.device)
    _check(lib.ks_pool_lastdim_f32(
        ctypes.c_void_p(x.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        ctypes.c_void_p(tap.data_ptr()), tap.numel(), c0, g, W, Q,
        ctypes.c_long(R), _stream()), "ks_pool_lastdim_f32")
    return out


def pool_middim(x: torch.Tensor, tap: torch.Tensor, c0: int, g: int,
                Qy: int) -> torch.Tensor:
    """Triangular lattice pooling of the MIDDLE dim with transpose:
    [B, H, C] -> [B, C, Qy] (the descriptor-gather layout)."""
    lib = _load()
    x = _prep(x, "pool_middim.x")
    tap = _prep(tap, "pool_middim.tap")
    B, H, C = x.shape
    out = torch.empty(B, C, Qy, dtype=torch.float32, device=x.device)
    _check(lib.ks_pool_middim_f32(
        ctypes.c_void_p(x.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        ctypes.c_void_p(tap.data_ptr()), tap.numel(), c0, g, H, C, Qy, B,
        _stream()), "ks_pool_middim_f32")
    return out


def woodbury_yz(T_all: torch.Tensor, LRt: torch.Tensor,
                starts: torch.Tensor, ncs: torch.Tensor):
    """Per-class Woodbury inner solve (woodbury.hip): one workgroup per
    class computes z = (I + T_c T_c^T)^{-1} (T_c LR_c) and returns
    Yz [G, db] = T_c^T z (zero rows for empty classes) plus a skipped
    mask for classes with nc > 32 (torch fallback)."""
    lib = _load()
    G, db = LRt.shape
    Yz = torch.empty(G, db, dtype=torch.float32, device=T_all.device)
    skipped = torch.zeros(G, dtype=torch.int32, device=T_all.device)
    _check(lib.ks_woodbury_f32(
        ctypes.c_void_p(T_all.data_ptr()), ctypes.c_void_p(LRt.data_ptr()),
        ctypes.c_void_p(starts.data_ptr()), ctypes.c_void_p(ncs.data_ptr()),
        ctypes.c_void_p(Yz.data_ptr()), ctypes.c_void_p(skipped.data_ptr()),
        G, db, _stream()), "ks_woodbury_f32")
    return Yz, skipped


def gmm_stats_mfma(X: torch.Tensor, mu: torch.Tensor, var: torch.Tensor,
                   w: torch.Tensor, n_blocks: int = 512):
    """GMM EM E-step sufficient statistics in ONE fused launch
    (fv_mfma.hip STATS mode): X [n, D] -> (Nk [K], Sx [K, D], Sxx [K, D],
    ll [1]) in fp64, never materializing gamma [n, K].  Grid-strided over
    32-row chunks; per-block fp32 partials are summed here in fp64."""
    lib = _load()
    X = _prep(X, "gmm_stats.X")
    n, D = X.shape
    K = mu.shape[0]
    a, b, c = _gmm_coeffs(_prep(mu, "mu"), _prep(var, "var"), _prep(w, "w"))
    P = _fv_P(a, b, c, K, X.device)
    nb = min(n_blocks, max(1, (n + 31) // 32))
    width = K + 2 * K * D + 1
    out = torch.empty(nb, width, dtype=torch.float32, device=X.device)
    _check(lib.ks_gmm_stats_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(P.data_ptr()),
        ctypes.c_void_p(out.data_ptr()), nb, n, D, K, _f16_stats(),
        _stream()), "ks_gmm_stats_f32")
    s = out.sum(0, dtype=torch.float64)
    Nk = s[:K].contiguous()
    Sx = s[K:K + K * D].view(K, D).contiguous()
    Sxx = s[K + K * D:K + 2 * K * D].view(K, D).contiguous()
    ll = s[K + 2 * K * D:].contiguous()
    return Nk, Sx, Sxx, ll


def fisher_vector(X: torch.Tensor, mu: torch.Tensor, var: torch.Tensor,
                  w: torch.Tensor) -> torch.Tensor:
    """Improved FV per image: X [B, n, D] -> [B, 2*K*D]."""
    lib = _load()
    X = _prep(X, "fisher_vector.X")
    B, n, D = X.shape
    K = mu.shape[0]
    mu = _prep(mu, "mu")
    var = _prep(var, "var")
    w = _prep(w, "w")
    a, b, c = _gmm_coeffs(mu, var, w)
    out = torch.empty(B, 2 * K * D, dtype=torch.float32, device=X.device)
    _check(lib.ks_fisher_vector_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(a.data_ptr()),
        ctypes.c_void_p(b.data_ptr()), ctypes.c_void_p(c.data_ptr()),
        ctypes.c_void_p(mu.data_ptr()), ctypes.c_void_p(var.data_ptr()),
        ctypes.c_void_p(w.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        B, n, D, K, _stream()), "ks_fisher_vector_f32")
    return out


_gauss_cache = {}

#: kernel-side Gaussian radius limit (sift.hip MAXR, dynamic-LDS bound);
#: SIFTExtractor falls back to the torch blur chain beyond it — the blur
#: must never be silently truncated
SIFT_OMAP_MAX_RADIUS = 48


def sift_omap_radius(sigma: float) -> int:
    import math as _m
    return max(1, int(_m.ceil(3.0 * sigma)))


def sift_omap(imgs: torch.Tensor, sigma: float) -> torch.Tensor:
    """Fused blur+gradient+orientation-binning: [B, H, W] -> [B, 8, H, W]."""
    lib = _load()
    imgs = _prep(imgs, "sift_omap.imgs")
    B, H, W = imgs.shape
    key = (round(sigma, 6), str(imgs.device))
    gk = _gauss_cache.get(key)
    if gk is None:
        radius = sift_omap_radius(sigma)
        if radius > SIFT_OMAP_MAX_RADIUS:
            raise RuntimeError(f"sift_omap: radius {radius} > "
                               f"{SIFT_OMAP_MAX_RADIUS} (caller must route "
                               "to the torch blur path)")
        x = torch.arange(-radius, radius + 1, dtype=torch.float32)
        k = torch.exp(-0.5 * (x / sigma) ** 2)
        gk = (k / k.sum()).to(imgs.device)
        _gauss_cache[key] = gk
    radius = (gk.numel() - 1) // 2
    out = torch.empty(B, 8, H, W, dtype=torch.float32, device=imgs.device)
    _check(lib.ks_sift_omap_f32(
        ctypes.c_void_p(imgs.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        ctypes.c_void_p(gk.data_ptr()), radius, B, H, W, _stream()),
        "ks_sift_omap_f32")
    return out


def sift_omap_pooled(imgs: torch.Tensor, sigma: float, bin_size: int,
                     g: int, Qx: int) -> torch.Tensor:
    """Fused blur+gradient+orientation + horizontal triangle pooling on
    the gcd lattice: [B, H, W] -> [B, 8, H, Qx] (column q pools at center
    ``bin//2 + q*g``, zero boundary — the merged-matrix A1 operand without
    the 17 GB omap round-trip or the 36x-waste banded GEMM)."""
    lib = _load()
    imgs = _prep(imgs, "sift_omap_pooled.imgs")
    B, H, W = imgs.shape
    key = (round(sigma, 6), str(imgs.device))
    gk = _gauss_cache.get(key)
    if gk is None:
        radius = min(sift_omap_radius(sigma), SIFT_OMAP_MAX_RADIUS)
        x = torch.arange(-radius, radius + 1, dtype=torch.float32)
        k = torch.exp(-0.5 * (x / sigma) ** 2)
        gk = (k / k.sum()).to(imgs.device)
        _gauss_cache[key] = gk
    radius = (gk.numel() - 1) // 2
    out = torch.empty(B, 8, H, Qx, dtype=torch.float32, device=imgs.device)
    _check(lib.ks_sift_omap_pooled_f32(
        ctypes.c_void_p(imgs.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        ctypes.c_void_p(gk.data_ptr()), radius, bin_size, g, Qx, B, H, W,
        _stream()), "ks_sift_omap_pooled_f32")
    return out


def sift_omap_pooled_supported(bin_size: int) -> bool:
    return 2 <= bin_size <= 9


def sift_norm_(D: torch.Tensor, clamp: float = 0.2,
               eps: float = 1e-20) -> torch.Tensor:
    """In-place SIFT descriptor normalization (L2 -> clamp -> L2) per row."""
    lib = _load()
    D = _prep_inplace(D, "sift_norm.D")
    rows, dim = D.shape
    _check(lib.ks_sift_norm_f32(
        ctypes.c_void_p(D.data_ptr()), ctypes.c_long(rows), dim, clamp, eps,
        _stream()), "ks_sift_norm_f32")
    return D


def sift_norm_slice_(D: torch.Tensor, clamp: float = 0.2,
                     eps: float = 1e-20) -> torch.Tensor:
    """In-place SIFT normalization of a [B, n, dim] SLICE of a larger
    [B, n_total, dim] buffer (rows contiguous, batch pitch arbitrary) —
    lets multi-scale descriptor assembly write each scale straight into
    the final buffer with no torch.cat pass."""
    lib = _load()
    if D.dtype != torch.float32 or not D.is_cuda:
        raise RuntimeError("sift_norm_slice_: needs CUDA float32")
    B, nrows, dim = D.shape
    if D.stride(2) != 1 or D.stride(1) != dim:
        raise RuntimeError("sift_norm_slice_: rows must be contiguous")
    _check(lib.ks_sift_norm2_f32(
        ctypes.c_void_p(D.data_ptr()), ctypes.c_long(D.stride(0)), nrows,
        ctypes.c_long(B * nrows), dim, clamp, eps, _stream()),
        "ks_sift_norm2_f32")
    return D


def sym_rectify(X: torch.Tensor, alpha: float = 0.0) -> torch.Tensor:
    lib = _load()
    X = _prep(X, "sym_rectify.X")
    rows, cols = X.shape
    out = torch.empty(rows, 2 * cols, dtype=torch.float32, device=X.device)
    _check(lib.ks_sym_rectify_f32(
        ctypes.c_void_p(X.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        rows, cols, alpha, _stream()), "ks_sym_rectify_f32")
    return out
