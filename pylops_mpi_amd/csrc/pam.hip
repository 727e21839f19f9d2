// pam — pylops-mpi hot path, MI355X-native (gfx950/CDNA4) kernels.
//
// Design notes (see also /root/repo/DESIGN.md):
//  * Every kernel is HBM-bound (stencils: 16 B/pt algorithmic fp64;
//    axpy: 24 B/pt; dot: 16 B/pt).  They are written as coalesced
//    grid-stride loops with 16-byte vector accesses (double2 / float4),
//    block = 256 threads (4 waves of 64), grids capped so blocks ≫ 256 CUs
//    but bounded (Guideline 11).
//  * The finite-difference stencils are FUSED: one pass, reading the
//    neighbour halo planes directly from small exchange buffers instead of
//    materializing a concatenated ghosted copy the way the reference does
//    (ref DistributedArray.py:974,992-994,1028 — an extra full read+write
//    per apply there).
//  * Stencil math is table-driven: each (operator, kind, direction) is a
//    compile-time list of (row offset, coefficient, global-row interval)
//    terms — the closed forms of the reference's slice algebra
//    (ref basicoperators/FirstDerivative.py:141-318,
//     SecondDerivative.py:124-256), verified against the oracle's
//    rank-simulated restatement and dense-transpose adjoints.
//  * Reductions accumulate in float64 with a fixed tree shape for a given
//    n (grid-stride per-thread accumulation -> 64-lane __shfl_down wave
//    reduction -> LDS tree -> fixed-size partial buffer -> one-block
//    deterministic combine), so CGLS traces are reproducible run-to-run.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 -ffp-contract=off
//        -fPIC -shared pam.hip -o ../libpam.so
// (-ffp-contract=off: keep mul+add roundings separate so results track the
//  reference NumPy op-for-op at the ulp level.)

#include <hip/hip_runtime.h>
#include <math.h>
#include <stdint.h>
#include <stdlib.h>
#include <utility>

#include "../../include/pam.h"
#include "fd_defs.h"

#define PAM_ABI_VERSION 1
#define BLK 256
#define NPARTIAL 1024  // fixed stage-1 partial count cap (determinism)

extern "C" int64_t pam_version(void) { return PAM_ABI_VERSION; }
extern "C" int64_t pam_reduce_ws_elems(void) { return NPARTIAL; }

// ---------------------------------------------------------------------------
// helpers
// ---------------------------------------------------------------------------
static inline int check(hipError_t e) { return (int)e; }

static inline int64_t grid_1d(int64_t work) {
  // One block per BLK work items — NO grid-stride looping by default.
  // A/B at the bench size (fp64 axpy, single box): cap=4096 (the old
  // default) 4.52 TB/s, 32768 4.98, 65536 5.14, uncapped 5.78 TB/s
  // (+28%) — the stride loop was the bottleneck, not launch width.
  // (The deterministic dot/norm reductions are unaffected: their
  // stage-1 grid is the fixed NPARTIAL tree, not grid_1d.)
  // PAM_EW_CAP restores a cap for A/Bs.
  static int64_t cap = [] {
    const char* e = getenv("PAM_EW_CAP");
    return (int64_t)(e ? atoll(e) : 0x7FFFFFFF);
  }();
  int64_t g = (work + BLK - 1) / BLK;
  if (g > cap) g = cap;
  if (g < 1) g = 1;
  return g;
}

// ---------------------------------------------------------------------------
// element-wise kernels (ref DistributedArray.py:605-683 local math)
// EW op codes: 0 fill, 1 neg, 2 add, 3 sub, 4 mul, 5 scale, 6 axpy, 7 xpby
// ---------------------------------------------------------------------------
template <typename T, int OP, int V>
__device__ __forceinline__ void ew_body(T* __restrict__ y,
                                        const T* __restrict__ a,
                                        const T* __restrict__ b,
                                        T alpha, int64_t n) {
  const int64_t nv = n / V;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
       i += stride) {
    T va[V], vb[V], vy[V];
    if constexpr (OP == 1 || OP == 5) {  // unary on a
      loadv<T, V>(a + i * V, va);
    } else if constexpr (OP >= 2 && OP <= 4) {  // binary a,b
      loadv<T, V>(a + i * V, va);
      loadv<T, V>(b + i * V, vb);
    } else if constexpr (OP == 6 || OP == 7) {  // y & a
      loadv<T, V>(a + i * V, va);
      loadv<T, V>(y + i * V, vy);
    }
#pragma unroll
    for (int k = 0; k < V; ++k) {
      if constexpr (OP == 0) vy[k] = alpha;
      else if constexpr (OP == 1) vy[k] = -va[k];
      else if constexpr (OP == 2) vy[k] = va[k] + vb[k];
      else if constexpr (OP == 3) vy[k] = va[k] - vb[k];
      else if constexpr (OP == 4) vy[k] = va[k] * vb[k];
      else if constexpr (OP == 5) vy[k] = alpha * va[k];
      else if constexpr (OP == 6) vy[k] = vy[k] + alpha * va[k];
      else if constexpr (OP == 7) vy[k] = va[k] + alpha * vy[k];
    }
    storev<T, V>(y + i * V, vy);
  }
  // scalar tail
  const int64_t tail0 = nv * V;
  const int64_t gid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (int64_t i = tail0 + gid; i < n; i += stride) {
    T va = 0, vb = 0;
    if constexpr (OP != 0) va = a[i];
    if constexpr (OP >= 2 && OP <= 4) vb = b[i];
    if constexpr (OP == 0) y[i] = alpha;
    else if constexpr (OP == 1) y[i] = -va;
    else if constexpr (OP == 2) y[i] = va + vb;
    else if constexpr (OP == 3) y[i] = va - vb;
    else if constexpr (OP == 4) y[i] = va * vb;
    else if constexpr (OP == 5) y[i] = alpha * va;
    else if constexpr (OP == 6) y[i] = y[i] + alpha * va;
    else if constexpr (OP == 7) y[i] = va + alpha * y[i];
  }
}

template <typename T, int OP, int V>
__global__ void __launch_bounds__(BLK) ew_kernel(T* __restrict__ y,
                                                 const T* __restrict__ a,
                                                 const T* __restrict__ b,
                                                 T alpha, int64_t n) {
  ew_body<T, OP, V>(y, a, b, alpha, n);
}

// device-scalar variant: alpha lives in HBM (1-elem f64 written by a prior
// reduction / pam_scalar_* kernel on the same stream) so the CG/CGLS
// recurrence (ref cls_basic.py:370-404) runs with no host round-trip per
// update.  scale is a host constant (+-1.0): (T)(scale * *ap) is exact for
// sign flips, so results are bit-identical to the host-scalar kernels.
template <typename T, int OP, int V>
__global__ void __launch_bounds__(BLK) ewd_kernel(T* __restrict__ y,
                                                  const T* __restrict__ a,
                                                  const double* __restrict__ ap,
                                                  double scale, int64_t n) {
  ew_body<T, OP, V>(y, a, nullptr, (T)(scale * ap[0]), n);
}

template <typename T, int OP>
static int ew_launch(void* stream, void* y, const void* a, const void* b,
                     double alpha, int64_t n) {
  if (n < 0) return PAM_EARG;
  if (n == 0) return 0;
  constexpr int V = VecW<T>::value;
  const bool aligned = ((uintptr_t)y % 16 == 0) &&
                       (a == nullptr || (uintptr_t)a % 16 == 0) &&
                       (b == nullptr || (uintptr_t)b % 16 == 0);
  hipStream_t s = (hipStream_t)stream;
  const int64_t grid = grid_1d(n / (aligned ? V : 1) + 1);
  if (aligned)
    hipLaunchKernelGGL((ew_kernel<T, OP, V>), dim3(grid), dim3(BLK), 0, s,
                       (T*)y, (const T*)a, (const T*)b, (T)alpha, n);
  else
    hipLaunchKernelGGL((ew_kernel<T, OP, 1>), dim3(grid), dim3(BLK), 0, s,
                       (T*)y, (const T*)a, (const T*)b, (T)alpha, n);
  return check(hipGetLastError());
}

#define EW_ENTRY(name, OP, A, B, ALPHA)                                       \
  extern "C" int name {                                                       \
    if (dtype == PAM_F64)                                                     \
      return ew_launch<double, OP>(stream, y, A, B, ALPHA, n);                \
    if (dtype == PAM_F32)                                                     \
      return ew_launch<float, OP>(stream, y, A, B, ALPHA, n);                 \
    return PAM_EDTYPE;                                                        \
  }

EW_ENTRY(pam_fill(void* stream, void* y, int64_t n, double value, int dtype),
         0, nullptr, nullptr, value)
EW_ENTRY(pam_neg(void* stream, void* y, const void* x, int64_t n, int dtype),
         1, x, nullptr, 0.0)
EW_ENTRY(pam_add(void* stream, void* y, const void* a, const void* b,
                 int64_t n, int dtype),
         2, a, b, 0.0)
EW_ENTRY(pam_sub(void* stream, void* y, const void* a, const void* b,
                 int64_t n, int dtype),
         3, a, b, 0.0)
EW_ENTRY(pam_mul(void* stream, void* y, const void* a, const void* b,
                 int64_t n, int dtype),
         4, a, b, 0.0)
EW_ENTRY(pam_scale(void* stream, void* y, const void* x, double alpha,
                   int64_t n, int dtype),
         5, x, nullptr, alpha)
EW_ENTRY(pam_axpy(void* stream, void* y, const void* x, double alpha,
                  int64_t n, int dtype),
         6, x, nullptr, alpha)
EW_ENTRY(pam_xpby(void* stream, void* y, const void* x, double beta,
                  int64_t n, int dtype),
         7, x, nullptr, beta)

// ---------------------------------------------------------------------------
// device-scalar solver fast path (ref optimization/cls_basic.py:370-404):
// axpy/xpby whose scalar is read from device memory, plus the two tiny
// recurrence-scalar kernels.  Together these let a whole CG/CGLS iteration
// be launched with a single host synchronization (the stop-test readback).
// ---------------------------------------------------------------------------
template <typename T, int OP>
static int ewd_launch(void* stream, void* y, const void* a, const void* ap,
                      double scale, int64_t n) {
  if (n < 0 || !y || !a || !ap) return PAM_EARG;
  if (n == 0) return 0;
  constexpr int V = VecW<T>::value;
  const bool aligned =
      ((uintptr_t)y % 16 == 0) && ((uintptr_t)a % 16 == 0);
  hipStream_t s = (hipStream_t)stream;
  const int64_t grid = grid_1d(n / (aligned ? V : 1) + 1);
  if (aligned)
    hipLaunchKernelGGL((ewd_kernel<T, OP, V>), dim3(grid), dim3(BLK), 0, s,
                       (T*)y, (const T*)a, (const double*)ap, scale, n);
  else
    hipLaunchKernelGGL((ewd_kernel<T, OP, 1>), dim3(grid), dim3(BLK), 0, s,
                       (T*)y, (const T*)a, (const double*)ap, scale, n);
  return check(hipGetLastError());
}

// y += scale * (*alpha) * x
extern "C" int pam_axpy_d(void* stream, void* y, const void* x,
                          const void* alpha, double scale, int64_t n,
                          int dtype) {
  if (dtype == PAM_F64) return ewd_launch<double, 6>(stream, y, x, alpha, scale, n);
  if (dtype == PAM_F32) return ewd_launch<float, 6>(stream, y, x, alpha, scale, n);
  return PAM_EDTYPE;
}

// y = x + scale * (*beta) * y
extern "C" int pam_xpby_d(void* stream, void* y, const void* x,
                          const void* beta, double scale, int64_t n,
                          int dtype) {
  if (dtype == PAM_F64) return ewd_launch<double, 7>(stream, y, x, beta, scale, n);
  if (dtype == PAM_F32) return ewd_launch<float, 7>(stream, y, x, beta, scale, n);
  return PAM_EDTYPE;
}

// *out = |num[0] / (den[0] + damp * den[1])| — the CG/CGLS step scalars
// a = |kold / (q.q + damp^2 c.c)| (ref cls_basic.py:379-383; CG a with
// damp==0, ref :121-123).  |x/y| == |x|/|y| in IEEE, so this matches the
// reference's np.abs placement for the non-negative numerators involved.
__global__ void scalar_alpha_kernel(double* __restrict__ out,
                                    const double* __restrict__ num,
                                    const double* __restrict__ den,
                                    double damp) {
  out[0] = fabs(num[0] / (den[0] + damp * den[1]));
}

// *out = |num[0] / den[0]| — b = k/kold (ref cls_basic.py:394-395)
__global__ void scalar_div_kernel(double* __restrict__ out,
                                  const double* __restrict__ num,
                                  const double* __restrict__ den) {
  out[0] = fabs(num[0] / den[0]);
}

extern "C" int pam_scalar_alpha(void* stream, void* out, const void* num,
                                const void* den, double damp) {
  if (!out || !num || !den) return PAM_EARG;
  hipLaunchKernelGGL(scalar_alpha_kernel, dim3(1), dim3(1), 0,
                     (hipStream_t)stream, (double*)out, (const double*)num,
                     (const double*)den, damp);
  return check(hipGetLastError());
}

extern "C" int pam_scalar_div(void* stream, void* out, const void* num,
                              const void* den) {
  if (!out || !num || !den) return PAM_EARG;
  hipLaunchKernelGGL(scalar_div_kernel, dim3(1), dim3(1), 0,
                     (hipStream_t)stream, (double*)out, (const double*)num,
                     (const double*)den);
  return check(hipGetLastError());
}

// ---------------------------------------------------------------------------
// complex element-wise (interleaved re,im).  EWC op: 0 cmul, 1 cscale, 2 conj
// (ref DistributedArray.py:661-683, :840-854).  add/neg/real-axpy on complex
// arrays reuse the real kernels on the 2n view (solver scalars are real).
// ---------------------------------------------------------------------------
template <typename T, int OP>
__global__ void __launch_bounds__(BLK) ewc_kernel(T* __restrict__ y,
                                                  const T* __restrict__ a,
                                                  const T* __restrict__ b,
                                                  T ar, T ai, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const T xr = a[2 * i], xi = a[2 * i + 1];
    T zr, zi;
    if constexpr (OP == 0) {
      const T br = b[2 * i], bi = b[2 * i + 1];
      zr = xr * br - xi * bi;
      zi = xr * bi + xi * br;
    } else if constexpr (OP == 1) {
      zr = xr * ar - xi * ai;
      zi = xr * ai + xi * ar;
    } else {
      zr = xr;
      zi = -xi;
    }
    y[2 * i] = zr;
    y[2 * i + 1] = zi;
  }
}

template <typename T, int OP>
static int ewc_launch(void* stream, void* y, const void* a, const void* b,
                      double ar, double ai, int64_t n) {
  if (n < 0) return PAM_EARG;
  if (n == 0) return 0;
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL((ewc_kernel<T, OP>), dim3(grid_1d(n)), dim3(BLK), 0, s,
                     (T*)y, (const T*)a, (const T*)b, (T)ar, (T)ai, n);
  return check(hipGetLastError());
}

#define EWC_ENTRY(name, OP, B, AR, AI)                                        \
  extern "C" int name {                                                       \
    if (dtype == PAM_C128)                                                    \
      return ewc_launch<double, OP>(stream, y, a, B, AR, AI, n);              \
    if (dtype == PAM_C64)                                                     \
      return ewc_launch<float, OP>(stream, y, a, B, AR, AI, n);               \
    return PAM_EDTYPE;                                                        \
  }

EWC_ENTRY(pam_cmul(void* stream, void* y, const void* a, const void* b,
                   int64_t n, int dtype),
          0, b, 0.0, 0.0)
EWC_ENTRY(pam_cscale(void* stream, void* y, const void* a, double alpha_re,
                     double alpha_im, int64_t n, int dtype),
          1, nullptr, alpha_re, alpha_im)
EWC_ENTRY(pam_conj(void* stream, void* y, const void* a, int64_t n,
                   int dtype),
          2, nullptr, 0.0, 0.0)

// ---------------------------------------------------------------------------
// complex <-> real (de)interleave: the MDC chain's real extraction
// feeding rfft (ref MDC.py:55-69 — pylops' real FFT discards imag) and
// the real->complex(0 imag) output carrier.  torch's strided
// elementwise copies run these at ~3.7 TB/s (r02 MDC kernel trace); the
// vectorized forms below stream at the plain copy rate.  unzip reads V
// complex pairs per lane and writes V reals; zip the reverse.
// ---------------------------------------------------------------------------
template <typename T, int V>
__global__ void __launch_bounds__(BLK) unzip_kernel(
    T* __restrict__ dst, const T* __restrict__ src, int64_t n) {
  const int64_t nv = n / V;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t iv = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; iv < nv;
       iv += stride) {
    T pair[2 * V], out[V];
    loadv<T, V>(src + 2 * iv * V, pair);
    loadv<T, V>(src + 2 * iv * V + V, pair + V);
#pragma unroll
    for (int k = 0; k < V; ++k) out[k] = pair[2 * k];
    storev<T, V>(dst + iv * V, out);
  }
  const int64_t gid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (int64_t i = nv * V + gid; i < n; i += stride)
    dst[i] = src[2 * i];
}

template <typename T, int V>
__global__ void __launch_bounds__(BLK) zip_kernel(
    T* __restrict__ dst, const T* __restrict__ src, int64_t n) {
  const int64_t nv = n / V;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t iv = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; iv < nv;
       iv += stride) {
    T in[V], pair[2 * V];
    loadv<T, V>(src + iv * V, in);
#pragma unroll
    for (int k = 0; k < V; ++k) {
      pair[2 * k] = in[k];
      pair[2 * k + 1] = (T)0;
    }
    storev<T, V>(dst + 2 * iv * V, pair);
    storev<T, V>(dst + 2 * iv * V + V, pair + V);
  }
  const int64_t gid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (int64_t i = nv * V + gid; i < n; i += stride) {
    dst[2 * i] = src[i];
    dst[2 * i + 1] = (T)0;
  }
}

template <typename T>
static int zip_launch(void* stream, void* dst, const void* src, int64_t n,
                      bool unzip) {
  if (n < 0 || !dst || !src) return PAM_EARG;
  if (n == 0) return 0;
  constexpr int V = VecW<T>::value;
  const bool aligned = ((uintptr_t)dst % 16 == 0) &&
                       ((uintptr_t)src % 16 == 0);
  hipStream_t s = (hipStream_t)stream;
  const int64_t grid = grid_1d(n / (aligned ? V : 1) + 1);
  if (unzip) {
    if (aligned)
      hipLaunchKernelGGL((unzip_kernel<T, V>), dim3(grid), dim3(BLK), 0, s,
                         (T*)dst, (const T*)src, n);
    else
      hipLaunchKernelGGL((unzip_kernel<T, 1>), dim3(grid), dim3(BLK), 0, s,
                         (T*)dst, (const T*)src, n);
  } else {
    if (aligned)
      hipLaunchKernelGGL((zip_kernel<T, V>), dim3(grid), dim3(BLK), 0, s,
                         (T*)dst, (const T*)src, n);
    else
      hipLaunchKernelGGL((zip_kernel<T, 1>), dim3(grid), dim3(BLK), 0, s,
                         (T*)dst, (const T*)src, n);
  }
  return check(hipGetLastError());
}

extern "C" int pam_unzip(void* stream, void* dst_real, const void* src_cplx,
                         int64_t n, int dtype) {
  if (dtype == PAM_C128)
    return zip_launch<double>(stream, dst_real, src_cplx, n, true);
  if (dtype == PAM_C64)
    return zip_launch<float>(stream, dst_real, src_cplx, n, true);
  return PAM_EDTYPE;
}

extern "C" int pam_zip(void* stream, void* dst_cplx, const void* src_real,
                       int64_t n, int dtype) {
  if (dtype == PAM_C128)
    return zip_launch<double>(stream, dst_cplx, src_real, n, false);
  if (dtype == PAM_C64)
    return zip_launch<float>(stream, dst_cplx, src_real, n, false);
  return PAM_EDTYPE;
}

// ---------------------------------------------------------------------------
// transpose-fused (de)interleave for the MDC chain's FFT layout (r02):
// rocFFT's strided real plans insert their own full pack/unpack copy
// kernels (~500 us/direction at the cfg5 shape, r02 trace), so the
// chain transposes to the time-contiguous layout itself — fused with
// the complex<->real conversion it already needs, at the LDS-tiled
// transpose rate.  unzipT: complex (nt, m) -> real (m, nt) (real parts
// only).  zipT: real (m, nt) -> complex (nt, m) with zero imag.
// 32x32 LDS tiles (+1 pad), 256 threads as 8x32.
// ---------------------------------------------------------------------------
template <typename T>
__global__ void __launch_bounds__(BLK) unzipT_kernel(
    T* __restrict__ dst, const T* __restrict__ src, int64_t nt, int64_t m) {
  __shared__ T tile[32][33];
  const int64_t k0 = (int64_t)blockIdx.y * 32;  // time rows
  const int64_t j0 = (int64_t)blockIdx.x * 32;  // columns
  const int tr = threadIdx.x / 32;
  const int tc = threadIdx.x % 32;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int64_t k = k0 + tr + 8 * i;
    if (k < nt && j0 + tc < m)
      tile[tr + 8 * i][tc] = src[2 * (k * m + j0 + tc)];
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int64_t j = j0 + tr + 8 * i;
    if (j < m && k0 + tc < nt)
      dst[j * nt + k0 + tc] = tile[tc][tr + 8 * i];
  }
}

template <typename T>
__global__ void __launch_bounds__(BLK) zipT_kernel(
    T* __restrict__ dst, const T* __restrict__ src, int64_t nt, int64_t m) {
  __shared__ T tile[32][33];
  const int64_t j0 = (int64_t)blockIdx.y * 32;  // columns (src rows)
  const int64_t k0 = (int64_t)blockIdx.x * 32;  // time
  const int tr = threadIdx.x / 32;
  const int tc = threadIdx.x % 32;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int64_t j = j0 + tr + 8 * i;
    if (j < m && k0 + tc < nt)
      tile[tr + 8 * i][tc] = src[j * nt + k0 + tc];
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int64_t k = k0 + tr + 8 * i;
    if (k < nt && j0 + tc < m) {
      const int64_t off = 2 * (k * m + j0 + tc);
      dst[off] = tile[tc][tr + 8 * i];
      dst[off + 1] = (T)0;
    }
  }
}

extern "C" int pam_unzip_t(void* stream, void* dst_real,
                           const void* src_cplx, int64_t nt, int64_t m,
                           int dtype) {
  if (nt <= 0 || m <= 0 || !dst_real || !src_cplx) return PAM_EARG;
  dim3 grid((uint32_t)((m + 31) / 32), (uint32_t)((nt + 31) / 32));
  hipStream_t s = (hipStream_t)stream;
  if (dtype == PAM_C128) {
    hipLaunchKernelGGL((unzipT_kernel<double>), grid, dim3(BLK), 0, s,
                       (double*)dst_real, (const double*)src_cplx, nt, m);
    return check(hipGetLastError());
  }
  if (dtype == PAM_C64) {
    hipLaunchKernelGGL((unzipT_kernel<float>), grid, dim3(BLK), 0, s,
                       (float*)dst_real, (const float*)src_cplx, nt, m);
    return check(hipGetLastError());
  }
  return PAM_EDTYPE;
}

extern "C" int pam_zip_t(void* stream, void* dst_cplx, const void* src_real,
                         int64_t nt, int64_t m, int dtype) {
  if (nt <= 0 || m <= 0 || !dst_cplx || !src_real) return PAM_EARG;
  dim3 grid((uint32_t)((nt + 31) / 32), (uint32_t)((m + 31) / 32));
  hipStream_t s = (hipStream_t)stream;
  if (dtype == PAM_C128) {
    hipLaunchKernelGGL((zipT_kernel<double>), grid, dim3(BLK), 0, s,
                       (double*)dst_cplx, (const double*)src_real, nt, m);
    return check(hipGetLastError());
  }
  if (dtype == PAM_C64) {
    hipLaunchKernelGGL((zipT_kernel<float>), grid, dim3(BLK), 0, s,
                       (float*)dst_cplx, (const float*)src_real, nt, m);
    return check(hipGetLastError());
  }
  return PAM_EDTYPE;
}

// ---------------------------------------------------------------------------
// thresholding for ISTA/FISTA (the pylops _softthreshold/_hardthreshold
// formulas the reference imports, ref optimization/cls_sparsity.py:10):
//   soft real   : sign(x) * max(|x|-t, 0)
//   soft complex: z * max(|z|-t, 0)/|z|
//   hard        : x * (|x| >= sqrt(2 t))
//   half        : L1/2 prox (Xu et al. 2012, the published formula pylops'
//                 _halfthreshold implements — the EXACT prox of
//                 (t/2)|v|^(1/2), matching ISTA's thresh = eps*alpha*0.5):
//                 for |x| > (54^(1/3)/4) t^(2/3),
//                 y = (2/3) x (1 + cos(2pi/3 - (2/3) acos((t/8)(|x|/3)^-1.5)))
//                 else 0; complex by the magnitude rule.  Pinned by the prox
//                 optimality property test (tests/test_oracle_proximal.py) —
//                 pylops itself is absent from /root/reference.
// kind: 0 soft, 1 hard, 2 half.  In-place safe (y may alias x).
// ---------------------------------------------------------------------------
template <typename T>
__device__ __forceinline__ T half_factor(T a_, T thresh_) {
  // multiplicative factor of the L1/2 prox at magnitude a (>= 0).
  // Always evaluated in double so the cutoff decision (a discontinuity)
  // and the acos/pow chain agree with the f64 oracle for every dtype.
  const double a = (double)a_, thresh = (double)thresh_;
  const double cut = 0.9449407874211548 * pow(thresh, 2.0 / 3.0);
  if (!(a > cut)) return (T)0;  // 54^(1/3)/4 = 0.944940787421...
  const double phi = acos((thresh / 8.0) * pow(a / 3.0, -1.5));
  return (T)((2.0 / 3.0) *
             (1.0 + cos(2.0 * M_PI / 3.0 - (2.0 / 3.0) * phi)));
}

template <typename T, bool CPLX, int KIND>
__global__ void __launch_bounds__(BLK) thresh_kernel(T* __restrict__ y,
                                                     const T* __restrict__ x,
                                                     T thresh, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if constexpr (!CPLX) {
      const T v = x[i];
      if constexpr (KIND == 0) {
        const T m = fabs(v) - thresh;
        y[i] = m > (T)0 ? copysign(m, v) : (T)0;
      } else if constexpr (KIND == 1) {
        y[i] = fabs(v) >= sqrt((T)2 * thresh) ? v : (T)0;
      } else {
        y[i] = v * half_factor(fabs(v), thresh);
      }
    } else {
      const T zr = x[2 * i], zi = x[2 * i + 1];
      const T a = hypot(zr, zi);
      T s;
      if constexpr (KIND == 0) {
        const T m = a - thresh;
        s = (m > (T)0 && a > (T)0) ? m / a : (T)0;
      } else if constexpr (KIND == 1) {
        s = a >= sqrt((T)2 * thresh) ? (T)1 : (T)0;
      } else {
        s = half_factor(a, thresh);
      }
      y[2 * i] = zr * s;
      y[2 * i + 1] = zi * s;
    }
  }
}

template <typename T, bool CPLX>
static int thresh_launch(void* stream, void* y, const void* x, int kind,
                         double t, int64_t n) {
  if (n < 0) return PAM_EARG;
  if (n == 0) return 0;
  hipStream_t s = (hipStream_t)stream;
  if (kind == 0)
    hipLaunchKernelGGL((thresh_kernel<T, CPLX, 0>), dim3(grid_1d(n)),
                       dim3(BLK), 0, s, (T*)y, (const T*)x, (T)t, n);
  else if (kind == 1)
    hipLaunchKernelGGL((thresh_kernel<T, CPLX, 1>), dim3(grid_1d(n)),
                       dim3(BLK), 0, s, (T*)y, (const T*)x, (T)t, n);
  else if (kind == 2)
    hipLaunchKernelGGL((thresh_kernel<T, CPLX, 2>), dim3(grid_1d(n)),
                       dim3(BLK), 0, s, (T*)y, (const T*)x, (T)t, n);
  else
    return PAM_EOP;
  return check(hipGetLastError());
}

extern "C" int pam_thresh(void* stream, void* y, const void* x, int64_t n,
                          int kind, double thresh, int dtype) {
  switch (dtype) {
    case PAM_F64: return thresh_launch<double, false>(stream, y, x, kind, thresh, n);
    case PAM_F32: return thresh_launch<float, false>(stream, y, x, kind, thresh, n);
    case PAM_C128: return thresh_launch<double, true>(stream, y, x, kind, thresh, n);
    case PAM_C64: return thresh_launch<float, true>(stream, y, x, kind, thresh, n);
    default: return PAM_EDTYPE;
  }
}

// ---------------------------------------------------------------------------
// reductions (ref DistributedArray.py:685-717 dot, :719-838 norms)
// RED codes: 0 dot, 1 powsum |x^p|, 2 max|x|, 3 min|x|, 4 count_nonzero
// ---------------------------------------------------------------------------
template <int RED>
__device__ __forceinline__ double red_combine(double u, double v) {
  if constexpr (RED == 2) return fmax(u, v);
  else if constexpr (RED == 3) return fmin(u, v);
  else return u + v;
}

template <int RED> __device__ __forceinline__ double red_init() {
  if constexpr (RED == 2) return 0.0;          // max over |x| >= 0
  else if constexpr (RED == 3) return INFINITY;
  else return 0.0;
}

template <typename T, int RED, bool CPLX = false>
__global__ void __launch_bounds__(BLK) reduce_stage1(
    const T* __restrict__ x, const T* __restrict__ y, int64_t n, double p,
    double* __restrict__ partials) {
  double acc = red_init<RED>();
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double v;
    if constexpr (CPLX) {
      // reduce over |z| (norms only; |z^p| == |z|^p for real p)
      const double zr = (double)x[2 * i], zi = (double)x[2 * i + 1];
      if constexpr (RED == 1) {
        // fast paths: pow with a runtime exponent is transcendental-bound
        // (measured 3.4x a dot pass); p==2/p==1 cover every solver norm
        if (p == 2.0) v = zr * zr + zi * zi;
        else if (p == 1.0) v = hypot(zr, zi);
        else v = pow(hypot(zr, zi), p);
      }
      else if constexpr (RED == 2 || RED == 3) v = hypot(zr, zi);
      else v = (zr != 0.0 || zi != 0.0) ? 1.0 : 0.0;
    } else {
      const double xv = (double)x[i];
      if constexpr (RED == 0) v = xv * (double)y[i];
      else if constexpr (RED == 1) {
        // ref :786 float_power; pow(x,2) is correctly rounded == x*x and
        // |pow(x,1)| == |x|, so the fast paths are bit-identical
        if (p == 2.0) v = xv * xv;
        else if (p == 1.0) v = fabs(xv);
        else v = fabs(pow(xv, p));
      }
      else if constexpr (RED == 2 || RED == 3) v = fabs(xv);
      else v = (xv != 0.0) ? 1.0 : 0.0;
    }
    acc = red_combine<RED>(acc, v);
  }
  // 64-lane wavefront shuffle reduction
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    acc = red_combine<RED>(acc, __shfl_down(acc, off, 64));
  __shared__ double lds[BLK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) lds[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    double r = lds[0];
#pragma unroll
    for (int w = 1; w < BLK / 64; ++w) r = red_combine<RED>(r, lds[w]);
    partials[blockIdx.x] = r;
  }
}

template <int RED>
__global__ void __launch_bounds__(BLK) reduce_stage2(
    const double* __restrict__ partials, int64_t np, double* __restrict__ out) {
  double acc = red_init<RED>();
  for (int64_t i = threadIdx.x; i < np; i += BLK)
    acc = red_combine<RED>(acc, partials[i]);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    acc = red_combine<RED>(acc, __shfl_down(acc, off, 64));
  __shared__ double lds[BLK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) lds[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    double r = lds[0];
#pragma unroll
    for (int w = 1; w < BLK / 64; ++w) r = red_combine<RED>(r, lds[w]);
    *out = r;
  }
}

template <typename T, int RED, bool CPLX = false>
static int reduce_launch(void* stream, const void* x, const void* y, int64_t n,
                         double p, void* ws, void* out) {
  if (n < 0 || !x || !ws || !out) return PAM_EARG;
  hipStream_t s = (hipStream_t)stream;
  if (n == 0) {
    double z = (RED == 3) ? INFINITY : 0.0;
    // degenerate case: blocking copy (no stack-lifetime hazard)
    return check(hipMemcpy(out, &z, sizeof(double), hipMemcpyHostToDevice));
  }
  (void)s;
  int64_t nb = (n + BLK - 1) / BLK;
  if (nb > NPARTIAL) nb = NPARTIAL;
  hipLaunchKernelGGL((reduce_stage1<T, RED, CPLX>), dim3(nb), dim3(BLK), 0, s,
                     (const T*)x, (const T*)y, n, p, (double*)ws);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return (int)e;
  hipLaunchKernelGGL((reduce_stage2<RED>), dim3(1), dim3(BLK), 0, s,
                     (const double*)ws, nb, (double*)out);
  return check(hipGetLastError());
}

// ---- complex dot (out = sum x*y or sum conj(x)*y; 2-double result)
template <typename T, bool CONJX>
__global__ void __launch_bounds__(BLK) cdot_stage1(
    const T* __restrict__ x, const T* __restrict__ y, int64_t n,
    double* __restrict__ pre, double* __restrict__ pim) {
  double ar = 0.0, ai = 0.0;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const double xr = (double)x[2 * i];
    const double xi = CONJX ? -(double)x[2 * i + 1] : (double)x[2 * i + 1];
    const double yr = (double)y[2 * i], yi = (double)y[2 * i + 1];
    ar += xr * yr - xi * yi;
    ai += xr * yi + xi * yr;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    ar += __shfl_down(ar, off, 64);
    ai += __shfl_down(ai, off, 64);
  }
  __shared__ double lr[BLK / 64], li[BLK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) {
    lr[wave] = ar;
    li[wave] = ai;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    double sr = lr[0], si = li[0];
#pragma unroll
    for (int w = 1; w < BLK / 64; ++w) {
      sr += lr[w];
      si += li[w];
    }
    pre[blockIdx.x] = sr;
    pim[blockIdx.x] = si;
  }
}

__global__ void __launch_bounds__(BLK) cdot_stage2(
    const double* __restrict__ pre, const double* __restrict__ pim,
    int64_t np, double* __restrict__ out) {
  double ar = 0.0, ai = 0.0;
  for (int64_t i = threadIdx.x; i < np; i += BLK) {
    ar += pre[i];
    ai += pim[i];
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    ar += __shfl_down(ar, off, 64);
    ai += __shfl_down(ai, off, 64);
  }
  __shared__ double lr[BLK / 64], li[BLK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) {
    lr[wave] = ar;
    li[wave] = ai;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    double sr = lr[0], si = li[0];
#pragma unroll
    for (int w = 1; w < BLK / 64; ++w) {
      sr += lr[w];
      si += li[w];
    }
    out[0] = sr;
    out[1] = si;
  }
}

template <typename T>
static int cdot_launch(void* stream, const void* x, const void* y, int64_t n,
                       int conjx, void* ws, void* out) {
  if (n < 0 || !x || !y || !ws || !out) return PAM_EARG;
  hipStream_t s = (hipStream_t)stream;
  if (n == 0) {
    double z[2] = {0.0, 0.0};
    return check(hipMemcpy(out, z, sizeof(z), hipMemcpyHostToDevice));
  }
  int64_t nb = (n + BLK - 1) / BLK;
  if (nb > NPARTIAL) nb = NPARTIAL;
  double* pre = (double*)ws;
  double* pim = pre + NPARTIAL;
  if (conjx)
    hipLaunchKernelGGL((cdot_stage1<T, true>), dim3(nb), dim3(BLK), 0, s,
                       (const T*)x, (const T*)y, n, pre, pim);
  else
    hipLaunchKernelGGL((cdot_stage1<T, false>), dim3(nb), dim3(BLK), 0, s,
                       (const T*)x, (const T*)y, n, pre, pim);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return (int)e;
  hipLaunchKernelGGL(cdot_stage2, dim3(1), dim3(BLK), 0, s, pre, pim, nb,
                     (double*)out);
  return check(hipGetLastError());
}

extern "C" int pam_cdot(void* stream, const void* x, const void* y, int64_t n,
                        int conjx, void* ws, void* out, int dtype) {
  if (dtype == PAM_C128)
    return cdot_launch<double>(stream, x, y, n, conjx, ws, out);
  if (dtype == PAM_C64)
    return cdot_launch<float>(stream, x, y, n, conjx, ws, out);
  return PAM_EDTYPE;
}

extern "C" int pam_dot(void* stream, const void* x, const void* y, int64_t n,
                       void* ws, void* out, int dtype) {
  if (!y) return PAM_EARG;
  if (dtype == PAM_F64)
    return reduce_launch<double, 0>(stream, x, y, n, 0.0, ws, out);
  if (dtype == PAM_F32)
    return reduce_launch<float, 0>(stream, x, y, n, 0.0, ws, out);
  return PAM_EDTYPE;
}

extern "C" int pam_norm_local(void* stream, const void* x, int64_t n, int op,
                              double p, void* ws, void* out, int dtype) {
  if (op < 0 || op > 3) return PAM_EOP;
#define NORM_CASE(T)                                                          \
  switch (op) {                                                               \
    case 0: return reduce_launch<T, 1>(stream, x, nullptr, n, p, ws, out);    \
    case 1: return reduce_launch<T, 2>(stream, x, nullptr, n, p, ws, out);    \
    case 2: return reduce_launch<T, 3>(stream, x, nullptr, n, p, ws, out);    \
    default: return reduce_launch<T, 4>(stream, x, nullptr, n, p, ws, out);   \
  }
  if (dtype == PAM_F64) { NORM_CASE(double) }
  if (dtype == PAM_F32) { NORM_CASE(float) }
#undef NORM_CASE
#define CNORM_CASE(T)                                                         \
  switch (op) {                                                               \
    case 0: return reduce_launch<T, 1, true>(stream, x, nullptr, n, p, ws, out); \
    case 1: return reduce_launch<T, 2, true>(stream, x, nullptr, n, p, ws, out); \
    case 2: return reduce_launch<T, 3, true>(stream, x, nullptr, n, p, ws, out); \
    default: return reduce_launch<T, 4, true>(stream, x, nullptr, n, p, ws, out); \
  }
  if (dtype == PAM_C128) { CNORM_CASE(double) }
  if (dtype == PAM_C64) { CNORM_CASE(float) }
#undef CNORM_CASE
  return PAM_EDTYPE;
}

// ---------------------------------------------------------------------------
// dense GEMV (local apply of MPIBlockDiag dense operators,
// ref basicoperators/BlockDiag.py:122-144).  HBM-bound: the matrix read
// dominates (nr*nc elements, read once).
// ---------------------------------------------------------------------------
#define GEMV_CHUNKS 64  // deterministic row-chunk count for the trans path

extern "C" int64_t pam_gemv_ws_elems(int64_t nr, int64_t nc) {
  (void)nr;
  return GEMV_CHUNKS * nc;
}

// y = A @ x: one 64-lane wave per output row, lanes stride the row with
// 16-B vector loads, wavefront shuffle reduction.
template <typename T, int V>
__global__ void __launch_bounds__(BLK) gemv_n_kernel(
    const T* __restrict__ A, const T* __restrict__ x, T* __restrict__ y,
    int64_t nr, int64_t nc) {
  const int64_t wave = ((int64_t)blockIdx.x * (BLK / 64))
                       + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  const int64_t nwaves = (int64_t)gridDim.x * (BLK / 64);
  const int64_t ncv = nc / V;
  for (int64_t r = wave; r < nr; r += nwaves) {
    const T* __restrict__ row = A + r * nc;
    double acc = 0.0;
    for (int64_t cv = lane; cv < ncv; cv += 64) {
      T a[V], b[V];
      loadv<T, V>(row + cv * V, a);
      loadv<T, V>(x + cv * V, b);
#pragma unroll
      for (int k = 0; k < V; ++k) acc += (double)a[k] * (double)b[k];
    }
    for (int64_t c = ncv * V + lane; c < nc; c += 64)
      acc += (double)row[c] * (double)x[c];
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if (lane == 0) y[r] = (T)acc;
  }
}

// y = A^T @ x, stage 1: block (cb, chunk) accumulates its row chunk into
// partials[chunk*nc + c] — fixed chunk count => deterministic combine.
template <typename T, int V>
__global__ void __launch_bounds__(BLK) gemv_t_stage1(
    const T* __restrict__ A, const T* __restrict__ x,
    double* __restrict__ partials, int64_t nr, int64_t nc) {
  // V consecutive columns per lane => 16-B row reads.  Measured: NO rate
  // change vs the scalar form (4096: 4.59 TB/s either way) — the t-path's
  // gap to the n-kernel is the fixed ~8 us two-stage cost (stage-2 launch
  // + 2 MB of partials), not access width.  Kept: same bits, wider loads.
  const int64_t c0 = ((int64_t)blockIdx.x * BLK + threadIdx.x) * V;
  const int chunk = blockIdx.y;
  const int64_t r0 = (nr * chunk) / GEMV_CHUNKS;
  const int64_t r1 = (nr * (chunk + 1)) / GEMV_CHUNKS;
  if (c0 >= nc) return;
  double acc[V] = {};
  for (int64_t r = r0; r < r1; ++r) {
    T v[V];
    loadv<T, V>(A + r * nc + c0, v);
#pragma unroll
    for (int j = 0; j < V; ++j)
      acc[j] += (double)v[j] * (double)x[r];
  }
#pragma unroll
  for (int j = 0; j < V; ++j)
    partials[(int64_t)chunk * nc + c0 + j] = acc[j];
}

template <typename T>
__global__ void __launch_bounds__(BLK) gemv_t_stage2(
    const double* __restrict__ partials, T* __restrict__ y, int64_t nc) {
  const int64_t c = (int64_t)blockIdx.x * BLK + threadIdx.x;
  if (c >= nc) return;
  double acc = 0.0;
  for (int ch = 0; ch < GEMV_CHUNKS; ++ch)
    acc += partials[(int64_t)ch * nc + c];
  y[c] = (T)acc;
}

template <typename T>
static int gemv_launch(void* stream, int trans, const void* A, const void* x,
                       void* y, int64_t nr, int64_t nc, void* ws) {
  if (nr <= 0 || nc <= 0 || !A || !x || !y) return PAM_EARG;
  hipStream_t s = (hipStream_t)stream;
  if (!trans) {
    constexpr int V = VecW<T>::value;
    const bool vec_ok = (nc % V == 0) && ((uintptr_t)A % 16 == 0) &&
                        ((uintptr_t)x % 16 == 0);
    int64_t nb = (nr + (BLK / 64) - 1) / (BLK / 64);
    if (nb > 4096) nb = 4096;
    if (vec_ok)
      hipLaunchKernelGGL((gemv_n_kernel<T, V>), dim3((uint32_t)nb), dim3(BLK),
                         0, s, (const T*)A, (const T*)x, (T*)y, nr, nc);
    else
      hipLaunchKernelGGL((gemv_n_kernel<T, 1>), dim3((uint32_t)nb), dim3(BLK),
                         0, s, (const T*)A, (const T*)x, (T*)y, nr, nc);
    return check(hipGetLastError());
  }
  if (!ws) return PAM_EARG;
  constexpr int V = VecW<T>::value;
  const bool vec_ok = (nc % V == 0) && ((uintptr_t)A % 16 == 0);
  const int64_t ncl = vec_ok ? nc / V : nc;
  dim3 g1((uint32_t)((ncl + BLK - 1) / BLK), GEMV_CHUNKS);
  if (vec_ok)
    hipLaunchKernelGGL((gemv_t_stage1<T, V>), g1, dim3(BLK), 0, s,
                       (const T*)A, (const T*)x, (double*)ws, nr, nc);
  else
    hipLaunchKernelGGL((gemv_t_stage1<T, 1>), g1, dim3(BLK), 0, s,
                       (const T*)A, (const T*)x, (double*)ws, nr, nc);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return (int)e;
  hipLaunchKernelGGL((gemv_t_stage2<T>),
                     dim3((uint32_t)((nc + BLK - 1) / BLK)), dim3(BLK), 0, s,
                     (const double*)ws, (T*)y, nc);
  return check(hipGetLastError());
}

extern "C" int pam_gemv(void* stream, int trans, const void* A, const void* x,
                        void* y, int64_t nr, int64_t nc, void* ws, int dtype) {
  if (dtype == PAM_F64)
    return gemv_launch<double>(stream, trans, A, x, y, nr, nc, ws);
  if (dtype == PAM_F32)
    return gemv_launch<float>(stream, trans, A, x, y, nr, nc, ws);
  return PAM_EDTYPE;
}

// ---------------------------------------------------------------------------
// fused finite-difference stencils
// (ref FirstDerivative.py:141-318, SecondDerivative.py:124-256 — closed
//  forms of the slice algebra; masks are global-row intervals)
// ---------------------------------------------------------------------------

extern "C" int64_t pam_fd_halo_width(int op) {
  switch (op) {
    case 0: case 1: case 2: case 3: case 4: case 5: return 1;
    case 6: case 7: case 8: case 9: case 10: case 11: case 12: case 13:
      return 2;
    default: return PAM_EOP;
  }
}

template <typename T, int OP, int V, bool NTS = false, bool RSWAP = false>
__global__ void __launch_bounds__(BLK) fd_kernel(Rows<T> R, T* __restrict__ y,
                                                 int64_t row0, int64_t N, T c,
                                                 int edge, int64_t rbegin,
                                                 int64_t rend) {
  // RSWAP (PAM_FD_RSWAP=1 A/B): row index from blockIdx.x instead of .y,
  // so XCD round-robin dispatch (linear id mod 8) lands CONSECUTIVE ROWS
  // on consecutive XCDs instead of consecutive column chunks — probes
  // whether the long-row deficit is an XCD/L2 block-mapping effect.
  const int64_t brow = RSWAP ? blockIdx.x : blockIdx.y;
  const int64_t grows = RSWAP ? gridDim.x : gridDim.y;
  const int64_t bcol = RSWAP ? blockIdx.y : blockIdx.x;
  const int64_t gcols = RSWAP ? gridDim.y : gridDim.x;
  const int64_t m = R.m, mv = m / V;
  const int64_t cstride = (int64_t)gcols * blockDim.x;
  for (int64_t i = rbegin + brow; i < rend; i += grows) {
    const int64_t g = row0 + i;
    T* __restrict__ yrow = y + i * m;
    for (int64_t jv = (int64_t)bcol * blockDim.x + threadIdx.x; jv < mv;
         jv += cstride) {
      const int64_t j = jv * V;
      T acc[V];
#pragma unroll
      for (int k = 0; k < V; ++k) acc[k] = (T)0;
#pragma unroll
      for (int t = 0; t < FDDef<OP>::NT; ++t) {
        const Term tm = FDDef<OP>::TERMS[t];
        if (g >= tm.lo && g <= N - 1 - tm.hi) {
          T v[V];
          loadv<T, V>(R.row(i + tm.off) + j, v);
#pragma unroll
          for (int k = 0; k < V; ++k) acc[k] += (T)tm.coeff * v[k];
        }
      }
      if (edge) fd_edge<T, OP, V>(R, i, j, g, N, acc);
#pragma unroll
      for (int k = 0; k < V; ++k) acc[k] *= c;
      storev_p<T, V, NTS>(yrow + j, acc);
    }
    // scalar tail columns (m not divisible by V)
    if constexpr (V > 1) {
      for (int64_t j = mv * V + (int64_t)bcol * blockDim.x + threadIdx.x;
           j < m; j += cstride) {
        T acc[1] = {(T)0};
#pragma unroll
        for (int t = 0; t < FDDef<OP>::NT; ++t) {
          const Term tm = FDDef<OP>::TERMS[t];
          if (g >= tm.lo && g <= N - 1 - tm.hi)
            acc[0] += (T)tm.coeff * R.row(i + tm.off)[j];
        }
        if (edge) fd_edge<T, OP, 1>(R, i, j, g, N, acc);
        const T o = acc[0] * c;
        if constexpr (NTS)
          __builtin_nontemporal_store(o, yrow + j);
        else
          yrow[j] = o;
      }
    }
  }
}


// constexpr-folded term accumulation for the rolling window: `tm` is a
// constexpr LOCAL here, so buf[tm.off + W] is a CONSTANT register index
// (the plain `for (t) { Term tm = TERMS[t]; buf[tm.off+W] }` form keeps
// the window in scratch — measured 2.0 TB/s vs in-register)
template <typename T, int OP, int V, int W, int t>
__device__ __forceinline__ void roll_term(const T (&win)[2 * W + 1][V],
                                          T (&acc)[V], int64_t g,
                                          int64_t N) {
  constexpr Term tm = FDDef<OP>::TERMS[t];
  if (g >= tm.lo && g <= N - 1 - tm.hi) {
#pragma unroll
    for (int k = 0; k < V; ++k)
      acc[k] += (T)tm.coeff * win[tm.off + W][k];
  }
}

template <typename T, int OP, int V, int W, int... Ts>
__device__ __forceinline__ void roll_terms(const T (&win)[2 * W + 1][V],
                                           T (&acc)[V], int64_t g,
                                           int64_t N,
                                           std::integer_sequence<int, Ts...>) {
  (roll_term<T, OP, V, W, Ts>(win, acc, g, N), ...);
}

// Rolling-window stencil (EXPERIMENTAL, PAM_FD_ROLL=1; r01 status
// below): each block walks a row range carrying the 2W+1 input rows in
// registers — every x element loaded from HBM exactly once, where the
// row-parallel fd_kernel relies on L2 absorbing neighbour-row re-reads
// (which fades at long rows: the (512,4096,256) N=8 per-rank shape
// measures 4.9 TB/s row-parallel vs 5.7 at the bench shape; a pad/
// non-pow2 probe — scripts/probe_rowstride.hip — ruled out row-stride
// channel aliasing, leaving re-read absorption as the cause).
// r01 measurements of THIS kernel: single chain/thread 4.8-5.2 TB/s;
// CV=4 chains with a runtime-indexed window spilled to scratch
// (~2.0 TB/s); with the constexpr-folded roll_terms below the window
// stays in registers and CV=4 reaches 4.9 (long rows) / 5.2 (bench
// shape) TB/s — STILL below the row-parallel kernel (4.9 / 5.7).  Key
// datum: rolling reads each x element exactly once, so its 4.9 TB/s on
// the long-row shape is a TRUE 16 B/pt streaming rate — the long-row
// gap is therefore NOT neighbour-row re-read traffic; something about
// the 1-load+1-store-per-row-step pattern (vs axpy's 5.78 on the same
// mix) is the round-2 question.
// Guards: load indices are clamped to existing rows (halo planes
// included); term/edge masks ignore the clamped garbage.
template <typename T, int OP, int V, int CV, bool NTS = false,
          bool NTL = false>
__global__ void __launch_bounds__(BLK) fd_roll_kernel(
    Rows<T> R, T* __restrict__ y, int64_t row0, int64_t N, T c, int edge,
    int64_t rbegin, int64_t rend) {
  constexpr int W = FDDef<OP>::W;
  constexpr int NROLL = 2 * W + 1;
  const int64_t m = R.m, mv = m / V;
  // CV independent column-vectors per thread (separate rolling windows)
  // => CV outstanding loads per row step instead of 1: the single-chain
  // form was memory-latency-bound (one dependent load per row).  Chains
  // are BLOCK-interleaved (chain q of lane t at column block*BLK*CV +
  // q*BLK + t) so each chain's wave accesses stay fully coalesced.
  const int64_t base = (int64_t)blockIdx.x * blockDim.x * CV + threadIdx.x;
  const int64_t nr = rend - rbegin;
  const int64_t c0 = rbegin + (nr * blockIdx.y) / gridDim.y;
  const int64_t c1 = rbegin + (nr * (blockIdx.y + 1)) / gridDim.y;
  if (c1 <= c0 || base >= mv) return;
  const int64_t lo = R.gf ? -(int64_t)R.w : 0;
  const int64_t hi = R.nloc - 1 + (R.gb ? R.w : 0);
  auto clamp_row = [&](int64_t i) {
    return i < lo ? lo : (i > hi ? hi : i);
  };
  // constant-indexed per-chain state (a runtime-compacted chain array
  // made the windows spill to scratch: 5.2 -> 1.9 TB/s)
  int64_t js[CV];
  bool vq[CV];
#pragma unroll
  for (int q = 0; q < CV; ++q) {
    const int64_t jq = base + (int64_t)q * blockDim.x;
    vq[q] = jq < mv;
    js[q] = (vq[q] ? jq : mv - 1) * V;  // clamped: loads stay in range
  }
  T buf[CV][NROLL][V];  // rows i-W .. i+W of each owned column vector
#pragma unroll
  for (int q = 0; q < CV; ++q)
#pragma unroll
    for (int k = 0; k < NROLL; ++k)
      if constexpr (NTL) loadv_nt<T, V>(R.row(clamp_row(c0 - W + k)) + js[q], buf[q][k]);
      else loadv<T, V>(R.row(clamp_row(c0 - W + k)) + js[q], buf[q][k]);
  for (int64_t i = c0; i < c1; ++i) {
    const int64_t g = row0 + i;
    // issue ALL next-row loads first (CV independent chains)
    T nxt[CV][V];
    const int64_t rn = clamp_row(i + 1 + W);
#pragma unroll
    for (int q = 0; q < CV; ++q)
      if constexpr (NTL) loadv_nt<T, V>(R.row(rn) + js[q], nxt[q]);
      else loadv<T, V>(R.row(rn) + js[q], nxt[q]);
#pragma unroll
    for (int q = 0; q < CV; ++q) {
      T acc[V];
#pragma unroll
      for (int k = 0; k < V; ++k) acc[k] = (T)0;
      roll_terms<T, OP, V, W>(
          buf[q], acc, g, N,
          std::make_integer_sequence<int, FDDef<OP>::NT>{});
      if (edge) fd_edge<T, OP, V>(R, i, js[q], g, N, acc);
#pragma unroll
      for (int k = 0; k < V; ++k) acc[k] *= c;
      if (vq[q]) storev_p<T, V, NTS>(y + i * m + js[q], acc);
#pragma unroll
      for (int k = 0; k < NROLL - 1; ++k)
#pragma unroll
        for (int v = 0; v < V; ++v) buf[q][k][v] = buf[q][k + 1][v];
#pragma unroll
      for (int v = 0; v < V; ++v) buf[q][NROLL - 1][v] = nxt[q][v];
    }
  }
}

template <int... Us, typename F>
__device__ __forceinline__ void fd_static_for(
    std::integer_sequence<int, Us...>, F&& f) {
  (f(std::integral_constant<int, Us>{}), ...);
}

// roll2 term accumulation: window slot is ((off + W + ROT) mod NROLL) —
// all constexpr, so the window lives in registers with NO per-row shift
template <typename T, int OP, int V, int W, int ROT, int t>
__device__ __forceinline__ void roll2_term(const T (&win)[2 * W + 1][V],
                                           T (&acc)[V], int64_t g,
                                           int64_t N) {
  constexpr Term tm = FDDef<OP>::TERMS[t];
  constexpr int NR = 2 * W + 1;
  constexpr int slot = ((tm.off + W + ROT) % NR + NR) % NR;
  if (g >= tm.lo && g <= N - 1 - tm.hi) {
#pragma unroll
    for (int k = 0; k < V; ++k)
      acc[k] += (T)tm.coeff * win[slot][k];
  }
}

template <typename T, int OP, int V, int W, int ROT, int... Ts>
__device__ __forceinline__ void roll2_terms(
    const T (&win)[2 * W + 1][V], T (&acc)[V], int64_t g, int64_t N,
    std::integer_sequence<int, Ts...>) {
  (roll2_term<T, OP, V, W, ROT, Ts>(win, acc, g, N), ...);
}

// Statically-rotated rolling stencil (PAM_FD_ROLL=2, r02): like
// fd_roll_kernel (1 load + 1 store per point — the PMC-measured fix for
// the long-row shape, where the row-parallel kernel's neighbour re-read
// really does go to HBM: FETCH 8.53 GB vs 4.29 algorithmic at
// 512x4096x256, r02 session 1), but the row loop is unrolled by the
// window depth so the slot indices rotate at COMPILE time — no
// register-to-register window shift per row (fd_roll_kernel spends
// 2W*V*CV moves per row step; this kernel spends zero).  Loads for row
// i+W+1 issue before row i's arithmetic, giving CV independent loads
// in flight per thread.
template <typename T, int OP, int V, int CV, bool NTS = false>
__global__ void __launch_bounds__(BLK) fd_roll2_kernel(
    Rows<T> R, T* __restrict__ y, int64_t row0, int64_t N, T c, int edge,
    int64_t rbegin, int64_t rend) {
  constexpr int W = FDDef<OP>::W;
  constexpr int NROLL = 2 * W + 1;
  const int64_t m = R.m, mv = m / V;
  const int64_t base = (int64_t)blockIdx.x * blockDim.x * CV + threadIdx.x;
  const int64_t nr = rend - rbegin;
  const int64_t c0 = rbegin + (nr * blockIdx.y) / gridDim.y;
  const int64_t c1 = rbegin + (nr * (blockIdx.y + 1)) / gridDim.y;
  if (c1 <= c0 || base >= mv) return;
  const int64_t lo = R.gf ? -(int64_t)R.w : 0;
  const int64_t hi = R.nloc - 1 + (R.gb ? R.w : 0);
  auto clamp_row = [&](int64_t i) {
    return i < lo ? lo : (i > hi ? hi : i);
  };
  int64_t js[CV];
  bool vq[CV];
#pragma unroll
  for (int q = 0; q < CV; ++q) {
    const int64_t jq = base + (int64_t)q * blockDim.x;
    vq[q] = jq < mv;
    js[q] = (vq[q] ? jq : mv - 1) * V;
  }
  T buf[CV][NROLL][V];
#pragma unroll
  for (int q = 0; q < CV; ++q)
#pragma unroll
    for (int k = 0; k < NROLL; ++k)
      loadv<T, V>(R.row(clamp_row(c0 - W + k)) + js[q], buf[q][k]);
  int64_t i = c0;
  auto step = [&](auto rotc) {
    constexpr int ROT = decltype(rotc)::value;
    constexpr int oldest = ROT % NROLL;
    const int64_t rn = clamp_row(i + W + 1);
    T nxt[CV][V];
#pragma unroll
    for (int q = 0; q < CV; ++q)
      loadv<T, V>(R.row(rn) + js[q], nxt[q]);
    const int64_t g = row0 + i;
#pragma unroll
    for (int q = 0; q < CV; ++q) {
      T acc[V];
#pragma unroll
      for (int k = 0; k < V; ++k) acc[k] = (T)0;
      roll2_terms<T, OP, V, W, ROT>(
          buf[q], acc, g, N,
          std::make_integer_sequence<int, FDDef<OP>::NT>{});
      if (edge) fd_edge<T, OP, V>(R, i, js[q], g, N, acc);
#pragma unroll
      for (int k = 0; k < V; ++k) acc[k] *= c;
      if (vq[q]) storev_p<T, V, NTS>(y + i * m + js[q], acc);
#pragma unroll
      for (int v = 0; v < V; ++v) buf[q][oldest][v] = nxt[q][v];
    }
    ++i;
  };
  while (i + NROLL <= c1)
    fd_static_for(std::make_integer_sequence<int, NROLL>{},
                  [&](auto rc) { step(rc); });
  fd_static_for(std::make_integer_sequence<int, NROLL>{}, [&](auto rc) {
    if (i < c1) step(rc);
  });
}

static int fd_vec_override() {
  static int v = [] {
    const char* e = getenv("PAM_FD_VEC");
    return e ? atoi(e) : 0;
  }();
  return v;
}

// PAM_FD_NT=1 opts the output stores into the nt cache hint.  Measured
// NEGATIVE on the production kernel (bench A/B r01: 291.2 plain vs 290.5
// nt pairs/s — no change): the +9% the standalone probe saw
// (scripts/probe_nt_store.hip) does not transfer once the launch geometry
// gives the row re-use its L2 locality; the kernel is already at the
// platform's read+write mix ceiling.  Default: plain stores.
static int fd_nt_override() {
  static int v = [] {
    const char* e = getenv("PAM_FD_NT");
    return e ? atoi(e) : 0;
  }();
  return v;
}

template <typename T, int OP>
static int fd_launch(void* stream, int edge, const void* x, const void* gf,
                     const void* gb, void* y, int64_t nloc, int64_t m,
                     int64_t row0, int64_t nglob, int64_t rbegin,
                     int64_t rend, double coeff) {
  if (nloc < 0 || m <= 0 || rbegin < 0 || rend > nloc) return PAM_EARG;
  const int64_t nrows = rend - rbegin;
  if (nloc == 0 || nrows <= 0) return 0;
  Rows<T> R{(const T*)x, (const T*)gf, (const T*)gb, nloc, m, FDDef<OP>::W};
  const bool align16 = ((uintptr_t)x % 16 == 0) && ((uintptr_t)y % 16 == 0) &&
                       (gf == nullptr || (uintptr_t)gf % 16 == 0) &&
                       (gb == nullptr || (uintptr_t)gb % 16 == 0);
  // default: 16 B/lane (double2 / float4) — the measured optimum; 32 B
  // (V=4 double) was -14% in an A/B (PAM_FD_VEC keeps the knob)
  int V = 1;
  if (align16) {
    if (sizeof(T) == 8)
      V = (m % 2 == 0) ? 2 : 1;
    else
      V = (m % 4 == 0) ? 4 : 1;
    const int ov = fd_vec_override();
    if ((ov == 1 || ov == 2 || ov == 4) && m % ov == 0) V = ov;
  }
  const int64_t mv = m / V;
  // Grid shape (PAM_FD_GY / PAM_FD_CAP override for A/Bs): one block-row
  // per local row (gy = nrows), column blocks sized so each block runs
  // ~4 vector iterations (total blocks ~= work/(BLK*4)).  Bench A/Bs on
  // one box, 2-3 reps each (pairs/s at 2048x2048x128 fp64):
  //   old default gy<=512/cap=4096: 286.7
  //   gy=nrows cap=16384: 295.8   cap=131072: 316.6
  //   cap=262144: 332.0 (the peak — frac 0.719)   cap=524288: 319.6
  //   cap=1048576 (no loop at all): 216.6 (hard regression)
  static int gycap = [] {
    const char* e = getenv("PAM_FD_GY");
    return e ? atoi(e) : 65535;
  }();
  static int totcap = [] {
    const char* e = getenv("PAM_FD_CAP");
    return e ? atoi(e) : 262144;
  }();
  int gy = (int)(nrows < gycap ? nrows : gycap);
  if (gy < 1) gy = 1;
  int64_t gx64 = (mv + BLK - 1) / BLK;
  int64_t cap = totcap / gy;
  if (cap < 1) cap = 1;
  if (gx64 > cap) gx64 = cap;
  dim3 grid((uint32_t)gx64, (uint32_t)gy);
  hipStream_t s = (hipStream_t)stream;
  // rolling-window path: DEFAULT for long rows (r02).  r01's negative
  // A/B predates the constexpr-folded window; the r02 PMC runs show the
  // row-parallel kernel's neighbour-row re-reads really go to HBM once
  // a row outgrows L2 absorption (FETCH 8.53 GB vs 4.29 algorithmic at
  // 512x4096x256 => 4.85 TB/s algorithmic), while the rolling kernel
  // moves exactly 1R+1W (PMC 1.035x) and measures 5.35 TB/s there with
  // V=4/CV=8/TGT=2048 (the swept optimum; NT stores -15%, roll2's
  // static rotation spills at CV8V4).  The bench shape (2 MiB rows)
  // keeps the row-parallel kernel (5.78 vs 5.22).  PAM_FD_ROLL: unset =
  // auto, -1 = force row-parallel, 1/2 = force rolling (2 = the
  // statically-rotated variant).  The r02s10/s11 12-shape crossover
  // sweep (profiles/) refined the auto rule: the rolling kernel is
  // flat (~5.0-5.3 TB/s) while row-parallel is peaky (4.8-5.7), and
  // row-parallel loses not only above 4 MiB rows but ALSO at mid-size
  // rows whenever the local row count is not a multiple of 1024 (its
  // gy row-grid then maps raggedly across the 8 XCDs and the L2
  // neighbour-row grouping degrades: 1536-row shapes lose 4-7% at
  // 2.5-3 MiB rows).  Rule (fits all 12 measured shapes): roll when
  // rowbytes > PAM_FD_LONGROW (default 4 MiB, strict — the 4 MiB
  // power-of-two shape prefers row-parallel), or when rowbytes >=
  // 5/8 of it (2.5 MiB) and nrows % 1024 != 0.
  static int rollov = [] {
    const char* e = getenv("PAM_FD_ROLL");
    return e ? atoi(e) : 0;
  }();
  static int64_t longrow = [] {
    const char* e = getenv("PAM_FD_LONGROW");
    return (int64_t)(e ? atoll(e) : (4 << 20));
  }();
  const int64_t rowbytes = m * (int64_t)sizeof(T);
  const bool roll_auto =
      rollov == 0 && V > 1 &&
      (rowbytes > longrow ||
       (rowbytes * 8 >= longrow * 5 && (nrows % 1024) != 0));
  static int rolltgt = [] {
    const char* e = getenv("PAM_FD_ROLL_TGT");
    return e ? atoi(e) : 2048;
  }();
  if ((rollov > 0 && V > 1) || roll_auto) {
    // upgrade to 32-B lanes for the rolling form (its swept optimum;
    // the row-parallel kernel's optimum stays 16 B)
    if (roll_auto && fd_vec_override() == 0 && m % 4 == 0 && align16)
      V = 4;
    static int cvov = [] {
      const char* e = getenv("PAM_FD_ROLL_CV");
      return e ? atoi(e) : 8;
    }();
    const int CV = (cvov == 2 || cvov == 4) ? cvov : 8;
    const int64_t mv2 = m / V;
    int64_t gx = (mv2 + (int64_t)BLK * CV - 1) / ((int64_t)BLK * CV);
    int64_t gyr = rolltgt / (gx ? gx : 1);
    const int64_t minchunk = 16;  // amortize the 2W+1-row preload
    int64_t maxgy = (nrows + minchunk - 1) / minchunk;
    if (gyr > maxgy) gyr = maxgy;
    if (gyr < 1) gyr = 1;
    if (gyr > 65535) gyr = 65535;
    dim3 gridr((uint32_t)gx, (uint32_t)gyr);
    const bool rnt = fd_nt_override() != 0;
    static int ntlov = [] {
      const char* e = getenv("PAM_FD_NTL");
      return e ? atoi(e) : 0;
    }();
    const bool rntl = ntlov != 0;
#define ROLL_LAUNCH(KER, VV, CC)                                              \
  do {                                                                        \
    if (rnt)                                                                  \
      hipLaunchKernelGGL((KER<T, OP, VV, CC, true>), gridr, dim3(BLK), 0, s,  \
                         R, (T*)y, row0, nglob, (T)coeff, edge, rbegin,       \
                         rend);                                               \
    else                                                                      \
      hipLaunchKernelGGL((KER<T, OP, VV, CC, false>), gridr, dim3(BLK), 0, s, \
                         R, (T*)y, row0, nglob, (T)coeff, edge, rbegin,       \
                         rend);                                               \
  } while (0)
#define ROLL_LAUNCH1(VV, CC)                                                  \
  do {                                                                        \
    if (rntl)                                                                 \
      hipLaunchKernelGGL((fd_roll_kernel<T, OP, VV, CC, false, true>),        \
                         gridr, dim3(BLK), 0, s, R, (T*)y, row0, nglob,       \
                         (T)coeff, edge, rbegin, rend);                       \
    else                                                                      \
      ROLL_LAUNCH(fd_roll_kernel, VV, CC);                                    \
  } while (0)
    if (rollov == 2) {        // statically-rotated variant
      if (V == 4 && CV == 2) ROLL_LAUNCH(fd_roll2_kernel, 4, 2);
      else if (V == 4 && CV == 8) ROLL_LAUNCH(fd_roll2_kernel, 4, 8);
      else if (V == 4) ROLL_LAUNCH(fd_roll2_kernel, 4, 4);
      else if (CV == 2) ROLL_LAUNCH(fd_roll2_kernel, 2, 2);
      else if (CV == 8) ROLL_LAUNCH(fd_roll2_kernel, 2, 8);
      else ROLL_LAUNCH(fd_roll2_kernel, 2, 4);
    } else {
      if (V == 4 && CV == 2) ROLL_LAUNCH1(4, 2);
      else if (V == 4 && CV == 8) ROLL_LAUNCH1(4, 8);
      else if (V == 4) ROLL_LAUNCH1(4, 4);
      else if (CV == 2) ROLL_LAUNCH1(2, 2);
      else if (CV == 8) ROLL_LAUNCH1(2, 8);
      else ROLL_LAUNCH1(2, 4);
    }
#undef ROLL_LAUNCH1
#undef ROLL_LAUNCH
    return check(hipGetLastError());
  }
  // PAM_FD_RSWAP=1: row index from blockIdx.x (XCD-mapping A/B; see
  // fd_kernel docstring).  Grid dims swap with it.
  static int rswap = [] {
    const char* e = getenv("PAM_FD_RSWAP");
    return e ? atoi(e) : 0;
  }();
  if (rswap) {
    dim3 grid_s((uint32_t)gy, (uint32_t)gx64);
    if (V == 4)
      hipLaunchKernelGGL((fd_kernel<T, OP, 4, false, true>), grid_s,
                         dim3(BLK), 0, s, R, (T*)y, row0, nglob, (T)coeff,
                         edge, rbegin, rend);
    else if (V == 2 && sizeof(T) == 8)
      hipLaunchKernelGGL((fd_kernel<T, OP, 2, false, true>), grid_s,
                         dim3(BLK), 0, s, R, (T*)y, row0, nglob, (T)coeff,
                         edge, rbegin, rend);
    else
      hipLaunchKernelGGL((fd_kernel<T, OP, 1, false, true>), grid_s,
                         dim3(BLK), 0, s, R, (T*)y, row0, nglob, (T)coeff,
                         edge, rbegin, rend);
    return check(hipGetLastError());
  }
  const bool nt = fd_nt_override() != 0;
  if (nt) {
    if (V == 4)
      hipLaunchKernelGGL((fd_kernel<T, OP, 4, true>), grid, dim3(BLK), 0, s,
                         R, (T*)y, row0, nglob, (T)coeff, edge, rbegin, rend);
    else if (V == 2 && sizeof(T) == 8)
      hipLaunchKernelGGL((fd_kernel<T, OP, 2, true>), grid, dim3(BLK), 0, s,
                         R, (T*)y, row0, nglob, (T)coeff, edge, rbegin, rend);
    else
      hipLaunchKernelGGL((fd_kernel<T, OP, 1, true>), grid, dim3(BLK), 0, s,
                         R, (T*)y, row0, nglob, (T)coeff, edge, rbegin, rend);
  } else if (V == 4)
    hipLaunchKernelGGL((fd_kernel<T, OP, 4>), grid, dim3(BLK), 0, s, R, (T*)y,
                       row0, nglob, (T)coeff, edge, rbegin, rend);
  else if (V == 2 && sizeof(T) == 8)
    hipLaunchKernelGGL((fd_kernel<T, OP, 2>), grid, dim3(BLK), 0, s, R, (T*)y,
                       row0, nglob, (T)coeff, edge, rbegin, rend);
  else
    hipLaunchKernelGGL((fd_kernel<T, OP, 1>), grid, dim3(BLK), 0, s, R, (T*)y,
                       row0, nglob, (T)coeff, edge, rbegin, rend);
  return check(hipGetLastError());
}

extern "C" int pam_fd_apply(void* stream, int op, int edge, const void* x,
                            const void* gf, const void* gb, void* y,
                            int64_t nloc, int64_t m, int64_t row0,
                            int64_t nglob, int64_t rbegin, int64_t rend,
                            double coeff, int dtype) {
#define FD_CASE(T)                                                            \
  switch (op) {                                                               \
    case 0: return fd_launch<T, 0>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 1: return fd_launch<T, 1>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 2: return fd_launch<T, 2>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 3: return fd_launch<T, 3>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 4: return fd_launch<T, 4>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 5: return fd_launch<T, 5>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 6: return fd_launch<T, 6>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 7: return fd_launch<T, 7>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 8: return fd_launch<T, 8>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 9: return fd_launch<T, 9>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 10: return fd_launch<T, 10>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 11: return fd_launch<T, 11>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 12: return fd_launch<T, 12>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    case 13: return fd_launch<T, 13>(stream, edge, x, gf, gb, y, nloc, m, row0, nglob, rbegin, rend, coeff); \
    default: return PAM_EOP;                                                  \
  }
  if (dtype == PAM_F64) { FD_CASE(double) }
  if (dtype == PAM_F32) { FD_CASE(float) }
#undef FD_CASE
  return PAM_EDTYPE;
}
