// pam — serial (single-rank) strided finite-difference kernels.
// Separate translation unit: co-compiling these with pam.hip's kernels
// reproducibly segfaults clang-22/ROCm 7.2's gfx950 backend (instantiation
// interaction; the same code compiles alone), so they live here.

#include <hip/hip_runtime.h>
#include <stdint.h>

#include "../../include/pam.h"
#include "fd_defs.h"

#define BLK 256

static inline int check_s(hipError_t e) { return (int)e; }

static inline int64_t grid_1d_s(int64_t work) {
  int64_t g = (work + BLK - 1) / BLK;
  if (g > 4096) g = 4096;
  if (g < 1) g = 1;
  return g;
}

// serial (single-rank) stencil along an arbitrary axis of a local block,
// viewed as [batch, d, m] with the derivative along d (offsets of +-m
// elements).  The local ops inside MPIBlockDiag for Gradient/Laplacian's
// axes >= 1 (ref basicoperators/Gradient.py:101-118, Laplacian.py:98-126
// wrap serial pylops FirstDerivative/SecondDerivative).  Fully coalesced
// grid-stride over the flat index; no ghosts (global edges inside d).
// edge fixups in flat index space (mirrors fd_edge in fd_defs.h; kept
// separate because constructing Rows<T> (restrict-qualified members) in
// device code segfaults clang-22/ROCm 7.2's gfx950 backend — see the
// header comment).  AT(off) = element (i+off, j) of this batch's block.
template <typename T, int OP>
__device__ __forceinline__ void fd_edge_serial(const T* __restrict__ Xb,
                                               int64_t i, int64_t j,
                                               int64_t d, int64_t m,
                                               T& acc) {
#define AT(off) Xb[(i + (off)) * m + (j)]
  const int64_t g = i, N = d;
  if constexpr (OP == 4) {
    if (g == 0) acc = AT(1) - AT(0);
    else if (g == N - 1) acc = AT(0) - AT(-1);
  } else if constexpr (OP == 5) {
    if (g == 0) acc -= AT(0);
    else if (g == 1) acc += AT(-1);
    if (g == N - 2) acc -= AT(1);
    else if (g == N - 1) acc += AT(0);
  } else if constexpr (OP == 6) {
    if (g == 0) acc = AT(1) - AT(0);
    else if (g == N - 1) acc = AT(0) - AT(-1);
    else if (g == 1 || g == N - 2) acc = (T)0.5 * (AT(1) - AT(-1));
  } else if constexpr (OP == 7) {
    if (g == 0) acc -= AT(0) + (T)0.5 * AT(1);
    else if (g == 1) acc += AT(-1);
    else if (g == 2) acc += (T)0.5 * AT(-1);
    if (g == N - 3) acc -= (T)0.5 * AT(1);
    else if (g == N - 2) acc -= AT(1);
    else if (g == N - 1) acc += (T)0.5 * AT(-1) + AT(0);
  } else if constexpr (OP == 12) {
    if (g == 0) acc = AT(0) - (T)2 * AT(1) + AT(2);
    else if (g == N - 1) acc = AT(-2) - (T)2 * AT(-1) + AT(0);
  } else if constexpr (OP == 13) {
    if (g == 0) acc += AT(0);
    else if (g == 1) acc -= (T)2 * AT(-1);
    else if (g == 2) acc += AT(-2);
    if (g == N - 3) acc += AT(2);
    else if (g == N - 2) acc -= (T)2 * AT(1);
    else if (g == N - 1) acc += AT(0);
  }
#undef AT
}

template <typename T, int OP>
__global__ void __launch_bounds__(BLK) fd_serial_kernel(
    const T* __restrict__ x, T* __restrict__ y, int64_t batch, int64_t d,
    int64_t m, T c, int edge) {
  const int64_t total = batch * d * m;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const int64_t j = idx % m;
    const int64_t rem = idx / m;
    const int64_t i = rem % d;
    const T* Xb = x + (rem - i) * m;  // batch base
    T acc = (T)0;
#pragma unroll
    for (int t = 0; t < FDDef<OP>::NT; ++t) {
      const Term tm = FDDef<OP>::TERMS[t];
      if (i >= tm.lo && i <= d - 1 - tm.hi)
        acc += (T)tm.coeff * Xb[(i + tm.off) * m + j];
    }
    if (edge) fd_edge_serial<T, OP>(Xb, i, j, d, m, acc);
    y[idx] = acc * c;
  }
}

template <typename T, int OP>
static int fd_serial_launch(void* stream, int edge, const void* x, void* y,
                            int64_t batch, int64_t d, int64_t m,
                            double coeff) {
  if (batch <= 0 || d <= 0 || m <= 0) return PAM_EARG;
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL((fd_serial_kernel<T, OP>),
                     dim3(grid_1d_s(batch * d * m)), dim3(BLK), 0, s,
                     (const T*)x, (T*)y, batch, d, m, (T)coeff, edge);
  return check_s(hipGetLastError());
}

extern "C" int pam_fd_serial(void* stream, int op, int edge, const void* x,
                             void* y, int64_t batch, int64_t d, int64_t m,
                             double coeff, int dtype) {
#define FDS_CASE(T)                                                           \
  switch (op) {                                                               \
    case 0: return fd_serial_launch<T, 0>(stream, edge, x, y, batch, d, m, coeff); \
    case 1: return fd_serial_launch<T, 1>(stream, edge, x, y, batch, d, m, coeff); \
    case 2: return fd_serial_launch<T, 2>(stream, edge, x, y, batch, d, m, coeff); \
    case 3: return fd_serial_launch<T, 3>(stream, edge, x, y, batch, d, m, coeff); \
    case 4: return fd_serial_launch<T, 4>(stream, edge, x, y, batch, d, m, coeff); \
    case 5: return fd_serial_launch<T, 5>(stream, edge, x, y, batch, d, m, coeff); \
    case 6: return fd_serial_launch<T, 6>(stream, edge, x, y, batch, d, m, coeff); \
    case 7: return fd_serial_launch<T, 7>(stream, edge, x, y, batch, d, m, coeff); \
    case 8: return fd_serial_launch<T, 8>(stream, edge, x, y, batch, d, m, coeff); \
    case 9: return fd_serial_launch<T, 9>(stream, edge, x, y, batch, d, m, coeff); \
    case 10: return fd_serial_launch<T, 10>(stream, edge, x, y, batch, d, m, coeff); \
    case 11: return fd_serial_launch<T, 11>(stream, edge, x, y, batch, d, m, coeff); \
    case 12: return fd_serial_launch<T, 12>(stream, edge, x, y, batch, d, m, coeff); \
    case 13: return fd_serial_launch<T, 13>(stream, edge, x, y, batch, d, m, coeff); \
    default: return PAM_EOP;                                                  \
  }
  if (dtype == PAM_F64) { FDS_CASE(double) }
  if (dtype == PAM_F32) { FDS_CASE(float) }
#undef FDS_CASE
  return PAM_EDTYPE;
}


// ---------------------------------------------------------------------------
// serial non-stationary 1-D convolution along axis d of a [batch, d, m]
// block (the local operator inside MPINonStationaryConvolve1D's
// Halo.H @ BlockDiag @ Halo sandwich, ref signalprocessing/
// NonStatConvolve1d.py:129-168).  pylops' published convention, re-derived
// (pylops is not vendored; locked by dense-adjoint/dottest tests):
//   filters hs[nf][hsize] (odd hsize, centred) anchored at ih = oh + q*dh;
//   the filter at position ix is the linear interpolation of the two
//   nearest anchors, clamped to hs[0] / hs[nf-1] outside;
//   forward (scatter form): y[n] += x[ix] * h_ix[n - ix + hh]
//   => gather: y[n] = sum_t x[n+hh-t] * h_{n+hh-t}[t]
//   adjoint: z[n] = sum_t x[n-hh+t] * h_n[t]
// ---------------------------------------------------------------------------
template <typename T>
__device__ __forceinline__ T nsc_h(const T* __restrict__ hs, int64_t nf,
                                   int64_t hsize, double oh, double dh,
                                   int64_t ix, int64_t t) {
  const double q = ((double)ix - oh) / dh;
  const int64_t ic = (int64_t)floor(q);
  if (ic < 0) return hs[t];
  if (ic >= nf - 1) return hs[(nf - 1) * hsize + t];
  const T w = (T)(q - (double)ic);
  return ((T)1 - w) * hs[ic * hsize + t] + w * hs[(ic + 1) * hsize + t];
}

template <typename T, bool FWD>
__global__ void __launch_bounds__(BLK) nsconv_kernel(
    const T* __restrict__ x, T* __restrict__ y, const T* __restrict__ hs,
    int64_t batch, int64_t d, int64_t m, int64_t nf, int64_t hsize,
    double oh, double dh) {
  const int64_t hh = hsize / 2;
  const int64_t total = batch * d * m;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += stride) {
    const int64_t j = idx % m;
    const int64_t rem = idx / m;
    const int64_t n = rem % d;
    const T* Xb = x + (rem - n) * m;
    T acc = (T)0;
    for (int64_t t = 0; t < hsize; ++t) {
      const int64_t ix = FWD ? (n + hh - t) : (n - hh + t);
      if (ix < 0 || ix >= d) continue;
      const T h = nsc_h<T>(hs, nf, hsize, oh, dh, FWD ? ix : n, t);
      acc += Xb[ix * m + j] * h;
    }
    y[idx] = acc;
  }
}

extern "C" int pam_nsconv(void* stream, int forward, const void* x, void* y,
                          const void* hs, int64_t batch, int64_t d,
                          int64_t m, int64_t nf, int64_t hsize, double oh,
                          double dh, int dtype) {
  if (batch <= 0 || d <= 0 || m <= 0 || nf <= 0 || hsize <= 0 || !x || !y ||
      !hs)
    return PAM_EARG;
  hipStream_t s = (hipStream_t)stream;
#define NSC_CASE(T)                                                           \
  do {                                                                        \
    if (forward)                                                              \
      hipLaunchKernelGGL((nsconv_kernel<T, true>),                            \
                         dim3(grid_1d_s(batch * d * m)), dim3(BLK), 0, s,     \
                         (const T*)x, (T*)y, (const T*)hs, batch, d, m, nf,   \
                         hsize, oh, dh);                                      \
    else                                                                      \
      hipLaunchKernelGGL((nsconv_kernel<T, false>),                           \
                         dim3(grid_1d_s(batch * d * m)), dim3(BLK), 0, s,     \
                         (const T*)x, (T*)y, (const T*)hs, batch, d, m, nf,   \
                         hsize, oh, dh);                                      \
    return check_s(hipGetLastError());                                        \
  } while (0)
  if (dtype == PAM_F64) NSC_CASE(double);
  if (dtype == PAM_F32) NSC_CASE(float);
#undef NSC_CASE
  return PAM_EDTYPE;
}
