// pam — 256^2-tile f32 GEMM with async global->LDS staging (glds), the
// deep-pipeline schedule of the CDNA guide's "glds, 2 LDS buffers"
// recipe (cdna_hip_programming.md §async-copy table): one block per CU,
// AGPR-resident 256-accumulator tile, both operands DMA'd straight to
// LDS while the MFMA loop runs on the previous panel.
//
// A is consumed K-MAJOR (At = A^T, [K, M] row-major): glds writes LDS
// lane-linearly (wave-uniform base + lane x 16 B), so the LDS image of a
// panel row must be contiguous in the order lanes fetch it — with At,
// one wave instruction fetches At[k][m0 + lane*4 .. +3] (16 contiguous
// bytes per lane) and lands exactly one [k][0..255] LDS row.  A
// row-major A cannot be staged transposed this way (the per-lane global
// source must be contiguous), which is why the general pam_gemm keeps
// its register-staged transpose; SUMMA callers amortize one
// pam_transpose per broadcast tile (~0.3% of the panel GEMM).
//
// Fragment reads are conflict-free without padding: lanes read
// consecutive m (or n) within an unpadded 256-float LDS row.
//
// Preconditions (fast path only — caller falls back to pam_gemm):
// M % 256 == 0, N % 256 == 0, K % 32 == 0, 16-byte aligned pointers.

#include <hip/hip_runtime.h>
#include <stdint.h>

#include "../../include/pam.h"

#define G256_BLK 256

typedef float f32x16 __attribute__((ext_vector_type(16)));

template <bool ACC>
__global__ void __launch_bounds__(G256_BLK, 1) gemm256_f32_kernel(
    const float* __restrict__ At, const float* __restrict__ B,
    float* __restrict__ C, int64_t M, int64_t N, int64_t K) {
  constexpr int BM = 256, BN = 256, BK = 32;
  __shared__ float As[2][BK][BM];
  __shared__ float Bs[2][BK][BN];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;          // 4 waves as 2x2 over the 256^2 tile
  const int wr = wave >> 1, wc = wave & 1;
  const int li = lane & 31;           // MFMA row/col within 32x32 tile
  const int lk = lane >> 5;           // MFMA k (0..1)
  const int64_t m0 = (int64_t)blockIdx.y * BM;
  const int64_t n0 = (int64_t)blockIdx.x * BN;

  f32x16 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {};

  // one wave instruction DMAs one 256-float LDS row; each wave owns 8
  // rows of the 32-row panel (interleaved by wave id)
  auto glds_panel = [&](int buf, int64_t k0) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int kk = wave * 8 + e;
      const float* ga = At + (k0 + kk) * M + m0 + lane * 4;
      const float* gb = B + (k0 + kk) * N + n0 + lane * 4;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)ga,
          (__attribute__((address_space(3))) uint32_t*)&As[buf][kk][0],
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)gb,
          (__attribute__((address_space(3))) uint32_t*)&Bs[buf][kk][0],
          16, 0, 0);
    }
  };

  // Schedule (guide §pipelining-across-barriers): the p+1 DMA spans the
  // whole MFMA(p) loop.  Per iteration: issue glds(p+1) [16 instrs per
  // wave], then wait vmcnt(16) — drains glds(p), LEAVES p+1 in flight —
  // raw s_barrier (NOT __syncthreads: its fence would emit vmcnt(0) and
  // drain the span), MFMA(p), raw s_barrier before the next overwrite.
  const int64_t NP = K / BK;
  glds_panel(0, 0);
  for (int64_t p = 0; p < NP; ++p) {
    const int cur = (int)(p & 1);
    if (p + 1 < NP) {
      glds_panel(cur ^ 1, (p + 1) * BK);
      asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    // MFMA on panel p (reads LDS only) overlaps the p+1 DMA.  Fragments
    // are register-double-buffered one k-pair ahead (same scheme as
    // gemm_kernel): at 1 wave/SIMD there is no cross-wave latency
    // hiding, so the next ds_reads must issue under the current MFMAs.
    float a[2][4], b[2][4];
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      a[0][mi] = As[cur][lk][wr * 128 + mi * 32 + li];
#pragma unroll
    for (int nj = 0; nj < 4; ++nj)
      b[0][nj] = Bs[cur][lk][wc * 128 + nj * 32 + li];
#pragma unroll
    for (int kp = 0; kp < BK / 2; ++kp) {
      const int cf = kp & 1, nf = (kp + 1) & 1;
      if (kp + 1 < BK / 2) {
        const int krow = (kp + 1) * 2 + lk;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
          a[nf][mi] = As[cur][krow][wr * 128 + mi * 32 + li];
#pragma unroll
        for (int nj = 0; nj < 4; ++nj)
          b[nf][nj] = Bs[cur][krow][wc * 128 + nj * 32 + li];
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int nj = 0; nj < 4; ++nj)
          acc[mi][nj] = __builtin_amdgcn_mfma_f32_32x32x2f32(
              a[cf][mi], b[cf][nj], acc[mi][nj], 0, 0, 0);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: v_mfma_f32_32x32x2 C mapping — col = lane&31,
  // row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int nj = 0; nj < 4; ++nj) {
      const int64_t r0 = m0 + wr * 128 + mi * 32;
      const int64_t cc = n0 + wc * 128 + nj * 32 + li;
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int64_t rr =
            r0 + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
        if constexpr (ACC)
          C[rr * N + cc] += acc[mi][nj][reg];
        else
          C[rr * N + cc] = acc[mi][nj][reg];
      }
    }
}

extern "C" int pam_gemm_kt(void* stream, const void* At, const void* B,
                           void* C, int64_t M, int64_t N, int64_t K,
                           int accumulate, int dtype) {
  if (dtype != PAM_F32) return PAM_EDTYPE;
  if (M <= 0 || N <= 0 || K <= 0 || !At || !B || !C) return PAM_EARG;
  if (M % 256 || N % 256 || K % 32 ||
      ((uintptr_t)At | (uintptr_t)B | (uintptr_t)C) % 16)
    return PAM_EARG;  // fast path only — caller falls back to pam_gemm
  dim3 grid((uint32_t)(N / 256), (uint32_t)(M / 256));
  hipStream_t s = (hipStream_t)stream;
  if (accumulate)
    hipLaunchKernelGGL((gemm256_f32_kernel<true>), grid, dim3(G256_BLK), 0,
                       s, (const float*)At, (const float*)B, (float*)C, M,
                       N, K);
  else
    hipLaunchKernelGGL((gemm256_f32_kernel<false>), grid, dim3(G256_BLK), 0,
                       s, (const float*)At, (const float*)B, (float*)C, M,
                       N, K);
  return (int)hipGetLastError();
}
