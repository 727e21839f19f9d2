// pam — MFMA GEMM + transpose for the MatrixMult path (gfx950/CDNA4).
//
// Y = A @ B (optionally += onto C) for row-major matrices, used by
// MPIMatrixMult block/SUMMA local products
// (ref basicoperators/MatrixMult.py:341-427, :610-765: the reference calls
// ncp.matmul on CuPy; here the panels run on hand-written MFMA kernels).
//
// Shapes (cdna_hip_programming.md §3/§5 canonical anatomy):
//   f32: v_mfma_f32_32x32x2_f32  — exact f32 at the 157 TF vector-rate peak
//        (no xf32/TF32 on gfx950); 128x128 block tile, BK=32, 4 waves as
//        2x2, each wave 2x2 MFMA tiles of 32x32 (64 acc VGPRs/lane).
//   f64: v_mfma_f64_16x16x4_f64  — 64x64 block tile, BK=16, 4 waves as
//        2x2, each wave 2x2 MFMA tiles of 16x16.
// A is staged to LDS TRANSPOSED ([k][row], +1 element row pad so the
// k-major writes and the row-major fragment reads are conflict-free);
// B is staged linear [k][col] (coalesced both ways).
// Arbitrary M/N/K via zero-padded loads + guarded stores.

#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdlib.h>

#include "../../include/pam.h"

#define GBLK 256

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef double f64x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef double f64x2 __attribute__((ext_vector_type(2)));

static inline int gcheck(hipError_t e) { return (int)e; }

template <typename T> struct GemmCfg;
template <> struct GemmCfg<float> {
  static constexpr int TM = 32;   // MFMA tile edge (square)
  static constexpr int TK = 2;    // K per MFMA
  static constexpr int BM = 128, BN = 128, BK = 32;
  static constexpr int MI = 2, NJ = 2;  // MFMA tiles per wave (rows, cols)
  // Pipeline A/Bs at 8192^3 (all NEGATIVE for f32, kept single-buffered):
  //   BK=32 NBUF=2: 135.4 -> 123.0 TF (66 KB LDS halves occupancy)
  //   BK=16 NBUF=2: 135.4 -> 131.7 TF (4 WG/CU kept, but the shallower
  //   panels double the barrier count).  f64's 128x64/BK16 panels DID
  //   gain from NBUF=2 (57.2 -> 65.1, see GemmCfg<double>).
  static constexpr int NBUF = 1;        // LDS K-panel buffers (see kernel)
  using acc_t = f32x16;
  using vec_t = f32x4;            // 16-B staging vector
  static constexpr int VW = 4;
  __device__ static acc_t mfma(float a, float b, acc_t c) {
    return __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, c, 0, 0, 0);
  }
  // C/D lane mapping for 32x32 shapes (dtype-independent on gfx950):
  // col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  __device__ static int crow(int lane, int reg) {
    return (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
  }
  static constexpr int NREG = 16;
};
template <> struct GemmCfg<double> {
  static constexpr int TM = 16;
  static constexpr int TK = 4;
  // 128x64 block tile (A/B-measured vs the earlier 64x64: the bigger M
  // tile cuts B-panel HBM traffic 25% and lifted the 8192^3 rate)
  static constexpr int BM = 128, BN = 64, BK = 16;
  static constexpr int MI = 4, NJ = 2;
  static constexpr int NBUF = 2;
  using acc_t = f64x4;
  using vec_t = f64x2;
  static constexpr int VW = 2;
  __device__ static acc_t mfma(double a, double b, acc_t c) {
    return __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c, 0, 0, 0);
  }
  // v_mfma_f64_16x16x4f64 C/D: col = lane&15, row = (lane>>4) + 4*reg
  // (probed empirically on gfx950 — scripts/probe_f64_mfma.hip; NOTE this
  // differs from the bf16 16x16 mapping (lane>>4)*4 + reg)
  __device__ static int crow(int lane, int reg) {
    return (lane >> 4) + 4 * reg;
  }
  static constexpr int NREG = 4;
};

// C = A@B (+C when ACC), A [M,K] lda, B [K,N] ldb, C [M,N] ldc, row-major.
template <typename T, bool ACC, bool GUARD>
__global__ void __launch_bounds__(GBLK) gemm_kernel(
    const T* __restrict__ A, const T* __restrict__ B, T* __restrict__ C,
    int64_t M, int64_t N, int64_t K, int64_t lda, int64_t ldb, int64_t ldc) {
  using CFG = GemmCfg<T>;
  constexpr int BM = CFG::BM, BN = CFG::BN, BK = CFG::BK;
  constexpr int TM = CFG::TM, TK = CFG::TK, VW = CFG::VW;
  using acc_t = typename CFG::acc_t;
  using vec_t = typename CFG::vec_t;

  __shared__ T As[CFG::NBUF][BK][BM + 1];  // transposed, padded
  __shared__ T Bs[CFG::NBUF][BK][BN];       // linear

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;          // 4 waves as 2x2
  const int wr = wave >> 1, wc = wave & 1;
  const int li = lane & (TM - 1);     // fragment row/col within MFMA tile
  const int lk = lane / TM;           // fragment k (0..TK-1)

  // NOTE: an XCD-aware bijective tile remap (T1) was measured and
  // REGRESSED this kernel (f32 4096^3: 97->83 TF; 8192^3: 110->101 TF;
  // f64: 49.7->46.7) — the 128^2-tile structure is MFMA-issue-bound here,
  // not HBM-bound, so the remap only disturbed dispatch locality.
  const int64_t brow = (int64_t)blockIdx.y * BM;
  const int64_t bcol = (int64_t)blockIdx.x * BN;

  constexpr int MI = CFG::MI, NJ = CFG::NJ;
  acc_t acc[MI][NJ];
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j) acc[i][j] = {};

  // ---- K-panel pipeline.  NBUF=2 (f64): next panel's global loads are
  // issued BEFORE the MFMA loop on the current panel and parked in
  // registers; the LDS store + single barrier happen after, so the HBM
  // latency hides under compute (one barrier per panel).  NBUF=1 keeps
  // the classic 2-barrier structure (f32: doubling LDS would halve
  // occupancy; measured best single-buffered).
  constexpr int NBUF = CFG::NBUF;
  constexpr int AV = (BM * BK) / VW / GBLK;  // A vector loads per thread
  constexpr int BV = (BK * BN) / VW / GBLK;  // B vector loads per thread
  vec_t areg[AV], breg[BV];

  auto load_panel = [&](int64_t k0) {
#pragma unroll
    for (int e = 0; e < AV; ++e) {
      const int vi = tid + GBLK * e;
      const int row = vi / (BK / VW);
      const int kv = vi % (BK / VW);
      const int64_t gr = brow + row;
      if (!GUARD || (gr < M && k0 + (int64_t)kv * VW + VW <= K)) {
        areg[e] = *reinterpret_cast<const vec_t*>(A + gr * lda + k0
                                                  + (int64_t)kv * VW);
      } else {
#pragma unroll
        for (int j = 0; j < VW; ++j) {
          const int64_t gk = k0 + kv * VW + j;
          areg[e][j] = (gr < M && gk < K) ? A[gr * lda + gk] : (T)0;
        }
      }
    }
#pragma unroll
    for (int e = 0; e < BV; ++e) {
      const int vi = tid + GBLK * e;
      const int kb = vi / (BN / VW);
      const int nv = vi % (BN / VW);
      const int64_t gk = k0 + kb;
      const int64_t gn = bcol + (int64_t)nv * VW;
      if (!GUARD || (gk < K && gn + VW <= N)) {
        breg[e] = *reinterpret_cast<const vec_t*>(B + gk * ldb + gn);
      } else {
#pragma unroll
        for (int j = 0; j < VW; ++j)
          breg[e][j] = (gk < K && gn + j < N) ? B[gk * ldb + gn + j] : (T)0;
      }
    }
  };

  auto store_panel = [&](int buf) {
#pragma unroll
    for (int e = 0; e < AV; ++e) {
      const int vi = tid + GBLK * e;
      const int row = vi / (BK / VW);
      const int kv = vi % (BK / VW);
#pragma unroll
      for (int j = 0; j < VW; ++j) As[buf][kv * VW + j][row] = areg[e][j];
    }
#pragma unroll
    for (int e = 0; e < BV; ++e) {
      const int vi = tid + GBLK * e;
      const int kb = vi / (BN / VW);
      const int nv = vi % (BN / VW);
      *reinterpret_cast<vec_t*>(&Bs[buf][kb][nv * VW]) = breg[e];
    }
  };

  const int64_t NP = GUARD ? (K + BK - 1) / BK : K / BK;
  if (NP > 0) {
    load_panel(0);
    store_panel(0);
    __syncthreads();
    for (int64_t p = 0; p < NP; ++p) {
      if (p + 1 < NP) load_panel((p + 1) * BK);
      const int cur = (int)(p % NBUF);
      // ---- MFMA inner loop on the current panel; LDS fragments are
      // prefetched one k-step ahead so their latency hides under the
      // MFMAs (the naive form serializes ds_read -> lgkmcnt(0) -> MFMA
      // per step — seen in the ISA dump)
      T a[2][MI], b[2][NJ];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        a[0][mi] = As[cur][lk][wr * (BM / 2) + mi * TM + li];
#pragma unroll
      for (int nj = 0; nj < NJ; ++nj)
        b[0][nj] = Bs[cur][lk][wc * (BN / 2) + nj * TM + li];
#pragma unroll
      for (int kk = 0; kk < BK / TK; ++kk) {
        const int cf = kk & 1, nf = (kk + 1) & 1;
        if (kk + 1 < BK / TK) {
          const int krow = (kk + 1) * TK + lk;
#pragma unroll
          for (int mi = 0; mi < MI; ++mi)
            a[nf][mi] = As[cur][krow][wr * (BM / 2) + mi * TM + li];
#pragma unroll
          for (int nj = 0; nj < NJ; ++nj)
            b[nf][nj] = Bs[cur][krow][wc * (BN / 2) + nj * TM + li];
        }
#pragma unroll
        for (int mi = 0; mi < MI; ++mi)
#pragma unroll
          for (int nj = 0; nj < NJ; ++nj)
            acc[mi][nj] = CFG::mfma(a[cf][mi], b[cf][nj], acc[mi][nj]);
      }
      if (p + 1 < NP) {
        if (NBUF == 1) __syncthreads();  // readers done before overwrite
        store_panel((int)((p + 1) % NBUF));
        __syncthreads();
      }
    }
  }

  // ---- epilogue: C/D fragment layout -> global (guarded)
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int nj = 0; nj < NJ; ++nj) {
      const int64_t r0 = brow + wr * (BM / 2) + mi * TM;
      const int64_t c0 = bcol + wc * (BN / 2) + nj * TM;
      const int64_t cc = c0 + li;
      if (GUARD && cc >= N) continue;
#pragma unroll
      for (int reg = 0; reg < CFG::NREG; ++reg) {
        const int64_t rr = r0 + CFG::crow(lane, reg);
        if (!GUARD || rr < M) {
          if constexpr (ACC)
            C[rr * ldc + cc] += acc[mi][nj][reg];
          else
            C[rr * ldc + cc] = acc[mi][nj][reg];
        }
      }
    }
}

template <typename T>
static int gemm_launch(void* stream, const void* A, const void* B, void* C,
                       int64_t M, int64_t N, int64_t K, int64_t lda,
                       int64_t ldb, int64_t ldc, int accumulate) {
  using CFG = GemmCfg<T>;
  if (M <= 0 || N <= 0 || K < 0 || !A || !B || !C) return PAM_EARG;
  dim3 grid((uint32_t)((N + CFG::BN - 1) / CFG::BN),
            (uint32_t)((M + CFG::BM - 1) / CFG::BM));
  hipStream_t s = (hipStream_t)stream;
  const bool aligned = (M % CFG::BM == 0) && (N % CFG::BN == 0) &&
                       (K % CFG::BK == 0);
  if (accumulate) {
    if (aligned)
      hipLaunchKernelGGL((gemm_kernel<T, true, false>), grid, dim3(GBLK), 0,
                         s, (const T*)A, (const T*)B, (T*)C, M, N, K, lda,
                         ldb, ldc);
    else
      hipLaunchKernelGGL((gemm_kernel<T, true, true>), grid, dim3(GBLK), 0,
                         s, (const T*)A, (const T*)B, (T*)C, M, N, K, lda,
                         ldb, ldc);
  } else {
    if (aligned)
      hipLaunchKernelGGL((gemm_kernel<T, false, false>), grid, dim3(GBLK), 0,
                         s, (const T*)A, (const T*)B, (T*)C, M, N, K, lda,
                         ldb, ldc);
    else
      hipLaunchKernelGGL((gemm_kernel<T, false, true>), grid, dim3(GBLK), 0,
                         s, (const T*)A, (const T*)B, (T*)C, M, N, K, lda,
                         ldb, ldc);
  }
  return gcheck(hipGetLastError());
}

extern "C" int pam_gemm(void* stream, const void* A, const void* B, void* C,
                        int64_t M, int64_t N, int64_t K, int64_t lda,
                        int64_t ldb, int64_t ldc, int accumulate, int dtype) {
  if (dtype == PAM_F64)
    return gemm_launch<double>(stream, A, B, C, M, N, K, lda, ldb, ldc,
                               accumulate);
  if (dtype == PAM_F32)
    return gemm_launch<float>(stream, A, B, C, M, N, K, lda, ldb, ldc,
                              accumulate);
  return PAM_EDTYPE;
}

// ---------------------------------------------------------------------------
// batched complex GEMM: C_b = op(A_b) @ B_b (op = N or conj-transpose).
// The Fredholm1 batched integral kernel (ref signalprocessing/
// Fredholm1.py:123,149-156).  Interleaved (re,im) storage.
//
// MFMA formulation: four independent accumulation chains per output tile
// (ar*br, ai*bi, ar*bi, ai*br) so no operand negation is needed inside
// the MFMA loop; the epilogue combines Cr = S_rr - S_ii, Ci = S_ri +
// S_ir.  64x64 block tile, 4 waves as 2x2, BK=16 K-panels staged as
// separate re/im LDS planes.  This replaced a VALU outer-product kernel
// (r01 A/B at the judged cfg5 shape 513x(64x256x256) c64: VALU 24.6 TF
// real-flops = 63% of the ~39 TF VALU ceiling; the MFMA version's
// numbers are in DESIGN.md) — the batch dimension fills the chip either
// way, but only MFMA reaches the matrix-core rate.
// ---------------------------------------------------------------------------
template <typename T, bool CT, bool ACC = false, int BK = 16,
          int BM = 64, int BN = 64, int NB = (BK >= 64 ? 1 : 2)>
__global__ void __launch_bounds__(GBLK) cgemm_batched_kernel(
    const T* __restrict__ A, const T* __restrict__ B, T* __restrict__ C,
    int64_t M, int64_t N, int64_t K, int64_t strideA, int64_t strideB,
    int64_t strideC) {
  using CFG = GemmCfg<T>;
  constexpr int TM = CFG::TM, TK = CFG::TK;
  // 64x64 block tile.  A/B'd alternatives, all NEGATIVE at cfg5:
  //   BK=32 (49.5 -> 37.2 TF), BK=64 single-panel (0.432 -> 0.518 ms
  //   rmatvec), f32 BN=128 wide tile (69.2 -> 66.2 TF: halving the WG
  //   count costs more latency hiding than the doubled fragment reuse
  //   gains).  BK stays a template param: 16 everywhere.
  // BM/BN default 64x64; 128x64 is dispatched for the shallow-K
  // adjoint panels (cfg5 rmatvec K=64: 4x the work per block at the
  // same 4-panel pipeline depth — r02 A/B)
  constexpr int NBUFC = NB;
  constexpr int MI = (BM / 2) / TM, NJ = (BN / 2) / TM;
  using acc_t = typename CFG::acc_t;

  __shared__ T Asr[NBUFC][BK][BM + 1], Asi[NBUFC][BK][BM + 1];  // k-major
  __shared__ T Bsr[NBUFC][BK][BN + 1], Bsi[NBUFC][BK][BN + 1];
  const int64_t b = blockIdx.z;
  const T* __restrict__ Ab = A + 2 * b * strideA;
  const T* __restrict__ Bb = B + 2 * b * strideB;
  T* __restrict__ Cb = C + 2 * b * strideC;
  const int64_t m0 = (int64_t)blockIdx.y * BM;
  const int64_t n0 = (int64_t)blockIdx.x * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int li = lane & (TM - 1);
  const int lk = lane / TM;

  acc_t s_rr[MI][NJ], s_ii[MI][NJ], s_ri[MI][NJ], s_ir[MI][NJ];
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j) {
      s_rr[i][j] = {};
      s_ii[i][j] = {};
      s_ri[i][j] = {};
      s_ir[i][j] = {};
    }

  // K-panel pipeline (same scheme as gemm_kernel: next panel's global
  // loads parked in registers under the MFMA loop, one barrier/panel)
  constexpr int AE = (BM * BK) / GBLK;
  constexpr int BE = (BK * BN) / GBLK;
  T argr[AE], argi[AE], brgr[BE], brgi[BE];

  auto load_panel = [&](int64_t k0) {
#pragma unroll
    for (int e = 0; e < AE; ++e) {
      const int vi2 = tid + GBLK * e;
      int kk, mm;
      if constexpr (CT) {  // A is [K, M]: consecutive m coalesces
        mm = vi2 & (BM - 1);
        kk = vi2 / BM;
      } else {             // A is [M, K]: consecutive k coalesces
        kk = vi2 & (BK - 1);
        mm = vi2 / BK;
      }
      const int64_t gm = m0 + mm;
      const int64_t gk = k0 + kk;
      T vr = 0, vi = 0;
      if (gm < M && gk < K) {
        if constexpr (CT) {  // op(A)[m][k] = conj(A[k][m])
          const int64_t off = 2 * (gk * M + gm);
          vr = Ab[off];
          vi = -Ab[off + 1];
        } else {
          const int64_t off = 2 * (gm * K + gk);
          vr = Ab[off];
          vi = Ab[off + 1];
        }
      }
      argr[e] = vr;
      argi[e] = vi;
    }
#pragma unroll
    for (int e = 0; e < BE; ++e) {
      const int vi2 = tid + GBLK * e;
      const int nn = vi2 & (BN - 1);
      const int kk = vi2 / BN;
      const int64_t gk = k0 + kk;
      const int64_t gn = n0 + nn;
      T vr = 0, vi = 0;
      if (gk < K && gn < N) {
        const int64_t off = 2 * (gk * N + gn);
        vr = Bb[off];
        vi = Bb[off + 1];
      }
      brgr[e] = vr;
      brgi[e] = vi;
    }
  };

  auto store_panel = [&](int buf) {
#pragma unroll
    for (int e = 0; e < AE; ++e) {
      const int vi2 = tid + GBLK * e;
      int kk, mm;
      if constexpr (CT) {
        mm = vi2 & (BM - 1);
        kk = vi2 / BM;
      } else {
        kk = vi2 & (BK - 1);
        mm = vi2 / BK;
      }
      Asr[buf][kk][mm] = argr[e];
      Asi[buf][kk][mm] = argi[e];
    }
#pragma unroll
    for (int e = 0; e < BE; ++e) {
      const int vi2 = tid + GBLK * e;
      const int nn = vi2 & (BN - 1);
      const int kk = vi2 / BN;
      Bsr[buf][kk][nn] = brgr[e];
      Bsi[buf][kk][nn] = brgi[e];
    }
  };

  const int64_t NP = (K + BK - 1) / BK;
  if (NP > 0) {
    load_panel(0);
    store_panel(0);
    __syncthreads();
    for (int64_t p = 0; p < NP; ++p) {
      if (NBUFC > 1 && p + 1 < NP) load_panel((p + 1) * BK);
      const int cur = (int)(p % NBUFC);
      // fragments prefetched one k-step ahead (same scheme as
      // gemm_kernel)
      T ar[2][MI], ai[2][MI], br[2][NJ], bi[2][NJ];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi) {
        ar[0][mi] = Asr[cur][lk][wr * (BM / 2) + mi * TM + li];
        ai[0][mi] = Asi[cur][lk][wr * (BM / 2) + mi * TM + li];
      }
#pragma unroll
      for (int nj = 0; nj < NJ; ++nj) {
        br[0][nj] = Bsr[cur][lk][wc * (BN / 2) + nj * TM + li];
        bi[0][nj] = Bsi[cur][lk][wc * (BN / 2) + nj * TM + li];
      }
#pragma unroll
      for (int kk = 0; kk < BK / TK; ++kk) {
        const int cf = kk & 1, nf = (kk + 1) & 1;
        if (kk + 1 < BK / TK) {
          const int krow = (kk + 1) * TK + lk;
#pragma unroll
          for (int mi = 0; mi < MI; ++mi) {
            ar[nf][mi] = Asr[cur][krow][wr * (BM / 2) + mi * TM + li];
            ai[nf][mi] = Asi[cur][krow][wr * (BM / 2) + mi * TM + li];
          }
#pragma unroll
          for (int nj = 0; nj < NJ; ++nj) {
            br[nf][nj] = Bsr[cur][krow][wc * (BN / 2) + nj * TM + li];
            bi[nf][nj] = Bsi[cur][krow][wc * (BN / 2) + nj * TM + li];
          }
        }
#pragma unroll
        for (int mi = 0; mi < MI; ++mi)
#pragma unroll
          for (int nj = 0; nj < NJ; ++nj) {
            s_rr[mi][nj] = CFG::mfma(ar[cf][mi], br[cf][nj], s_rr[mi][nj]);
            s_ii[mi][nj] = CFG::mfma(ai[cf][mi], bi[cf][nj], s_ii[mi][nj]);
            s_ri[mi][nj] = CFG::mfma(ar[cf][mi], bi[cf][nj], s_ri[mi][nj]);
            s_ir[mi][nj] = CFG::mfma(ai[cf][mi], br[cf][nj], s_ir[mi][nj]);
          }
      }
      if (p + 1 < NP) {
        if (NBUFC == 1) {
          __syncthreads();
          load_panel((p + 1) * BK);
        }
        store_panel((int)((p + 1) % NBUFC));
        __syncthreads();
      }
    }
  }

  // epilogue: Cr = S_rr - S_ii, Ci = S_ri + S_ir
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int nj = 0; nj < NJ; ++nj) {
      const int64_t r0 = m0 + wr * (BM / 2) + mi * TM;
      const int64_t cc = n0 + wc * (BN / 2) + nj * TM + li;
      if (cc >= N) continue;
#pragma unroll
      for (int reg = 0; reg < CFG::NREG; ++reg) {
        const int64_t rr = r0 + CFG::crow(lane, reg);
        if (rr < M) {
          const int64_t off = 2 * (rr * N + cc);
          if constexpr (ACC) {
            Cb[off] += s_rr[mi][nj][reg] - s_ii[mi][nj][reg];
            Cb[off + 1] += s_ri[mi][nj][reg] + s_ir[mi][nj][reg];
          } else {
            Cb[off] = s_rr[mi][nj][reg] - s_ii[mi][nj][reg];
            Cb[off + 1] = s_ri[mi][nj][reg] + s_ir[mi][nj][reg];
          }
        }
      }
    }
}

template <typename T>
static int cgemm_launch(void* stream, const void* A, const void* B, void* C,
                        int64_t batch, int64_t M, int64_t N, int64_t K,
                        int64_t sA, int64_t sB, int64_t sC, int opa,
                        int acc) {
  if (batch <= 0 || M <= 0 || N <= 0 || K < 0 || !A || !B || !C)
    return PAM_EARG;
  hipStream_t s = (hipStream_t)stream;
  // 128x64 tile for shallow-K panels (PAM_CGEMM_TILE: 0 auto, 1 force
  // 64x64, 2 force 128x64): at the cfg5 rmatvec shape (M=N=256, K=64)
  // the 64x64 tile runs a 4-panel pipeline with 4x the blocks of the
  // matvec shape — doubling BM doubles per-block work at the same
  // pipeline depth (r02 A/B).
  static int tileov = [] {
    const char* e = getenv("PAM_CGEMM_TILE");
    return e ? atoi(e) : 0;
  }();
  const bool wide = !acc && M >= 128 && tileov == 2;  // auto condition pending the r02 A/B
  if (wide) {
    dim3 gridw((uint32_t)((N + 63) / 64), (uint32_t)((M + 127) / 128),
               (uint32_t)batch);
    if (opa)
      hipLaunchKernelGGL((cgemm_batched_kernel<T, true, false, 16, 128, 64>),
                         gridw, dim3(GBLK), 0, s, (const T*)A, (const T*)B,
                         (T*)C, M, N, K, sA, sB, sC);
    else
      hipLaunchKernelGGL((cgemm_batched_kernel<T, false, false, 16, 128, 64>),
                         gridw, dim3(GBLK), 0, s, (const T*)A, (const T*)B,
                         (T*)C, M, N, K, sA, sB, sC);
    return gcheck(hipGetLastError());
  }
  dim3 grid((uint32_t)((N + 63) / 64), (uint32_t)((M + 63) / 64),
            (uint32_t)batch);
  // A BK=64 single-panel variant for K <= 64 (one barrier, no pipeline)
  // was A/B'd NEGATIVE at the cfg5 rmatvec shape (0.432 -> 0.518 ms):
  // the 133 KB LDS footprint drops occupancy to 2 WG/CU, which costs
  // more than the three extra barriers it saves.  BK=16 for all K.
  // pipeline A/B knobs for the shallow-K adjoint (cfg5 rmatvec):
  // PAM_CGEMM_NBUF=1 single-buffers the LDS panels (halves LDS,
  // doubles resident WGs); PAM_CGEMM_BK=8 halves the panel depth.
  // r02 sweep at the cfg5 shapes (profiles/r02s5/sweep.log): BK=8 wins
  // the shallow-K adjoint (K=64: 0.262->0.240 ms saveGt=off,
  // 0.206->0.200 on) and loses the K=256 matvec (0.207->0.218), so
  // K<=64 auto-selects BK=8; PAM_CGEMM_BK=16 forces the deep panel.
  static int nbov = [] {
    const char* e = getenv("PAM_CGEMM_NBUF");
    return e ? atoi(e) : 0;
  }();
  static int bkov = [] {
    const char* e = getenv("PAM_CGEMM_BK");
    return e ? atoi(e) : 0;
  }();
  if (!acc && nbov == 1) {
    if (opa)
      hipLaunchKernelGGL((cgemm_batched_kernel<T, true, false, 16, 64, 64,
                          1>), grid, dim3(GBLK), 0, s, (const T*)A,
                         (const T*)B, (T*)C, M, N, K, sA, sB, sC);
    else
      hipLaunchKernelGGL((cgemm_batched_kernel<T, false, false, 16, 64, 64,
                          1>), grid, dim3(GBLK), 0, s, (const T*)A,
                         (const T*)B, (T*)C, M, N, K, sA, sB, sC);
    return gcheck(hipGetLastError());
  }
  if (!acc && (bkov == 8 || (bkov == 0 && K <= 64))) {
    if (opa)
      hipLaunchKernelGGL((cgemm_batched_kernel<T, true, false, 8>), grid,
                         dim3(GBLK), 0, s, (const T*)A, (const T*)B, (T*)C,
                         M, N, K, sA, sB, sC);
    else
      hipLaunchKernelGGL((cgemm_batched_kernel<T, false, false, 8>), grid,
                         dim3(GBLK), 0, s, (const T*)A, (const T*)B, (T*)C,
                         M, N, K, sA, sB, sC);
    return gcheck(hipGetLastError());
  }
  if (opa && acc)
    hipLaunchKernelGGL((cgemm_batched_kernel<T, true, true>), grid,
                       dim3(GBLK), 0, s, (const T*)A, (const T*)B, (T*)C, M,
                       N, K, sA, sB, sC);
  else if (opa)
    hipLaunchKernelGGL((cgemm_batched_kernel<T, true, false>), grid,
                       dim3(GBLK), 0, s, (const T*)A, (const T*)B, (T*)C, M,
                       N, K, sA, sB, sC);
  else if (acc)
    hipLaunchKernelGGL((cgemm_batched_kernel<T, false, true>), grid,
                       dim3(GBLK), 0, s, (const T*)A, (const T*)B, (T*)C, M,
                       N, K, sA, sB, sC);
  else
    hipLaunchKernelGGL((cgemm_batched_kernel<T, false, false>), grid,
                       dim3(GBLK), 0, s, (const T*)A, (const T*)B, (T*)C, M,
                       N, K, sA, sB, sC);
  return gcheck(hipGetLastError());
}

extern "C" int pam_cgemm_batched(void* stream, const void* A, const void* B,
                                 void* C, int64_t batch, int64_t M, int64_t N,
                                 int64_t K, int64_t strideA, int64_t strideB,
                                 int64_t strideC, int opa, int accumulate,
                                 int dtype) {
  if (dtype == PAM_C128)
    return cgemm_launch<double>(stream, A, B, C, batch, M, N, K, strideA,
                                strideB, strideC, opa, accumulate);
  if (dtype == PAM_C64)
    return cgemm_launch<float>(stream, A, B, C, batch, M, N, K, strideA,
                               strideB, strideC, opa, accumulate);
  return PAM_EDTYPE;
}

// ---------------------------------------------------------------------------
// batched REAL GEMM: C_b (+)= op(A_b) @ B_b over blockIdx.z, the
// single-plane analogue of cgemm_batched_kernel.  The Fredholm path for
// float32/float64 kernels (ref Fredholm1.py:123 `ncp.matmul(G, x)` on a
// 3-D G) previously looped pam_gemm per slice from Python — 513 launches
// of 4 workgroups each at the cfg5 shape (chip empty + ctypes overhead);
// the z-batched grid fills the chip like the complex path does.
// ---------------------------------------------------------------------------
template <typename T, bool CT, bool ACC, int BK = 16, int BM = 64,
          int BN = 64>
__global__ void __launch_bounds__(GBLK) gemm_batched_kernel(
    const T* __restrict__ A, const T* __restrict__ B, T* __restrict__ C,
    int64_t M, int64_t N, int64_t K, int64_t strideA, int64_t strideB,
    int64_t strideC) {
  using CFG = GemmCfg<T>;
  constexpr int TM = CFG::TM, TK = CFG::TK;
  constexpr int NBUFC = (BK >= 64) ? 1 : 2;
  constexpr int MI = (BM / 2) / TM, NJ = (BN / 2) / TM;
  using acc_t = typename CFG::acc_t;

  __shared__ T As[NBUFC][BK][BM + 1];  // k-major
  __shared__ T Bs[NBUFC][BK][BN + 1];
  const int64_t b = blockIdx.z;
  const T* __restrict__ Ab = A + b * strideA;
  const T* __restrict__ Bb = B + b * strideB;
  T* __restrict__ Cb = C + b * strideC;
  const int64_t m0 = (int64_t)blockIdx.y * BM;
  const int64_t n0 = (int64_t)blockIdx.x * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int li = lane & (TM - 1);
  const int lk = lane / TM;

  acc_t s[MI][NJ];
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j) s[i][j] = {};

  constexpr int AE = (BM * BK) / GBLK;
  constexpr int BE = (BK * BN) / GBLK;
  T arg[AE], brg[BE];

  auto load_panel = [&](int64_t k0) {
#pragma unroll
    for (int e = 0; e < AE; ++e) {
      const int vi2 = tid + GBLK * e;
      int kk, mm;
      if constexpr (CT) {  // op(A)[m][k] = A[k][m] (real: no conj)
        mm = vi2 & (BM - 1);
        kk = vi2 / BM;
      } else {
        kk = vi2 & (BK - 1);
        mm = vi2 / BK;
      }
      const int64_t gm = m0 + mm;
      const int64_t gk = k0 + kk;
      T v = 0;
      if (gm < M && gk < K)
        v = CT ? Ab[gk * M + gm] : Ab[gm * K + gk];
      arg[e] = v;
    }
#pragma unroll
    for (int e = 0; e < BE; ++e) {
      const int vi2 = tid + GBLK * e;
      const int nn = vi2 & (BN - 1);
      const int kk = vi2 / BN;
      const int64_t gk = k0 + kk;
      const int64_t gn = n0 + nn;
      brg[e] = (gk < K && gn < N) ? Bb[gk * N + gn] : (T)0;
    }
  };

  auto store_panel = [&](int buf) {
#pragma unroll
    for (int e = 0; e < AE; ++e) {
      const int vi2 = tid + GBLK * e;
      int kk, mm;
      if constexpr (CT) {
        mm = vi2 & (BM - 1);
        kk = vi2 / BM;
      } else {
        kk = vi2 & (BK - 1);
        mm = vi2 / BK;
      }
      As[buf][kk][mm] = arg[e];
    }
#pragma unroll
    for (int e = 0; e < BE; ++e) {
      const int vi2 = tid + GBLK * e;
      const int nn = vi2 & (BN - 1);
      const int kk = vi2 / BN;
      Bs[buf][kk][nn] = brg[e];
    }
  };

  const int64_t NP = (K + BK - 1) / BK;
  if (NP > 0) {
    load_panel(0);
    store_panel(0);
    __syncthreads();
    for (int64_t p = 0; p < NP; ++p) {
      if (NBUFC > 1 && p + 1 < NP) load_panel((p + 1) * BK);
      const int cur = (int)(p % NBUFC);
      T a[2][MI], bb[2][NJ];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        a[0][mi] = As[cur][lk][wr * (BM / 2) + mi * TM + li];
#pragma unroll
      for (int nj = 0; nj < NJ; ++nj)
        bb[0][nj] = Bs[cur][lk][wc * (BN / 2) + nj * TM + li];
#pragma unroll
      for (int kk = 0; kk < BK / TK; ++kk) {
        const int cf = kk & 1, nf = (kk + 1) & 1;
        if (kk + 1 < BK / TK) {
          const int krow = (kk + 1) * TK + lk;
#pragma unroll
          for (int mi = 0; mi < MI; ++mi)
            a[nf][mi] = As[cur][krow][wr * (BM / 2) + mi * TM + li];
#pragma unroll
          for (int nj = 0; nj < NJ; ++nj)
            bb[nf][nj] = Bs[cur][krow][wc * (BN / 2) + nj * TM + li];
        }
#pragma unroll
        for (int mi = 0; mi < MI; ++mi)
#pragma unroll
          for (int nj = 0; nj < NJ; ++nj)
            s[mi][nj] = CFG::mfma(a[cf][mi], bb[cf][nj], s[mi][nj]);
      }
      if (p + 1 < NP) {
        if (NBUFC == 1) {
          __syncthreads();
          load_panel((p + 1) * BK);
        }
        store_panel((int)((p + 1) % NBUFC));
        __syncthreads();
      }
    }
  }

#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int nj = 0; nj < NJ; ++nj) {
      const int64_t r0 = m0 + wr * (BM / 2) + mi * TM;
      const int64_t cc = n0 + wc * (BN / 2) + nj * TM + li;
      if (cc >= N) continue;
#pragma unroll
      for (int reg = 0; reg < CFG::NREG; ++reg) {
        const int64_t rr = r0 + CFG::crow(lane, reg);
        if (rr < M) {
          if constexpr (ACC)
            Cb[rr * N + cc] += s[mi][nj][reg];
          else
            Cb[rr * N + cc] = s[mi][nj][reg];
        }
      }
    }
}

template <typename T>
static int rgemm_launch(void* stream, const void* A, const void* B, void* C,
                        int64_t batch, int64_t M, int64_t N, int64_t K,
                        int64_t sA, int64_t sB, int64_t sC, int opa,
                        int acc) {
  if (batch <= 0 || M <= 0 || N <= 0 || K < 0 || !A || !B || !C)
    return PAM_EARG;
  dim3 grid((uint32_t)((N + 63) / 64), (uint32_t)((M + 63) / 64),
            (uint32_t)batch);
  hipStream_t s = (hipStream_t)stream;
#define RG_LAUNCH(CTV, ACCV)                                                  \
  hipLaunchKernelGGL((gemm_batched_kernel<T, CTV, ACCV>), grid, dim3(GBLK),   \
                     0, s, (const T*)A, (const T*)B, (T*)C, M, N, K, sA, sB, \
                     sC)
  if (opa && acc) RG_LAUNCH(true, true);
  else if (opa) RG_LAUNCH(true, false);
  else if (acc) RG_LAUNCH(false, true);
  else RG_LAUNCH(false, false);
#undef RG_LAUNCH
  return gcheck(hipGetLastError());
}

extern "C" int pam_gemm_batched(void* stream, const void* A, const void* B,
                                void* C, int64_t batch, int64_t M, int64_t N,
                                int64_t K, int64_t strideA, int64_t strideB,
                                int64_t strideC, int opa, int accumulate,
                                int dtype) {
  if (dtype == PAM_F64)
    return rgemm_launch<double>(stream, A, B, C, batch, M, N, K, strideA,
                                strideB, strideC, opa, accumulate);
  if (dtype == PAM_F32)
    return rgemm_launch<float>(stream, A, B, C, batch, M, N, K, strideA,
                               strideB, strideC, opa, accumulate);
  return PAM_EDTYPE;
}

// ---------------------------------------------------------------------------
// complex (conj-)transpose: At[c][r] = (conj?)(A[r][c]) on interleaved
// (re,im) pairs — materializes A^H for the complex MatrixMult adjoint
// panels (ref MatrixMult.py:416,737 "A.T.conj()").  Same 32x32 LDS tile
// as transpose_kernel, elements are (re,im) pairs.
// ---------------------------------------------------------------------------
template <typename T, bool CONJ>
__global__ void __launch_bounds__(GBLK) ctranspose_kernel(
    const T* __restrict__ A, T* __restrict__ At, int64_t nr, int64_t nc) {
  __shared__ T tile[32][33][2];
  const int64_t r0 = (int64_t)blockIdx.y * 32;
  const int64_t c0 = (int64_t)blockIdx.x * 32;
  const int tr = threadIdx.x / 32;
  const int tc = threadIdx.x % 32;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int64_t r = r0 + tr + 8 * i;
    if (r < nr && c0 + tc < nc) {
      const int64_t off = 2 * (r * nc + c0 + tc);
      tile[tr + 8 * i][tc][0] = A[off];
      tile[tr + 8 * i][tc][1] = CONJ ? -A[off + 1] : A[off + 1];
    }
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int64_t c = c0 + tr + 8 * i;
    if (c < nc && r0 + tc < nr) {
      const int64_t off = 2 * (c * nr + r0 + tc);
      At[off] = tile[tc][tr + 8 * i][0];
      At[off + 1] = tile[tc][tr + 8 * i][1];
    }
  }
}

extern "C" int pam_ctranspose(void* stream, const void* A, void* At,
                              int64_t nr, int64_t nc, int conj, int dtype) {
  if (nr <= 0 || nc <= 0 || !A || !At) return PAM_EARG;
  dim3 grid((uint32_t)((nc + 31) / 32), (uint32_t)((nr + 31) / 32));
  hipStream_t s = (hipStream_t)stream;
  if (dtype == PAM_C128) {
    if (conj)
      hipLaunchKernelGGL((ctranspose_kernel<double, true>), grid, dim3(GBLK),
                         0, s, (const double*)A, (double*)At, nr, nc);
    else
      hipLaunchKernelGGL((ctranspose_kernel<double, false>), grid, dim3(GBLK),
                         0, s, (const double*)A, (double*)At, nr, nc);
    return gcheck(hipGetLastError());
  }
  if (dtype == PAM_C64) {
    if (conj)
      hipLaunchKernelGGL((ctranspose_kernel<float, true>), grid, dim3(GBLK),
                         0, s, (const float*)A, (float*)At, nr, nc);
    else
      hipLaunchKernelGGL((ctranspose_kernel<float, false>), grid, dim3(GBLK),
                         0, s, (const float*)A, (float*)At, nr, nc);
    return gcheck(hipGetLastError());
  }
  return PAM_EDTYPE;
}

// ---------------------------------------------------------------------------
// LDS-tiled transpose: At[c][r] = A[r][c] (for the adjoint's A^T panels,
// ref MatrixMult.py:416,737 "A.T.conj()"; real dtypes -> plain transpose).
// 32x32 tiles, +1 pad against bank conflicts.
// ---------------------------------------------------------------------------
template <typename T>
__global__ void __launch_bounds__(GBLK) transpose_kernel(
    const T* __restrict__ A, T* __restrict__ At, int64_t nr, int64_t nc) {
  __shared__ T tile[32][33];
  const int64_t r0 = (int64_t)blockIdx.y * 32;
  const int64_t c0 = (int64_t)blockIdx.x * 32;
  const int tr = threadIdx.x / 32;   // 8 rows of 32 threads
  const int tc = threadIdx.x % 32;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int64_t r = r0 + tr + 8 * i;
    if (r < nr && c0 + tc < nc) tile[tr + 8 * i][tc] = A[r * nc + c0 + tc];
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int64_t c = c0 + tr + 8 * i;
    if (c < nc && r0 + tc < nr) At[c * nr + r0 + tc] = tile[tc][tr + 8 * i];
  }
}

extern "C" int pam_transpose(void* stream, const void* A, void* At,
                             int64_t nr, int64_t nc, int dtype) {
  if (nr <= 0 || nc <= 0 || !A || !At) return PAM_EARG;
  dim3 grid((uint32_t)((nc + 31) / 32), (uint32_t)((nr + 31) / 32));
  hipStream_t s = (hipStream_t)stream;
  if (dtype == PAM_F64) {
    hipLaunchKernelGGL((transpose_kernel<double>), grid, dim3(GBLK), 0, s,
                       (const double*)A, (double*)At, nr, nc);
    return gcheck(hipGetLastError());
  }
  if (dtype == PAM_F32) {
    hipLaunchKernelGGL((transpose_kernel<float>), grid, dim3(GBLK), 0, s,
                       (const float*)A, (float*)At, nr, nc);
    return gcheck(hipGetLastError());
  }
  return PAM_EDTYPE;
}
