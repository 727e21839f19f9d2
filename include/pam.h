/* pam — pylops-mpi hot path, MI355X-native (gfx950) C-ABI.
 *
 * Stateless kernel entry points over raw device pointers + HIP streams.
 * Device memory ownership, rank partitioning and RCCL collectives live in
 * the host layer (pylops_mpi_amd, PyTorch-ROCm device tensors +
 * torch.distributed over RCCL); this library is pure compute.
 *
 * Each entry point names the reference interface it replaces
 * (paths relative to /root/reference/pylops_mpi/).
 *
 * Conventions:
 *  - `stream` is a hipStream_t passed as void* (the caller's current
 *    stream; the library never synchronizes).
 *  - all functions return 0 on success, a hipError_t (>0) on launch
 *    failure, or a PAM_E* code (<0) on argument errors.
 *  - `dtype`: 0 = float64, 1 = float32, 2 = complex128, 3 = complex64
 *    (complex: interleaved re,im pairs of the base type).
 *  - scalar outputs (`out`) are single-element float64 device buffers.
 */
#ifndef PAM_H
#define PAM_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define PAM_F64 0
#define PAM_F32 1
#define PAM_C128 2 /* complex: interleaved (re, im) pairs of the base type */
#define PAM_C64 3

#define PAM_EARG   (-1)  /* bad argument */
#define PAM_EDTYPE (-2)  /* unsupported dtype */
#define PAM_EOP    (-3)  /* unknown op code */

/* ABI version, bumped on any signature change. */
int64_t pam_version(void);

/* Number of float64 workspace elements the reduction entry points need in
 * `ws` (device buffer). */
int64_t pam_reduce_ws_elems(void);

/* ------------------------------------------------------------------ *
 * Element-wise vector ops.
 * Replaces DistributedArray.add/iadd/multiply/__neg__/__setitem__
 * (ref DistributedArray.py:605-683,214-252) — per-rank local math on the
 * HBM-resident block.
 * ------------------------------------------------------------------ */
int pam_fill(void* stream, void* y, int64_t n, double value, int dtype);
int pam_neg(void* stream, void* y, const void* x, int64_t n, int dtype);
/* y = a + b */
int pam_add(void* stream, void* y, const void* a, const void* b, int64_t n,
            int dtype);
/* y = a - b (bitwise equal to the reference's add(-b), ref :624-628) */
int pam_sub(void* stream, void* y, const void* a, const void* b, int64_t n,
            int dtype);
/* y = a * b */
int pam_mul(void* stream, void* y, const void* a, const void* b, int64_t n,
            int dtype);
/* y = alpha * x */
int pam_scale(void* stream, void* y, const void* x, double alpha, int64_t n,
              int dtype);
/* y += alpha * x — the fused CGLS update (x += a*c / s -= a*q,
 * ref optimization/cls_basic.py:390-391: the reference materializes two
 * temporaries per axpy; this is one pass). */
int pam_axpy(void* stream, void* y, const void* x, double alpha, int64_t n,
             int dtype);
/* y = x + beta * y — the CGLS direction update c = r + b*c
 * (ref optimization/cls_basic.py:396). */
int pam_xpby(void* stream, void* y, const void* x, double beta, int64_t n,
             int dtype);

/* Device-scalar solver fast path.  alpha/beta are 1-element f64 DEVICE
 * buffers written by a prior pam_dot / pam_scalar_* launch on the same
 * stream; scale is a host constant (+-1.0, exact).  Together with the two
 * recurrence-scalar kernels below, a whole CG/CGLS iteration
 * (ref optimization/cls_basic.py:370-404) is launched with a single host
 * synchronization — the stop-test readback of k — instead of one blocking
 * readback per dot/norm. */
/* y += scale * (*alpha) * x */
int pam_axpy_d(void* stream, void* y, const void* x, const void* alpha,
               double scale, int64_t n, int dtype);
/* y = x + scale * (*beta) * y */
int pam_xpby_d(void* stream, void* y, const void* x, const void* beta,
               double scale, int64_t n, int dtype);
/* *out = |num[0] / (den[0] + damp * den[1])| — a = |kold / (q.q + damp^2
 * c.c)|, ref cls_basic.py:379-383 (CG: damp == 0, ref :121-123). */
int pam_scalar_alpha(void* stream, void* out, const void* num,
                     const void* den, double damp);
/* *out = |num[0] / den[0]| — b = k / kold, ref cls_basic.py:394-395. */
int pam_scalar_div(void* stream, void* out, const void* num, const void* den);

/* Complex element-wise ops (interleaved storage).  add/sub/neg/fill and
 * real-alpha axpy/xpby on complex arrays are the REAL kernels above on the
 * 2n-float view (the solvers' a,b scalars are real, ref cls_basic.py:389).
 * These cover the truly complex cases (ref DistributedArray.py:661-683
 * multiply, :840-854 conj). */
int pam_cmul(void* stream, void* y, const void* a, const void* b, int64_t n,
             int dtype);
int pam_cscale(void* stream, void* y, const void* x, double alpha_re,
               double alpha_im, int64_t n, int dtype);
int pam_conj(void* stream, void* y, const void* x, int64_t n, int dtype);

/* Thresholding for ISTA/FISTA (pylops _softthreshold/_hardthreshold/
 * _halfthreshold formulas, imported by ref optimization/cls_sparsity.py:10;
 * half = the published Xu et al. 2012 L1/2 prox, see pam.hip).
 * kind: 0 = soft, 1 = hard, 2 = half.  In-place safe (y may equal x). */
int pam_thresh(void* stream, void* y, const void* x, int64_t n, int kind,
               double thresh, int dtype);

/* ------------------------------------------------------------------ *
 * Reductions (wavefront-shuffle + LDS tree, deterministic tree shape for
 * a given n — fixed partial count, fixed combine order).
 * `ws` is a device float64 scratch of >= pam_reduce_ws_elems() elements;
 * `out` is a 1-element float64 device buffer (the caller allreduces it
 * across ranks, replacing ref Distributed.py:35-112).
 * ------------------------------------------------------------------ */
/* out = sum(x[i] * y[i]) in float64.
 * Replaces the local part of DistributedArray.dot (ref :685-717). */
int pam_dot(void* stream, const void* x, const void* y, int64_t n, void* ws,
            void* out, int dtype);

/* out = sum(x[i] * y[i]) for complex arrays (conjx != 0: vdot,
 * conj(x)*y); `out` is a 2-element float64 device buffer (re, im) and
 * `ws` needs 2*pam_reduce_ws_elems() float64 elements.
 * Replaces the complex branch of DistributedArray.dot (ref :685-717). */
int pam_cdot(void* stream, const void* x, const void* y, int64_t n,
             int conjx, void* ws, void* out, int dtype);

/* Local part of DistributedArray.norm / _compute_vector_norm
 * (ref :719-838).  op: 0 = sum(|x^p|) (float_power semantics, ref :786),
 * 1 = max(|x|), 2 = min(|x|), 3 = count_nonzero.  Complex dtypes reduce
 * over |z| (|z^p| == |z|^p for real p). */
int pam_norm_local(void* stream, const void* x, int64_t n, int op, double p,
                   void* ws, void* out, int dtype);

/* ------------------------------------------------------------------ *
 * Fused finite-difference stencils along axis 0 of a SCATTER-partitioned
 * array.  One kernel per (operator, kind, direction): no materialized
 * ghost concatenation (the reference copies the whole array per apply,
 * ref DistributedArray.py:974,992-994,1028).
 *
 * Replaces MPIFirstDerivative._matvec_* and ._rmatvec_*
 * (ref basicoperators/FirstDerivative.py:141-318) and
 * MPISecondDerivative (ref basicoperators/SecondDerivative.py:124-256).
 *
 * Layout: x, y are [nloc, m] row-major (m = prod(dims[1:]) elements);
 * gf/gb hold pam_fd_halo_width(op) neighbour planes ([w, m]; gf = last
 * planes of rank-1, gb = first planes of rank+1; may be NULL at the
 * global edges).  row0 = global row of local row 0, nglob = dims[0].
 * coeff = 1/sampling (first derivative) or 1/sampling^2 (second).
 *
 * op codes:
 *   0 fd1 forward  matvec   1 fd1 forward  rmatvec
 *   2 fd1 backward matvec   3 fd1 backward rmatvec
 *   4 fd1 cent3    matvec   5 fd1 cent3    rmatvec
 *   6 fd1 cent5    matvec   7 fd1 cent5    rmatvec
 *   8 fd2 forward  matvec   9 fd2 forward  rmatvec
 *  10 fd2 backward matvec  11 fd2 backward rmatvec
 *  12 fd2 centered matvec  13 fd2 centered rmatvec
 * ------------------------------------------------------------------ */
int64_t pam_fd_halo_width(int op);
/* rbegin/rend restrict the computed LOCAL row range [rbegin, rend) so the
 * interior (halo-independent) rows can be launched concurrently with the
 * RCCL halo exchange and the boundary rows after it completes. */
int pam_fd_apply(void* stream, int op, int edge, const void* x,
                 const void* gf, const void* gb, void* y, int64_t nloc,
                 int64_t m, int64_t row0, int64_t nglob, int64_t rbegin,
                 int64_t rend, double coeff, int dtype);

/* Serial non-stationary 1-D convolution along axis d of [batch, d, m]
 * (the local operator of MPINonStationaryConvolve1D's Halo sandwich,
 * ref signalprocessing/NonStatConvolve1d.py:129-168): filters hs[nf][hsize]
 * anchored at oh + q*dh, linearly interpolated per position, clamped at
 * the ends.  forward != 0: y[n] = sum_t x[n+hh-t] h_{n+hh-t}[t];
 * adjoint: y[n] = sum_t x[n-hh+t] h_n[t]. */
int pam_nsconv(void* stream, int forward, const void* x, void* y,
               const void* hs, int64_t batch, int64_t d, int64_t m,
               int64_t nf, int64_t hsize, double oh, double dh, int dtype);

/* Serial (single-rank) stencil along an arbitrary axis: the block is
 * viewed as [batch, d, m] with the derivative along d.  The local
 * operators inside MPIBlockDiag for Gradient/Laplacian axes >= 1
 * (ref basicoperators/Gradient.py:101-118, Laplacian.py:98-126, which
 * wrap serial pylops First/SecondDerivative). */
int pam_fd_serial(void* stream, int op, int edge, const void* x, void* y,
                  int64_t batch, int64_t d, int64_t m, double coeff,
                  int dtype);

/* ------------------------------------------------------------------ *
 * Dense GEMV: y = A @ x (trans=0) or y = A^T @ x (trans=1) for a
 * row-major [nr, nc] matrix resident in HBM.
 *
 * The local apply of a dense operator inside MPIBlockDiag
 * (ref basicoperators/BlockDiag.py:122-144 calling a serial dense
 * MatrixMult per rank, as in examples/plot_cgls.py:30-33) — HBM-bound
 * matrix read, wave-per-row (trans=0) or column-tile partials with a
 * deterministic chunk combine (trans=1; `ws` is a float64 device scratch
 * of pam_gemv_ws_elems(nr, nc) elements, unused for trans=0).
 * ------------------------------------------------------------------ */
int64_t pam_gemv_ws_elems(int64_t nr, int64_t nc);
int pam_gemv(void* stream, int trans, const void* A, const void* x, void* y,
             int64_t nr, int64_t nc, void* ws, int dtype);

/* ------------------------------------------------------------------ *
 * Panel GEMM: C = A @ B (accumulate != 0: C += A @ B) for row-major
 * A [M,K] (lda), B [K,N] (ldb), C [M,N] (ldc).
 *
 * The local panel product of MPIMatrixMult — block kind
 * (ref basicoperators/MatrixMult.py:369-372 `ncp.matmul(A_local,
 * X_local)`) and the SUMMA accumulation step (ref :661-668
 * `Y_local += ncp.dot(Atemp, Xtemp)`) — on MFMA matrix cores:
 * v_mfma_f32_32x32x2_f32 (exact f32) / v_mfma_f64_16x16x4_f64.
 *
 * pam_transpose materializes A^T for the adjoint panels
 * (ref MatrixMult.py:416,737 `A.T.conj()`; real dtypes).
 * ------------------------------------------------------------------ */
int pam_gemm(void* stream, const void* A, const void* B, void* C, int64_t M,
             int64_t N, int64_t K, int64_t lda, int64_t ldb, int64_t ldc,
             int accumulate, int dtype);
int pam_transpose(void* stream, const void* A, void* At, int64_t nr,
                  int64_t nc, int dtype);

/* Batched complex GEMM: for b in [0, batch):
 *   C_b (+)= op(A_b) @ B_b  with op = N (opa=0) or conjugate-transpose
 *   (opa=1; then A_b is [K, M]); accumulate != 0 adds into C.
 * A_b = A + b*strideA etc. (strides in ELEMENTS = complex pairs).
 * The Fredholm1 batched integral kernel (ref signalprocessing/
 * Fredholm1.py:123 `ncp.matmul(G, x)` and :149-156 adjoint), and —
 * with batch=1 — the complex MatrixMult panel product
 * (ref MatrixMult.py:341-427,610-765 complex dtypes). */
int pam_cgemm_batched(void* stream, const void* A, const void* B, void* C,
                      int64_t batch, int64_t M, int64_t N, int64_t K,
                      int64_t strideA, int64_t strideB, int64_t strideC,
                      int opa, int accumulate, int dtype);

/* Batched REAL GEMM: the single-plane analogue of pam_cgemm_batched
 * (op = N or transpose; accumulate != 0 adds into C).  The Fredholm1
 * path for float32/float64 kernels (ref Fredholm1.py:123 on a real G)
 * — one z-batched launch instead of a per-slice host loop. */
int pam_gemm_batched(void* stream, const void* A, const void* B, void* C,
                     int64_t batch, int64_t M, int64_t N, int64_t K,
                     int64_t strideA, int64_t strideB, int64_t strideC,
                     int opa, int accumulate, int dtype);

/* Complex <-> real (de)interleave (the MDC chain's real extraction and
 * complex output carrier, ref waveeqprocessing/MDC.py:55-69): unzip
 * writes the n real parts of an interleaved complex array; zip writes n
 * (re, 0) pairs from a real array.  dtype names the COMPLEX type. */
int pam_unzip(void* stream, void* dst_real, const void* src_cplx, int64_t n,
              int dtype);
int pam_zip(void* stream, void* dst_cplx, const void* src_real, int64_t n,
            int dtype);

/* Strided-batched real FFTs along dim 0 of a row-major (nt, m) array
 * (rocFFT native layout — no permute copies; UNSCALED; the MDC chain
 * folds the ortho norms into its kernel).  dtype is the REAL element
 * type.  pam_irfft_strided may clobber its input (rocFFT scratch). */
int pam_rfft_strided(void* stream, const void* in_real, void* out_cplx,
                     int64_t nt, int64_t m, int dtype);
int pam_irfft_strided(void* stream, void* in_cplx, void* out_real,
                      int64_t nt, int64_t m, int dtype);

/* Contiguous-batch variants ((m, nt) layout, transform along the last
 * dim) — the MDC chain transposes itself (pam_unzip_t / pam_zip_t /
 * pam_ctranspose) because rocFFT's strided real plans insert full
 * pack/unpack copies. */
int pam_rfft_contig(void* stream, const void* in_real, void* out_cplx,
                    int64_t nt, int64_t m, int dtype);
int pam_irfft_contig(void* stream, void* in_cplx, void* out_real,
                     int64_t nt, int64_t m, int dtype);

/* Transpose-fused (de)interleave: unzip_t = complex (nt, m) -> real
 * (m, nt) (real parts); zip_t = real (m, nt) -> complex (nt, m) with
 * zero imag.  dtype names the COMPLEX type. */
int pam_unzip_t(void* stream, void* dst_real, const void* src_cplx,
                int64_t nt, int64_t m, int dtype);
int pam_zip_t(void* stream, void* dst_cplx, const void* src_real,
              int64_t nt, int64_t m, int dtype);

/* 256^2-tile f32 GEMM with async global->LDS staging (glds deep
 * pipeline): C (+)= At^T @ B for a K-MAJOR A (At = A^T, [K, M]
 * row-major) and row-major B [K, N], C [M, N].  Fast path only:
 * returns PAM_EARG unless M,N % 256 == 0, K % 32 == 0 and pointers are
 * 16-B aligned — callers fall back to pam_gemm. */
int pam_gemm_kt(void* stream, const void* At, const void* B, void* C,
                int64_t M, int64_t N, int64_t K, int accumulate,
                int dtype);

/* Complex (conj-)transpose on interleaved (re,im) pairs: At = A^T
 * (conj=0) or A^H (conj=1) — the complex MatrixMult adjoint panels
 * (ref MatrixMult.py:416,737 `A.T.conj()`). */
int pam_ctranspose(void* stream, const void* A, void* At, int64_t nr,
                   int64_t nc, int conj, int dtype);

#ifdef __cplusplus
}
#endif
#endif /* PAM_H */
