// pam — strided-batched real FFTs for the MDC chain (hipFFT/rocFFT).
//
// torch.fft.rfft(dim=0) on a (nt, m) tensor moves the transform dim to
// the innermost position with TWO full permute copies per call (r02 MDC
// kernel trace: 2 x ~200 us of at::native elementwise per apply at the
// cfg5 shape).  rocFFT handles the (stride=m, dist=1) layout natively;
// these entry points plan it once per (nt, m, dtype, direction) and run
// the transform in place in the chain's natural frequency-major layout.
//
// Transforms are UNSCALED (rocFFT convention); the MDC chain folds the
// ortho 1/sqrt(nt) of both directions into the Fredholm kernel G (a
// per-frequency diagonal scale commuting with the per-frequency
// block-diagonal G — waveeqprocessing/MDC.py:41-43 already prescales G
// by dr*dt*sqrt(nt)).
//
// NOTE hipfftExecZ2D/C2R may overwrite the input buffer (rocFFT uses it
// as scratch) — callers pass intermediates they own.

#include <hip/hip_runtime.h>
#include <hipfft/hipfft.h>
#include <stdint.h>

#include <map>
#include <mutex>
#include <tuple>

#include "../../include/pam.h"

namespace {

using PlanKey = std::tuple<int64_t, int64_t, int, int>;  // nt, m, dt, mode

struct PlanCache {
  std::map<PlanKey, hipfftHandle> plans;
  std::mutex mu;
};

PlanCache& cache() {
  static PlanCache c;
  return c;
}

// mode: 0/1 = strided inverse/forward ((stride=m, dist=1) layout);
//       2/3 = contiguous-batch inverse/forward ((stride=1, dist=len))
int get_plan(int64_t nt, int64_t m, int dtype, int mode,
             hipfftHandle* out) {
  PlanCache& c = cache();
  std::lock_guard<std::mutex> lock(c.mu);
  const PlanKey key{nt, m, dtype, mode};
  auto it = c.plans.find(key);
  if (it != c.plans.end()) {
    *out = it->second;
    return 0;
  }
  hipfftHandle plan;
  const int fwd = mode & 1;
  const bool contig = mode >= 2;
  const int64_t nfft = nt / 2 + 1;
  int n[1] = {(int)nt};
  int inembed[1] = {(int)(fwd ? nt : nfft)};
  int onembed[1] = {(int)(fwd ? nfft : nt)};
  hipfftType type;
  if (dtype == PAM_F32)
    type = fwd ? HIPFFT_R2C : HIPFFT_C2R;
  else
    type = fwd ? HIPFFT_D2Z : HIPFFT_Z2D;
  hipfftResult r;
  if (contig)
    r = hipfftPlanMany(&plan, 1, n, inembed, 1, inembed[0], onembed, 1,
                       onembed[0], type, (int)m);
  else
    r = hipfftPlanMany(&plan, 1, n, inembed, (int)m, 1, onembed, (int)m, 1,
                       type, (int)m);
  if (r != HIPFFT_SUCCESS) return PAM_EARG;
  c.plans.emplace(key, plan);
  *out = plan;
  return 0;
}

}  // namespace

/* Forward R2C along dim 0 of a row-major (nt, m) real array into a
 * (nt/2+1, m) complex array (same column-strided layout), UNSCALED.
 * dtype = the REAL element type (PAM_F32/PAM_F64). */
extern "C" int pam_rfft_strided(void* stream, const void* in_real,
                                void* out_cplx, int64_t nt, int64_t m,
                                int dtype) {
  if (nt <= 0 || m <= 0 || !in_real || !out_cplx) return PAM_EARG;
  hipfftHandle plan;
  int rc = get_plan(nt, m, dtype, 1, &plan);
  if (rc) return rc;
  hipfftSetStream(plan, (hipStream_t)stream);
  hipfftResult r;
  if (dtype == PAM_F32)
    r = hipfftExecR2C(plan, (hipfftReal*)in_real,
                      (hipfftComplex*)out_cplx);
  else if (dtype == PAM_F64)
    r = hipfftExecD2Z(plan, (hipfftDoubleReal*)in_real,
                      (hipfftDoubleComplex*)out_cplx);
  else
    return PAM_EDTYPE;
  return r == HIPFFT_SUCCESS ? 0 : PAM_EARG;
}

/* Inverse C2R along dim 0: (nt/2+1, m) complex -> (nt, m) real,
 * UNSCALED (a full forward+inverse round trip multiplies by nt).
 * MAY clobber the input buffer. */
extern "C" int pam_irfft_strided(void* stream, void* in_cplx,
                                 void* out_real, int64_t nt, int64_t m,
                                 int dtype) {
  if (nt <= 0 || m <= 0 || !in_cplx || !out_real) return PAM_EARG;
  hipfftHandle plan;
  int rc = get_plan(nt, m, dtype, 0, &plan);
  if (rc) return rc;
  hipfftSetStream(plan, (hipStream_t)stream);
  hipfftResult r;
  if (dtype == PAM_F32)
    r = hipfftExecC2R(plan, (hipfftComplex*)in_cplx,
                      (hipfftReal*)out_real);
  else if (dtype == PAM_F64)
    r = hipfftExecZ2D(plan, (hipfftDoubleComplex*)in_cplx,
                      (hipfftDoubleReal*)out_real);
  else
    return PAM_EDTYPE;
  return r == HIPFFT_SUCCESS ? 0 : PAM_EARG;
}


/* Contiguous-batch R2C: (m, nt) real row-major -> (m, nt/2+1) complex,
 * UNSCALED (the MDC chain transposes to this layout itself — fused into
 * its unzip/zip kernels — because rocFFT's strided real plans insert
 * full pack/unpack copies). */
extern "C" int pam_rfft_contig(void* stream, const void* in_real,
                               void* out_cplx, int64_t nt, int64_t m,
                               int dtype) {
  if (nt <= 0 || m <= 0 || !in_real || !out_cplx) return PAM_EARG;
  hipfftHandle plan;
  int rc = get_plan(nt, m, dtype, 3, &plan);
  if (rc) return rc;
  hipfftSetStream(plan, (hipStream_t)stream);
  hipfftResult r;
  if (dtype == PAM_F32)
    r = hipfftExecR2C(plan, (hipfftReal*)in_real,
                      (hipfftComplex*)out_cplx);
  else if (dtype == PAM_F64)
    r = hipfftExecD2Z(plan, (hipfftDoubleReal*)in_real,
                      (hipfftDoubleComplex*)out_cplx);
  else
    return PAM_EDTYPE;
  return r == HIPFFT_SUCCESS ? 0 : PAM_EARG;
}

/* Contiguous-batch C2R: (m, nt/2+1) complex -> (m, nt) real, UNSCALED.
 * MAY clobber the input buffer. */
extern "C" int pam_irfft_contig(void* stream, void* in_cplx,
                                void* out_real, int64_t nt, int64_t m,
                                int dtype) {
  if (nt <= 0 || m <= 0 || !in_cplx || !out_real) return PAM_EARG;
  hipfftHandle plan;
  int rc = get_plan(nt, m, dtype, 2, &plan);
  if (rc) return rc;
  hipfftSetStream(plan, (hipStream_t)stream);
  hipfftResult r;
  if (dtype == PAM_F32)
    r = hipfftExecC2R(plan, (hipfftComplex*)in_cplx,
                      (hipfftReal*)out_real);
  else if (dtype == PAM_F64)
    r = hipfftExecZ2D(plan, (hipfftDoubleComplex*)in_cplx,
                      (hipfftDoubleReal*)out_real);
  else
    return PAM_EDTYPE;
  return r == HIPFFT_SUCCESS ? 0 : PAM_EARG;
}
