// Empirically determine the lane<->element mapping of v_mfma_f64_16x16x4f64.
#include <hip/hip_runtime.h>
#include <stdio.h>
typedef double f64x4 __attribute__((ext_vector_type(4)));
__global__ void probe(const double* A, const double* B, double* out) {
  int l = threadIdx.x;
  f64x4 acc = {0, 0, 0, 0};
  acc = __builtin_amdgcn_mfma_f64_16x16x4f64(A[l], B[l], acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) out[l * 4 + r] = acc[r];
}
int main() {
  // candidate input mapping: lane l supplies A[i=l%16][k=l/16], B[k=l/16][j=l%16]
  double hA[64], hB[64];
  for (int l = 0; l < 64; ++l) {
    int i = l % 16, k = l / 16;
    hA[l] = 1.0 + i * 4 + k;          // A[i][k] = 1 + i*4+k
    hB[l] = 1.0 / (1.0 + k * 16 + i); // B[k][j] = 1/(1+k*16+j), j=l%16
  }
  double *dA, *dB, *dO;
  hipMalloc(&dA, 64 * 8); hipMalloc(&dB, 64 * 8); hipMalloc(&dO, 256 * 8);
  hipMemcpy(dA, hA, 64 * 8, hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, 64 * 8, hipMemcpyHostToDevice);
  probe<<<1, 64>>>(dA, dB, dO);
  double hO[256];
  hipMemcpy(hO, dO, 256 * 8, hipMemcpyDeviceToHost);
  // host reference D[i][j] = sum_k A[i][k]*B[k][j]
  double D[16][16];
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      double s = 0;
      for (int k = 0; k < 4; ++k)
        s += (1.0 + i * 4 + k) * (1.0 / (1.0 + k * 16 + j));
      D[i][j] = s;
    }
  // find (i,j) for each (lane, reg)
  for (int l = 0; l < 64; ++l)
    for (int r = 0; r < 4; ++r) {
      double v = hO[l * 4 + r];
      int fi = -1, fj = -1, nm = 0;
      for (int i = 0; i < 16; ++i)
        for (int j = 0; j < 16; ++j)
          if (fabs(v - D[i][j]) < 1e-12 * (1 + fabs(v))) { fi = i; fj = j; nm++; }
      if (l < 20 || nm != 1)
        printf("lane %2d reg %d -> i=%2d j=%2d (matches=%d) v=%.6f\n",
               l, r, fi, fj, nm, v);
    }
  return 0;
}
