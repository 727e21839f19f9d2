// Does the stencil's long-row wall come from power-of-two row strides
// (DRAM channel aliasing of the x[r-1]/x[r+1]/y[r] streams)?  Scan row
// length m around 1 MiB elements with and without a pad.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#define BLK 256
typedef double T;
struct V2 { T x, y; };

__global__ void __launch_bounds__(BLK) stencil(const T* __restrict__ x,
                                               T* __restrict__ y,
                                               int64_t rows, int64_t m,
                                               int64_t ld) {
  const int64_t mv = m / 2;
  for (int64_t r = blockIdx.y; r < rows; r += gridDim.y) {
    const T* xm = x + (r > 0 ? r - 1 : r) * ld;
    const T* xp = x + (r + 1 < rows ? r + 1 : r) * ld;
    T* yr = y + r * ld;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < mv;
         i += stride) {
      V2 a = *(const V2*)(xm + 2 * i);
      V2 b = *(const V2*)(xp + 2 * i);
      V2 o;
      o.x = 0.5 * (b.x - a.x);
      o.y = 0.5 * (b.y - a.y);
      *(V2*)(yr + 2 * i) = o;
    }
  }
}

static double run(int64_t rows, int64_t m, int64_t ld) {
  T *x, *y;
  (void)hipMalloc(&x, rows * ld * sizeof(T));
  (void)hipMalloc(&y, rows * ld * sizeof(T));
  (void)hipMemset(x, 0x11, rows * ld * sizeof(T));
  int64_t gx = (m / 2 + BLK - 1) / BLK;
  int64_t cap = 262144 / rows;
  if (cap < 1) cap = 1;
  if (gx > cap) gx = cap;
  dim3 grid((unsigned)gx, (unsigned)(rows < 65535 ? rows : 65535));
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  for (int w = 0; w < 3; ++w)
    hipLaunchKernelGGL(stencil, grid, dim3(BLK), 0, 0, x, y, rows, m, ld);
  (void)hipDeviceSynchronize();
  (void)hipEventRecord(e0);
  for (int r = 0; r < 15; ++r)
    hipLaunchKernelGGL(stencil, grid, dim3(BLK), 0, 0, x, y, rows, m, ld);
  (void)hipEventRecord(e1);
  (void)hipEventSynchronize(e1);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, e0, e1);
  (void)hipFree(x);
  (void)hipFree(y);
  return (double)rows * m * 16.0 * 15 / (ms * 1e-3) / 1e12;
}

int main() {
  const int64_t rows = 512;
  struct { const char* name; int64_t m, ld; } cases[] = {
      {"m=1Mi   ld=1Mi   (the N=8 slab)", 1048576, 1048576},
      {"m=1Mi   ld=1Mi+256 (padded)", 1048576, 1048576 + 256},
      {"m=1Mi-4Ki ld=same (non-pow2)", 1048576 - 4096, 1048576 - 4096},
      {"m=256Ki ld=256Ki (bench-like)", 262144, 262144},
      {"m=256Ki ld=256Ki+256 (padded)", 262144, 262144 + 256},
  };
  for (auto& c : cases) printf("%-34s %6.2f TB/s\n", c.name, run(rows, c.m, c.ld));
  return 0;
}
