// Probe: does the nt (non-temporal) cache hint on stores/loads move HBM
// bandwidth for the stencil's access pattern (stream-read x, stream-write
// y, 8R+8B/pt fp64)?  Standalone: hipcc --offload-arch=gfx950 -O3
// -ffp-contract=off scripts/probe_nt_store.hip -o gpurun_out/probe_nt
// Informs fd_kernel store policy (pylops_mpi_amd/csrc/fd_defs.h).
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>

#define BLK 256
typedef double T;
struct V2 { T x, y; };

template <int MODE>  // 0 plain, 1 nt store, 2 nt load, 3 nt both
__global__ void __launch_bounds__(BLK) stencil(const T* __restrict__ x,
                                               T* __restrict__ y,
                                               int64_t rows, int64_t m) {
  // centered3 interior body on a [rows, m] field, vector width 2 (16 B/lane)
  const int64_t mv = m / 2;
  for (int64_t r = blockIdx.y; r < rows; r += gridDim.y) {
    const T* xm = x + (r > 0 ? r - 1 : r) * m;
    const T* xp = x + (r + 1 < rows ? r + 1 : r) * m;
    T* yr = y + r * m;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < mv;
         i += stride) {
      V2 a, b, o;
      if constexpr (MODE & 2) {
        a.x = __builtin_nontemporal_load(xm + 2 * i);
        a.y = __builtin_nontemporal_load(xm + 2 * i + 1);
        b.x = __builtin_nontemporal_load(xp + 2 * i);
        b.y = __builtin_nontemporal_load(xp + 2 * i + 1);
      } else {
        a = *(const V2*)(xm + 2 * i);
        b = *(const V2*)(xp + 2 * i);
      }
      o.x = 0.5 * (b.x - a.x);
      o.y = 0.5 * (b.y - a.y);
      if constexpr (MODE & 1) {
        __builtin_nontemporal_store(o.x, yr + 2 * i);
        __builtin_nontemporal_store(o.y, yr + 2 * i + 1);
      } else {
        *(V2*)(yr + 2 * i) = o;
      }
    }
  }
}

int main() {
  const int64_t rows = 2048, m = 2048LL * 128;
  const int64_t n = rows * m;
  T *x, *y;
  (void)hipMalloc(&x, n * sizeof(T));
  (void)hipMalloc(&y, n * sizeof(T));
  (void)hipMemset(x, 0x11, n * sizeof(T));
  dim3 grid(8, 512);
  {
    int64_t g = (m / 2 + BLK - 1) / BLK;
    grid.x = (unsigned)((g < 4096 / 512) ? g : 4096 / 512 * 8);
    if (grid.x < 1) grid.x = 1;
  }
  // match the production launch shape: gy<=512, gx capped so gx*gy<=4096*?
  grid = dim3(8, 512);
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  const char* names[4] = {"plain      ", "nt-store   ", "nt-load    ",
                          "nt-both    "};
  for (int mode = 0; mode < 4; ++mode) {
    auto launch = [&](int md) {
      switch (md) {
        case 0: hipLaunchKernelGGL(stencil<0>, grid, dim3(BLK), 0, 0, x, y, rows, m); break;
        case 1: hipLaunchKernelGGL(stencil<1>, grid, dim3(BLK), 0, 0, x, y, rows, m); break;
        case 2: hipLaunchKernelGGL(stencil<2>, grid, dim3(BLK), 0, 0, x, y, rows, m); break;
        default: hipLaunchKernelGGL(stencil<3>, grid, dim3(BLK), 0, 0, x, y, rows, m); break;
      }
    };
    for (int w = 0; w < 3; ++w) launch(mode);
    (void)hipDeviceSynchronize();
    (void)hipEventRecord(e0);
    const int reps = 20;
    for (int r = 0; r < reps; ++r) launch(mode);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    double tbs = (double)n * 16.0 * reps / (ms * 1e-3) / 1e12;
    printf("%s %8.3f ms/launch  %6.2f TB/s algorithmic\n", names[mode],
           ms / reps, tbs);
  }
  return 0;
}
