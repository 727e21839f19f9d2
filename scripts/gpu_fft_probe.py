"""Time MPIFFTND (1 GPU) against raw torch.fft on the same data — the
operator adds reshaped I/O + (at world 1) trivial redistributes; this
probe quantifies that overhead and the rocFFT throughput itself.
"""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pylops_mpi_amd as pm
from pylops_mpi_amd.comm import init_default_comm


def timeit(fn, reps=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / reps


def main():
    init_default_comm(torch.device("cuda:0"))
    for dims, axes, real, dt in (
        ((1024, 512, 512), (0, 1, 2), False, np.complex128),
        ((1024, 512, 512), (0, 1, 2), True, np.float64),
        ((4096, 4096), (0, 1), False, np.complex128),
    ):
        n = int(np.prod(dims))
        op = pm.MPIFFTND(dims=dims, axes=axes, real=real, dtype=dt)
        g = torch.Generator(device="cuda").manual_seed(1)
        tdt = torch.complex128 if np.dtype(dt).kind == "c" else torch.float64
        if tdt == torch.complex128:
            x = (torch.randn(n, generator=g, dtype=torch.float64,
                             device="cuda")
                 + 1j * torch.randn(n, generator=g, dtype=torch.float64,
                                    device="cuda"))
        else:
            x = torch.randn(n, generator=g, dtype=torch.float64,
                            device="cuda")
        xd = pm.DistributedArray.to_dist(x)
        y = op.matvec(xd)
        t_mv = timeit(lambda: op.matvec(xd))
        t_rmv = timeit(lambda: op.rmatvec(y))
        xs = x.reshape(dims)
        if real:
            t_raw = timeit(lambda: torch.fft.rfftn(xs, dim=list(axes)))
        else:
            t_raw = timeit(lambda: torch.fft.fftn(xs, dim=list(axes)))
        gb = n * np.dtype(dt).itemsize / 1e9
        print(f"dims={dims} real={real} {np.dtype(dt).name}: "
              f"matvec {t_mv * 1e3:8.2f} ms, rmatvec {t_rmv * 1e3:8.2f} ms, "
              f"raw fftn {t_raw * 1e3:8.2f} ms "
              f"(overhead x{t_mv / t_raw:4.2f}); array {gb:.2f} GB")


if __name__ == "__main__":
    main()
