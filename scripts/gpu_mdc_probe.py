"""MDC-only timing/profiling helper (cfg5 shape)."""
import os, sys, time
import numpy as np, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pylops_mpi_amd as pm
from pylops_mpi_amd.comm import init_default_comm


def timeit(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / iters


def main():
    init_default_comm(torch.device("cuda:0"))
    g = torch.Generator(device="cuda").manual_seed(42)
    nf, ns, nr, nv, nt = 513, 64, 256, 256, 1024
    G = (torch.rand((nf, ns, nr), generator=g, device="cuda") - 0.5
         + 1j * (torch.rand((nf, ns, nr), generator=g, device="cuda") - 0.5)
         ).to(torch.complex64)
    mdc = pm.MPIMDC(G, nt, nv, nf, dt=0.004, dr=1.0, twosided=False)
    xm = pm.DistributedArray((mdc.shape[1],),
                             partition=pm.Partition.BROADCAST,
                             dtype=np.complex64)
    xm[:] = (torch.rand(mdc.shape[1], generator=g, device="cuda") - 0.5
             ).to(torch.complex64)
    print(f"mdc matvec  {timeit(lambda: mdc.matvec(xm)) * 1e3:.3f} ms")
    y = mdc.matvec(xm)
    print(f"mdc rmatvec {timeit(lambda: mdc.rmatvec(y)) * 1e3:.3f} ms")


if __name__ == "__main__":
    main()
