"""Fredholm1-only timing at the cfg5 judged shape (cgemm tile A/Bs)."""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pylops_mpi_amd as pm  # noqa: E402
from pylops_mpi_amd.comm import init_default_comm  # noqa: E402


def timeit(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / iters


def main():
    init_default_comm(torch.device("cuda:0"))
    g = torch.Generator(device="cuda").manual_seed(7)
    nf, ns, nr, nv = 513, 64, 256, 256
    G = (torch.rand((nf, ns, nr), generator=g, device="cuda") - 0.5
         + 1j * (torch.rand((nf, ns, nr), generator=g, device="cuda") - 0.5)
         ).to(torch.complex64)
    for saveGt in (False, True):
        op = pm.MPIFredholm1(G, nv, saveGt=saveGt, dtype="complex64")
        x = pm.DistributedArray((op.shape[1],),
                                partition=pm.Partition.BROADCAST,
                                dtype=np.complex64)
        x[:] = (torch.rand(op.shape[1], generator=g, device="cuda") - 0.5
                ).to(torch.complex64)
        y = op.matvec(x)
        tm = timeit(lambda: op.matvec(x)) * 1e3
        tr = timeit(lambda: op.rmatvec(y)) * 1e3
        # real flops: 4 real mul-adds per cmac -> 8 flops per MAC
        fl = 8.0 * nf * ns * nr * nv
        print(f"saveGt={saveGt}: matvec {tm:.3f} ms ({fl/tm/1e9:.1f} TF) "
              f"rmatvec {tr:.3f} ms ({fl/tr/1e9:.1f} TF) "
              f"tile={os.environ.get('PAM_CGEMM_TILE','0')}")


if __name__ == "__main__":
    main()
