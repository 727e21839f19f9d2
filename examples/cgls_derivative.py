"""Smoothed-model inversion with MPIFirstDerivative + CGLS — the
reference's plot_cgls.py / derivative tutorial pattern on the MI355X
stack.  Run on N GPUs of one node with:

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 examples/cgls_derivative.py
"""
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pylops_mpi_amd as pm
from pylops_mpi_amd.comm import init_default_comm


def main():
    comm = init_default_comm()
    dims = (512, 256, 16)
    n = int(np.prod(dims))
    Fop = pm.MPIFirstDerivative(dims, kind="centered", order=3,
                                dtype=np.float64)

    # synthetic model + data, model resident in HBM from the start
    g = torch.Generator(device=comm.device).manual_seed(0)
    x = pm.DistributedArray((n,), comm)
    x[:] = torch.randn(x.local_shape, generator=g, dtype=torch.float64,
                       device=comm.device)
    y = Fop @ x

    x0 = x.zeros_like()
    xinv, istop, niter, r1, r2, cost = pm.cgls(Fop, y, x0, niter=40,
                                               damp=1e-4, tol=0.0)
    res = (Fop @ xinv - y).norm() / y.norm()
    if comm.rank == 0:
        print(f"cgls: {niter} iters, relative data residual {res:.3e}")


if __name__ == "__main__":
    main()
