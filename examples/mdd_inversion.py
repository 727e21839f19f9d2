"""Multi-dimensional deconvolution (MDD): invert the MPIMDC chain with
CGLS — the reference's tutorials/mdd.py pattern at a small size.
Single-GPU: python examples/mdd_inversion.py
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
    nt, nv, nfreq, ns, nr = 64, 4, 33, 16, 12  # ns > nr: overdetermined
    g = torch.Generator(device=comm.device).manual_seed(1)
    G = (torch.rand((nfreq, ns, nr), generator=g, device=comm.device) - 0.5
         + 1j * (torch.rand((nfreq, ns, nr), generator=g,
                            device=comm.device) - 0.5)).to(torch.complex64)
    MDCop = pm.MPIMDC(G, nt, nv, nfreq, dt=0.004, dr=1.0, twosided=False)

    xt = pm.DistributedArray((MDCop.shape[1],),
                             partition=pm.Partition.BROADCAST,
                             dtype=np.complex64)
    xt[:] = (torch.rand(MDCop.shape[1], generator=g,
                        device=comm.device) - 0.5).to(torch.complex64)
    d = MDCop @ xt

    x0 = xt.zeros_like()
    xinv, *_ = pm.cgls(MDCop, d, x0, niter=30, tol=0.0)
    err = (xinv - xt).norm() / xt.norm()
    if comm.rank == 0:
        print(f"mdd: relative model error after 30 CGLS iters {err:.3e}")


if __name__ == "__main__":
    main()
