"""Stencil launch-shape sweep at a given dims (env PAM_FD_*)."""
import os, sys, time
import numpy as np, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pylops_mpi_amd as pm
from pylops_mpi_amd.comm import init_default_comm

def main():
    init_default_comm(torch.device("cuda:0"))
    dims = tuple(int(v) for v in os.environ.get("DIMS", "512x4096x256").split("x"))
    n = int(np.prod(dims))
    op = pm.MPIFirstDerivative(dims, kind="centered", order=3)
    g = torch.Generator(device="cuda").manual_seed(1)
    x = pm.DistributedArray((n,))
    x[:] = torch.randn(n, generator=g, dtype=torch.float64, device="cuda")
    rmv = os.environ.get("RMATVEC") == "1"
    apply = op.rmatvec if rmv else op.matvec
    for _ in range(3):
        y = apply(x)
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(20):
        y = apply(x)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t) / 20
    print(f"dims={dims} {'rmv' if rmv else 'mv'} gy={os.environ.get('PAM_FD_GY','-')} "
          f"cap={os.environ.get('PAM_FD_CAP','-')}: {dt*1e3:7.3f} ms "
          f"{16*n/dt/1e12:5.2f} TB/s")

main()
