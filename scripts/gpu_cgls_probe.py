"""Time CGLS end-to-end at the bench config (1 GPU), device-scalar vs
host-scalar iteration (solvers.py module docstring), and check the two
cost traces agree bitwise.  Run via gpurun; prints one line per path.
"""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pylops_mpi_amd as pm
from pylops_mpi_amd.comm import init_default_comm


def run_config(dims, niter):
    n = int(np.prod(dims))
    op = pm.MPIFirstDerivative(dims, kind="centered", order=3)
    g = torch.Generator(device="cuda").manual_seed(1)
    y = pm.DistributedArray((n,))
    y[:] = torch.randn(n, generator=g, dtype=torch.float64, device="cuda")
    x0 = pm.DistributedArray((n,))
    x0[:] = 0.0

    traces = {}
    for tag in ("dev", "host"):
        if tag == "host":
            os.environ["PAM_DISABLE_DEVSCALARS"] = "1"
        else:
            os.environ.pop("PAM_DISABLE_DEVSCALARS", None)
        pm.cgls(op, y, x0.copy(), niter=3, damp=0.1, tol=0.0)  # warmup
        torch.cuda.synchronize()
        t = time.perf_counter()
        _, _, _, _, _, cost = pm.cgls(op, y, x0.copy(), niter=niter,
                                      damp=0.1, tol=0.0)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t
        traces[tag] = np.asarray(cost)
        print(f"cgls[{tag:4s}] niter={niter} dims={dims}: {dt:.3f} s total, "
              f"{dt / niter * 1e3:.3f} ms/iter, cost[-1]={cost[-1]:.17e}")
    os.environ.pop("PAM_DISABLE_DEVSCALARS", None)
    same = np.array_equal(traces["dev"], traces["host"])
    print(f"  traces bitwise equal: {same}")
    assert same


def main():
    init_default_comm(torch.device("cuda:0"))
    # bench config (GPU-bound: 20 ms of kernels/iter hides host latency)
    run_config((2048, 2048, 128), 50)
    # small configs, where per-iteration sync/launch latency dominates
    run_config((256, 256, 64), 200)
    run_config((64, 64, 32), 500)


if __name__ == "__main__":
    main()
