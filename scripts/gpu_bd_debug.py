"""Diagnose the BlockDiag saveAt=False pair (r02 session-12 anomaly)."""
import os, sys, time
import numpy as np, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pylops_mpi_amd as pm
from pylops_mpi_amd.comm import init_default_comm

def timeit(fn, iters=50):
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
    g = torch.Generator(device="cuda").manual_seed(42)
    n = 4096
    Ad = torch.rand((n, n), generator=g, dtype=torch.float64,
                    device="cuda") * 2 - 1
    xv = torch.rand(n, generator=g, dtype=torch.float64, device="cuda")
    op = pm.DenseLocal(Ad, saveAt=False)
    print(f"raw mv   {timeit(lambda: op.matvec(xv))*1e6:8.1f} us")
    print(f"raw rmv  {timeit(lambda: op.rmatvec(xv))*1e6:8.1f} us")
    bd = pm.MPIBlockDiag([op])
    x = pm.DistributedArray((n,))
    x[:] = xv
    y0 = bd.matvec(x)
    print(f"bd mv    {timeit(lambda: bd.matvec(x))*1e6:8.1f} us")
    print(f"bd rmv   {timeit(lambda: bd.rmatvec(y0))*1e6:8.1f} us")
    def pair():
        yy = bd.matvec(x)
        bd.rmatvec(yy)
    print(f"bd pair  {timeit(pair)*1e6:8.1f} us")
    op2 = pm.DenseLocal(Ad, saveAt=True)
    bd2 = pm.MPIBlockDiag([op2])
    y0b = bd2.matvec(x)
    def pair2():
        yy = bd2.matvec(x)
        bd2.rmatvec(yy)
    print(f"bd pair saveAt=True {timeit(pair2)*1e6:8.1f} us")

main()

def more():
    init_default_comm(torch.device("cuda:0"))
    g = torch.Generator(device="cuda").manual_seed(42)
    n = 4096
    Ad = torch.rand((n, n), generator=g, dtype=torch.float64,
                    device="cuda") * 2 - 1
    xv = torch.rand(n, generator=g, dtype=torch.float64, device="cuda")
    op = pm.DenseLocal(Ad, saveAt=False)
    bd = pm.MPIBlockDiag([op])
    x = pm.DistributedArray((n,))
    x[:] = xv
    y0 = bd.matvec(x)
    def pair_sync():
        yy = bd.matvec(x)
        torch.cuda.synchronize()
        bd.rmatvec(yy)
    print(f"pair w/ mid-sync {timeit(pair_sync)*1e6:8.1f} us")
    def pair_fixed_input():
        bd.matvec(x)
        bd.rmatvec(y0)     # rmv input NOT the fresh mv output
    print(f"pair fixed-input {timeit(pair_fixed_input)*1e6:8.1f} us")
    def rr():
        bd.rmatvec(y0)
        bd.rmatvec(y0)
    print(f"rmv+rmv          {timeit(rr)*1e6:8.1f} us")
    def mm():
        bd.matvec(x)
        bd.matvec(x)
    print(f"mv+mv            {timeit(mm)*1e6:8.1f} us")
    op_raw = pm.DenseLocal(Ad, saveAt=False)
    def raw_pair():
        yy = op_raw.matvec(xv)
        op_raw.rmatvec(yy)
    print(f"raw mv->rmv      {timeit(raw_pair)*1e6:8.1f} us")

if __name__ == "__main__" or True:
    more()
