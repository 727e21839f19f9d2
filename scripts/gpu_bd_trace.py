import os, sys, time
import numpy as np, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pylops_mpi_amd as pm
from pylops_mpi_amd.comm import init_default_comm
init_default_comm(torch.device("cuda:0"))
g = torch.Generator(device="cuda").manual_seed(42)
n = 4096
Ad = torch.rand((n, n), generator=g, dtype=torch.float64, device="cuda") * 2 - 1
xv = torch.rand(n, generator=g, dtype=torch.float64, device="cuda")
bd = pm.MPIBlockDiag([pm.DenseLocal(Ad, saveAt=False)])
x = pm.DistributedArray((n,)); x[:] = xv
for _ in range(3):
    yy = bd.matvec(x); bd.rmatvec(yy)
torch.cuda.synchronize()
t0 = time.perf_counter()
host = 0.0
for _ in range(50):
    h0 = time.perf_counter()
    yy = bd.matvec(x); bd.rmatvec(yy)
    host += time.perf_counter() - h0
torch.cuda.synchronize()
wall = time.perf_counter() - t0
print(f"wall {wall*1e3:.2f} ms  host-enqueue {host*1e3:.2f} ms for 50 pairs")
