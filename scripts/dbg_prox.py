import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
import oracle
import pylops_mpi_amd as pm
from pylops_mpi_amd.proximal import MPIL2
from pylops_mpi_amd.proximal.operators import _identity_op
from pylops_mpi_amd.comm import init_default_comm
init_default_comm(torch.device("cuda:0"))
dev = lambda a: torch.as_tensor(np.ascontiguousarray(a), device="cuda")
host = lambda t: t.cpu().numpy()
rng = np.random.default_rng(10)
ny, nx = 36, 24
A = rng.standard_normal((ny, nx)); x = rng.standard_normal(nx); b = rng.standard_normal(ny)
Op = pm.MPIBlockDiag([pm.DenseLocal(dev(A))])
bd = pm.DistributedArray.to_dist(dev(b))
x0 = pm.DistributedArray((nx,)); x0[:] = 0.0
xd = pm.DistributedArray.to_dist(dev(x))
tau, sigma = 0.4, 1.3
# stage 1: OpTb
OpTb = sigma * (Op.H @ bd)
print("OpTb diff:", np.abs(host(OpTb.asarray()) - sigma*(A.T@b)).max())
# stage 2: y
y = xd + tau * OpTb
yref = x + tau*(sigma*(A.T@b))
print("y diff:", np.abs(host(y.asarray()) - yref).max())
# stage 3: Op1 matvec
Iop = _identity_op(xd, Op.dtype)
Op1 = Iop + float(tau*sigma) * (Op.H @ Op)
v = rng.standard_normal(nx)
vd = pm.DistributedArray.to_dist(dev(v))
got = host(Op1.matvec(vd).asarray())
want = v.copy() + (tau*sigma)*(A.T@(A@v))
print("Op1 mv diff:", np.abs(got-want).max())
# stage 4: cg traces
xs, it, cost = pm.cg(Op1, y, x0.copy(), niter=15, tol=0.0)
xr = oracle.dense_cg(lambda w: w.copy() + (tau*sigma)*(A.T@(A@w)), yref, np.zeros(nx), niter=15, tol=0.0)
print("cg diff:", np.abs(host(xs.asarray())-xr).max(), "iters", it)
# full prox
l2d = MPIL2(Op=Op, b=bd, x0=x0, sigma=sigma, niter=15, solver="cg", kwargs_solver={"tol":0.0})
l2s = oracle.SerL2(Op=A, b=b, x0=np.zeros(nx), sigma=sigma, niter=15, solver="cg", kwargs_solver={"tol":0.0})
g1 = host(l2d.prox(xd, tau).asarray()); w1 = l2s.prox(x, tau)
print("prox diff:", np.abs(g1-w1).max(), " norm", np.linalg.norm(w1))
