"""CG / CGLS solvers — same recurrences as the reference
(/root/reference/pylops_mpi/optimization/cls_basic.py and basic.py), driving
the operator + DistributedArray surface unchanged ("cgls drives
the loop unchanged").

MI355X-first difference: the three per-iteration vector updates use the
fused one-pass HIP axpy/xpby kernels (DistributedArray.iaxpy_/xpby_)
instead of the reference's two fresh temporaries per update
(ref cls_basic.py:390-391,396 via DistributedArray.py:618-683) — same
arithmetic per element (mul-round, add-round; -ffp-contract=off), one HBM
pass instead of three.
"""
from typing import Optional, Tuple

import numpy as np

from .distributedarray import DistributedArray


class CG:
    """ref cls_basic.py:12-249."""

    def __init__(self, Op):
        self.Op = Op

    def setup(self, y: DistributedArray, x0: DistributedArray,
              niter: Optional[int] = None, tol: float = 1e-4,
              show: bool = False) -> DistributedArray:
        # ref :57-108
        self.y = y
        self.niter = niter
        self.tol = tol
        x = x0.copy()
        self.r = self.y - self.Op.matvec(x)
        self.rank = x.rank
        self.c = self.r.copy()
        self.kold = float(np.abs(self.r.dot(self.r.conj())))
        self.cost = [float(np.sqrt(self.kold))]
        self.iiter = 0
        return x

    def step(self, x: DistributedArray, show: bool = False):
        # ref :110-141
        Opc = self.Op.matvec(self.c)
        cOpc = np.abs(self.c.dot(Opc.conj()))
        a = float(self.kold / cOpc)
        x.iaxpy_(a, self.c)          # x += a * c
        self.r.iaxpy_(-a, Opc)       # r -= a * Opc
        k = float(np.abs(self.r.dot(self.r.conj())))
        b = float(k / self.kold)
        self.c.xpby_(self.r, b)      # c = r + b * c
        self.kold = k
        self.iiter += 1
        self.cost.append(float(np.sqrt(self.kold)))
        return x

    def run(self, x, niter=None, show=False, itershow=(10, 10, 10)):
        niter = self.niter if niter is None else niter
        if niter is None:
            raise ValueError("niter must not be None")
        while self.iiter < niter and self.kold > self.tol:
            x = self.step(x, show)
        return x

    def finalize(self, show: bool = False):
        self.cost = np.array(self.cost)

    def solve(self, y, x0, niter: int = 10, tol: float = 1e-4,
              show: bool = False, itershow=(10, 10, 10)):
        x = self.setup(y=y, x0=x0, niter=niter, tol=tol, show=show)
        x = self.run(x, niter, show=show, itershow=itershow)
        self.finalize(show)
        return x, self.iiter, self.cost


class CGLS:
    """ref cls_basic.py:252-531."""

    def __init__(self, Op):
        self.Op = Op

    def setup(self, y: DistributedArray, x0: DistributedArray,
              niter: Optional[int] = None, damp: float = 0.0,
              tol: float = 1e-4, show: bool = False) -> DistributedArray:
        # ref :308-368
        self.y = y
        self.damp = damp ** 2
        self.tol = tol
        self.niter = niter
        x = x0.copy()
        self.s = self.y - self.Op.matvec(x)
        r = self.Op.rmatvec(self.s)
        if damp != 0.0:
            r.iaxpy_(-damp, x)  # r -= damp * x, fused (ref :347-348)
        self.rank = x.rank
        self.c = r.copy()
        self.q = self.Op.matvec(self.c)
        self.kold = float(np.abs(r.dot(r.conj())))
        self.cost = [float(self.s.norm())]
        # NB: the reference uses the raw damp here and damp**2 in step
        # (ref :358 vs :401) — reproduced faithfully.
        self.cost1 = [float(np.sqrt(self.cost[0] ** 2
                                    + damp * np.abs(x.dot(x.conj()))))]
        self.iiter = 0
        return x

    def step(self, x: DistributedArray, show: bool = False):
        # ref :370-404; vector updates fused (module docstring)
        a = float(np.abs(self.kold / (self.q.dot(self.q.conj())
                                      + self.damp * self.c.dot(self.c.conj()))))
        x.iaxpy_(a, self.c)          # x += a * c        (ref :390)
        self.s.iaxpy_(-a, self.q)    # s -= a * q        (ref :391)
        r = self.Op.rmatvec(self.s)
        if self.damp != 0.0:
            r.iaxpy_(-self.damp, x)  # r -= damp^2 * x, fused (ref :392-393)
        k = float(np.abs(r.dot(r.conj())))
        b = float(k / self.kold)
        self.c.xpby_(r, b)           # c = r + b * c     (ref :396)
        self.q = self.Op.matvec(self.c)
        self.kold = k
        self.iiter += 1
        self.cost.append(float(self.s.norm()))
        self.cost1.append(float(np.sqrt(self.cost[self.iiter] ** 2
                                        + self.damp
                                        * np.abs(x.dot(x.conj())))))
        return x

    def run(self, x, niter=None, show=False, itershow=(10, 10, 10)):
        # ref :406-449
        niter = self.niter if niter is None else niter
        if niter is None:
            raise ValueError("niter must not be None")
        while self.iiter < niter and self.kold > self.tol:
            x = self.step(x, show)
        return x

    def finalize(self, show: bool = False):
        # ref :451-469
        self.istop = 1 if self.kold < self.tol else 2
        self.r1norm = self.kold
        self.r2norm = self.cost1[self.iiter]
        self.cost = np.array(self.cost)

    def solve(self, y, x0, niter: int = 10, damp: float = 0.0,
              tol: float = 1e-4, show: bool = False,
              itershow=(10, 10, 10)):
        # ref :471-531
        x = self.setup(y=y, x0=x0, niter=niter, damp=damp, tol=tol,
                       show=show)
        x = self.run(x, niter, show=show, itershow=itershow)
        self.finalize(show)
        return x, self.istop, self.iiter, self.r1norm, self.r2norm, self.cost


def cg(Op, y, x0, niter: int = 10, tol: float = 1e-4, show: bool = False,
       itershow: Tuple[int, int, int] = (10, 10, 10)):
    """ref optimization/basic.py:13-70."""
    cgsolve = CG(Op)
    x, iiter, cost = cgsolve.solve(y=y, x0=x0, niter=niter, tol=tol,
                                   show=show, itershow=itershow)
    return x, iiter, cost


def cgls(Op, y, x0, niter: int = 10, damp: float = 0.0, tol: float = 1e-4,
         show: bool = False, itershow: Tuple[int, int, int] = (10, 10, 10)):
    """ref optimization/basic.py:73-148."""
    cgsolve = CGLS(Op)
    x, istop, iiter, r1norm, r2norm, cost = cgsolve.solve(
        y=y, x0=x0, niter=niter, damp=damp, tol=tol, show=show,
        itershow=itershow)
    return x, istop, iiter, r1norm, r2norm, cost


def power_iteration(Op, b_k, niter: int = 10, tol: float = 1e-5,
                    dtype="float64"):
    """Largest-eigenpair power iteration,
    ref optimization/eigs.py:10-100 (random re-init of b_k, vdot Rayleigh
    quotient, renormalization via the HIP scale kernel)."""
    import torch

    from .stacked import StackedDistributedArray

    cmpx = np.issubdtype(np.dtype(dtype), np.complexfloating)

    def _randomize(d):
        t = torch.rand(d.local_shape, dtype=torch.float64,
                       device=d.local_array.device)
        if cmpx:
            t = t + 1j * torch.rand(d.local_shape, dtype=torch.float64,
                                    device=d.local_array.device)
        d[:] = t.to(d.local_array.dtype)

    if isinstance(b_k, StackedDistributedArray):
        for d in b_k.distarrays:
            _randomize(d)
    else:
        _randomize(b_k)
    b_k = b_k * (1.0 / float(b_k.norm()))
    maxeig_old = 0.0
    iiter = 0
    for iiter in range(niter):
        b1_k = Op.matvec(b_k)
        maxeig = b_k.dot(b1_k, vdot=True)
        maxeig = complex(maxeig) if cmpx else float(maxeig)
        b1_k_norm = float(b1_k.norm())
        b_k = b1_k * (1.0 / b1_k_norm)
        if abs(maxeig - maxeig_old) < tol * abs(maxeig):
            break
        maxeig_old = maxeig
    return maxeig, b_k, iiter + 1
