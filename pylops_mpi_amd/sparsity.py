"""ISTA / FISTA sparse solvers.

Drop-in for /root/reference/pylops_mpi/optimization/cls_sparsity.py:50-718
and sparsity.py wrappers: proximal-gradient iterations with the pylops
soft/hard thresholding formulas (HIP pam_thresh kernel, in place) and the
power-iteration step-size estimate ('soft'/'hard'/'half' kinds).
"""
from math import sqrt
from typing import Optional, Tuple, Union

import numpy as np

from . import _ffi
from .distributedarray import DistributedArray
from .solvers import power_iteration
from .stacked import StackedDistributedArray

_THRESH_KINDS = {"soft": 0, "hard": 1, "half": 2}


def _thresh_inplace(d: DistributedArray, kind: int, thresh: float):
    d._require_compute()
    flat = d._flat()
    _ffi.checked(_ffi.lib().pam_thresh(
        d._stream(), flat.data_ptr(), flat.data_ptr(),
        d.local_array.numel(), kind, float(thresh), d._dt()), "thresh")
    return d


def _apply_thresh(x, kind: int, thresh: float):
    # ref cls_sparsity.py:22-48
    if isinstance(x, StackedDistributedArray):
        for d in x.distarrays:
            _thresh_inplace(d, kind, thresh)
        return x
    return _thresh_inplace(x, kind, thresh)


class ISTA:
    """ref cls_sparsity.py:50-487."""

    def __init__(self, Op):
        self.Op = Op

    def setup(self, y, x0, niter: Optional[int] = None, SOp=None,
              eps: float = 0.1, alpha: Optional[float] = None,
              eigsdict=None, tol: float = 1e-10, threshkind: str = "soft",
              decay=None, monitorres: bool = False, show: bool = False):
        # ref :143-271
        self.y = y
        self.SOp = SOp
        self.niter = niter
        self.eps = eps
        self.eigsdict = {} if eigsdict is None else eigsdict
        self.tol = tol
        self.monitorres = monitorres
        if threshkind not in ("hard", "soft", "half"):
            raise ValueError(
                f"threshkind must be hard, soft, half, got {threshkind}")
        self.threshkind = _THRESH_KINDS[threshkind]
        self.decay = np.ones(niter) if decay is None else decay
        if alpha is not None:
            self.alpha = alpha
        else:
            # 1/lambda_max(Op^H Op) via power iteration (ref :245-259)
            Op1 = self.Op.H * self.Op
            maxeig = np.abs(power_iteration(
                Op1, b_k=x0.empty_like(), dtype=Op1.dtype,
                **self.eigsdict)[0])
            self.alpha = float(1.0 / maxeig)
        self.thresh = eps * self.alpha * 0.5
        x = x0.copy()
        self.rank = x.rank
        self.cost = []
        self.iiter = 0
        self.normresold = np.inf
        return x

    def step(self, x, show: bool = False):
        # ref :273-346
        xold = x.copy()
        res = self.y - self.Op.matvec(x)
        if self.monitorres:
            self.normres = res.norm()
            if self.normres > self.normresold:
                raise ValueError(
                    f"ISTA stopped at iteration {self.iiter} due to "
                    "residual increasing, consider modifying "
                    "eps and/or alpha...")
            self.normresold = self.normres
        grad = self.alpha * self.Op.rmatvec(res)
        x_unthresh = x + grad
        if self.SOp is not None:
            x_unthresh = self.SOp.rmatvec(x_unthresh)
        x = _apply_thresh(x_unthresh, self.threshkind,
                          self.decay[self.iiter] * self.thresh)
        if self.SOp is not None:
            x = self.SOp.matvec(x)
        xupdate = float((x - xold).norm())
        costdata = 0.5 * float(res.norm()) ** 2
        costreg = self.eps * float(x.norm(ord=1))
        self.cost.append(float(costdata + costreg))
        self.iiter += 1
        return x, xupdate

    def run(self, x, niter=None, show=False, itershow=(10, 10, 10)):
        # ref :348-397
        xupdate = np.inf
        niter = self.niter if niter is None else niter
        if niter is None:
            raise ValueError("niter must not be None")
        while self.iiter < niter and xupdate > self.tol:
            x, xupdate = self.step(x, show)
        return x

    def finalize(self, show: bool = False):
        self.cost = np.array(self.cost)

    def solve(self, y, x0, niter=None, SOp=None, eps=0.1, alpha=None,
              eigsdict=None, tol=1e-10, threshkind="soft", decay=None,
              monitorres=False, show=False, itershow=(10, 10, 10)):
        x = self.setup(y=y, x0=x0, niter=niter, SOp=SOp, eps=eps,
                       alpha=alpha, eigsdict=eigsdict, tol=tol,
                       threshkind=threshkind, decay=decay,
                       monitorres=monitorres, show=show)
        x = self.run(x, niter, show=show, itershow=itershow)
        self.finalize(show)
        return x, self.iiter, self.cost


class FISTA(ISTA):
    """ref cls_sparsity.py:489-718."""

    def setup(self, y, x0, niter=None, SOp=None, eps=0.1, alpha=None,
              eigsdict=None, tol=1e-10, threshkind="soft", decay=None,
              monitorres=False, show=False):
        x = super().setup(y=y, x0=x0, niter=niter, SOp=SOp, eps=eps,
                          alpha=alpha, eigsdict=eigsdict, tol=tol,
                          threshkind=threshkind, decay=decay,
                          monitorres=monitorres, show=show)
        self.t = 1.0
        self.z = x.copy()
        return x

    def step(self, x, z, show: bool = False):
        # ref :581-666
        xold = x.copy()
        res = self.y - self.Op.matvec(z)
        if self.monitorres:
            self.normres = res.norm()
            if self.normres > self.normresold:
                raise ValueError(
                    f"FISTA stopped at iteration {self.iiter} due to "
                    "residual increasing, consider modifying "
                    "eps and/or alpha...")
            self.normresold = self.normres
        grad = self.alpha * self.Op.rmatvec(res)
        x_unthresh = z + grad
        if self.SOp is not None:
            x_unthresh = self.SOp.rmatvec(x_unthresh)
        x = _apply_thresh(x_unthresh, self.threshkind,
                          self.decay[self.iiter] * self.thresh)
        if self.SOp is not None:
            x = self.SOp.matvec(x)
        told = self.t
        self.t = (1.0 + sqrt(1.0 + 4.0 * self.t ** 2)) / 2.0
        z = x + ((told - 1.0) / self.t) * (x - xold)
        xupdate = float((x - xold).norm())
        costdata = 0.5 * float((self.y - self.Op.matvec(x)).norm()) ** 2
        costreg = self.eps * float(x.norm(ord=1))
        self.cost.append(float(costdata + costreg))
        self.iiter += 1
        return x, z, xupdate

    def run(self, x, niter=None, show=False, itershow=(10, 10, 10)):
        # ref :667-718
        xupdate = np.inf
        niter = self.niter if niter is None else niter
        if niter is None:
            raise ValueError("niter must not be None")
        z = self.z
        while self.iiter < niter and xupdate > self.tol:
            x, z, xupdate = self.step(x, z, show)
        return x


def ista(Op, y, x0, niter: int = 10, **kwargs
         ) -> Tuple[Union[DistributedArray, StackedDistributedArray], int,
                    np.ndarray]:
    """ref optimization/sparsity.py:11 wrapper."""
    solver = ISTA(Op)
    return solver.solve(y=y, x0=x0, niter=niter, **kwargs)


def fista(Op, y, x0, niter: int = 10, **kwargs):
    """ref optimization/sparsity.py:136 wrapper."""
    solver = FISTA(Op)
    return solver.solve(y=y, x0=x0, niter=niter, **kwargs)
