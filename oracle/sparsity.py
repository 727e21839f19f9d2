"""Rank-simulating restatements of the reference's sparse solvers and
power iteration (TEST INFRASTRUCTURE — see oracle/__init__.py header:
only tests/, smoke() and bench.py's cpu_baseline may use this package).

Each function follows the reference implementation line for line:

* :func:`sim_ista` / :func:`sim_fista` — ref
  ``optimization/cls_sparsity.py`` ISTA.setup/step/run (:243-416) and
  FISTA.step/run (:581-718): threshold ``eps*alpha/2`` scaled by
  ``decay[iiter]``, cost ``0.5*||res||^2 + eps*||x||_1`` with the
  residual taken BEFORE the update (ISTA) / recomputed at the new ``x``
  (FISTA), stop on ``iiter < niter and xupdate > tol``.
* :func:`sim_power_iteration` — ref ``optimization/eigs.py:10-102``:
  random re-init of ``b_k`` (here the rank-deterministic
  :func:`powerit_rand` draws, which the golden generator also patches
  into the reference's ``np.random.rand``), Rayleigh quotient via
  ``vdot``, renormalisation each step, relative-tolerance early stop.
"""
from math import sqrt
from typing import List, Optional, Sequence, Tuple

import numpy as np

from .proximal import hard_threshold, half_threshold, soft_threshold
from .ranksim import SimArray

_THRESH = {"soft": soft_threshold, "hard": hard_threshold,
           "half": half_threshold}


def powerit_rand(rank: int, n: int) -> np.ndarray:
    """Deterministic stand-in for the per-rank ``np.random.rand(n)``
    draw in ref eigs.py:72-75 (the golden generator patches the
    reference side with this same function, keyed on the shim rank)."""
    return np.random.default_rng(12000 + rank).random(n)


def sim_power_iteration(Op, counts: Sequence[int], niter: int = 10,
                        tol: float = 1e-5
                        ) -> Tuple[float, SimArray, int]:
    """ref optimization/eigs.py:10-102 on a square operator whose
    solver-space rank splits are ``counts``."""
    locs: List[np.ndarray] = [powerit_rand(r, int(c)).astype(np.float64)
                              for r, c in enumerate(counts)]
    n = int(np.sum(counts))
    b_k = SimArray(locs, (n,))
    b_k_norm = b_k.norm()
    b_k = b_k * (1.0 / b_k_norm)
    maxeig_old = 0.0
    maxeig = 0.0
    iiter = 0
    for iiter in range(niter):
        b1_k = Op.matvec(b_k)
        maxeig = float(b_k.dot(b1_k, vdot=True))
        b1_k_norm = b1_k.norm()
        b_k = b1_k * (1.0 / b1_k_norm)
        if np.abs(maxeig - maxeig_old) < tol * maxeig:
            break
        maxeig_old = maxeig
    return maxeig, b_k, iiter + 1


def _sparse_setup(eps, alpha, niter, threshkind, decay):
    threshf = _THRESH[threshkind]
    thresh = eps * alpha * 0.5
    if decay is None:
        decay = np.ones(niter)
    return threshf, thresh, decay


def sim_ista(Op, y: SimArray, x0: SimArray, niter: int, eps: float,
             alpha: float, threshkind: str = "soft",
             decay: Optional[np.ndarray] = None, tol: float = 1e-10
             ) -> Tuple[SimArray, int, np.ndarray]:
    """ref cls_sparsity.py ISTA.setup/step/run (:243-416)."""
    threshf, thresh, decay = _sparse_setup(eps, alpha, niter, threshkind,
                                           decay)
    x = x0.copy()
    cost: List[float] = []
    iiter = 0
    xupdate = np.inf
    while iiter < niter and xupdate > tol:
        xold = x.copy()
        res = y - Op.matvec(x)
        grad = alpha * Op.rmatvec(res)
        x = x + grad
        x.locals = [threshf(a, decay[iiter] * thresh) for a in x.locals]
        xupdate = float((x - xold).norm())
        costdata = 0.5 * float(res.norm()) ** 2
        costreg = eps * float(x.norm(1))
        cost.append(costdata + costreg)
        iiter += 1
    return x, iiter, np.asarray(cost)


def sim_fista(Op, y: SimArray, x0: SimArray, niter: int, eps: float,
              alpha: float, threshkind: str = "soft",
              decay: Optional[np.ndarray] = None, tol: float = 1e-10
              ) -> Tuple[SimArray, int, np.ndarray]:
    """ref cls_sparsity.py FISTA.step/run (:581-718): gradient at the
    auxiliary z, Nesterov t-sequence, cost recomputed at the new x."""
    threshf, thresh, decay = _sparse_setup(eps, alpha, niter, threshkind,
                                           decay)
    x = x0.copy()
    z = x.copy()
    t = 1.0
    cost: List[float] = []
    iiter = 0
    xupdate = np.inf
    while iiter < niter and xupdate > tol:
        xold = x.copy()
        res = y - Op.matvec(z)
        grad = alpha * Op.rmatvec(res)
        x = z + grad
        x.locals = [threshf(a, decay[iiter] * thresh) for a in x.locals]
        told = t
        t = (1.0 + sqrt(1.0 + 4.0 * t ** 2)) / 2.0
        z = x + ((told - 1.0) / t) * (x - xold)
        xupdate = float((x - xold).norm())
        costdata = 0.5 * float((y - Op.matvec(x)).norm()) ** 2
        costreg = eps * float(x.norm(1))
        cost.append(costdata + costreg)
        iiter += 1
    return x, iiter, np.asarray(cost)
