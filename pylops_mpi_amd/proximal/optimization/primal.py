"""Distributed proximal solvers — restate ref
proximal/optimization/primal.py:22-347 (ProximalGradient, ADMML2) on the
RCCL/HIP stack.  The `_x0z0_init` helper the reference imports from
pyproximal is restated inline (x required; z defaults to A @ x0).

Note: the reference accepts `niterback` but its released code never
branches into backtracking (`tau` is used directly at primal.py:146-149),
so `tau` is required here too and `niterback` is kept for signature
parity only.
"""
import sys
import time
from math import sqrt
from typing import Any, Callable, Optional

import numpy as np

from ...distributedarray import DistributedArray
from ...stacked import StackedDistributedArray
from ...vstack import MPIStackedVStack
from ...solvers import cgls


def _x0z0_init(x0, z0, A, Opname: str = "A"):
    # pyproximal.optimization.primal._x0z0_init, restated
    if x0 is None:
        raise ValueError("x0 must be provided")
    x = x0.copy()
    z = z0.copy() if z0 is not None else A @ x
    return x, z


def ProximalGradient(
    proxf,
    proxg,
    x0,
    epsg=1.0,
    tau: Optional[float] = None,
    eta: float = 1.0,
    niter: int = 10,
    niterback: int = 100,
    acceleration: Optional[str] = None,
    tol: Optional[float] = None,
    callback: Optional[Callable] = None,
    show: bool = False,
):
    """(Accelerated) proximal gradient, ref primal.py:22-201."""
    rank = x0.rank
    epsg = np.asarray(epsg, dtype=float)
    if epsg.size == 1:
        epsg = epsg * np.ones(niter)
    if acceleration not in [None, "None", "vandenberghe", "fista"]:
        raise NotImplementedError(
            "Acceleration should be None, vandenberghe or fista")
    if tau is None:
        raise NotImplementedError(
            "backtracking (tau=None) is not implemented by the reference's "
            "released solver; pass an explicit tau")
    if show and rank == 0:
        tstart = time.time()
        print("Accelerated Proximal Gradient\n"
              "---------------------------------------------------------\n"
              f"Proximal operator (f): {proxf}\n"
              f"Proximal operator (g): {proxg}\n"
              f"tau = {tau}\tniter = {niter}\ttol = {tol}\n"
              f"acceleration = {acceleration}\n")
        sys.stdout.flush()

    # ref :133-139
    t = 1.0
    x = x0.copy()
    y = x.copy()
    pfg = np.inf
    tolbreak = False
    for iiter in range(niter):
        xold = x.copy()
        # proximal step (ref :144-150)
        if eta == 1.0:
            x = proxg.prox(y - tau * proxf.grad(y), epsg[iiter] * tau)
        else:
            x = x + eta * (
                proxg.prox(x - tau * proxf.grad(x), epsg[iiter] * tau) - x)
        # update y (ref :152-161)
        if acceleration == "vandenberghe":
            omega = iiter / (iiter + 3)
        elif acceleration == "fista":
            told = t
            t = (1.0 + np.sqrt(1.0 + 4.0 * t ** 2)) / 2.0
            omega = (told - 1.0) / t
        else:
            omega = 0
        y = x + omega * (x - xold)
        if callback is not None:
            callback(x)
        # tolerance on the objective (ref :166-172)
        if tol is not None:
            pfgold = pfg
            pf, pg = proxf(x), proxg(x)
            pfg = pf + np.sum(epsg[iiter] * pg)
            if np.abs(1.0 - pfg / pfgold) < tol:
                tolbreak = True
        if show and rank == 0 and (iiter < 10 or niter - iiter < 10
                                   or (niter // 10 > 0
                                       and iiter % (niter // 10) == 0)):
            pf, pg = proxf(x), proxg(x)
            print(f"{iiter + 1:6g}  {pf:10.3e}  {pg:10.3e}  "
                  f"{pf + np.sum(epsg[iiter] * pg):10.3e}")
            sys.stdout.flush()
        if tolbreak:
            break
    if show and rank == 0:
        print(f"\nTotal time (s) = {time.time() - tstart:.2f}\n")
        sys.stdout.flush()
    return x


def ADMML2(
    proxg,
    Op,
    b,
    A,
    x0,
    tau: float,
    niter: int = 10,
    z0=None,
    gfirst: bool = False,
    callback: Optional[Callable] = None,
    show: bool = False,
    kwargs_solver: Optional[dict] = None,
):
    """ADMM with an L2 misfit term, ref primal.py:209-347."""
    kwargs_solver = {} if kwargs_solver is None else dict(kwargs_solver)
    rank = x0.rank
    x, z = _x0z0_init(x0, z0, A, Opname="A")
    u = z.zeros_like()
    if show and rank == 0:
        print("ADMM\n"
              "---------------------------------------------------------\n"
              f"Proximal operator (g): {proxg}\n"
              f"tau = {tau:10e}\tniter = {niter}\n")
        sys.stdout.flush()
    sqrttau = 1.0 / sqrt(tau)
    for iiter in range(niter):
        if gfirst:  # ref :308-316
            Ax = A @ x
            z = proxg.prox(Ax + u, tau)
            Opreg = MPIStackedVStack([Op, sqrttau * A])
            breg = StackedDistributedArray([b, sqrttau * (z - u)])
            x = cgls(Opreg, breg, x, **kwargs_solver)[0]
        else:  # ref :317-323
            Opreg = MPIStackedVStack([Op, sqrttau * A])
            breg = StackedDistributedArray([b, sqrttau * (z - u)])
            x = cgls(Opreg, breg, x, **kwargs_solver)[0]
            Ax = A @ x
            z = proxg.prox(Ax + u, tau)
        u = u + Ax - z
        if callback is not None:
            callback(x)
        if show and rank == 0 and (iiter < 10 or niter - iiter < 10
                                   or (niter // 10 > 0
                                       and iiter % (niter // 10) == 0)):
            pf = 0.5 * float((Op @ x - b).norm()) ** 2
            pg = proxg(Ax)
            print(f"{iiter + 1:6g}  {pf:10.3e}  {pg:10.3e}  "
                  f"{pf + pg:10.3e}")
            sys.stdout.flush()
    if show and rank == 0:
        print("---------------------------------------------------------\n")
        sys.stdout.flush()
    return x, z
