"""Rank-simulated CG / CGLS recurrences — TEST INFRASTRUCTURE ONLY.

Restates /root/reference/pylops_mpi/optimization/cls_basic.py:
  CG.setup/step   -> :86-141
  CGLS.setup/step -> :308-404
on oracle SimArray vectors (P sequulated ranks, one process).
Returns (x, cost_history) so CGLS iterate traces can be pinned.
"""
from typing import Tuple

import numpy as np

from .ranksim import SimArray


def sim_cg(Op, y: SimArray, x0: SimArray, niter: int,
           tol: float = 1e-4) -> Tuple[SimArray, list]:
    """ref cls_basic.py:86-141."""
    x = x0.copy()
    r = y - Op.matvec(x)
    c = r.copy()
    kold = float(np.abs(r.dot(r.conj())))
    cost = [float(np.sqrt(kold))]
    iiter = 0
    while iiter < niter and kold > tol:
        Opc = Op.matvec(c)
        cOpc = np.abs(c.dot(Opc.conj()))
        a = float(kold / cOpc)
        x += a * c
        r -= a * Opc
        k = float(np.abs(r.dot(r.conj())))
        b = float(k / kold)
        c = r + b * c
        kold = k
        iiter += 1
        cost.append(float(np.sqrt(kold)))
    return x, cost


def sim_cgls(Op, y: SimArray, x0: SimArray, niter: int, damp: float = 0.0,
             tol: float = 1e-4) -> Tuple[SimArray, list]:
    """ref cls_basic.py:308-404.

    ``cost`` is the residual-norm history (``self.cost``, ref :357,400),
    the trace the north-star pins to 1e-6.
    """
    damp2 = damp ** 2
    x = x0.copy()
    s = y - Op.matvec(x)
    damped_x = x * damp
    r = Op.rmatvec(s) - damped_x
    c = r.copy()
    q = Op.matvec(c)
    kold = float(np.abs(r.dot(r.conj())))
    cost = [float(s.norm())]
    cost1 = [float(np.sqrt(cost[0] ** 2 + damp * np.abs(x.dot(x.conj()))))]
    iiter = 0
    while iiter < niter and kold > tol:
        a = float(np.abs(kold / (q.dot(q.conj()) + damp2 * c.dot(c.conj()))))
        x += a * c
        s -= a * q
        damped_x = damp2 * x
        r = Op.rmatvec(s) - damped_x
        k = float(np.abs(r.dot(r.conj())))
        b = float(k / kold)
        c = r + b * c
        q = Op.matvec(c)
        kold = k
        iiter += 1
        cost.append(float(s.norm()))
        cost1.append(float(np.sqrt(cost[iiter] ** 2
                                   + damp2 * np.abs(x.dot(x.conj())))))
    return x, cost
