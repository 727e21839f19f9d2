"""Distributed adjoint dot-test — ref /root/reference/pylops_mpi/utils/dottest.py:11-107."""
from typing import Optional

import numpy as np

from .distributedarray import DistributedArray


def dottest(Op, u: DistributedArray, v: DistributedArray,
            nr: Optional[int] = None, nc: Optional[int] = None,
            rtol: float = 1e-6, atol: float = 1e-21,
            raiseerror: bool = True, verb: bool = False) -> bool:
    """(Op u)^H v == u^H (Op^H v) within tolerance (ref :76-107)."""
    if nr is None:
        nr = Op.shape[0]
    if nc is None:
        nc = Op.shape[1]
    if (nr, nc) != Op.shape:
        raise AssertionError("Provided nr and nc do not match operator shape")
    y = Op.matvec(u)
    x = Op.rmatvec(v)
    yy = y.dot(v, vdot=True)
    xx = u.dot(x, vdot=True)
    passed = bool(np.isclose(xx, yy, rtol, atol))
    if (not passed and raiseerror) or verb:
        status = "passed" if passed else "failed"
        msg = f"Dot test {status}, v^H(Opu)={yy} - u^H(Op^Hv)={xx}"
        if not passed and raiseerror:
            raise AssertionError(msg)
        print(msg)
    return passed
