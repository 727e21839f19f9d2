"""Independent serial stencil restatements — TEST INFRASTRUCTURE ONLY.

Closed-form per-index restatements of the serial operators the reference's
tests compare against (pylops FirstDerivative / SecondDerivative as used by
/root/reference/tests/test_derivative.py:220-229).  These are written from
the stencil formulas (ref basicoperators/FirstDerivative.py:61-80 notes and
the P=1 limit of the distributed code), NOT by calling the rank-simulated
code — so they pin oracle/stencils.py independently.

All stencils act along axis 0 of an array of shape ``dims`` and broadcast
over the remaining axes.
"""
import numpy as np


def serial_fd1_matvec(x: np.ndarray, sampling: float = 1.0,
                      kind: str = "centered", edge: bool = False,
                      order: int = 3) -> np.ndarray:
    n = x.shape[0]
    y = np.zeros_like(x)
    if kind == "forward":
        y[:-1] = x[1:] - x[:-1]
    elif kind == "backward":
        y[1:] = x[1:] - x[:-1]
    elif kind == "centered" and order == 3:
        y[1:-1] = 0.5 * (x[2:] - x[:-2])
        if edge:
            y[0] = x[1] - x[0]
            y[-1] = x[-1] - x[-2]
    elif kind == "centered" and order == 5:
        y[2:-2] = (x[:-4] / 12.0 - 2 * x[1:-3] / 3.0
                   + 2 * x[3:-1] / 3.0 - x[4:] / 12.0)
        if edge:
            y[0] = x[1] - x[0]
            y[1] = 0.5 * (x[2] - x[0])
            y[-1] = x[-1] - x[-2]
            y[-2] = 0.5 * (x[-1] - x[-3])
    else:
        raise NotImplementedError(kind)
    assert n == y.shape[0]
    return y / sampling


def serial_fd2_matvec(x: np.ndarray, sampling: float = 1.0,
                      kind: str = "centered", edge: bool = False) -> np.ndarray:
    y = np.zeros_like(x)
    if kind == "forward":
        y[:-2] = x[2:] - 2 * x[1:-1] + x[:-2]
    elif kind == "backward":
        y[2:] = x[2:] - 2 * x[1:-1] + x[:-2]
    elif kind == "centered":
        y[1:-1] = x[2:] - 2 * x[1:-1] + x[:-2]
        if edge:
            y[0] = x[0] - 2 * x[1] + x[2]
            y[-1] = x[-3] - 2 * x[-2] + x[-1]
    else:
        raise NotImplementedError(kind)
    return y / sampling ** 2


def dense_matrix_from_matvec(matvec, n: int, dtype=np.float64) -> np.ndarray:
    """Explicit dense matrix of a linear map on R^n (column by column).

    Used to pin every adjoint: rmatvec must equal ``A.conj().T @ x``.
    """
    A = np.zeros((n, n), dtype=dtype)
    for j in range(n):
        e = np.zeros(n, dtype=dtype)
        e[j] = 1.0
        A[:, j] = matvec(e)
    return A
