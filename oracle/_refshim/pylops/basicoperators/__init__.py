"""Serial operators the reference imports.  MatrixMult/Identity are
faithful numpy restatements of pylops' published semantics (used live
in the ref-parity suites, e.g. as MPIBlockDiag local blocks per
examples/plot_cgls.py:30-33); FirstDerivative/SecondDerivative are
import-time-only here (the distributed derivative parity is pinned on
the reference's own MPIFirstDerivative instead) and raise on use."""
import numpy as np

from ..linearoperator import LinearOperator


class MatrixMult(LinearOperator):
    """y = A @ x for a dense serial block (pylops.MatrixMult, dense
    otherdims=None case)."""

    def __init__(self, A, dtype="float64"):
        self.A = np.asarray(A)
        super().__init__(dtype=np.dtype(dtype), shape=self.A.shape)

    def _matvec(self, x):
        return self.A @ x

    def _rmatvec(self, x):
        return self.A.conj().T @ x


class Identity(LinearOperator):
    """pylops.Identity including the rectangular form (N < M truncates,
    the adjoint zero-pads) — MDC's frequency mask (ref MDC.py:61-64)."""

    def __init__(self, N, M=None, inplace=True, dtype="float64"):
        M = N if M is None else M
        super().__init__(dtype=np.dtype(dtype), shape=(N, M))

    def _matvec(self, x):
        N, M = self.shape
        return x[:N].copy() if N <= M else np.concatenate(
            [x, np.zeros(N - M, dtype=x.dtype)])

    def _rmatvec(self, x):
        N, M = self.shape
        return x[:M].copy() if M <= N else np.concatenate(
            [x, np.zeros(M - N, dtype=x.dtype)])


class _SerialDerivative(LinearOperator):
    """Serial pylops FirstDerivative/SecondDerivative stand-in for the
    reference's Gradient/Laplacian local blocks (ref Gradient.py:108-116,
    Laplacian.py:97-126).  pylops itself is absent, so the
    per-axis stencil is supplied by the repo oracle's rank-1 simulation
    of the SAME published formulas (oracle/stencils.py — already pinned
    against the reference's distributed axis-0 operators at P=1..8);
    what the ref-parity suite then pins is the DISTRIBUTED composition:
    StackedVStack / BlockDiag mechanics and the scaled/summed composite
    algebra."""

    def __init__(self, dims, axis=0, sampling=1.0, kind="centered",
                 edge=False, dtype="float64"):
        dims = (dims,) if isinstance(dims, (int, np.integer)) \
            else tuple(int(d) for d in dims)
        n = int(np.prod(dims))
        super().__init__(dtype=np.dtype(dtype), shape=(n, n))
        self.dims = dims        # after super(): the base derives a flat
        self.axis = int(axis) % len(dims)
        moved = (dims[self.axis],) + tuple(
            d for i, d in enumerate(dims) if i != self.axis)
        self._sim = self._make_sim(moved, sampling, kind, edge)

    def _roll(self, x, fwd):
        from oracle.ranksim import to_dist
        arr = np.moveaxis(np.asarray(x).reshape(self.dims), self.axis, 0)
        shp = arr.shape
        d = to_dist(arr.ravel(), 1)
        res = (self._sim.matvec(d) if fwd else self._sim.rmatvec(d))
        out = np.moveaxis(res.asarray().reshape(shp), 0, self.axis)
        return np.ascontiguousarray(out).ravel()

    def _matvec(self, x):
        return self._roll(x, True)

    def _rmatvec(self, x):
        return self._roll(x, False)


class FirstDerivative(_SerialDerivative):
    def _make_sim(self, moved, sampling, kind, edge):
        from oracle.stencils import SimFirstDerivative
        return SimFirstDerivative(moved, sampling, kind, edge, order=3)


class SecondDerivative(_SerialDerivative):
    def _make_sim(self, moved, sampling, kind, edge):
        from oracle.stencils import SimSecondDerivative
        return SimSecondDerivative(moved, sampling, kind, edge)
