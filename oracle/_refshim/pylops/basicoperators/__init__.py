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
    """pylops.Identity (N == M case; MDC's freq mask uses the
    rectangular form — not exercised through this stub)."""

    def __init__(self, N, M=None, dtype="float64"):
        M = N if M is None else M
        super().__init__(dtype=np.dtype(dtype), shape=(N, M))

    def _matvec(self, x):
        return x.copy()

    def _rmatvec(self, x):
        return x.copy()


class _ImportOnly(LinearOperator):
    _name = "stub"

    def __init__(self, *a, **k):
        raise NotImplementedError(
            f"pylops stub: serial {self._name} is not implemented — the "
            "ref-parity suite does not construct it")


class FirstDerivative(_ImportOnly):
    _name = "FirstDerivative"


class SecondDerivative(_ImportOnly):
    _name = "SecondDerivative"
