"""Serial LinearOperator base: the subset of pylops.LinearOperator the
reference package touches (ref LinearOperator.py:10 subclassing and
BlockDiag.py:129,141 ``oper.matvec``/``oper.rmatvec`` on 1-D arrays)."""
import numpy as np


class LinearOperator:
    def __init__(self, Op=None, dtype=None, shape=None, dims=None,
                 dimsd=None):
        if Op is not None:
            self.Op = Op
            shape = Op.shape if shape is None else shape
            dtype = Op.dtype if dtype is None else dtype
        self.shape = tuple(shape) if shape is not None else None
        self.dtype = np.dtype(dtype) if dtype is not None else None
        self.dims = dims if dims is not None else (
            (self.shape[1],) if self.shape else None)
        self.dimsd = dimsd if dimsd is not None else (
            (self.shape[0],) if self.shape else None)

    def matvec(self, x):
        if len(x) != self.shape[1]:
            raise ValueError("dimension mismatch")
        return self._matvec(np.asarray(x))

    def rmatvec(self, x):
        if len(x) != self.shape[0]:
            raise ValueError("dimension mismatch")
        return self._rmatvec(np.asarray(x))

    def _matvec(self, x):
        return self.Op._matvec(x)

    def _rmatvec(self, x):
        return self.Op._rmatvec(x)

    @property
    def H(self):
        return _Adjoint(self)

    def dot(self, x):
        return self.matvec(x)

    def __mul__(self, x):
        if np.isscalar(x):
            return _Scaled(self, x)
        return self.matvec(x)

    __matmul__ = __mul__


class _Adjoint(LinearOperator):
    def __init__(self, op):
        super().__init__(dtype=op.dtype, shape=(op.shape[1], op.shape[0]))
        self._op = op

    def _matvec(self, x):
        return self._op._rmatvec(x)

    def _rmatvec(self, x):
        return self._op._matvec(x)


class _Scaled(LinearOperator):
    def __init__(self, op, alpha):
        super().__init__(dtype=op.dtype, shape=op.shape)
        self._op, self._alpha = op, alpha

    def _matvec(self, x):
        return self._alpha * self._op._matvec(x)

    def _rmatvec(self, x):
        return np.conj(self._alpha) * self._op._rmatvec(x)


def aslinearoperator(Op):
    return Op if isinstance(Op, LinearOperator) else LinearOperator(Op)
