"""StackedDistributedArray — a list of DistributedArrays with the same
math surface, plus the stacked operator base.

Drop-in for /root/reference/pylops_mpi/DistributedArray.py:1041-1320 and
the operator side of StackedLinearOperator.py (thin Python over the member
arrays — no new kernels).  Adds the fused iaxpy_/xpby_
used by our CG/CGLS so stacked systems drive the same solver loop.
"""
from typing import List, Optional

import numpy as np

from .comm import PamComm, get_default_comm
from .distributedarray import DistributedArray


class StackedDistributedArray:
    """ref DistributedArray.py:1041-1320."""

    def __init__(self, distarrays: List[DistributedArray],
                 base_comm: Optional[PamComm] = None):
        self.distarrays = distarrays
        self.narrays = len(distarrays)
        self.base_comm = base_comm if base_comm is not None \
            else get_default_comm()
        self.rank = self.base_comm.rank
        self.size = self.base_comm.size
        # ref :1077-1082 (sum of global shapes, element-wise)
        self._global_shape = distarrays[0].global_shape
        for iarr in range(1, self.narrays):
            self._global_shape = tuple(
                g1 + g2 for g1, g2 in zip(self._global_shape,
                                          distarrays[iarr].global_shape))

    @property
    def global_shape(self):
        return self._global_shape

    def __getitem__(self, index):
        return self.distarrays[index]

    def __setitem__(self, index, value):
        # ref :1091-1102
        target = self.distarrays[index]
        if isinstance(value, DistributedArray):
            target[:] = value[:]
        else:
            target[:] = value

    def asarray(self):
        # ref :1131-1144
        import torch
        return torch.hstack([d.asarray().reshape(-1)
                             for d in self.distarrays])

    def _check_stacked_size(self, other):
        # ref :1146-1156
        if self.narrays != other.narrays:
            raise ValueError("Stacked arrays must be composed the same "
                             "number of of distributed arrays")
        for iarr in range(self.narrays):
            if self.distarrays[iarr].global_shape != \
                    other[iarr].global_shape:
                raise ValueError(
                    f"Stacked arrays {iarr} have different global shape:"
                    f"{self.distarrays[iarr].global_shape} / "
                    f"{other[iarr].global_shape}")

    # ------------------------------------------------------------ math
    def __neg__(self):
        return StackedDistributedArray([-d for d in self.distarrays],
                                       self.base_comm)

    def add(self, other):
        self._check_stacked_size(other)
        return StackedDistributedArray(
            [a + b for a, b in zip(self.distarrays, other.distarrays)],
            self.base_comm)

    def iadd(self, other):
        self._check_stacked_size(other)
        for a, b in zip(self.distarrays, other.distarrays):
            a.iadd(b)
        return self

    def multiply(self, x):
        if isinstance(x, StackedDistributedArray):
            self._check_stacked_size(x)
            return StackedDistributedArray(
                [a * b for a, b in zip(self.distarrays, x.distarrays)],
                self.base_comm)
        return StackedDistributedArray([a * x for a in self.distarrays],
                                       self.base_comm)

    def __add__(self, x):
        return self.add(x)

    def __iadd__(self, x):
        return self.iadd(x)

    def __sub__(self, x):
        return self.__add__(-x)

    def __isub__(self, x):
        return self.iaxpy_(-1.0, x)

    def __mul__(self, x):
        return self.multiply(x)

    __rmul__ = __mul__

    def iaxpy_(self, alpha: float, x: "StackedDistributedArray"):
        for a, b in zip(self.distarrays, x.distarrays):
            a.iaxpy_(alpha, b)
        return self

    def xpby_(self, x: "StackedDistributedArray", beta: float):
        for a, b in zip(self.distarrays, x.distarrays):
            a.xpby_(b, beta)
        return self

    def dot(self, other, vdot: bool = False):
        # ref :1231-1255
        self._check_stacked_size(other)
        dotprod = 0.0
        for iarr in range(self.narrays):
            dotprod += self[iarr].dot(other[iarr], vdot=vdot)
        return dotprod

    def norm(self, ord: Optional[float] = None):
        # ref :1257-1281
        norms = np.array([d.norm(ord) for d in self.distarrays])
        ord = 2 if ord is None else ord
        if ord in ("fro", "nuc"):
            raise ValueError(f"norm-{ord} not possible for vectors")
        if ord == 0:
            return np.float64(np.sum(norms))
        if ord == np.inf:
            return np.float64(np.max(norms))
        if ord == -np.inf:
            return np.float64(np.min(norms))
        return np.float64(np.power(np.sum(np.power(norms, ord)), 1.0 / ord))

    def conj(self):
        return StackedDistributedArray([d.conj() for d in self.distarrays],
                                       self.base_comm)

    def copy(self):
        return StackedDistributedArray([d.copy() for d in self.distarrays],
                                       self.base_comm)

    def zeros_like(self):
        return StackedDistributedArray(
            [d.zeros_like() for d in self.distarrays], self.base_comm)

    def empty_like(self):
        return StackedDistributedArray(
            [d.empty_like() for d in self.distarrays], self.base_comm)

    def __repr__(self):
        repr_dist = "\n".join(d.__repr__() for d in self.distarrays)
        return (f"<StackedDistributedArray with {self.narrays} distributed "
                f"arrays: \n" + repr_dist)


class MPIStackedLinearOperator:
    """Minimal stacked-operator base (ref StackedLinearOperator.py): shape
    protocol + matvec/rmatvec on StackedDistributedArray + adjoint."""

    def __init__(self, shape=None, dims=None, dimsd=None, dtype=None,
                 base_comm: Optional[PamComm] = None):
        if shape is not None:
            self.shape = tuple(shape)
        else:
            self.shape = (int(np.prod(dimsd)), int(np.prod(dims)))
        self.dims = dims if dims is not None else (self.shape[1],)
        self.dimsd = dimsd if dimsd is not None else (self.shape[0],)
        self.dtype = dtype
        self.base_comm = base_comm if base_comm is not None \
            else get_default_comm()
        self.rank = self.base_comm.rank
        self.size = self.base_comm.size

    def matvec(self, x):
        return self._matvec(x)

    def rmatvec(self, x):
        return self._rmatvec(x)

    def adjoint(self):
        return _StackedAdjoint(self)

    H = property(adjoint)

    def transpose(self):
        # ref StackedLinearOperator.py:406-426
        return _StackedTransposed(self)

    T = property(transpose)

    def conj(self):
        # ref :543-568
        return _StackedConj(self)

    def __mul__(self, x):
        # ref :223-262 dot dispatch
        if isinstance(x, MPIStackedLinearOperator):
            return _StackedProduct(self, x)
        if np.isscalar(x):
            return _StackedScaled(self, x)
        return self.matvec(x)

    def __rmul__(self, x):
        if np.isscalar(x):
            return _StackedScaled(self, x)
        return NotImplemented

    def __matmul__(self, x):
        return self.__mul__(x)

    def __pow__(self, p):
        # ref :515-540
        return _StackedPower(self, p)

    def __add__(self, x):
        # ref :486-512
        return _StackedSum(self, x)

    def __neg__(self):
        return _StackedScaled(self, -1)

    def __sub__(self, x):
        return self.__add__(-x)


class _StackedAdjoint(MPIStackedLinearOperator):
    def __init__(self, A):
        self.A = A
        super().__init__(shape=(A.shape[1], A.shape[0]), dtype=A.dtype,
                         base_comm=A.base_comm)

    def _matvec(self, x):
        return self.A.rmatvec(x)

    def _rmatvec(self, x):
        return self.A.matvec(x)


class _StackedScaled(MPIStackedLinearOperator):
    def __init__(self, A, alpha):
        # ref StackedLinearOperator.py:456-484
        if not np.isscalar(alpha):
            raise ValueError('scalar expected as alpha')
        self.A = A
        self.alpha = alpha
        super().__init__(shape=A.shape, dtype=A.dtype, base_comm=A.base_comm)

    def _matvec(self, x):
        return self.A.matvec(x) * self.alpha

    def _rmatvec(self, x):
        return self.A.rmatvec(x) * np.conj(self.alpha)


class _StackedTransposed(MPIStackedLinearOperator):
    """ref StackedLinearOperator.py:406-426 (Aᵀ = conj ∘ Aᴴ ∘ conj)."""

    def __init__(self, A):
        self.A = A
        super().__init__(shape=(A.shape[1], A.shape[0]), dtype=A.dtype,
                         base_comm=A.base_comm)

    def _matvec(self, x):
        return self.A.rmatvec(x.conj()).conj()

    def _rmatvec(self, x):
        return self.A.matvec(x.conj()).conj()


class _StackedConj(MPIStackedLinearOperator):
    """ref StackedLinearOperator.py:543-568."""

    def __init__(self, A):
        if not isinstance(A, MPIStackedLinearOperator):
            raise TypeError('A must be a MPIStackedLinearOperator')
        self.A = A
        super().__init__(shape=A.shape, dtype=A.dtype, base_comm=A.base_comm)

    def _matvec(self, x):
        return self.A.matvec(x.conj()).conj()

    def _rmatvec(self, x):
        return self.A.rmatvec(x.conj()).conj()


class _StackedProduct(MPIStackedLinearOperator):
    """ref StackedLinearOperator.py:428-454."""

    def __init__(self, A, B):
        if not isinstance(A, MPIStackedLinearOperator) \
                or not isinstance(B, MPIStackedLinearOperator):
            raise ValueError(
                'both operands have to be a MPIStackedLinearOperator')
        if A.shape[1] != B.shape[0]:
            raise ValueError('cannot multiply %r and %r: shape mismatch'
                             % (A, B))
        self.args = (A, B)
        super().__init__(shape=(A.shape[0], B.shape[1]), dtype=A.dtype,
                         base_comm=A.base_comm)

    def _matvec(self, x):
        return self.args[0].matvec(self.args[1].matvec(x))

    def _rmatvec(self, x):
        return self.args[1].rmatvec(self.args[0].rmatvec(x))


class _StackedSum(MPIStackedLinearOperator):
    """ref StackedLinearOperator.py:486-512."""

    def __init__(self, A, B):
        if not isinstance(A, MPIStackedLinearOperator) \
                or not isinstance(B, MPIStackedLinearOperator):
            raise ValueError(
                'both operands have to be a MPIStackedLinearOperator')
        if A.shape != B.shape:
            raise ValueError("cannot add %r and %r: shape mismatch"
                             % (A, B))
        self.args = (A, B)
        super().__init__(shape=A.shape, dtype=A.dtype, base_comm=A.base_comm)

    def _matvec(self, x):
        return self.args[0].matvec(x) + self.args[1].matvec(x)

    def _rmatvec(self, x):
        return self.args[0].rmatvec(x) + self.args[1].rmatvec(x)


class _StackedPower(MPIStackedLinearOperator):
    """ref StackedLinearOperator.py:515-540."""

    def __init__(self, A, p):
        if A.shape[0] != A.shape[1]:
            raise ValueError("square MPIStackedLinearOperator expected, "
                             "got %r" % A)
        if not isinstance(p, (int, np.integer)) or p < 0:
            raise ValueError("non-negative integer expected as p")
        self.args = (A, int(p))
        super().__init__(shape=A.shape, dtype=A.dtype, base_comm=A.base_comm)

    def _power(self, fun, x):
        res = x.copy()
        for _ in range(self.args[1]):
            res = fun(res)
        return res

    def _matvec(self, x):
        return self._power(self.args[0].matvec, x)

    def _rmatvec(self, x):
        return self._power(self.args[0].rmatvec, x)
