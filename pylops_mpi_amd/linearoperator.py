"""MPILinearOperator — the drop-in operator boundary.

Mirrors /root/reference/pylops_mpi/LinearOperator.py:16-602: attribute
protocol (shape/dims/dimsd/dtype/base_comm/rank/size), matvec/rmatvec with
the reference's "dimension mismatch" ValueError, operator algebra
(dot/H/T/conj/*,+,**,-) and the composite operators.  Solvers only ever
touch this surface plus DistributedArray math, so they drive the HIP path
unchanged.
"""
from typing import Optional

import numpy as np

from .comm import PamComm, get_default_comm
from .distributedarray import DistributedArray


class MPILinearOperator:
    """ref LinearOperator.py:16-405.  ``Op=`` wraps a serial
    :class:`~pylops_mpi_amd.localops.LocalOperator` applied identically on
    every rank (the reference's pylops-operator wrap, ref :22-27)."""

    def __init__(self, Op=None, shape=None, dims=None, dimsd=None,
                 dtype=None, base_comm: Optional[PamComm] = None):
        if Op is not None:
            # wrap a serial (local) operator, ref :60-66: it is applied
            # identically on every rank's local array (meant for
            # BROADCAST-partitioned arrays, ref :22-27)
            self.Op = Op
            dtype = Op.dtype if dtype is None else dtype
            shape = Op.shape if shape is None else shape
            dims = getattr(Op, "dims", (Op.shape[1],)) if dims is None \
                else dims
            dimsd = getattr(Op, "dimsd", (Op.shape[0],)) if dimsd is None \
                else dimsd
        elif not hasattr(self, "Op"):
            # don't clobber an Op a subclass set before calling super()
            # (e.g. MPILaplacian's composed operator)
            self.Op = None
        if shape is not None:
            self.shape = shape
        if dims is not None:
            self.dims = dims
        if dimsd is not None:
            self.dimsd = dimsd
        if dtype is not None:
            self.dtype = dtype
        self.base_comm = base_comm if base_comm is not None \
            else get_default_comm()
        self.size = self.base_comm.size
        self.rank = self.base_comm.rank

    # ------------------------------------------------ shape/dims protocol
    # (ref :80-168)
    @property
    def shape(self):
        _shape = getattr(self, "_shape", None)
        if _shape is None:
            dims = getattr(self, "_dims", None)
            dimsd = getattr(self, "_dimsd", None)
            if dims is None or dimsd is None:
                raise AttributeError(
                    f"'{self.__class__.__name__}' object has no attribute "
                    "'shape' nor both fallback attributes ('dims', 'dimsd')")
            _shape = (int(np.prod(dimsd)), int(np.prod(dims)))
            self._shape = _shape
        return _shape

    @shape.setter
    def shape(self, new_shape):
        new_shape = tuple(new_shape)
        if len(new_shape) != 2:
            raise ValueError(
                f"Invalid shape; must be 2-d tuple of integers, "
                f"got {new_shape}")
        self._shape = new_shape

    @property
    def dims(self):
        _dims = getattr(self, "_dims", None)
        if _dims is None:
            _dims = (self.shape[1],)
        return _dims

    @dims.setter
    def dims(self, new_dims):
        self._dims = tuple(new_dims)

    @property
    def dimsd(self):
        _dimsd = getattr(self, "_dimsd", None)
        if _dimsd is None:
            _dimsd = (self.shape[0],)
        return _dimsd

    @dimsd.setter
    def dimsd(self, new_dimsd):
        self._dimsd = tuple(new_dimsd)

    # ----------------------------------------------------- matvec/rmatvec
    def matvec(self, x: DistributedArray) -> DistributedArray:
        # ref :170-192
        M, N = self.shape
        if x.global_shape != (N,):
            raise ValueError("dimension mismatch")
        return self._matvec(x)

    def rmatvec(self, x: DistributedArray) -> DistributedArray:
        # ref :206-230
        M, N = self.shape
        if x.global_shape != (M,):
            raise ValueError("dimension mismatch")
        return self._rmatvec(x)

    def _matvec(self, x):
        # ref :194-204 — serial-op wrap applies locally on every rank
        if getattr(self, "Op", None) is not None:
            y = DistributedArray(self.shape[0], x.base_comm, x.partition,
                                 x.axis, dtype=self.dtype)
            y[:] = self.Op.matvec(x.local_array)
            return y
        raise NotImplementedError

    def _rmatvec(self, x):
        # ref :232-242
        if getattr(self, "Op", None) is not None:
            y = DistributedArray(self.shape[1], x.base_comm, x.partition,
                                 x.axis, dtype=self.dtype)
            y[:] = self.Op.rmatvec(x.local_array)
            return y
        raise NotImplementedError

    # ------------------------------------------------------------ algebra
    # (ref :244-383)
    def dot(self, x):
        if isinstance(x, MPILinearOperator):
            Op = _ProductLinearOperator(self, x)
            self._copy_attributes(Op, exclude=["dims"])
            Op.dims = x.dims
            return Op
        elif np.isscalar(x):
            Op = _ScaledLinearOperator(self, x)
            self._copy_attributes(Op)
            return Op
        else:
            if x is None or x.ndim == 1:
                return self.matvec(x)
            raise ValueError(
                "expected 1-d DistributedArray, got %r" % (x.global_shape,))

    def adjoint(self):
        return self._adjoint()

    H = property(adjoint)

    def transpose(self):
        return self._transpose()

    T = property(transpose)

    def conj(self):
        return _ConjLinearOperator(self)

    def __mul__(self, x):
        return self.dot(x)

    def __rmul__(self, x):
        if np.isscalar(x):
            Op = _ScaledLinearOperator(self, x)
            self._copy_attributes(Op)
            return Op
        return NotImplemented

    def __matmul__(self, x):
        if np.isscalar(x):
            raise ValueError("Scalar not allowed, use * instead")
        return self.__mul__(x)

    def __rmatmul__(self, x):
        if np.isscalar(x):
            raise ValueError("Scalar not allowed, use * instead")
        return self.__rmul__(x)

    def __pow__(self, p):
        Op = _PowerLinearOperator(self, p)
        self._copy_attributes(Op)
        return Op

    def __add__(self, x):
        Op = _SumLinearOperator(self, x)
        self._copy_attributes(Op)
        return Op

    def __neg__(self):
        Op = _ScaledLinearOperator(self, -1)
        self._copy_attributes(Op)
        return Op

    def __sub__(self, x):
        return self.__add__(-x)

    def _adjoint(self):
        Op = _AdjointLinearOperator(self)
        self._copy_attributes(Op, exclude=["dims", "dimsd"])
        Op.dims = self.dimsd
        Op.dimsd = self.dims
        return Op

    def _transpose(self):
        Op = _TransposedLinearOperator(self)
        self._copy_attributes(Op, exclude=["dims", "dimsd"])
        Op.dims = self.dimsd
        Op.dimsd = self.dims
        return Op

    def _copy_attributes(self, dest, exclude=None):
        # ref :385-397
        attrs = ["dims", "dimsd"]
        if exclude is not None:
            for item in exclude:
                attrs.remove(item)
        for attr in attrs:
            if hasattr(self, attr):
                setattr(dest, attr, getattr(self, attr))

    def __repr__(self):
        M, N = self.shape
        dt = "unspecified dtype" if getattr(self, "dtype", None) is None \
            else f"dtype={self.dtype}"
        return f"<{M}x{N} {self.__class__.__name__} with {dt}>"


class _AdjointLinearOperator(MPILinearOperator):
    # ref :408-421
    def __init__(self, A):
        self.A = A
        self.args = (A,)
        super().__init__(shape=(A.shape[1], A.shape[0]), dtype=A.dtype,
                         base_comm=A.base_comm)

    def _matvec(self, x):
        return self.A.rmatvec(x)

    def _rmatvec(self, x):
        return self.A.matvec(x)


class _TransposedLinearOperator(MPILinearOperator):
    # ref :424-443
    def __init__(self, A):
        self.A = A
        self.args = (A,)
        super().__init__(shape=(A.shape[1], A.shape[0]), dtype=A.dtype,
                         base_comm=A.base_comm)

    def _matvec(self, x):
        return self.A.rmatvec(x.conj()).conj()

    def _rmatvec(self, x):
        return self.A.matvec(x.conj()).conj()


class _ProductLinearOperator(MPILinearOperator):
    # ref :446-466
    def __init__(self, A, B):
        if not isinstance(A, MPILinearOperator) \
                or not isinstance(B, MPILinearOperator):
            raise ValueError("both operands have to be a LinearOperator")
        if A.shape[1] != B.shape[0]:
            raise ValueError(
                "cannot multiply %r and %r: shape mismatch" % (A, B))
        self.args = (A, B)
        super().__init__(shape=(A.shape[0], B.shape[1]),
                         dtype=np.promote_types(A.dtype, B.dtype),
                         base_comm=A.base_comm)

    def _matvec(self, x):
        return self.args[0].matvec(self.args[1].matvec(x))

    def _rmatvec(self, x):
        return self.args[1].rmatvec(self.args[0].rmatvec(x))

    def _adjoint(self):
        A, B = self.args
        return B.H * A.H


class _ScaledLinearOperator(MPILinearOperator):
    # ref :469-496
    def __init__(self, A, alpha):
        if not isinstance(A, MPILinearOperator):
            raise ValueError("MPILinearOperator expected as A")
        if not np.isscalar(alpha):
            raise ValueError("scalar expected as alpha")
        self.args = (A, alpha)
        super().__init__(shape=A.shape, dtype=A.dtype, base_comm=A.base_comm)

    def _matvec(self, x):
        y = self.args[0].matvec(x)
        if y is not None:
            y = y * self.args[1]
        return y

    def _rmatvec(self, x):
        y = self.args[0].rmatvec(x)
        if y is not None:
            y = y * np.conj(self.args[1])
        return y

    def _adjoint(self):
        A, alpha = self.args
        return A.H * np.conj(alpha)


class _SumLinearOperator(MPILinearOperator):
    # ref :499-524
    def __init__(self, A, B):
        if not isinstance(A, MPILinearOperator) \
                or not isinstance(B, MPILinearOperator):
            raise ValueError("both operands have to be a MPILinearOperator")
        if A.shape != B.shape:
            raise ValueError(
                "cannot add %r and %r: shape mismatch" % (A, B))
        self.args = (A, B)
        super().__init__(shape=A.shape, dtype=A.dtype, base_comm=A.base_comm)

    def _matvec(self, x):
        return self.args[0].matvec(x) + self.args[1].matvec(x)

    def _rmatvec(self, x):
        return self.args[0].rmatvec(x) + self.args[1].rmatvec(x)

    def _adjoint(self):
        A, B = self.args
        return A.H + B.H


class _PowerLinearOperator(MPILinearOperator):
    # ref :527-552
    def __init__(self, A, p):
        if not isinstance(A, MPILinearOperator):
            raise ValueError("LinearOperator expected as A")
        if A.shape[0] != A.shape[1]:
            raise ValueError("square LinearOperator expected, got %r" % A)
        if not isinstance(p, (int, np.integer)) or p < 0:
            raise ValueError("non-negative integer expected as p")
        super().__init__(shape=A.shape, dtype=A.dtype, base_comm=A.base_comm)
        self.args = (A, p)

    def _power(self, fun, x):
        res = x.copy()
        for _ in range(self.args[1]):
            res[:] = fun(res).local_array
        return res

    def _matvec(self, x):
        return self._power(self.args[0].matvec, x)

    def _rmatvec(self, x):
        return self._power(self.args[0].rmatvec, x)


class _ConjLinearOperator(MPILinearOperator):
    # ref :555-580
    def __init__(self, A):
        if not isinstance(A, MPILinearOperator):
            raise TypeError("A must be a MPILinearOperator")
        self.A = A
        super().__init__(shape=A.shape, dtype=A.dtype, base_comm=A.base_comm)

    def _matvec(self, x):
        y = self.A.matvec(x.conj())
        if y is not None:
            y = y.conj()
        return y

    def _rmatvec(self, x):
        y = self.A.rmatvec(x.conj())
        if y is not None:
            y = y.conj()
        return y

    def _adjoint(self):
        return _ConjLinearOperator(self.A.H)


def asmpilinearoperator(Op):
    """Return Op as an MPILinearOperator (ref LinearOperator.py:583-602):
    pass-through for MPI operators, serial-wrap for local operators."""
    if isinstance(Op, MPILinearOperator):
        return Op
    return MPILinearOperator(Op=Op)
