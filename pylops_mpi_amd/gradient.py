"""MPIGradient / MPILaplacian — compositions of the first/second
derivative operators (ref basicoperators/Gradient.py:12-118,
Laplacian.py:12-126): axis 0 uses the distributed halo-exchange operator,
axes >= 1 use serial per-rank derivatives inside MPIBlockDiag."""
from typing import Tuple

import numpy as np

from .blockdiag import MPIBlockDiag
from .comm import PamComm, get_default_comm
from .derivative import MPIFirstDerivative, MPISecondDerivative
from .distributedarray import local_split, Partition
from .fdlocal import FirstDerivativeLocal, SecondDerivativeLocal
from .linearoperator import MPILinearOperator
from .stacked import MPIStackedLinearOperator
from .vstack import MPIStackedVStack


def _tuple_per_axis(v, n):
    if isinstance(v, (int, float, np.integer, np.floating)):
        return (v,) * n
    v = tuple(v)
    return v if len(v) == n else v * n


class MPIGradient(MPIStackedLinearOperator):
    """ref Gradient.py:12-118."""

    def __init__(self, dims, sampling=1, edge: bool = False,
                 kind: str = "centered", base_comm: PamComm = None,
                 dtype=np.float64):
        comm = base_comm if base_comm is not None else get_default_comm()
        dims = (dims,) if isinstance(dims, (int, np.integer)) \
            else tuple(dims)
        self.sampling = _tuple_per_axis(sampling, len(dims))
        self.edge = edge
        self.kind = kind
        self.dtype = np.dtype(dtype)
        # ref :100-118
        local_dims = local_split(dims, comm.size, comm.rank,
                                 Partition.SCATTER, 0)
        grad_ops = [MPIFirstDerivative(dims=dims, sampling=self.sampling[0],
                                       kind=self.kind, edge=self.edge,
                                       base_comm=comm, dtype=self.dtype)]
        for iax in range(1, len(dims)):
            grad_ops.append(MPIBlockDiag(
                [FirstDerivativeLocal(local_dims, axis=iax,
                                      sampling=self.sampling[iax],
                                      edge=self.edge, kind=self.kind,
                                      dtype=self.dtype)], base_comm=comm))
        self.Op = MPIStackedVStack(grad_ops, base_comm=comm)
        super().__init__(dims=dims, dimsd=self.Op.dimsd, dtype=self.dtype,
                         base_comm=comm)

    def _matvec(self, x):
        return self.Op._matvec(x)

    def _rmatvec(self, x):
        return self.Op._rmatvec(x)


class MPILaplacian(MPILinearOperator):
    """ref Laplacian.py:12-126."""

    def __init__(self, dims, axes=(-2, -1), weights: Tuple = (1, 1),
                 sampling: Tuple = (1, 1), edge: bool = False,
                 kind: str = "centered", base_comm: PamComm = None,
                 dtype=np.float64):
        comm = base_comm if base_comm is not None else get_default_comm()
        dims = tuple(dims)
        axes = tuple(ax if ax >= 0 else len(dims) + ax for ax in axes)
        if not (len(axes) == len(weights) == len(sampling)):
            raise ValueError(
                "axes, weights, and sampling have different size")
        self.axes, self.weights, self.sampling = axes, weights, sampling
        self.edge, self.kind = edge, kind
        self.dtype = np.dtype(dtype)
        # ref :97-126
        local_dims = local_split(dims, comm.size, comm.rank,
                                 Partition.SCATTER, 0)

        def term(ax, samp, weight):
            if ax == 0:
                return weight * MPISecondDerivative(
                    dims=dims, sampling=samp, kind=kind, edge=edge,
                    base_comm=comm, dtype=self.dtype)
            return weight * MPIBlockDiag(
                [SecondDerivativeLocal(local_dims, axis=ax, sampling=samp,
                                       kind=kind, edge=edge,
                                       dtype=self.dtype)], base_comm=comm)

        l2op = term(axes[0], sampling[0], weights[0])
        for ax, samp, weight in zip(axes[1:], sampling[1:], weights[1:]):
            l2op = l2op + term(ax, samp, weight)
        self.Op = l2op
        super().__init__(shape=l2op.shape, dims=dims, dimsd=dims,
                         dtype=self.dtype, base_comm=comm)

    def _matvec(self, x):
        return self.Op.matvec(x)

    def _rmatvec(self, x):
        return self.Op.rmatvec(x)
