"""MPIVStack / MPIHStack / stacked operators.

Drop-in for /root/reference/pylops_mpi/basicoperators/VStack.py:16-203 and
HStack.py:11-106: per-rank local operators stacked vertically (forward:
every rank applies its ops to the BROADCAST model and keeps its slice;
adjoint: per-op rmatvecs summed locally then allreduced over RCCL) —
HStack is the adjoint-flip of VStack.
"""
from typing import List, Optional, Sequence

import numpy as np
import torch

from .blockdiag import MPIBlockDiag  # noqa: F401  (re-export convenience)
from .comm import PamComm, get_default_comm
from .distributedarray import DistributedArray, Partition, as_torch_dtype
from .linearoperator import MPILinearOperator
from .localops import AdjointLocal, LocalOperator
from .rebalance import rebalance_1d
from .stacked import MPIStackedLinearOperator, StackedDistributedArray


class MPIVStack(MPILinearOperator):
    """ref VStack.py:16-150."""

    def __init__(self, ops: Sequence[LocalOperator],
                 base_comm: Optional[PamComm] = None, dtype=None):
        comm = base_comm if base_comm is not None else get_default_comm()
        self.ops = list(ops)
        nops = np.array([op.shape[0] for op in self.ops], dtype=np.int64)
        self.nops = int(nops.sum())
        self.local_shapes_n = comm.allgather_obj((self.nops,))
        mops_all = [int(m) for ms in comm.allgather_obj(
            [op.shape[1] for op in self.ops]) for m in ms]
        if len(set(mops_all)) > 1:
            # ref :112-113
            raise ValueError("Operators have different number of columns")
        self.mops = int(mops_all[0])
        self.nnops = np.insert(np.cumsum(nops), 0, 0)
        dimsd = (int(sum(s[0] for s in self.local_shapes_n)),)
        dims = (self.mops,)
        dtype = self.ops[0].dtype if dtype is None else np.dtype(dtype)
        super().__init__(dims=dims, dimsd=dimsd, dtype=dtype, base_comm=comm)

    def _matvec(self, x: DistributedArray) -> DistributedArray:
        # ref :121-133
        if x.partition not in (Partition.BROADCAST,
                               Partition.UNSAFE_BROADCAST):
            raise ValueError(
                f"x should have partition={Partition.BROADCAST},"
                f"{Partition.UNSAFE_BROADCAST}"
                f"Got  {x.partition} instead...")
        comm = x.base_comm
        y = DistributedArray(self.shape[0], comm, Partition.SCATTER, 0,
                             local_shapes=self.local_shapes_n,
                             dtype=self.dtype)
        pieces = [op.matvec(x.local_array.reshape(-1)) for op in self.ops]
        y[:] = torch.cat([p.reshape(-1) for p in pieces]).to(
            as_torch_dtype(self.dtype))
        return y

    def _rmatvec(self, x: DistributedArray) -> DistributedArray:
        # ref :135-150 (@reshaped(forward=False, stacking=True))
        if x.partition is not Partition.SCATTER:
            raise ValueError(
                f"x should have partition={Partition.SCATTER}, "
                f"{x.partition} != {Partition.SCATTER}")
        comm = x.base_comm
        counts = [int(s[0]) for s in self.local_shapes_n]
        local = rebalance_1d(x, counts)
        acc = None
        for iop, op in enumerate(self.ops):
            seg = local[int(self.nnops[iop]): int(self.nnops[iop + 1])]
            r = op.rmatvec(seg).reshape(-1)
            acc = r if acc is None else acc + r
        acc = acc.to(as_torch_dtype(self.dtype)).contiguous()
        comm.allreduce_(acc, "sum")  # ref :148-149
        y = DistributedArray(self.shape[1], comm, Partition.BROADCAST,
                             dtype=self.dtype)
        y[:] = acc
        return y


class MPIHStack(MPILinearOperator):
    """ref HStack.py:11-106 — the adjoint-flip of MPIVStack."""

    def __init__(self, ops: Sequence[LocalOperator],
                 base_comm: Optional[PamComm] = None, dtype=None):
        comm = base_comm if base_comm is not None else get_default_comm()
        self.ops = list(ops)
        nops_all = [int(n) for ns in comm.allgather_obj(
            [op.shape[0] for op in self.ops]) for n in ns]
        if len(set(nops_all)) > 1:
            raise ValueError("Operators have different number of rows")
        hops = [AdjointLocal(op) for op in self.ops]
        self.HStack = MPIVStack(hops, base_comm=comm, dtype=dtype).H
        super().__init__(dims=self.HStack.dims, dimsd=self.HStack.dimsd,
                         dtype=self.HStack.dtype, base_comm=comm)

    def _matvec(self, x: DistributedArray) -> DistributedArray:
        return self.HStack.matvec(x)

    def _rmatvec(self, x: DistributedArray) -> DistributedArray:
        return self.HStack.rmatvec(x)


class MPIStackedVStack(MPIStackedLinearOperator):
    """ref VStack.py:153-203 — vertical stack of MPILinearOperators."""

    def __init__(self, ops: List[MPILinearOperator],
                 base_comm: Optional[PamComm] = None, dtype=None):
        self.ops = ops
        if len(set(op.shape[1] for op in ops)) > 1:
            raise ValueError("Operators have different number of columns")
        dims = (ops[0].shape[1],)
        dimsd = (int(sum(op.shape[0] for op in ops)),)
        dtype = ops[0].dtype if dtype is None else np.dtype(dtype)
        super().__init__(dims=dims, dimsd=dimsd, dtype=dtype,
                         base_comm=base_comm)

    def _matvec(self, x: DistributedArray) -> StackedDistributedArray:
        return StackedDistributedArray([op.matvec(x) for op in self.ops])

    def _rmatvec(self, x: StackedDistributedArray) -> DistributedArray:
        y = self.ops[0].rmatvec(x[0])
        for xx, oper in zip(x.distarrays[1:], self.ops[1:]):
            y = y + oper.rmatvec(xx)
        return y


class MPIStackedBlockDiag(MPIStackedLinearOperator):
    """ref BlockDiag.py:147-188 — diagonal stack of MPILinearOperators."""

    def __init__(self, ops: List[MPILinearOperator],
                 base_comm: Optional[PamComm] = None, dtype=None):
        self.ops = ops
        dims = (int(sum(op.shape[1] for op in ops)),)
        dimsd = (int(sum(op.shape[0] for op in ops)),)
        dtype = ops[0].dtype if dtype is None else np.dtype(dtype)
        super().__init__(dims=dims, dimsd=dimsd, dtype=dtype,
                         base_comm=base_comm)

    def _matvec(self, x: StackedDistributedArray) -> StackedDistributedArray:
        return StackedDistributedArray(
            [op.matvec(xx) for xx, op in zip(x.distarrays, self.ops)])

    def _rmatvec(self, x: StackedDistributedArray) -> StackedDistributedArray:
        return StackedDistributedArray(
            [op.rmatvec(xx) for xx, op in zip(x.distarrays, self.ops)])
