"""MPIBlockDiag — each rank applies its own local operator(s) to its
slice; no inter-rank communication in the apply itself (only in the
reshaped input rebalance).

Drop-in for /root/reference/pylops_mpi/basicoperators/BlockDiag.py:16-144.
The reference stacks serial pylops operators; here the local operators
follow pylops_mpi_amd.localops.LocalOperator (HIP compute, e.g.
DenseLocal for the dense cfg-3 blocks).
"""
from typing import List, Optional, Sequence

import numpy as np
import torch

from .comm import PamComm, get_default_comm
from .distributedarray import DistributedArray, Partition, as_torch_dtype
from .linearoperator import MPILinearOperator
from .localops import LocalOperator
from .rebalance import rebalance_1d


class MPIBlockDiag(MPILinearOperator):

    def __init__(self, ops: Sequence[LocalOperator],
                 base_comm: Optional[PamComm] = None,
                 mask=None, dtype=None):
        comm = base_comm if base_comm is not None else get_default_comm()
        self.mask = mask  # ref :102-105: passed through to the outputs
        self.ops = list(ops)
        # per-rank row/col offsets, ref BlockDiag.py:106-116
        nops = np.array([op.shape[0] for op in self.ops], dtype=np.int64)
        mops = np.array([op.shape[1] for op in self.ops], dtype=np.int64)
        self.mops = int(mops.sum())
        self.nops = int(nops.sum())
        self.local_shapes_m = comm.allgather_obj((self.mops,))
        self.local_shapes_n = comm.allgather_obj((self.nops,))
        self.nnops = np.insert(np.cumsum(nops), 0, 0)
        self.mmops = np.insert(np.cumsum(mops), 0, 0)
        dims = (int(sum(s[0] for s in self.local_shapes_m)),)
        dimsd = (int(sum(s[0] for s in self.local_shapes_n)),)
        dtype = self.ops[0].dtype if dtype is None else np.dtype(dtype)
        super().__init__(dims=dims, dimsd=dimsd, dtype=dtype, base_comm=comm)

    def _apply(self, x: DistributedArray, forward: bool) -> DistributedArray:
        # the @reshaped(stacking=True) wrapper, ref decorators.py:47-52:
        # rebalance x to the per-rank operator sizes
        if x.partition is not Partition.SCATTER:
            raise ValueError(
                f"x should have partition={Partition.SCATTER}, "
                f"{x.partition} != {Partition.SCATTER}")
        comm = x.base_comm
        in_shapes = self.local_shapes_m if forward else self.local_shapes_n
        out_shapes = self.local_shapes_n if forward else self.local_shapes_m
        counts = [int(s[0]) for s in in_shapes]
        local = rebalance_1d(x, counts)
        offs = self.mmops if forward else self.nnops
        # per-op local applies, ref BlockDiag.py:122-144
        pieces: List[torch.Tensor] = []
        for iop, op in enumerate(self.ops):
            seg = local[int(offs[iop]): int(offs[iop + 1])]
            pieces.append(op.matvec(seg) if forward else op.rmatvec(seg))
        out = torch.cat(pieces) if len(pieces) != 1 else pieces[0].reshape(-1)
        out = out.to(as_torch_dtype(np.dtype(self.dtype)))
        gshape = self.shape[0] if forward else self.shape[1]
        return DistributedArray(
            int(gshape), comm, Partition.SCATTER, 0, local_array=out,
            local_shapes=out_shapes, mask=self.mask, dtype=self.dtype)

    def _matvec(self, x: DistributedArray) -> DistributedArray:
        return self._apply(x, forward=True)

    def _rmatvec(self, x: DistributedArray) -> DistributedArray:
        return self._apply(x, forward=False)
