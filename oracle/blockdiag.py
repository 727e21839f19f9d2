"""Rank-simulated MPIBlockDiag with dense local blocks — TEST ONLY.

Restates /root/reference/pylops_mpi/basicoperators/BlockDiag.py:100-144
(per-rank serial applies + the stacking reshaped rebalance,
ref utils/decorators.py:47-52) with dense NumPy matrices as the local
operators (the reference's examples/plot_cgls.py:30-33 recipe).
"""
from typing import List, Sequence

import numpy as np

from .ranksim import SimArray, SimStackedArray, reshaped_apply


class SimBlockDiag:
    def __init__(self, mats_per_rank: Sequence[Sequence[np.ndarray]]):
        self.mats = [list(ms) for ms in mats_per_rank]
        self.nops = [int(sum(A.shape[0] for A in ms)) for ms in self.mats]
        self.mops = [int(sum(A.shape[1] for A in ms)) for ms in self.mats]
        self.shape = (int(sum(self.nops)), int(sum(self.mops)))

    def _body(self, xs: List[np.ndarray], forward: bool) -> List[np.ndarray]:
        ys = []
        for r, ms in enumerate(self.mats):
            x = xs[r].ravel()
            pieces, off = [], 0
            for A in ms:
                w = A.shape[1] if forward else A.shape[0]
                seg = x[off: off + w]
                pieces.append(A @ seg if forward else A.T @ seg)
                off += w
            ys.append(np.concatenate(pieces))
        return ys

    def matvec(self, x: SimArray) -> SimArray:
        return reshaped_apply(lambda xs: self._body(xs, True), None, x,
                              target_counts=self.mops)

    def rmatvec(self, x: SimArray) -> SimArray:
        return reshaped_apply(lambda xs: self._body(xs, False), None, x,
                              target_counts=self.nops)

    def dense(self) -> np.ndarray:
        """The explicit global block-diagonal matrix (independent pin)."""
        blocks = [A for ms in self.mats for A in ms]
        n, m = self.shape
        out = np.zeros((n, m))
        r0 = c0 = 0
        for A in blocks:
            out[r0: r0 + A.shape[0], c0: c0 + A.shape[1]] = A
            r0 += A.shape[0]
            c0 += A.shape[1]
        return out


class SimStackedBlockDiag:
    """ref basicoperators/BlockDiag.py:147-189 MPIStackedBlockDiag:
    component-wise application of stacked MPI operators."""

    def __init__(self, ops):
        self.ops = list(ops)
        self.shape = (int(sum(op.shape[0] for op in self.ops)),
                      int(sum(op.shape[1] for op in self.ops)))

    def matvec(self, x: SimStackedArray) -> SimStackedArray:
        return SimStackedArray([op.matvec(xx) for op, xx
                                in zip(self.ops, x.arrays)])

    def rmatvec(self, x: SimStackedArray) -> SimStackedArray:
        return SimStackedArray([op.rmatvec(xx) for op, xx
                                in zip(self.ops, x.arrays)])


class SimStackedVStack:
    """ref basicoperators/VStack.py:153-203 MPIStackedVStack: matvec
    fans one model out to every operator; rmatvec folds component
    adjoints in operator order (:199-203)."""

    def __init__(self, ops):
        self.ops = list(ops)
        self.shape = (int(sum(op.shape[0] for op in self.ops)),
                      int(self.ops[0].shape[1]))

    def matvec(self, x: SimArray) -> SimStackedArray:
        return SimStackedArray([op.matvec(x) for op in self.ops])

    def rmatvec(self, x: SimStackedArray) -> SimArray:
        y = self.ops[0].rmatvec(x.arrays[0])
        for xx, op in zip(x.arrays[1:], self.ops[1:]):
            y = y + op.rmatvec(xx)
        return y
