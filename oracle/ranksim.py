"""Rank-simulated DistributedArray semantics — TEST INFRASTRUCTURE ONLY.

Pure-NumPy restatement of the reference's DistributedArray (P ranks run
sequentially in one process).  Reference citations are to
/root/reference/pylops_mpi/DistributedArray.py unless stated otherwise.
"""
from enum import Enum
from typing import List, Optional, Sequence, Tuple, Union

import numpy as np


class Partition(Enum):
    # ref DistributedArray.py:26-39
    BROADCAST = "Broadcast"
    UNSAFE_BROADCAST = "UnsafeBroadcast"
    SCATTER = "Scatter"


def local_split(global_shape: Tuple[int, ...], size: int, rank: int,
                partition: Partition = Partition.SCATTER,
                axis: int = 0) -> Tuple[int, ...]:
    """Local shape of ``rank`` out of ``size`` ranks.

    ref DistributedArray.py:42-71 — the first ``N % P`` ranks get one extra
    element along ``axis``.
    """
    if partition in (Partition.BROADCAST, Partition.UNSAFE_BROADCAST):
        return tuple(global_shape)
    local_shape = list(global_shape)
    if rank < (global_shape[axis] % size):
        local_shape[axis] = global_shape[axis] // size + 1
    else:
        local_shape[axis] = global_shape[axis] // size
    return tuple(local_shape)


def split_bounds(n: int, size: int) -> List[Tuple[int, int]]:
    """Start/stop of each rank's slice along the distributed axis.

    ref DistributedArray.py:483-490 (cumsum of allgathered local dims).
    """
    counts = [local_split((n,), size, r)[0] for r in range(size)]
    offs = np.concatenate([[0], np.cumsum(counts)])
    return [(int(offs[r]), int(offs[r + 1])) for r in range(size)]


def to_dist(x: np.ndarray, size: int, axis: int = 0,
            partition: Partition = Partition.SCATTER) -> "SimArray":
    """Scatter a global array into per-rank locals.  ref :438-491."""
    if partition in (Partition.BROADCAST, Partition.UNSAFE_BROADCAST):
        locals_ = [x.copy() for _ in range(size)]
        return SimArray(locals_, tuple(x.shape), axis=axis, partition=partition)
    bounds = split_bounds(x.shape[axis], size)
    locals_ = []
    for r in range(size):
        sl = [slice(None)] * x.ndim
        sl[axis] = slice(bounds[r][0], bounds[r][1])
        locals_.append(x[tuple(sl)].copy())
    return SimArray(locals_, tuple(x.shape), axis=axis, partition=partition)


def add_ghost_cells(locals_: Sequence[np.ndarray],
                    cells_front: Optional[Sequence[int]] = None,
                    cells_back: Optional[Sequence[int]] = None,
                    axis: int = 0) -> List[np.ndarray]:
    """Ghost-cell exchange over simulated ranks.  ref :955-1032.

    ``cells_front[r]`` / ``cells_back[r]`` are the per-rank request values
    (they differ per rank inside the ``reshaped`` rebalance,
    ref utils/decorators.py:71-74).  Rank r receives ``cells_front[r]``
    trailing cells of rank r-1 prepended, and ``cells_back[r]`` leading
    cells of rank r+1 appended, along ``axis``.
    """
    P = len(locals_)
    ghosted = [a.copy() for a in locals_]
    if cells_front is not None:
        cf = list(cells_front) + [0]
        for r in range(P):
            if r != 0 and cf[r] != 0:
                src = locals_[r - 1]
                # sender-side check, ref :996-1002
                if cf[r] > src.shape[axis]:
                    raise ValueError(
                        f"Local Shape at rank={r-1} along axis={axis} "
                        f"should be > {cf[r]}")
                take = np.take(src, np.arange(-cf[r], 0), axis=axis)
                ghosted[r] = np.concatenate([take, ghosted[r]], axis=axis)
    if cells_back is not None:
        cb = list(cells_back) + [0]
        for r in range(P):
            if r != P - 1 and cb[r] != 0:
                src = locals_[r + 1]
                # sender-side check, ref :1013-1019
                if cb[r] > src.shape[axis]:
                    raise ValueError(
                        f"Local Shape at rank={r+1} along axis={axis} "
                        f"should be > {cb[r]}")
                take = np.take(src, np.arange(cb[r]), axis=axis)
                ghosted[r] = np.append(ghosted[r], take, axis=axis)
    return ghosted


class SimArray:
    """List-of-locals stand-in for a DistributedArray (P sequential ranks)."""

    def __init__(self, locals_: List[np.ndarray], global_shape: Tuple[int, ...],
                 axis: int = 0, partition: Partition = Partition.SCATTER):
        self.locals = locals_
        self.global_shape = tuple(global_shape)
        self.axis = axis
        self.partition = partition

    @property
    def size(self) -> int:
        return len(self.locals)

    @property
    def local_counts(self) -> np.ndarray:
        return np.asarray([a.size for a in self.locals])

    def asarray(self) -> np.ndarray:
        # ref :401-436
        if self.partition in (Partition.BROADCAST, Partition.UNSAFE_BROADCAST):
            return self.locals[0]
        return np.concatenate(self.locals, axis=self.axis)

    def copy(self) -> "SimArray":
        return SimArray([a.copy() for a in self.locals], self.global_shape,
                        self.axis, self.partition)

    def zeros_like(self) -> "SimArray":
        return SimArray([np.zeros_like(a) for a in self.locals],
                        self.global_shape, self.axis, self.partition)

    def conj(self) -> "SimArray":
        return SimArray([a.conj() for a in self.locals], self.global_shape,
                        self.axis, self.partition)

    # elementwise math, ref :605-683
    def __neg__(self):
        return SimArray([-a for a in self.locals], self.global_shape,
                        self.axis, self.partition)

    def __add__(self, other: "SimArray"):
        return SimArray([a + b for a, b in zip(self.locals, other.locals)],
                        self.global_shape, self.axis, self.partition)

    def __sub__(self, other: "SimArray"):
        return self.__add__(-other)

    def __mul__(self, x: Union[float, "SimArray"]):
        if isinstance(x, SimArray):
            return SimArray([a * b for a, b in zip(self.locals, x.locals)],
                            self.global_shape, self.axis, self.partition)
        return SimArray([a * x for a in self.locals], self.global_shape,
                        self.axis, self.partition)

    __rmul__ = __mul__

    def __iadd__(self, other: "SimArray"):
        for r in range(self.size):
            self.locals[r] = self.locals[r] + other.locals[r]
        return self

    def __isub__(self, other: "SimArray"):
        return self.__iadd__(-other)

    def dot(self, other: "SimArray", vdot: bool = False):
        """ref :685-717 — per-rank flattened dot, then allreduce (sum)."""
        f = np.vdot if vdot else np.dot
        return sum(f(a.ravel(), b.ravel())
                   for a, b in zip(self.locals, other.locals))

    def norm(self, ord: Optional[int] = None):
        """ref :719-838 (axis=None path: flattened vector norm)."""
        ord = 2 if ord is None else ord
        if ord == 0:
            return float(sum(np.count_nonzero(a) for a in self.locals))
        if ord == np.inf:
            return float(max(np.max(np.abs(a.ravel())) for a in self.locals))
        if ord == -np.inf:
            return float(min(np.min(np.abs(a.ravel())) for a in self.locals))
        # ref :785-788: sum of |float_power(x, ord)| then ord-th root
        s = sum(np.sum(np.abs(np.float_power(a.ravel(), ord)))
                for a in self.locals)
        return np.power(s, 1.0 / ord)

    def ravel(self) -> "SimArray":
        # ref :872-897
        return SimArray([a.ravel() for a in self.locals],
                        (int(np.prod(self.global_shape)),), 0, self.partition)

    def add_ghost_cells(self, cells_front=None, cells_back=None):
        cf = None if cells_front is None else [cells_front] * self.size
        cb = None if cells_back is None else [cells_back] * self.size
        return add_ghost_cells(self.locals, cf, cb, axis=self.axis)


class SimStackedArray:
    """List-of-SimArrays stand-in for a StackedDistributedArray
    (ref DistributedArray.py:1041-1300): component-wise math, dot as
    the rank-ordered sum of component dots (:1231-1255), norm as the
    ord-power fold of component norms (:1257-1281)."""

    def __init__(self, arrays: List["SimArray"]):
        self.arrays = list(arrays)
        self.narrays = len(self.arrays)

    def copy(self) -> "SimStackedArray":
        return SimStackedArray([a.copy() for a in self.arrays])

    def conj(self) -> "SimStackedArray":
        return SimStackedArray([a.conj() for a in self.arrays])

    def zeros_like(self) -> "SimStackedArray":
        return SimStackedArray([a.zeros_like() for a in self.arrays])

    def __neg__(self):
        return SimStackedArray([-a for a in self.arrays])

    def __add__(self, other: "SimStackedArray"):
        return SimStackedArray([a + b for a, b
                                in zip(self.arrays, other.arrays)])

    def __sub__(self, other: "SimStackedArray"):
        return self.__add__(-other)

    def __mul__(self, x):
        if isinstance(x, SimStackedArray):
            return SimStackedArray([a * b for a, b
                                    in zip(self.arrays, x.arrays)])
        return SimStackedArray([a * x for a in self.arrays])

    __rmul__ = __mul__

    def __iadd__(self, other: "SimStackedArray"):
        for i in range(self.narrays):
            self.arrays[i] = self.arrays[i] + other.arrays[i]
        return self

    def __isub__(self, other: "SimStackedArray"):
        return self.__iadd__(-other)

    def dot(self, other: "SimStackedArray", vdot: bool = False):
        dotprod = 0.0
        for a, b in zip(self.arrays, other.arrays):
            dotprod += a.dot(b, vdot=vdot)
        return dotprod

    def norm(self, ord: Optional[int] = None):
        norms = np.hstack([a.norm(ord) for a in self.arrays])
        ord = 2 if ord is None else ord
        if ord == 0:
            return float(np.sum(norms))
        if ord == np.inf:
            return float(np.max(norms))
        if ord == -np.inf:
            return float(np.min(norms))
        return float(np.power(np.sum(np.power(norms, ord)), 1.0 / ord))

    def asarray(self) -> np.ndarray:
        return np.concatenate([a.asarray().ravel() for a in self.arrays])


def reshaped_apply(body, dims: Tuple[int, ...], x: SimArray,
                   target_counts: Optional[Sequence[int]] = None) -> SimArray:
    """The ``@reshaped`` wrapper, ref utils/decorators.py:44-82.

    Rebalances the flat 1-D input ``x`` to the plane-aligned split of
    ``dims`` (axis 0) — or, in the stacking form (ref decorators.py:47-52),
    to the explicit per-rank ``target_counts`` — reshapes, calls
    ``body(list_of_locals) -> list_of_locals`` and ravels the result
    back to 1-D.
    """
    if x.partition is not Partition.SCATTER:
        raise ValueError(f"x should have partition={Partition.SCATTER}")
    P = x.size
    if target_counts is None:
        arr_shapes = [local_split(dims, P, r) for r in range(P)]
    else:
        arr_shapes = [(int(c),) for c in target_counts]
    arr_counts = np.asarray([int(np.prod(s)) for s in arr_shapes])
    x_counts = np.asarray([int(a.size) for a in x.locals])
    # cumulative imbalance, ref decorators.py:69-73
    dif = np.cumsum(arr_counts - x_counts)
    cfs = [abs(min(0, dif[r - 1])) for r in range(P)]   # dif[-1] == 0 at r=0
    cbs = [max(0, dif[r]) for r in range(P)]
    ghosted = add_ghost_cells([a.ravel() for a in x.locals], cfs, cbs, axis=0)
    arr_locals = []
    for r in range(P):
        index = max(0, dif[r - 1])
        arr_locals.append(
            ghosted[r][index: arr_counts[r] + index].reshape(arr_shapes[r]))
    y_locals = body(arr_locals)
    gsize = int(np.prod(dims)) if dims is not None \
        else int(sum(y.size for y in y_locals))
    return SimArray([y.ravel() for y in y_locals],
                    (gsize,), 0, Partition.SCATTER)
