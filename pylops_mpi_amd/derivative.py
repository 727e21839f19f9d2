"""MPIFirstDerivative / MPISecondDerivative — the north-star operators.

Drop-in for /root/reference/pylops_mpi/basicoperators/FirstDerivative.py
and SecondDerivative.py, re-designed MI355X-first:

  * ONE fused HIP kernel per (kind, order, direction) (csrc/pam.hip
    fd_kernel) — the reference builds each apply from a materialized
    ghosted copy plus 1-4 full-array slice passes
    (ref FirstDerivative.py:201-246: ghost copy + slice-sub + divide);
  * the ghost exchange moves only max|stencil offset| planes per side over
    RCCL/xGMI and the kernel reads them in place;
  * the ``@reshaped`` input rebalance (ref utils/decorators.py:44-82) is
    the same arithmetic, but all ranks' local sizes are known locally so
    the reference's two object allgathers per apply disappear, and the
    balanced case is a zero-copy view.
"""
import numpy as np
import torch

from . import _ffi
from .distributedarray import (DistributedArray, Partition,
                               local_split)
from .linearoperator import MPILinearOperator
from .rebalance import rebalance_1d

_FD1_OPS = {("forward", 0): (0, 1), ("backward", 0): (2, 3),
            ("centered", 3): (4, 5), ("centered", 5): (6, 7)}
_FD2_OPS = {"forward": (8, 9), "backward": (10, 11), "centered": (12, 13)}
_NP_OF = {torch.float64: np.float64, torch.float32: np.float32,
          torch.complex128: np.complex128, torch.complex64: np.complex64}

# bench instrumentation: when enabled, each stencil kernel launch records a
# HIP event pair on the launch stream (torch's current stream, which is
# where pam_fd_apply is enqueued) keyed by op code.
KERNEL_TIMING = False
KERNEL_EVENTS: dict = {}


def _record_events(op):
    import torch as _t
    e0, e1 = (_t.cuda.Event(enable_timing=True),
              _t.cuda.Event(enable_timing=True))
    KERNEL_EVENTS.setdefault(op, []).append((e0, e1))
    return e0, e1


class _FDBase(MPILinearOperator):

    def __init__(self, dims, sampling, kind, edge, base_comm, dtype):
        dims = (dims,) if isinstance(dims, (int, np.integer)) else tuple(dims)
        super().__init__(dims=dims, dimsd=dims, dtype=np.dtype(dtype),
                         base_comm=base_comm)
        self.dims = dims
        self.sampling = sampling
        self.kind = kind
        self.edge = edge

    # --------------------------------------------------------------- plumbing
    def _plane_counts(self):
        """Balanced plane-aligned split of dims along axis 0 (what the
        reshaped wrapper redistributes to, ref decorators.py:61-67)."""
        P = self.base_comm.size
        shapes = [local_split(self.dims, P, r) for r in range(P)]
        counts = [int(np.prod(s)) for s in shapes]
        return shapes, counts

    def _rebalance(self, x: DistributedArray, counts):
        """ref decorators.py:66-77 — move the 1-D input to the plane-aligned
        split.  Balanced inputs are a zero-copy view."""
        return rebalance_1d(x, counts)

    def _halo(self, planes: torch.Tensor, w: int, comm):
        """Exchange w boundary planes of the [nloc, m] block with the
        neighbours over RCCL (values of ref DistributedArray.py:955-1032)."""
        r, P = comm.rank, comm.size
        if P == 1 or w == 0:
            return None, None
        nloc = planes.shape[0]
        send_prev = planes[:w].contiguous() if r > 0 else None
        send_next = planes[-w:].contiguous() if r < P - 1 else None
        gshape = (w,) + tuple(planes.shape[1:])
        gf = torch.empty(gshape, dtype=planes.dtype,
                         device=planes.device) if r > 0 else None
        gb = torch.empty(gshape, dtype=planes.dtype,
                         device=planes.device) if r < P - 1 else None
        comm.sendrecv_neighbors(send_prev, send_next, gf, gb)
        return gf, gb

    def _apply(self, x: DistributedArray, op: int) -> DistributedArray:
        # BROADCAST -> SCATTER conversion (ref FirstDerivative.py:128-138)
        if x.partition is Partition.BROADCAST:
            x = DistributedArray.to_dist(x.local_array, x.base_comm)
        if x.partition is not Partition.SCATTER:
            # ref decorators.py:45-46
            raise ValueError(
                f"x should have partition={Partition.SCATTER}, "
                f"{x.partition} != {Partition.SCATTER}")
        comm = x.base_comm
        shapes, counts = self._plane_counts()
        flat = self._rebalance(x, counts)
        nloc = shapes[comm.rank][0]
        m = int(np.prod(self.dims[1:], initial=1))
        planes = flat.view(nloc, m) if m > 1 else flat.view(nloc, 1)
        x._require_compute()
        w = int(_ffi.lib().pam_fd_halo_width(op))
        if comm.size > 1 and w > 0:
            # every rank with a neighbour sends w planes (rank 0 to the
            # next, rank P-1 to the previous), so EVERY rank's block must
            # hold >= w planes; shapes is deterministic on all ranks, so
            # this raises consistently everywhere instead of one rank
            # posting a mismatched irecv (ref :996-1002 sender guard)
            for rr in range(comm.size):
                if shapes[rr][0] < w:
                    raise ValueError(
                        f"Local Shape at rank={rr} along axis=0 should "
                        f"be > {w}")
        y_out = torch.empty_like(planes)
        y = y_out
        if planes.is_complex():
            # real-coefficient stencil acts componentwise: run the real
            # kernels on the interleaved (re,im) view with doubled columns
            planes = torch.view_as_real(planes).reshape(nloc, 2 * m)
            y = torch.view_as_real(y_out).reshape(nloc, 2 * m)
        row0 = int(np.sum([s[0] for s in shapes[: comm.rank]], initial=0))
        stream = torch.cuda.current_stream(planes.device).cuda_stream
        m = planes.shape[1]
        dt = _ffi.dtype_code(planes.dtype)
        edge = 1 if self.edge else 0

        def launch(gfp, gbp, r0, r1):
            _ffi.checked(_ffi.lib().pam_fd_apply(
                stream, op, edge, planes.data_ptr(),
                gfp.data_ptr() if gfp is not None else None,
                gbp.data_ptr() if gbp is not None else None,
                y.data_ptr(), nloc, m, row0, self.dims[0], r0, r1,
                self._coeff(), dt), "fd_apply")

        ev = _record_events(op) if KERNEL_TIMING else None
        if ev is not None:
            ev[0].record()
        from . import deps as _deps
        if (_deps.overlap_enabled and comm.size > 1 and w > 0
                and nloc > 2 * w):
            # overlap: post the RCCL halo exchange (its own stream), run the
            # halo-independent interior rows concurrently, then the
            # boundary rows once the planes have arrived
            r, P = comm.rank, comm.size
            gshape = (w,) + tuple(planes.shape[1:])
            gf = torch.empty(gshape, dtype=planes.dtype,
                             device=planes.device) if r > 0 else None
            gb = torch.empty(gshape, dtype=planes.dtype,
                             device=planes.device) if r < P - 1 else None
            works = comm.post_neighbors(
                planes[:w].contiguous() if r > 0 else None,
                planes[-w:].contiguous() if r < P - 1 else None, gf, gb)
            launch(None, None, w, nloc - w)      # interior
            for wk in works:
                wk.wait()
            launch(gf, gb, 0, w)                 # boundary rows
            launch(gf, gb, nloc - w, nloc)
        else:
            gf, gb = self._halo(planes, w, comm)
            launch(gf, gb, 0, nloc)
        if ev is not None:
            ev[1].record()
        # ravel back to a 1-D plane-aligned DistributedArray
        # (ref decorators.py:79-82; axis is already 0 so redistribute is a
        #  no-op, ref DistributedArray.py:516-517)
        return DistributedArray(
            int(np.prod(self.dims)), comm, Partition.SCATTER, 0,
            local_array=y_out.view(-1), local_shapes=[(c,) for c in counts],
            dtype=_NP_OF[y_out.dtype])

    def _matvec(self, x: DistributedArray) -> DistributedArray:
        return self._apply(x, self._op_mv)

    def _rmatvec(self, x: DistributedArray) -> DistributedArray:
        return self._apply(x, self._op_rmv)


class MPIFirstDerivative(_FDBase):
    """ref basicoperators/FirstDerivative.py:18-138 (ctor + dispatch)."""

    def __init__(self, dims, sampling: float = 1.0, kind: str = "centered",
                 edge: bool = False, order: int = 3, base_comm=None,
                 dtype=np.float64):
        super().__init__(dims, sampling, kind, edge, base_comm, dtype)
        self.order = order
        # kind dispatch, ref :100-126
        if kind == "centered" and order not in (3, 5):
            raise NotImplementedError("'order' must be '3, or '5'")
        key = (kind, order if kind == "centered" else 0)
        if key not in _FD1_OPS:
            raise NotImplementedError(
                "'kind' must be 'forward', 'centered', or 'backward'")
        self._op_mv, self._op_rmv = _FD1_OPS[key]

    def _coeff(self):
        return 1.0 / self.sampling


class MPISecondDerivative(_FDBase):
    """ref basicoperators/SecondDerivative.py:17-121."""

    def __init__(self, dims, sampling: float = 1.0, kind: str = "centered",
                 edge: bool = False, base_comm=None, dtype=np.float64):
        super().__init__(dims, sampling, kind, edge, base_comm, dtype)
        if kind not in _FD2_OPS:
            raise NotImplementedError(
                "'kind' must be 'forward', 'centered' or 'backward'")
        self._op_mv, self._op_rmv = _FD2_OPS[kind]

    def _coeff(self):
        return 1.0 / self.sampling ** 2
