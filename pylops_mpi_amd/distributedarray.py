"""HBM-resident DistributedArray — the drop-in surface of the reference's
pylops_mpi.DistributedArray (ref DistributedArray.py:103-1038) re-designed
MI355X-first:

  * local block lives in HBM as a torch device tensor (PyTorch-ROCm
    caching allocator), one process per GPU;
  * all element-wise math / dot / norm go through hand-written HIP/CDNA4
    kernels (libpam, see csrc/pam.hip) — there is NO CPU compute fallback:
    compute ops on a non-CUDA array, or without the extension, raise;
  * collectives go through RCCL over xGMI (comm.PamComm);
  * ghost-cell exchange moves only the halo planes; operators consume them
    directly (the reference materializes a concatenated ghosted copy,
    ref :974,992-994,1028).

Semantics (partition rules, remainder splits, error messages) mirror the
reference line for line; reference citations are to
/root/reference/pylops_mpi/DistributedArray.py unless stated otherwise.
"""
from enum import Enum
from numbers import Integral
from typing import List, Optional, Tuple, Union

import numpy as np
import torch

from . import _ffi
from .comm import PamComm, get_default_comm

_TORCH_DTYPES = {
    np.dtype(np.float64): torch.float64,
    np.dtype(np.float32): torch.float32,
    np.dtype(np.complex128): torch.complex128,
    np.dtype(np.complex64): torch.complex64,
}
_NP_DTYPES = {v: k for k, v in _TORCH_DTYPES.items()}


def as_torch_dtype(dtype) -> torch.dtype:
    if isinstance(dtype, torch.dtype):
        return dtype
    return _TORCH_DTYPES[np.dtype(dtype)]


class Partition(Enum):
    # ref :26-39
    BROADCAST = "Broadcast"
    UNSAFE_BROADCAST = "UnsafeBroadcast"
    SCATTER = "Scatter"


def local_split(global_shape: Tuple, size: int, rank: int,
                partition: "Partition" = Partition.SCATTER,
                axis: int = 0) -> Tuple:
    """ref :42-71 — the first ``N % P`` ranks get one extra element."""
    if partition in (Partition.BROADCAST, Partition.UNSAFE_BROADCAST):
        return tuple(global_shape)
    local_shape = list(global_shape)
    if rank < (global_shape[axis] % size):
        local_shape[axis] = global_shape[axis] // size + 1
    else:
        local_shape[axis] = global_shape[axis] // size
    return tuple(local_shape)


# per-device reduction scratch: (ws, out) float64 tensors
_red_scratch = {}

# sub-communicator cache: creating a torch process group is collective, so
# identical masks must map to one group (ref subcomm_split, :74-100)
_subcomm_cache = {}


def _get_subcomm(comm, mask_key):
    key = (id(comm), mask_key)
    if key not in _subcomm_cache:
        _subcomm_cache[key] = comm.split_by(list(mask_key),
                                            keys=list(range(comm.size)))
    return _subcomm_cache[key]


def _reduce_buffers(device):
    key = (device.type, device.index)
    if key not in _red_scratch:
        n = int(_ffi.lib().pam_reduce_ws_elems())
        _red_scratch[key] = (
            torch.empty(2 * n, dtype=torch.float64, device=device),
            torch.empty(2, dtype=torch.float64, device=device),
        )
    return _red_scratch[key]


class DistributedArray:
    """Drop-in for pylops_mpi.DistributedArray (ref :103-212 ctor).

    Differences from the reference surface, by design:
      * ``base_comm`` is a :class:`pylops_mpi_amd.comm.PamComm` (RCCL) —
        there is no separate ``base_comm_nccl``;
      * ``engine`` is ``"hip"`` (torch device tensor); ``local_array`` is a
        torch tensor;
      * ``local_shapes``, when given, must list every rank's shape (as in
        the reference); it is cached so no control-plane allgather is
        needed per op;
      * ``mask`` reductions run over a cached RCCL sub-group
        (ref subcomm_split :74-100).
    """

    def __init__(self, global_shape: Union[Tuple, Integral],
                 base_comm: Optional[PamComm] = None,
                 partition: Partition = Partition.SCATTER, axis: int = 0,
                 local_array: Optional[torch.Tensor] = None,
                 local_shapes: Optional[List[Tuple]] = None,
                 mask: Optional[List[Integral]] = None,
                 engine: str = "hip",
                 dtype=np.float64):
        if isinstance(global_shape, Integral):
            global_shape = (int(global_shape),)
        global_shape = tuple(int(s) for s in global_shape)
        if len(global_shape) <= axis:
            # ref :175-177
            raise IndexError(f"Axis {axis} out of range for DistributedArray "
                             f"of shape {global_shape}")
        if not isinstance(partition, Partition):
            # ref :178-180 ("partition not in Partition"; Python 3.10 enums
            # raise TypeError on non-member `in`, so spell it isinstance)
            raise ValueError(f"Should be either {Partition.BROADCAST}, "
                             f"{Partition.UNSAFE_BROADCAST} or "
                             f"{Partition.SCATTER}")
        self._engine = engine
        self._global_shape = global_shape
        self._base_comm = base_comm if base_comm is not None \
            else get_default_comm()
        self._partition = partition
        self._axis = axis
        # masked sub-communicator (ref :74-100,193-195): ranks sharing a
        # mask value form the group the 'global' reductions run over
        self._mask = None if mask is None else [int(v) for v in mask]
        if mask is None:
            self._sub_comm = self._base_comm
        else:
            if len(self._mask) != self._base_comm.size:
                raise ValueError("Mask length must equal communicator size")
            self._sub_comm = _get_subcomm(self._base_comm,
                                          tuple(self._mask))
        self.dtype = np.dtype(dtype) if local_array is None \
            else _NP_DTYPES[local_array.dtype]
        if local_shapes is not None:
            local_shapes = [tuple(int(v) for v in
                                  ((s,) if isinstance(s, Integral) else s))
                            for s in local_shapes]
            self._check_local_shapes(local_shapes)
            self._all_local_shapes = local_shapes
        else:
            self._all_local_shapes = [
                local_split(global_shape, self.size, r, partition, axis)
                for r in range(self.size)]
        self._local_shape = self._all_local_shapes[self.rank]

        if local_array is None:
            self._local_array = torch.empty(
                self._local_shape, dtype=as_torch_dtype(self.dtype),
                device=self.device)
        else:
            if tuple(local_array.shape) != self._local_shape:
                # ref :207-211
                raise ValueError(
                    f"local_array has shape {tuple(local_array.shape)}, "
                    f"expected {self._local_shape}")
            self._local_array = local_array

    # ------------------------------------------------------------ properties
    @property
    def global_shape(self):
        return self._global_shape

    @property
    def base_comm(self):
        return self._base_comm

    @property
    def local_shape(self):
        return self._local_shape

    @property
    def local_shapes(self):
        # ref :377-389 (allgather) — cached here, no comm needed
        return list(self._all_local_shapes)

    @property
    def local_array(self):
        return self._local_array

    @property
    def engine(self):
        return self._engine

    @property
    def rank(self):
        return self._base_comm.rank

    @property
    def size(self):
        return self._base_comm.size

    @property
    def axis(self):
        return self._axis

    @property
    def ndim(self):
        return len(self._global_shape)

    @property
    def partition(self):
        return self._partition

    @property
    def mask(self):
        return self._mask

    @property
    def sub_comm(self):
        # ref :392-399
        return self._sub_comm

    def _check_mask(self, other):
        # ref :581-585
        if not np.array_equal(self._mask, other._mask):
            raise ValueError("Mask of both the arrays must be same")

    @property
    def device(self):
        d = self._base_comm.device
        return d if d is not None else torch.device("cpu")

    # --------------------------------------------------------------- helpers
    def _require_compute(self):
        if self._local_array.device.type != "cuda":
            raise RuntimeError(
                "pam: compute ops require a CUDA (MI355X) device tensor — "
                "there is no CPU compute path")
        _ffi.lib()  # raises ImportError if the HIP extension is missing

    def _stream(self):
        return torch.cuda.current_stream(self._local_array.device).cuda_stream

    def _dt(self):
        return _ffi.dtype_code(self._local_array.dtype)

    def _flat(self) -> torch.Tensor:
        t = self._local_array
        return t.reshape(-1) if t.is_contiguous() else t.contiguous().view(-1)

    def _is_cplx(self) -> bool:
        return self._local_array.is_complex()

    def _ew(self, t: Optional[torch.Tensor] = None):
        """(flat, count, dtype_code) for element-wise kernels; complex
        arrays are handled as their 2n-float real view (valid for
        add/sub/neg/fill and REAL-alpha axpy/xpby/scale)."""
        t = self._local_array if t is None else t
        flat = t.reshape(-1) if t.is_contiguous() else t.contiguous().view(-1)
        if t.is_complex():
            rv = torch.view_as_real(flat).reshape(-1)
            return rv, rv.numel(), _ffi.dtype_code(rv.dtype)
        return flat, flat.numel(), _ffi.dtype_code(t.dtype)

    def _like(self, local: Optional[torch.Tensor] = None) -> "DistributedArray":
        return DistributedArray(self._global_shape, self._base_comm,
                                self._partition, self._axis,
                                local_array=local,
                                local_shapes=self._all_local_shapes,
                                mask=self._mask,
                                engine=self._engine, dtype=self.dtype)

    def _check_partition_shape(self, other):
        # ref :572-579
        if self._partition != other._partition:
            raise ValueError("Partition of both the arrays must be same")
        if self._local_shape != other._local_shape:
            raise ValueError(f"Local Array Shape Mismatch - "
                             f"{self._local_shape} != {other._local_shape}")

    def _check_local_shapes(self, local_shapes):
        # ref :554-570 (validated locally — shapes of all ranks are known)
        if len(local_shapes) != self.size:
            raise ValueError(
                f"Length of local shapes is not equal to number of "
                f"processes; {len(local_shapes)} != {self.size}")
        if self._partition in (Partition.BROADCAST,
                               Partition.UNSAFE_BROADCAST):
            if local_shapes[self.rank] != self._global_shape:
                raise ValueError(
                    f"Local shape is not equal to global shape at rank = "
                    f"{self.rank};{local_shapes[self.rank]} != "
                    f"{self._global_shape}")
        else:
            total = sum(s[self._axis] for s in local_shapes)
            ls = local_shapes[self.rank]
            ok_other = (np.array_equal(np.delete(ls, self._axis),
                                       np.delete(self._global_shape,
                                                 self._axis)))
            if total != self._global_shape[self._axis] or not ok_other:
                raise ValueError(
                    f"Local shapes don't align with the global shape;"
                    f"{local_shapes} != {self._global_shape}")

    # -------------------------------------------------------------- indexing
    def __getitem__(self, index):
        return self._local_array[index]

    def __setitem__(self, index, value):
        # ref :217-252 — BROADCAST re-broadcasts rank 0's assignment
        if self._partition is Partition.BROADCAST and self.size > 1:
            view = self._local_array[index]
            buf = torch.empty_like(view)
            if self.rank == 0:
                buf[...] = self._coerce(value, buf)
            self._base_comm.broadcast_(buf, root=0)
            self._local_array[index] = buf
        else:
            self._local_array[index] = self._coerce(
                value, self._local_array[index])

    @staticmethod
    def _coerce(value, like: torch.Tensor):
        if isinstance(value, torch.Tensor):
            return value.to(like.device, like.dtype)
        if isinstance(value, np.ndarray):
            return torch.as_tensor(value, dtype=like.dtype,
                                   device=like.device)
        return value

    # ----------------------------------------------------------- scatter etc
    @classmethod
    def to_dist(cls, x, base_comm: Optional[PamComm] = None,
                partition: Partition = Partition.SCATTER, axis: int = 0,
                local_shapes: Optional[List[Tuple]] = None,
                mask=None) -> "DistributedArray":
        """Scatter a globally-replicated array (ref :438-491 — every rank
        holds ``x`` and slices its own block by the cumsum rule)."""
        comm = base_comm if base_comm is not None else get_default_comm()
        if isinstance(x, np.ndarray):
            x = torch.as_tensor(x, device=comm.device)
        arr = cls(tuple(x.shape), comm, partition, axis,
                  local_shapes=local_shapes, mask=mask,
                  dtype=_NP_DTYPES[x.dtype])
        if partition in (Partition.BROADCAST, Partition.UNSAFE_BROADCAST):
            arr[:] = x
        else:
            counts = [s[axis] for s in arr._all_local_shapes]
            start = int(np.sum(counts[: arr.rank], initial=0))
            sl = [slice(None)] * x.ndim
            sl[axis] = slice(start, start + counts[arr.rank])
            arr[:] = x[tuple(sl)]
        return arr

    def asarray(self, masked: bool = False) -> torch.Tensor:
        """Gathered global view (ref :401-436).  Returns a torch tensor on
        this rank's device."""
        if self._partition in (Partition.BROADCAST,
                               Partition.UNSAFE_BROADCAST):
            return self._local_array
        if masked:
            comm = self._sub_comm
            shapes = [self._all_local_shapes[r] for r in comm.ranks]
        else:
            comm = self._base_comm
            shapes = self._all_local_shapes
        pieces = comm.allgather_tensors(
            self._local_array.contiguous(), shapes)
        return torch.cat(pieces, dim=self._axis)

    # ------------------------------------------------------------ halo moves
    def halo_exchange(self, width: int):
        """Exchange ``width`` boundary planes with the axis-0 neighbours.

        Returns ``(gf, gb)``: gf = last ``width`` planes of rank-1 (None at
        rank 0), gb = first ``width`` planes of rank+1 (None at the last
        rank).  This carries the same values as the reference's
        add_ghost_cells (ref :955-1032) without materializing the
        concatenated copy."""
        if self._axis != 0:
            raise NotImplementedError("halo_exchange requires axis=0")
        r, P = self.rank, self.size
        t = self._local_array
        nloc = t.shape[0]
        if width == 0 or P == 1:
            return None, None
        # ref :996-1002 sender-side guard, made deterministic: every rank
        # with a neighbour sends ``width`` planes, and all ranks know all
        # local shapes, so raise identically everywhere instead of one
        # rank posting a mismatched irecv
        for rr in range(P):
            if self._all_local_shapes[rr][0] < width:
                raise ValueError(
                    f"Local Shape at rank={rr} along axis=0 should be > "
                    f"{width}: dim(0) {self._all_local_shapes[rr][0]} < "
                    f"{width}; to achieve this use NUM_PROCESSES <= "
                    f"{max(1, self._global_shape[0] // width)}")
        send_prev = t[:width].contiguous() if r > 0 else None
        send_next = t[-width:].contiguous() if r < P - 1 else None
        gf = torch.empty_like(t[:width]) if r > 0 else None
        gb = torch.empty_like(t[:width]) if r < P - 1 else None
        self._base_comm.sendrecv_neighbors(send_prev, send_next, gf, gb)
        return gf, gb

    def add_ghost_cells(self, cells_front: Optional[int] = None,
                        cells_back: Optional[int] = None) -> torch.Tensor:
        """Reference-shaped ghosting (ref :955-1032): returns the local
        array with neighbour planes concatenated.  Kept for surface parity;
        operators use :meth:`halo_exchange` instead."""
        r, P = self.rank, self.size
        t = self._local_array
        parts = [t]
        if cells_front is not None and cells_front > 0 and P > 1:
            # value parity with ref :976-1005 for uniform requests
            if r > 0:
                gf = torch.empty((cells_front,) + tuple(t.shape[1:]),
                                 dtype=t.dtype, device=t.device)
            else:
                gf = None
            # senders are ranks < P-1; validate ALL of them on every rank
            # so the raise is deterministic (ref :996-1002)
            for rr in range(P - 1):
                if cells_front > self._all_local_shapes[rr][self._axis]:
                    raise ValueError(
                        f"Local Shape at rank={rr} along axis={self._axis} "
                        f"should be > {cells_front}")
            self._base_comm.sendrecv_neighbors(
                None, t[-cells_front:].contiguous() if r < P - 1 else None,
                gf, None)
            if gf is not None:
                parts.insert(0, gf)
        if cells_back is not None and cells_back > 0 and P > 1:
            if r < P - 1:
                gb = torch.empty((cells_back,) + tuple(t.shape[1:]),
                                 dtype=t.dtype, device=t.device)
            else:
                gb = None
            # senders are ranks > 0; validate ALL of them on every rank
            for rr in range(1, P):
                if cells_back > self._all_local_shapes[rr][self._axis]:
                    raise ValueError(
                        f"Local Shape at rank={rr} along axis={self._axis} "
                        f"should be > {cells_back}")
            self._base_comm.sendrecv_neighbors(
                t[:cells_back].contiguous() if r > 0 else None, None,
                None, gb)
            if gb is not None:
                parts.append(gb)
        return torch.cat(parts, dim=0) if len(parts) > 1 else t.clone()

    # ------------------------------------------------------------- math (HIP)
    def __neg__(self):
        self._require_compute()
        out = self._like()
        a, n, dt = self._ew()
        o, _, _ = self._ew(out._local_array)
        _ffi.checked(_ffi.lib().pam_neg(
            self._stream(), o.data_ptr(), a.data_ptr(), n, dt), "neg")
        return out

    def add(self, other: "DistributedArray") -> "DistributedArray":
        self._check_mask(other)
        # ref :636-651
        self._check_partition_shape(other)
        self._require_compute()
        out = self._like()
        a, n, dt = self._ew()
        b, _, _ = self._ew(other._local_array)
        o, _, _ = self._ew(out._local_array)
        _ffi.checked(_ffi.lib().pam_add(
            self._stream(), o.data_ptr(), a.data_ptr(), b.data_ptr(), n, dt),
            "add")
        return out

    def iadd(self, other: "DistributedArray") -> "DistributedArray":
        self._check_mask(other)
        # ref :653-659
        self._check_partition_shape(other)
        self._require_compute()
        a, n, dt = self._ew()
        b, _, _ = self._ew(other._local_array)
        _ffi.checked(_ffi.lib().pam_axpy(
            self._stream(), a.data_ptr(), b.data_ptr(), 1.0, n, dt), "iadd")
        return self

    def sub(self, other: "DistributedArray") -> "DistributedArray":
        # bitwise equal to ref __sub__ = add(-other) (ref :624-625),
        # including its mask validation (add() raises on mismatch)
        self._check_mask(other)
        self._check_partition_shape(other)
        self._require_compute()
        out = self._like()
        a, n, dt = self._ew()
        b, _, _ = self._ew(other._local_array)
        o, _, _ = self._ew(out._local_array)
        _ffi.checked(_ffi.lib().pam_sub(
            self._stream(), o.data_ptr(), a.data_ptr(), b.data_ptr(), n, dt),
            "sub")
        return out

    def multiply(self, x) -> "DistributedArray":
        # ref :661-683
        self._require_compute()
        out = self._like()
        if isinstance(x, DistributedArray):
            self._check_partition_shape(x)
            self._check_mask(x)
            if self._is_cplx():
                _ffi.checked(_ffi.lib().pam_cmul(
                    self._stream(), out._flat().data_ptr(),
                    self._flat().data_ptr(), x._flat().data_ptr(),
                    self._local_array.numel(), self._dt()), "cmul")
            else:
                _ffi.checked(_ffi.lib().pam_mul(
                    self._stream(), out._flat().data_ptr(),
                    self._flat().data_ptr(), x._flat().data_ptr(),
                    self._local_array.numel(), self._dt()), "mul")
        elif self._is_cplx() and isinstance(x, complex) and x.imag != 0.0:
            _ffi.checked(_ffi.lib().pam_cscale(
                self._stream(), out._flat().data_ptr(),
                self._flat().data_ptr(), float(x.real), float(x.imag),
                self._local_array.numel(), self._dt()), "cscale")
        else:
            # real scalar: componentwise on the (possibly complex) view
            a, n, dt = self._ew()
            o, _, _ = self._ew(out._local_array)
            _ffi.checked(_ffi.lib().pam_scale(
                self._stream(), o.data_ptr(), a.data_ptr(),
                float(np.real(x)), n, dt), "scale")
        return out

    # fused solver updates (not in the reference surface — the reference
    # allocates 2 temporaries per axpy, ref cls_basic.py:390-391 via
    # :618-683; these are the one-pass HIP equivalents)
    def iaxpy_(self, alpha: float, x: "DistributedArray") -> "DistributedArray":
        """self += alpha * x (REAL alpha), fused — componentwise on the
        real view for complex arrays (CG/CGLS scalars are real,
        ref cls_basic.py:389-395)."""
        self._check_mask(x)
        self._check_partition_shape(x)
        self._require_compute()
        a, n, dt = self._ew()
        b, _, _ = self._ew(x._local_array)
        _ffi.checked(_ffi.lib().pam_axpy(
            self._stream(), a.data_ptr(), b.data_ptr(), float(alpha), n, dt),
            "axpy")
        return self

    def xpby_(self, x: "DistributedArray", beta: float) -> "DistributedArray":
        """self = x + beta * self (REAL beta), fused (CGLS c = r + b*c)."""
        self._check_mask(x)
        self._check_partition_shape(x)
        self._require_compute()
        a, n, dt = self._ew()
        b, _, _ = self._ew(x._local_array)
        _ffi.checked(_ffi.lib().pam_xpby(
            self._stream(), a.data_ptr(), b.data_ptr(), float(beta), n, dt),
            "xpby")
        return self

    # device-scalar solver fast path: the scalar lives in a caller-owned
    # 1-element f64 CUDA buffer written by a prior dot_into / pam_scalar_*
    # launch on the same stream, so a CG/CGLS iteration needs one host sync
    # (the stop test) instead of one per dot (see solvers.py).
    def iaxpy_dev_(self, alpha_t: torch.Tensor, x: "DistributedArray",
                   scale: float = 1.0) -> "DistributedArray":
        """self += scale * alpha_t[0] * x (bit-identical to iaxpy_)."""
        self._require_compute()
        a, n, dt = self._ew()
        b, _, _ = self._ew(x._local_array)
        _ffi.checked(_ffi.lib().pam_axpy_d(
            self._stream(), a.data_ptr(), b.data_ptr(), alpha_t.data_ptr(),
            float(scale), n, dt), "axpy_d")
        return self

    def xpby_dev_(self, x: "DistributedArray", beta_t: torch.Tensor,
                  scale: float = 1.0) -> "DistributedArray":
        """self = x + scale * beta_t[0] * self (bit-identical to xpby_)."""
        self._require_compute()
        a, n, dt = self._ew()
        b, _, _ = self._ew(x._local_array)
        _ffi.checked(_ffi.lib().pam_xpby_d(
            self._stream(), a.data_ptr(), b.data_ptr(), beta_t.data_ptr(),
            float(scale), n, dt), "xpby_d")
        return self

    def dot_into(self, other: "DistributedArray",
                 out_t: torch.Tensor) -> torch.Tensor:
        """Local dot reduced into ``out_t`` (1-elem f64 device slice); the
        global allreduce is the CALLER's, so the solver can batch several
        dots into one collective.  Real SCATTER arrays only."""
        self._require_compute()
        ws, _ = _reduce_buffers(self._local_array.device)
        _ffi.checked(_ffi.lib().pam_dot(
            self._stream(), self._flat().data_ptr(), other._flat().data_ptr(),
            self._local_array.numel(), ws.data_ptr(), out_t.data_ptr(),
            self._dt()), "dot")
        return out_t

    def __add__(self, x):
        return self.add(x)

    def __iadd__(self, x):
        return self.iadd(x)

    def __sub__(self, x):
        return self.sub(x)

    def __isub__(self, x):
        return self.iaxpy_(-1.0, x)

    def __mul__(self, x):
        return self.multiply(x)

    def __rmul__(self, x):
        return self.multiply(x)

    # --------------------------------------------------------- reductions
    def dot(self, other: "DistributedArray", vdot: bool = False):
        """ref :685-717 — local dot + allreduce.  Real dtypes: vdot == dot."""
        self._check_partition_shape(other)
        self._check_mask(other)
        self._require_compute()
        if self._partition in (Partition.BROADCAST,
                               Partition.UNSAFE_BROADCAST):
            x = DistributedArray.to_dist(self._local_array, self._base_comm)
            y = DistributedArray.to_dist(other._local_array, self._base_comm)
            return x.dot(y, vdot=vdot)
        ws, out = _reduce_buffers(self._local_array.device)
        if self._is_cplx():
            _ffi.checked(_ffi.lib().pam_cdot(
                self._stream(), self._flat().data_ptr(),
                other._flat().data_ptr(), self._local_array.numel(),
                1 if vdot else 0, ws.data_ptr(), out.data_ptr(),
                self._dt()), "cdot")
            self._sub_comm.allreduce_(out, "sum")
            v = out.cpu()
            return np.complex128(complex(float(v[0]), float(v[1])))
        _ffi.checked(_ffi.lib().pam_dot(
            self._stream(), self._flat().data_ptr(), other._flat().data_ptr(),
            self._local_array.numel(), ws.data_ptr(), out.data_ptr(),
            self._dt()), "dot")
        self._sub_comm.allreduce_(out[:1], "sum")
        return np.float64(out[0].item())

    def _norm_local(self, op: int, p: float) -> torch.Tensor:
        ws, out = _reduce_buffers(self._local_array.device)
        _ffi.checked(_ffi.lib().pam_norm_local(
            self._stream(), self._flat().data_ptr(),
            self._local_array.numel(), op, p, ws.data_ptr(), out.data_ptr(),
            self._dt()), "norm")
        return out[:1]

    def norm(self, ord: Optional[float] = None,
             axis: Optional[int] = None):
        """ref :805-838 (axis=None: flattened vector norm via the HIP
        reduction kernels, ref :719-789; axis given: per-axis norms —
        np.linalg.norm semantics, pinned by the reference's own test
        tests/test_distributedarray.py:215-222)."""
        if axis is not None:
            return self._norm_axis(ord, int(axis))
        self._require_compute()
        if self._partition in (Partition.BROADCAST,
                               Partition.UNSAFE_BROADCAST):
            return DistributedArray.to_dist(
                self._local_array, self._base_comm).norm(ord)
        ord = 2 if ord is None else ord
        if ord in ("fro", "nuc"):
            raise ValueError(f"norm-{ord} not possible for vectors")
        if ord == 0:
            out = self._norm_local(3, 0.0)
            self._sub_comm.allreduce_(out, "sum")
            return np.float64(out.item())
        if ord == np.inf:
            out = self._norm_local(1, 0.0)
            self._sub_comm.allreduce_(out, "max")
            return np.float64(out.item())
        if ord == -np.inf:
            out = self._norm_local(2, 0.0)
            self._sub_comm.allreduce_(out, "min")
            return np.float64(out.item())
        out = self._norm_local(0, float(ord))
        self._sub_comm.allreduce_(out, "sum")
        return np.float64(out.item() ** (1.0 / ord))

    def _norm_axis(self, ord, axis: int) -> torch.Tensor:
        """Per-axis norms (ref :719-771 _compute_vector_norm + :828-838).
        Returns the GLOBAL result tensor on every rank, matching
        np.linalg.norm(x_global, ord, axis=axis) — small reduced outputs,
        computed with torch device ops + one collective."""
        if axis >= self.ndim:
            raise ValueError(f"axis={axis} is out of range for array of "
                             f"dimension {self.ndim}")
        if self._partition in (Partition.BROADCAST,
                               Partition.UNSAFE_BROADCAST):
            return DistributedArray.to_dist(
                self._local_array, self._base_comm)._norm_axis(ord, axis)
        t = self._local_array
        if self._axis != axis:
            # the reduced axis is local: per-rank norms, allgathered and
            # stitched along the (shifted) distributed axis (ref :830-836)
            norm_axis = self._axis - 1 if axis < self._axis else self._axis
            loc = torch.linalg.vector_norm(
                t, ord=(2 if ord is None else ord), dim=axis).to(
                    torch.float64)
            shapes = [tuple(s[:axis] + s[axis + 1:])
                      for s in self._all_local_shapes]
            parts = self._base_comm.allgather_tensors(loc.contiguous(),
                                                      shapes)
            return torch.cat([p.reshape(s) for p, s in zip(parts, shapes)],
                             dim=norm_axis)
        # the reduced axis IS the distributed axis: elementwise
        # cross-rank reduction (ref :736-771)
        ord = 2 if ord is None else ord
        if ord in ("fro", "nuc"):
            raise ValueError(f"norm-{ord} not possible for vectors")
        if ord == 0:
            out = torch.count_nonzero(t, dim=axis).to(torch.float64)
            self._sub_comm.allreduce_(out, "sum")
            return out
        if ord == np.inf:
            out = torch.amax(torch.abs(t), dim=axis).to(torch.float64)
            self._sub_comm.allreduce_(out, "max")
            return out
        if ord == -np.inf:
            out = torch.amin(torch.abs(t), dim=axis).to(torch.float64)
            self._sub_comm.allreduce_(out, "min")
            return out
        out = torch.sum(torch.abs(torch.float_power(t, ord)),
                        dim=axis).to(torch.float64)
        self._sub_comm.allreduce_(out, "sum")
        return torch.pow(out, 1.0 / ord)

    # ------------------------------------------------------------ structure
    def conj(self):
        # ref :840-854.  Real dtypes: conj is the identity, so (following
        # torch.conj) the result SHARES storage instead of copying — the
        # reference clones 4.3 GB per CGLS dot here (r.dot(r.conj()),
        # ref cls_basic.py:389-401); treat the result as read-only.
        if not self._is_cplx():
            return self._like(self._local_array)
        self._require_compute()
        out = self._like()
        _ffi.checked(_ffi.lib().pam_conj(
            self._stream(), out._flat().data_ptr(), self._flat().data_ptr(),
            self._local_array.numel(), self._dt()), "conj")
        return out

    def copy(self):
        return self._like(self._local_array.clone())

    def zeros_like(self):
        # ref :791-803 (NB: the reference drops the mask here)
        out = DistributedArray(self._global_shape, self._base_comm,
                               self._partition, self._axis,
                               local_shapes=self._all_local_shapes,
                               engine=self._engine, dtype=self.dtype)
        out._local_array.zero_()
        return out

    def empty_like(self):
        return self._like()

    def ravel(self, order: str = "C"):
        # ref :872-897
        local_shapes = [(int(np.prod(s)),) for s in self._all_local_shapes]
        flat = self._local_array.reshape(-1)
        return DistributedArray(
            int(np.prod(self._global_shape)), self._base_comm,
            self._partition, 0, local_array=flat,
            local_shapes=local_shapes, engine=self._engine, dtype=self.dtype)

    def reshape(self, local_shape, axis: int = 0):
        """ref :899-944 — re-view the local block; the new global shape is
        stitched from the allgathered local shapes."""
        local_shape = tuple(int(v) for v in (
            (local_shape,) if isinstance(local_shape, Integral)
            else local_shape))
        local_shapes = [tuple(int(v) for v in s) for s in
                        self._base_comm.allgather_obj(local_shape)]
        ref_shape = local_shapes[0]
        if self._partition is Partition.SCATTER:
            if (local_shape[:axis] != ref_shape[:axis]
                    or local_shape[axis + 1:] != ref_shape[axis + 1:]):
                raise ValueError(
                    f"All local shapes must match on every axis except "
                    f"axis={axis}. Got {local_shape} in rank {self.rank} "
                    f"and {ref_shape} in rank 0.")
            global_shape = list(ref_shape)
            global_shape[axis] = sum(ls[axis] for ls in local_shapes)
        else:
            if local_shape != ref_shape:
                raise ValueError(
                    "All local shapes must be identical for "
                    "Partition.BROADCAST and Partition.UNSAFE_BROADCAST. "
                    f"Got {local_shape} in rank {self.rank} and "
                    f"{ref_shape} in rank 0.")
            global_shape = list(ref_shape)
        return DistributedArray(
            tuple(global_shape), self._base_comm, self._partition, axis,
            local_array=self._local_array.reshape(local_shapes[self.rank]),
            local_shapes=local_shapes, mask=self._mask,
            engine=self._engine, dtype=self.dtype)

    def redistribute(self, axis: int):
        """ref :493-552 — all-to-all realignment of the distribution axis
        (the FFT pencil transpose).  The result is balanced along the new
        axis even if this array is unbalanced (ref :504-506).  The
        reference serializes P pairwise sendrecvs (ref :534-551); here
        every (src,dst) block is posted in ONE batched RCCL isend/irecv
        group so xGMI links run concurrently."""
        if self._axis == axis or self._partition is not Partition.SCATTER:
            return self
        if self.size == 1:
            # same block, new distribution label: zero-copy re-wrap
            return DistributedArray(
                self._global_shape, self._base_comm, self._partition, axis,
                local_array=self._local_array, mask=self._mask,
                engine=self._engine, dtype=self.dtype)
        counts_from = [s[self._axis] for s in self._all_local_shapes]
        counts_to = [local_split(self._global_shape, self.size, r,
                                 Partition.SCATTER, axis)[axis]
                     for r in range(self.size)]
        offs = np.cumsum([0] + counts_to[:-1])
        sends, recvs = [], []
        for r in range(self.size):
            sl = [slice(None)] * self._local_array.ndim
            sl[axis] = slice(int(offs[r]), int(offs[r]) + counts_to[r])
            sends.append(self._local_array[tuple(sl)].contiguous())
            shp = list(self._global_shape)
            shp[self._axis] = counts_from[r]
            shp[axis] = counts_to[self.rank]
            recvs.append(torch.empty(
                shp, dtype=self._local_array.dtype,
                device=self._local_array.device))
        recvs[self.rank].copy_(sends[self.rank])
        self._base_comm.exchange(
            [(t, r) for r, t in enumerate(sends) if r != self.rank],
            [(t, r) for r, t in enumerate(recvs) if r != self.rank])
        return DistributedArray(
            self._global_shape, self._base_comm, self._partition, axis,
            local_array=torch.cat(recvs, dim=self._axis), mask=self._mask,
            engine=self._engine, dtype=self.dtype)

    def __repr__(self):
        return (f"<DistributedArray with global shape={self.global_shape}, "
                f"local shape={self.local_shape}, dtype={self.dtype}, "
                f"processes={list(range(self.size))})> ")
