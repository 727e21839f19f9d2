"""Distributed N-D FFT operators — MPIFFTND / MPIFFT2D.

Restates ref signalprocessing/_baseffts.py:15-140 (parameter handling,
norms, frequency vectors, output dims/dtypes) and FFTND.py:154-316 /
FFT2D.py:147-176 (the transforms and the real-FFT sqrt(2) scaling).

The reference delegates the distributed transform to mpi4py-fft's PFFT
pencils (an OPTIONAL extra, absent from this stack).  MI355X-native
equivalent: local transforms run on rocFFT via torch.fft; whenever the
next transform axis is the distributed axis, the array is realigned with
ONE batched RCCL all-to-all (DistributedArray.redistribute — the pencil
transpose), keeping the canonical numpy rfftn/irfftn ordering (real
transform on ``axes[-1]`` first in the forward, last in the adjoint).
As in the reference, the flattened output is redistributed back to
axis 0 before raveling (ref utils/decorators.py:79-82), so internal
pencil layouts never leak to the caller.
"""
import warnings
from numbers import Integral
from typing import Optional, Sequence, Union

import numpy as np
import torch

from .comm import PamComm
from .distributedarray import DistributedArray, Partition
from .fft_helper import fftshift_nd, ifftshift_nd
from .linearoperator import MPILinearOperator
from .rebalance import rebalance_1d

_REAL_OF = {np.dtype(np.complex128): np.float64,
            np.dtype(np.complex64): np.float32,
            np.dtype(np.float64): np.float64,
            np.dtype(np.float32): np.float32}
_CPLX_OF = {np.dtype(np.float64): np.complex128,
            np.dtype(np.float32): np.complex64,
            np.dtype(np.complex128): np.complex128,
            np.dtype(np.complex64): np.complex64}


def _as_int_tuple(v) -> tuple:
    if isinstance(v, Integral):
        return (int(v),)
    return tuple(int(x) for x in v)


def _as_flag_array(v, n: int, name: str) -> np.ndarray:
    if isinstance(v, (bool, np.bool_)):
        out = np.full(n, bool(v))
    else:
        out = np.asarray([bool(x) for x in v])
    return out


class _MPIBaseFFTND(MPILinearOperator):
    """ref signalprocessing/_baseffts.py:15-140."""

    def __init__(self, dims, axes=None, sampling=1.0, norm="none",
                 real=False, ifftshift_before=False, fftshift_after=False,
                 dtype="complex128",
                 base_comm: Optional[PamComm] = None):
        dims = _as_int_tuple(dims)
        self.dims = tuple(dims)
        self.ndim = len(dims)
        axes = _as_int_tuple(axes)
        for d in axes:
            if not -self.ndim <= d < self.ndim:
                raise ValueError(f"axis {d} is out of bounds for array of "
                                 f"dimension {self.ndim}")
        self.axes = np.array([d % self.ndim for d in axes])
        self.naxes = len(self.axes)
        if self.naxes != len(np.unique(self.axes)):
            warnings.warn(
                "At least one direction is repeated. This may cause "
                "unexpected results.", stacklevel=2)
        self.nffts = tuple(int(dims[d]) for d in self.axes)
        if np.isscalar(sampling):
            sampling = np.full(self.naxes, float(sampling))
        else:
            sampling = np.asarray([float(s) for s in sampling])
        self.sampling = sampling
        self.ifftshift_before = _as_flag_array(ifftshift_before, self.naxes,
                                               "ifftshift_before")
        self.fftshift_after = _as_flag_array(fftshift_after, self.naxes,
                                             "fftshift_after")
        if (self.naxes != len(self.sampling)
                or self.naxes != len(self.ifftshift_before)
                or self.naxes != len(self.fftshift_after)):
            raise ValueError(
                "`axes`, `sampling`, `ifftshift_before` and "
                "`fftshift_after` must have the same number of elements. "
                f"Received {self.naxes}, {len(self.sampling)}, "
                f"{len(self.ifftshift_before)} and "
                f"{len(self.fftshift_after)}, respectively.")
        # norm handling, ref _baseffts.py:77-89 (exact error strings)
        if norm == "none":
            self.norm = "none"
        elif norm.lower() == "1/n":
            self.norm = "1/n"
        elif norm == "backward":
            raise ValueError(
                'To use no scaling on the forward transform, use "none". '
                "Note that in this case, the adjoint transform will *not* "
                "have a 1/n scaling.")
        elif norm == "forward":
            raise ValueError(
                'To use 1/n scaling on the forward transform, use "1/n". '
                "Note that in this case, the adjoint transform will *also* "
                "have a 1/n scaling.")
        else:
            raise ValueError(f"`norm`={norm} is not one of 'none' or '1/n'")
        self.real = real
        fs = [np.fft.fftshift(np.fft.fftfreq(n, d=s)) if shift
              else np.fft.fftfreq(n, d=s)
              for n, s, shift in zip(self.nffts, self.sampling,
                                     self.fftshift_after)]
        if self.real:
            fs[-1] = np.fft.rfftfreq(self.nffts[-1], d=self.sampling[-1])
            if self.fftshift_after[-1]:
                warnings.warn(
                    "Using real=True and fftshift_after on the last "
                    "direction.", stacklevel=2)
                fs[-1] = np.fft.fftshift(fs[-1])
        self.fs = tuple(fs)
        dimsd = np.array(dims)
        dimsd[self.axes] = self.nffts
        if self.real:
            dimsd[self.axes[-1]] = self.nffts[-1] // 2 + 1
        self.dimsd = tuple(int(d) for d in dimsd)
        dtype = np.dtype(dtype)
        self.rdtype = np.dtype(_REAL_OF[dtype]) if self.real else dtype
        self.cdtype = np.dtype(_CPLX_OF[dtype])
        self.clinear = not (self.real
                            or np.issubdtype(dtype, np.floating))
        super().__init__(dtype=self.cdtype, dims=tuple(self.dims),
                         dimsd=self.dimsd, base_comm=base_comm)


class MPIFFTND(_MPIBaseFFTND):
    """ref signalprocessing/FFTND.py:22-316 (see module docstring)."""

    def __init__(self, dims, axes=(0, 1, 2), sampling=1.0, norm="none",
                 real=False, ifftshift_before=False, fftshift_after=False,
                 dtype="complex128",
                 base_comm: Optional[PamComm] = None):
        super().__init__(dims=dims, axes=axes, sampling=sampling, norm=norm,
                         real=real, ifftshift_before=ifftshift_before,
                         fftshift_after=fftshift_after, dtype=dtype,
                         base_comm=base_comm)
        if self.ndim < 2 and self.size > 1:
            raise ValueError(
                "distributed FFTND requires at least 2 input dimensions")
        self._scale = float(np.prod(self.nffts)) if self.norm == "none" \
            else 1.0 / float(np.prod(self.nffts))

    # ------------------------------------------------------ reshaped I/O
    def _reshape_in(self, x: DistributedArray, shape,
                    copy: bool) -> DistributedArray:
        """The @reshaped input rebalance (ref utils/decorators.py:44-78):
        flat 1-D -> ``shape`` with a balanced axis-0 split.  When the
        split already matches and nothing downstream mutates in place
        (``copy=False``), the result is a zero-copy view of ``x``."""
        if x.partition is not Partition.SCATTER:
            raise ValueError(f"x should have partition={Partition.SCATTER}, "
                             f"{x.partition} != {Partition.SCATTER}")
        from .distributedarray import local_split
        size = x.base_comm.size
        lshapes = [local_split(tuple(shape), size, r, Partition.SCATTER, 0)
                   for r in range(size)]
        counts = [int(np.prod(s)) for s in lshapes]
        t = rebalance_1d(x, counts).reshape(lshapes[x.rank])
        if copy and t.data_ptr() == x.local_array.data_ptr():
            t = t.clone()
        return DistributedArray(tuple(shape), x.base_comm,
                                Partition.SCATTER, 0, local_array=t,
                                local_shapes=lshapes, engine="hip",
                                dtype=x.dtype)

    @staticmethod
    def _flatten_out(y: DistributedArray) -> DistributedArray:
        """ref utils/decorators.py:79-82: redistribute to axis 0, ravel."""
        return y.redistribute(axis=0).ravel()

    # ------------------------------------------------------- transforms
    def _other_axis(self, current: int, pending) -> int:
        """Pick the realignment target: any axis but ``current``,
        preferring one that needs no further transform."""
        cands = [a for a in range(self.ndim) if a != current]
        done = [a for a in cands if a not in pending]
        return (done or cands)[0]

    def _dist_fft(self, arr: DistributedArray, forward: bool
                  ) -> DistributedArray:
        """Transform over self.axes with rocFFT locals + all-to-all
        realignments; canonical rfftn/irfftn ordering."""
        last = int(self.axes[-1])
        pending = [int(a) for a in self.axes]
        t = arr.local_array
        world1 = arr.base_comm.size == 1
        if world1 and self.real:
            # one fused rocFFT real plan (canonical rfftn/irfftn order)
            dims_ = [int(a) for a in self.axes]
            if forward:
                t = torch.fft.rfftn(t, dim=dims_, norm="backward")
            else:
                t = torch.fft.irfftn(t, s=[self.nffts[-1]], dim=[last],
                                     norm="backward") \
                    if len(dims_) == 1 else torch.fft.irfftn(
                        t, s=[t.shape[a] for a in dims_[:-1]]
                        + [self.nffts[-1]], dim=dims_, norm="backward")
            return self._wrap(arr, t)
        while pending:
            # at world 1 every axis is local: one fused rocFFT plan
            cur = -1 if world1 else arr.axis
            if forward and self.real and last in pending:
                # real transform must run FIRST (np.fft.rfftn order)
                if cur == last:
                    arr = arr.redistribute(self._other_axis(cur, pending))
                    t, cur = arr.local_array, arr.axis
                t = torch.fft.rfft(t, n=self.nffts[-1], dim=last,
                                   norm="backward")
                pending.remove(last)
                arr = self._wrap(arr, t)
                continue
            # the real inverse transform (irfft) is NEVER part of the
            # batched complex ifftn — it runs in its dedicated branch below
            hold_last = (not forward) and self.real
            local_now = [a for a in pending
                         if a != cur and not (hold_last and a == last)]
            if local_now:
                if forward:
                    t = torch.fft.fftn(t, dim=local_now, norm="backward")
                else:
                    t = torch.fft.ifftn(t, dim=local_now, norm="backward")
                for a in local_now:
                    pending.remove(a)
                arr = self._wrap(arr, t)
                continue
            if not pending:
                break
            if pending == [last] and (not forward) and self.real:
                # real inverse transform runs LAST (np.fft.irfftn order)
                if cur == last:
                    arr = arr.redistribute(self._other_axis(cur, pending))
                    t = arr.local_array
                t = torch.fft.irfft(t, n=self.nffts[-1], dim=last,
                                    norm="backward")
                pending.remove(last)
                arr = self._wrap(arr, t)
                continue
            # remaining axis is the distributed one: realign and loop
            arr = arr.redistribute(self._other_axis(cur, pending))
            t = arr.local_array
        return arr

    def _wrap(self, like: DistributedArray, t: torch.Tensor
              ) -> DistributedArray:
        """Re-wrap a transformed local block (possibly new dtype/extent
        along a LOCAL axis) as a DistributedArray with the same
        distribution axis."""
        if like.base_comm.size == 1:
            # world 1: the local block IS the global array; the axis label
            # may even be the axis a real transform resized
            return DistributedArray(
                tuple(t.shape), like.base_comm, Partition.SCATTER,
                like.axis, local_array=t, engine="hip",
                dtype=_t2np(t.dtype))
        axis = like.axis
        gshape = list(like.global_shape)
        lshapes = [list(s) for s in like.local_shapes]
        for d in range(len(gshape)):
            if d == axis:
                continue
            if t.shape[d] != lshapes[like.rank][d]:
                gshape[d] = t.shape[d]
                for s in lshapes:
                    s[d] = t.shape[d]
        return DistributedArray(
            tuple(gshape), like.base_comm, Partition.SCATTER, axis,
            local_array=t, local_shapes=[tuple(s) for s in lshapes],
            engine="hip", dtype=_t2np(t.dtype))

    def _scale_real_fft(self, x: DistributedArray,
                        inverse: bool = False) -> None:
        """sqrt(2) scaling of the conjugate-twin bins along axes[-1]
        (ref FFTND.py:278-309, incl. the distributed-axis overlap)."""
        scale = 1 / np.sqrt(2) if inverse else np.sqrt(2)
        last = int(self.axes[-1])
        hi = 1 + (self.nffts[-1] - 1) // 2
        if x.axis == last:
            sizes = [s[last] for s in x.local_shapes]
            local_start = sum(sizes[:x.rank])
            local_stop = local_start + sizes[x.rank]
            fstart, fstop = max(1, local_start), min(hi, local_stop)
            if fstop > fstart:
                sl = [slice(None)] * x.ndim
                sl[last] = slice(fstart - local_start, fstop - local_start)
                x.local_array[tuple(sl)] *= scale
        else:
            sl = [slice(None)] * x.ndim
            sl[last] = slice(1, hi)
            x.local_array[tuple(sl)] *= scale

    # ----------------------------------------------------- matvec paths
    def _matvec(self, x: DistributedArray) -> DistributedArray:
        # ref FFTND.py:214-244
        arr = self._reshape_in(
            x, self.dims,
            copy=bool(self.ifftshift_before.any())
            or (not self.clinear and x.local_array.is_complex()))
        if self.ifftshift_before.any():
            arr = ifftshift_nd(
                arr, axes=[int(a) for a in
                           self.axes[self.ifftshift_before]])
        if not self.clinear and arr.local_array.is_complex():
            # the reference casts the complex carrier to REAL storage
            # before the real transform (FFTND.py:222-227 via the
            # real-dtype PFFT input array)
            arr = self._wrap(arr, arr.local_array.real.contiguous())
        y = self._dist_fft(arr, forward=True)
        if self.real:
            self._scale_real_fft(y, inverse=False)
        if self.norm == "1/n":
            y.local_array.mul_(self._scale)
        if np.dtype(_t2np(y.local_array.dtype)) != self.cdtype:
            y = self._wrap(y, y.local_array.to(_np2t(self.cdtype)))
        if self.fftshift_after.any():
            y = fftshift_nd(
                y, axes=[int(a) for a in self.axes[self.fftshift_after]])
        return self._flatten_out(y)

    def _rmatvec(self, x: DistributedArray) -> DistributedArray:
        # ref FFTND.py:246-276
        arr = self._reshape_in(
            x, self.dimsd,
            copy=bool(self.fftshift_after.any()) or self.real)
        if self.fftshift_after.any():
            arr = ifftshift_nd(
                arr, axes=[int(a) for a in self.axes[self.fftshift_after]])
        if self.real:
            self._scale_real_fft(arr, inverse=True)
        y = self._dist_fft(arr, forward=False)
        if self.norm == "none":
            y.local_array.mul_(self._scale)
        if not self.clinear and y.local_array.is_complex():
            y = self._wrap(y, y.local_array.real.contiguous())
        if np.dtype(_t2np(y.local_array.dtype)) != self.rdtype:
            y = self._wrap(y, y.local_array.to(_np2t(self.rdtype)))
        if self.ifftshift_before.any():
            y = fftshift_nd(
                y, axes=[int(a) for a in self.axes[self.ifftshift_before]])
        return self._flatten_out(y)

    def __truediv__(self, y) -> DistributedArray:
        # ref FFTND.py:311-316
        y_div = self._rmatvec(y)
        y_div.local_array.div_(self._scale)
        return y_div


class MPIFFT2D(MPIFFTND):
    """ref signalprocessing/FFT2D.py:11-176."""

    def __init__(self, dims, axes=(0, 1), sampling=1.0, norm="none",
                 real=False, ifftshift_before=False, fftshift_after=False,
                 dtype="complex128",
                 base_comm: Optional[PamComm] = None):
        if len(dims) < 2:
            raise ValueError("FFT2D requires at least two input dimensions")
        if len(_as_int_tuple(axes)) != 2:
            raise ValueError(
                "FFT2D must be applied along exactly two dimensions")
        super().__init__(dims=dims, axes=axes, sampling=sampling, norm=norm,
                         real=real, ifftshift_before=ifftshift_before,
                         fftshift_after=fftshift_after, dtype=dtype,
                         base_comm=base_comm)
        self.f1, self.f2 = self.fs
        del self.fs


_T2NP = {torch.float64: np.float64, torch.float32: np.float32,
         torch.complex128: np.complex128, torch.complex64: np.complex64}
_NP2T = {np.dtype(np.float64): torch.float64,
         np.dtype(np.float32): torch.float32,
         np.dtype(np.complex128): torch.complex128,
         np.dtype(np.complex64): torch.complex64}


def _t2np(td):
    return _T2NP[td]


def _np2t(npd):
    return _NP2T[np.dtype(npd)]
