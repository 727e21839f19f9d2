"""MPINonStationaryConvolve1D — distributed non-stationary 1-D convolution.

Drop-in for /root/reference/pylops_mpi/signalprocessing/
NonStatConvolve1d.py:12-168: the model is halo-padded so each rank's
extended block covers the neighbouring filters, a serial non-stationary
convolution (HIP pam_nsconv, filters linearly interpolated between
regularly-sampled anchors) runs per rank inside MPIBlockDiag, and the
halo's adjoint strips the pad:  Op = Halo.H @ BlockDiag([conv]) @ Halo.
"""
import math
from typing import Optional, Tuple, Union

import numpy as np
import torch

from . import _ffi
from .blockdiag import MPIBlockDiag
from .comm import PamComm, get_default_comm
from .halo import MPIHalo
from .localops import LocalOperator


def halo_block_split(global_shape: Tuple, comm: PamComm,
                     grid_shape: Optional[Tuple] = None) -> Tuple:
    """ref Halo.py:12-66 — this rank's Cartesian slice (ceil blocks)."""
    ndim = len(global_shape)
    size = comm.size
    if grid_shape is None:
        grid_shape = (1,) * (ndim - 1) + (size,)
    if math.prod(grid_shape) != size:
        raise ValueError(
            f"grid_shape {grid_shape} does not match comm size {size}")
    coords = np.unravel_index(comm.rank, grid_shape)
    slices = []
    for gdim, procs, coord in zip(global_shape, grid_shape, coords):
        blk = math.ceil(gdim / procs)
        start = int(coord) * blk
        slices.append(slice(start, min(start + blk, gdim)))
    return tuple(slices)


class NonStationaryConvolve1DLocal(LocalOperator):
    """Serial non-stationary convolution along ``axis`` of ``dims``
    (the pylops NonStationaryConvolve1D the reference wraps; pylops is not
    vendored, so the interpolation convention is re-derived — see
    csrc/fdserial.hip pam_nsconv — and locked by dense-adjoint tests)."""

    def __init__(self, dims: Union[int, Tuple], hs: torch.Tensor, ih,
                 axis: int = -1, dtype=np.float64):
        self.dims = (dims,) if isinstance(dims, (int, np.integer)) \
            else tuple(int(v) for v in dims)
        axis = axis if axis >= 0 else len(self.dims) + axis
        self.axis = axis
        ih = np.asarray(ih)
        if hs.shape[1] % 2 == 0:
            raise ValueError("filters hs must have odd length")
        if len(np.unique(np.diff(ih))) > 1:
            raise ValueError(
                "the indices of filters 'ih' are must be regularly sampled")
        if min(ih) < 0 or max(ih) >= self.dims[axis]:
            raise ValueError(
                "the indices of filters 'ih' must be larger than 0 and "
                "smaller than `dims`")
        self.hs = hs.contiguous()
        self.oh = float(ih[0])
        self.dh = float(ih[1] - ih[0]) if len(ih) > 1 else 1.0
        self.dtype = np.dtype(dtype)
        n = int(np.prod(self.dims))
        self.shape = (n, n)
        self.batch = int(np.prod(self.dims[:axis], initial=1))
        self.d = self.dims[axis]
        self.m = int(np.prod(self.dims[axis + 1:], initial=1))

    def _run(self, x: torch.Tensor, forward: bool) -> torch.Tensor:
        if x.device.type != "cuda":
            raise RuntimeError(
                "pam: compute ops require a CUDA (MI355X) device tensor — "
                "there is no CPU compute path")
        flat = x.reshape(-1).contiguous()
        y = torch.empty_like(flat)
        stream = torch.cuda.current_stream(x.device).cuda_stream
        _ffi.checked(_ffi.lib().pam_nsconv(
            stream, 1 if forward else 0, flat.data_ptr(), y.data_ptr(),
            self.hs.data_ptr(), self.batch, self.d, self.m,
            self.hs.shape[0], self.hs.shape[1], self.oh, self.dh,
            _ffi.dtype_code(flat.dtype)), "nsconv")
        return y

    def matvec(self, x):
        return self._run(x, True)

    def rmatvec(self, x):
        return self._run(x, False)


def MPINonStationaryConvolve1D(dims, hs: torch.Tensor, ih, axis: int = -1,
                               base_comm: Optional[PamComm] = None,
                               dtype="float64",
                               _local_factory=NonStationaryConvolve1DLocal):
    """ref NonStatConvolve1d.py:12-168 (``_local_factory`` is a test seam
    for the CPU gloo suite)."""
    comm = base_comm if base_comm is not None else get_default_comm()
    rank, size = comm.rank, comm.size
    dims = (dims,) if isinstance(dims, (int, np.integer)) else tuple(dims)
    axis = axis if axis >= 0 else len(dims) + axis
    ih = np.asarray(ih)
    hs_np = hs if isinstance(hs, np.ndarray) else hs.cpu().numpy()
    if hs_np.shape[1] % 2 == 0:
        raise ValueError("filters hs must have odd length")
    if len(np.unique(np.diff(ih))) > 1:
        raise ValueError(
            "the indices of filters 'ih' are must be regularly sampled")
    if min(ih) < 0 or max(ih) >= dims[axis]:
        raise ValueError(
            "the indices of filters 'ih' must be larger than 0 and "
            "smaller than `dims`")
    if dims[axis] % size:
        raise ValueError(
            f"number of input samples {dims[0]} is not divisible by "
            f"the number of ranks ({size})")
    # halo sizing, ref :101-117: distance to the closest out-of-partition
    # filter plus half the filter support
    dims_local = dims[axis] // size
    starts_local = np.arange(0, dims[axis], dims_local)
    start_local = int(starts_local[rank])
    end_local = start_local + dims_local - 1
    ihidx_local = np.where((ih >= start_local) & (ih <= end_local))[0]
    if len(ihidx_local) == 0:
        raise ValueError(f"rank {rank} has zerof filters!")
    ihdiff = int(np.diff(ih)[0]) if len(ih) > 1 else 1
    ih_local = ih[ihidx_local]
    dist_start_local = 0 if rank == 0 \
        else ihdiff - (int(ih_local[0]) - start_local)
    dist_end_local = 0 if rank == (size - 1) \
        else ihdiff - (end_local - int(ih_local[-1]))
    dists = comm.allgather_obj((dist_start_local, dist_end_local))
    dist_start = max(d[0] for d in dists)
    dist_end = max(d[1] for d in dists)
    halo = max(dist_start, dist_end) + (hs_np.shape[1] // 2 + 1)
    if size == 1:
        halo = 0  # ref :118-120
    proc_grid_shape = [1] * len(dims)
    proc_grid_shape[axis] = size
    HOp = MPIHalo(dims=dims, halo=halo, proc_grid_shape=proc_grid_shape,
                  comm=comm, dtype=dtype)
    # per-rank serial operator on the extended block with the filter
    # subset in extended-block coordinates, ref :129-165
    dims_ns = list(dims)
    if size == 1:
        dims_ns[axis] = dims_local + halo
        COp = _local_factory(dims=tuple(dims_ns), hs=hs, ih=ih, axis=axis,
                             dtype=dtype)
    else:
        x_slice = halo_block_split(
            dims if len(dims) == 1 else (dims[axis],), comm, (size,))
        if rank == 0:
            dims_ns[axis] = dims_local + halo
            COp = _local_factory(
                dims=tuple(dims_ns), hs=hs[: int(ihidx_local[-1]) + 2],
                ih=ih[: int(ihidx_local[-1]) + 2], axis=axis, dtype=dtype)
        elif rank == size - 1:
            dims_ns[axis] = dims_local + halo
            COp = _local_factory(
                dims=tuple(dims_ns), hs=hs[int(ihidx_local[0]) - 1:],
                ih=ih[int(ihidx_local[0]) - 1:] - x_slice[0].start + halo,
                axis=axis, dtype=dtype)
        else:
            dims_ns[axis] = dims_local + 2 * halo
            COp = _local_factory(
                dims=tuple(dims_ns),
                hs=hs[int(ihidx_local[0]) - 1: int(ihidx_local[-1]) + 2],
                ih=(ih[int(ihidx_local[0]) - 1: int(ihidx_local[-1]) + 2]
                    - x_slice[0].start + halo),
                axis=axis, dtype=dtype)
    COp_full = MPIBlockDiag([COp], base_comm=comm)
    return HOp.H @ COp_full @ HOp  # ref :166-168
