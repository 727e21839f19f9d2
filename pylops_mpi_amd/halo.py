"""MPIHalo — N-dimensional Cartesian ghost-cell pad/exchange operator.

Drop-in for /root/reference/pylops_mpi/basicoperators/Halo.py:138-423:
forward embeds each rank's Cartesian block into a zero-padded extended
block and fills the ghost regions from the one-hop neighbours (per-axis
sequential exchanges so corners propagate, ref :390-395); adjoint strips
the ghosts.  The Cartesian topology is computed locally (C-order rank ->
coords, the Create_cart default) and the exchanges are batched RCCL
isend/irecv rounds.
"""
import math
from typing import Optional, Tuple, Union

import numpy as np
import torch

from .comm import PamComm, get_default_comm
from .distributedarray import DistributedArray, Partition, as_torch_dtype
from .linearoperator import MPILinearOperator


class MPIHalo(MPILinearOperator):

    def __init__(self, dims: Tuple, halo: Union[int, Tuple],
                 proc_grid_shape: Optional[Tuple] = None,
                 comm: Optional[PamComm] = None, dtype=np.float64):
        comm = comm if comm is not None else get_default_comm()
        self.global_dims = tuple(int(d) for d in dims)
        self.ndim = len(self.global_dims)
        self.comm = comm
        if proc_grid_shape is None:
            # ref :152-153: all ranks on the last axis
            proc_grid_shape = (1,) * (self.ndim - 1) + (comm.size,)
        self.proc_grid_shape = tuple(int(p) for p in proc_grid_shape)
        if math.prod(self.proc_grid_shape) != comm.size:
            raise ValueError(
                f"grid_shape {self.proc_grid_shape} does not match comm "
                f"size {comm.size}")
        self._coords = [self._unravel(r) for r in range(comm.size)]
        self.neigh = self._neighbors(comm.rank)
        self.halo = self._parse_halo(halo, comm.rank)
        self.local_dims = self._block_dims(comm.rank)
        self.local_extent = tuple(
            self.local_dims[ax] + self.halo[2 * ax] + self.halo[2 * ax + 1]
            for ax in range(self.ndim))
        self._validate_exchange_widths(halo)
        self._local_dim_sizes = [int(np.prod(self._block_dims(r)))
                                 for r in range(comm.size)]
        self._local_extent_sizes = [
            int(np.prod(self._extent_dims(r, halo)))
            for r in range(comm.size)]
        dimsd = (int(sum(self._local_extent_sizes)),)
        super().__init__(dims=self.global_dims, dimsd=dimsd,
                         dtype=np.dtype(dtype), base_comm=comm)

    # ------------------------------------------------------------- topology
    def _unravel(self, rank: int) -> Tuple[int, ...]:
        return tuple(int(c) for c in
                     np.unravel_index(rank, self.proc_grid_shape))

    def _ravel(self, coords) -> int:
        return int(np.ravel_multi_index(coords, self.proc_grid_shape))

    def _neighbors(self, rank: int):
        coords = self._coords[rank]
        neigh = {}
        for ax in range(self.ndim):
            for sgn, d in (("-", -1), ("+", 1)):
                c = list(coords)
                c[ax] += d
                neigh[(sgn, ax)] = (self._ravel(c)
                                    if 0 <= c[ax] < self.proc_grid_shape[ax]
                                    else None)  # MPI.PROC_NULL analogue
        return neigh

    def _parse_halo(self, h, rank: int) -> Tuple[int, ...]:
        # ref :197-227 (scalar halos are trimmed at global borders)
        neigh = self._neighbors(rank)
        if isinstance(h, (int, np.integer)):
            trimmed = [int(h)] * (2 * self.ndim)
            for ax in range(self.ndim):
                if trimmed[2 * ax] and neigh[("-", ax)] is None:
                    trimmed[2 * ax] = 0
                if trimmed[2 * ax + 1] and neigh[("+", ax)] is None:
                    trimmed[2 * ax + 1] = 0
            halo = tuple(trimmed)
        else:
            h = tuple(int(v) for v in h)
            if len(h) == 1:
                halo = h * (2 * self.ndim)
            elif len(h) == self.ndim:
                halo = sum(tuple((d, d) for d in h), ())
            elif len(h) == 2 * self.ndim:
                halo = h
            else:
                raise ValueError(
                    f"Invalid halo length {len(h)} for ndim={self.ndim}")
        if any(v < 0 for v in halo):
            raise ValueError("Halo widths must be non-negative")
        return halo

    def _block_dims(self, rank: int) -> Tuple[int, ...]:
        # ref :243-255 (ceil blocks, last block clipped)
        coords = self._coords[rank]
        out = []
        for gdim, coord, nproc in zip(self.global_dims, coords,
                                      self.proc_grid_shape):
            blk = math.ceil(gdim / nproc)
            start = coord * blk
            out.append(min(start + blk, gdim) - start)
        return tuple(out)

    def _extent_dims(self, rank: int, h) -> Tuple[int, ...]:
        halo = self._parse_halo(h, rank)
        ld = self._block_dims(rank)
        return tuple(ld[ax] + halo[2 * ax] + halo[2 * ax + 1]
                     for ax in range(self.ndim))

    def _validate_exchange_widths(self, h) -> None:
        # ref :280-318, computed locally (the grid layout is deterministic)
        for r in range(self.comm.size):
            halo = self._parse_halo(h, r)
            neigh = self._neighbors(r)
            ld = self._block_dims(r)
            for ax in range(self.ndim):
                before, after = halo[2 * ax], halo[2 * ax + 1]
                if before > ld[ax] and neigh[("-", ax)] is not None:
                    raise ValueError(
                        "MPIHalo halo widths are not supported by the "
                        "current one-hop exchange: halo width exceeds local "
                        "block size")
                if after > ld[ax] and neigh[("+", ax)] is not None:
                    raise ValueError(
                        "MPIHalo halo widths are not supported by the "
                        "current one-hop exchange: halo width exceeds local "
                        "block size")
                pn = neigh[("+", ax)]
                if pn is not None:
                    if after != self._parse_halo(h, pn)[2 * ax]:
                        raise ValueError(
                            "MPIHalo halo widths are not supported by the "
                            "current one-hop exchange: halo width does not "
                            "match neighbor halo width")

    # ------------------------------------------------------------- exchange
    def _exchange_along_axis(self, arr: torch.Tensor, axis: int,
                             before: int, after: int) -> None:
        """ref :320-360 — two pairwise swaps per axis, batched isend/irecv."""
        minus, plus = self.neigh[("-", axis)], self.neigh[("+", axis)]
        comm = self.comm
        sl = [slice(None)] * self.ndim
        sends, recvs, places = [], [], []
        if before and minus is not None:
            s = sl.copy()
            s[axis] = slice(before, 2 * before)
            snd = arr[tuple(s)].contiguous()
            rcv = torch.empty_like(snd)
            sends.append((snd, minus))
            recvs.append((rcv, minus))
            d = sl.copy()
            d[axis] = slice(0, before)
            places.append((d, rcv))
        if after and plus is not None:
            s = sl.copy()
            s[axis] = slice(-2 * after, -after)
            snd = arr[tuple(s)].contiguous()
            rcv = torch.empty_like(snd)
            sends.append((snd, plus))
            recvs.append((rcv, plus))
            d = sl.copy()
            d[axis] = slice(-after, None) if after else sl.copy()
            places.append((d, rcv))
        comm.exchange(sends, recvs)
        for d, rcv in places:
            arr[tuple(d)] = rcv

    # -------------------------------------------------------------- applies
    def _matvec(self, x: DistributedArray) -> DistributedArray:
        # ref :362-398
        if x.partition != Partition.SCATTER:
            raise ValueError(
                f"x should have partition={Partition.SCATTER} "
                f"Got {x.partition} instead...")
        if x.local_array.numel() != int(np.prod(self.local_dims)):
            raise ValueError(
                "MPIHalo input local shapes do not match the Cartesian "
                "block decomposition")
        y = DistributedArray(
            self.shape[0], x.base_comm, Partition.SCATTER, 0,
            local_shapes=[(s,) for s in self._local_extent_sizes],
            dtype=self.dtype)
        core = x.local_array.reshape(self.local_dims)
        halo_arr = torch.zeros(self.local_extent,
                               dtype=as_torch_dtype(self.dtype),
                               device=core.device)
        core_slices = tuple(
            slice(self.halo[2 * ax], self.halo[2 * ax] + self.local_dims[ax])
            for ax in range(self.ndim))
        halo_arr[core_slices] = core
        for ax in range(self.ndim):
            self._exchange_along_axis(halo_arr, ax, self.halo[2 * ax],
                                      self.halo[2 * ax + 1])
        y[:] = halo_arr.reshape(-1)
        return y

    def _rmatvec(self, x: DistributedArray) -> DistributedArray:
        # ref :400-423 — strip the ghosts
        if x.partition != Partition.SCATTER:
            raise ValueError(
                f"x should have partition={Partition.SCATTER} "
                f"Got {x.partition} instead...")
        if x.local_array.numel() != int(np.prod(self.local_extent)):
            raise ValueError(
                "MPIHalo input local shapes do not match the Cartesian "
                "block decomposition")
        res = DistributedArray(
            self.shape[1], x.base_comm, Partition.SCATTER, 0,
            local_shapes=[(s,) for s in self._local_dim_sizes],
            dtype=self.dtype)
        arr = x.local_array.reshape(self.local_extent)
        core_slices = tuple(
            slice(self.halo[2 * ax], self.halo[2 * ax] + self.local_dims[ax])
            for ax in range(self.ndim))
        res[:] = arr[core_slices].reshape(-1).contiguous()
        return res
