"""MPIFredholm1 — batched Fredholm integral of the first kind, sliced
across ranks along the first (frequency) dimension.

Drop-in for /root/reference/pylops_mpi/signalprocessing/Fredholm1.py:14-169.
The per-slice products run on the batched complex GEMM kernel
(pam_cgemm_batched; real dtypes fall back to per-slice MFMA pam_gemm);
results are allgathered over RCCL and the output stays
BROADCAST-partitioned like the reference's.
"""
from typing import Optional

import numpy as np
import torch

from . import _ffi
from .comm import PamComm, get_default_comm
from .distributedarray import DistributedArray, Partition, as_torch_dtype
from .linearoperator import MPILinearOperator


def _stream(t):
    return torch.cuda.current_stream(t.device).cuda_stream


class MPIFredholm1(MPILinearOperator):

    def __init__(self, G: torch.Tensor, nz: int = 1, saveGt: bool = False,
                 usematmul: bool = True,
                 base_comm: Optional[PamComm] = None, dtype="float64"):
        comm = base_comm if base_comm is not None else get_default_comm()
        self.nz = nz
        self.nsl, self.nx, self.ny = (int(s) for s in G.shape)
        self.nsls = [int(v) for v in comm.allgather_obj(self.nsl)]
        if 1 in self.nsls:
            # ref :90-93 (raised at rank 0 there; deterministic here)
            raise NotImplementedError(
                f"All ranks must have at least 2 or more elements in the "
                f"first dimension: local split is instead {self.nsls}...")
        nslstot = int(sum(self.nsls))
        self.islstart = np.insert(np.cumsum(self.nsls)[:-1], 0, 0)
        self.islend = np.cumsum(self.nsls)
        dims = (nslstot, self.ny, self.nz)
        dimsd = (nslstot, self.nx, self.nz)
        super().__init__(dims=dims, dimsd=dimsd, dtype=np.dtype(dtype),
                         base_comm=comm)
        self.G = G.to(as_torch_dtype(np.dtype(dtype))).contiguous()
        if saveGt:
            self.GT = self.G.conj().transpose(1, 2).contiguous()
        self.usematmul = usematmul  # kept for surface parity; always batched

    # ------------------------------------------------------------ batched
    def _batched(self, A: torch.Tensor, X: torch.Tensor, opa: int
                 ) -> torch.Tensor:
        """Y_b = op(A_b) @ X_b for every local slice."""
        if A.device.type != "cuda":
            raise RuntimeError(
                "pam: compute ops require a CUDA (MI355X) device tensor — "
                "there is no CPU compute path")
        batch = A.shape[0]
        if opa:
            K, M = int(A.shape[1]), int(A.shape[2])
        else:
            M, K = int(A.shape[1]), int(A.shape[2])
        N = int(X.shape[2])
        Y = torch.empty((batch, M, N), dtype=A.dtype, device=A.device)
        A, X = A.contiguous(), X.contiguous()
        if A.is_complex():
            _ffi.checked(_ffi.lib().pam_cgemm_batched(
                _stream(A), A.data_ptr(), X.data_ptr(), Y.data_ptr(), batch,
                M, N, K, A.shape[1] * A.shape[2], K * N, M * N, opa, 0,
                _ffi.dtype_code(A.dtype)), "cgemm_batched")
        else:
            # real dtypes: one z-batched MFMA launch (opa=1 loads A
            # transposed in-kernel; conj is a no-op on reals).  The r01
            # form looped pam_gemm per slice from Python — 4-workgroup
            # launches, chip empty.
            _ffi.checked(_ffi.lib().pam_gemm_batched(
                _stream(A), A.data_ptr(), X.data_ptr(), Y.data_ptr(),
                batch, M, N, K, A.shape[1] * A.shape[2], K * N, M * N,
                opa, 0, _ffi.dtype_code(A.dtype)), "gemm_batched")
        return Y

    # ------------------------------------------------------------- applies
    def _apply(self, x: DistributedArray, forward: bool) -> DistributedArray:
        # ref :107-169
        if x.partition not in (Partition.BROADCAST,
                               Partition.UNSAFE_BROADCAST):
            raise ValueError(
                f"x should have partition={Partition.BROADCAST},"
                f"{Partition.UNSAFE_BROADCAST}"
                f"Got  {x.partition} instead...")
        comm = x.base_comm
        gshape = self.shape[0] if forward else self.shape[1]
        dims = self.dimsd if not forward else self.dims
        xl = x.local_array.reshape(dims)
        r = comm.rank
        xs = xl[int(self.islstart[r]): int(self.islend[r])]
        xs = xs.to(self.G.dtype)
        if forward:
            y1 = self._batched(self.G, xs, 0)          # ref :123
        elif hasattr(self, "GT"):
            y1 = self._batched(self.GT, xs, 0)         # ref :150
        else:
            y1 = self._batched(self.G, xs, 1)          # ref :152-156
        nmid = self.nx if forward else self.ny
        tiles = comm.allgather_tensors(
            y1.reshape(-1),
            [(int(n) * nmid * self.nz,) for n in self.nsls])  # ref :129,167
        # wrap the gathered tensor as the output's storage directly — it
        # is freshly allocated here, so no defensive copy is needed (the
        # r01 form cat'ed into a separate DistributedArray: two extra
        # full passes per apply)
        full = tiles[0] if len(tiles) == 1 else torch.cat(tiles)
        return DistributedArray(int(gshape), comm, x.partition,
                                local_array=full.reshape(-1),
                                dtype=self.dtype)

    def _matvec(self, x: DistributedArray) -> DistributedArray:
        return self._apply(x, forward=True)

    def _rmatvec(self, x: DistributedArray) -> DistributedArray:
        return self._apply(x, forward=False)
