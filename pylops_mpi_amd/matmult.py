"""MPIMatrixMult — distributed matrix-matrix multiplication (block-row and
SUMMA kinds) on MFMA panel GEMMs + RCCL grid collectives.

Drop-in for /root/reference/pylops_mpi/basicoperators/MatrixMult.py:
  active_grid_comm      -> ref :24-79
  local_block_split     -> ref :82-129
  block_gather          -> ref :132-175
  _MPIBlockMatrixMult   -> ref :178-427
  _MPISummaMatrixMult   -> ref :430-765
  MPIMatrixMult factory -> ref :768-872

MI355X-first differences:
  * local panels run on hand-written MFMA kernels (pam_gemm:
    v_mfma_f32_32x32x2_f32 exact-f32 / v_mfma_f64_16x16x4_f64; complex
    dtypes on the 4-chain MFMA pam_cgemm_batched with batch=1) instead
    of CuPy matmul; A^T/A^H panels are materialized by LDS-tiled
    transpose kernels (pam_transpose / pam_ctranspose; `saveAt=True`
    stores it once, ref :317-318);
  * row/col communicators are RCCL process groups (comm.split_by); the
    SUMMA adjoint's tag-routed A^T exchange (ref :745-760) is posted as
    one batched isend/irecv round per step — RCCL has no tags, and each
    (src,dst) pair carries exactly one message per round, so ordering is
    preserved by construction;
  * metadata (shape gathers) travels on the gloo control plane of the
    WORLD communicator (all memberships are deterministic).
Like the reference (ref :295-298,564-566), both kinds require a square
number of ranks; use active_grid_comm to carve an active square grid out
of a non-square world (e.g. 2x2 out of 8).
"""
import math
from typing import Optional, Tuple

import numpy as np
import torch

from . import _ffi
from .comm import PamComm, get_default_comm
from .distributedarray import DistributedArray, Partition, as_torch_dtype
from .linearoperator import MPILinearOperator

def _stream(t: torch.Tensor):
    return torch.cuda.current_stream(t.device).cuda_stream


def active_grid_comm(base_comm: PamComm, N: int, M: int):
    """ref :24-79 — carve the largest useful square grid; inactive ranks
    get (None, rank, row, col, False)."""
    rank, size = base_comm.rank, base_comm.size
    p_prime = math.isqrt(size)
    row, col = divmod(rank, p_prime)
    active_dim = min(N, M, p_prime)
    is_active = (row < active_dim and col < active_dim)
    # every rank participates in group creation (torch.distributed rule)
    colors = []
    for r in range(size):
        rr, cc = divmod(r, p_prime)
        colors.append(0 if (rr < active_dim and cc < active_dim) else 1 + r)
    new_comm = base_comm.split_by(colors, keys=list(range(size)))
    if not is_active:
        return None, rank, row, col, False
    p_new = math.isqrt(new_comm.size)
    new_rank = new_comm.rank
    new_row, new_col = divmod(new_rank, p_new)
    return new_comm, new_rank, new_row, new_col, True


def local_block_split(global_shape: Tuple[int, int], rank: int,
                      comm: PamComm) -> Tuple[slice, slice]:
    """ref :82-129 — this rank's (row_slice, col_slice) of a 2-D array on
    a square process grid."""
    size = comm.size
    p_prime = math.isqrt(size)
    if p_prime * p_prime != size:
        raise RuntimeError(f"Number of processes must be a square number, "
                           f"provided {size} instead...")
    if not (isinstance(rank, (int, np.integer)) and 0 <= rank < size):
        raise ValueError(
            f"rank must be an integer in [0, {size}), got {rank!r}")
    pr, pc = divmod(int(rank), p_prime)
    orig_r, orig_c = global_shape
    blkr = math.ceil(orig_r / p_prime)
    blkc = math.ceil(orig_c / p_prime)
    rs, cs = pr * blkr, pc * blkc
    re, ce = min(rs + blkr, orig_r), min(cs + blkc, orig_c)
    return slice(rs, re), slice(cs, ce)


def block_gather(x: DistributedArray, orig_shape: Tuple[int, int],
                 comm: PamComm) -> torch.Tensor:
    """ref :132-175 — reassemble a 2-D-block-distributed matrix."""
    p_prime = math.isqrt(comm.size)
    if p_prime * p_prime != comm.size:
        raise RuntimeError(
            f"Communicator size must be a perfect square, got {comm.size!r}")
    nr, nc = orig_shape
    shapes = [(int(np.prod(s)),) for s in x.local_shapes]
    blks = comm.allgather_tensors(x.local_array.reshape(-1), shapes)
    C = torch.zeros((nr, nc), dtype=x.local_array.dtype,
                    device=x.local_array.device)
    for rank in range(p_prime * p_prime):
        rs, cs = local_block_split(orig_shape, rank, comm)
        if blks[rank].numel() != 0:
            C[rs, cs] = blks[rank].reshape(rs.stop - rs.start,
                                           cs.stop - cs.start)
    return C


class _MatMultBase(MPILinearOperator):
    """Shared plumbing: grid comms + HIP panel GEMM/transpose."""

    def _make_grid(self, comm: PamComm):
        size = comm.size
        self._P_prime = math.isqrt(size)
        if self._P_prime * self._P_prime != size:
            raise Exception(f"Number of processes must be a square number, "
                            f"provided {size} instead...")
        return self._P_prime

    # local HIP panels (overridable in CPU comm-logic tests)
    def _local_gemm(self, A: torch.Tensor, B: torch.Tensor,
                    C: Optional[torch.Tensor] = None,
                    accumulate: bool = False) -> torch.Tensor:
        M, K = A.shape
        K2, N = B.shape
        assert K == K2
        if C is None:
            C = torch.empty((M, N), dtype=A.dtype, device=A.device)
        if A.device.type != "cuda":
            raise RuntimeError(
                "pam: compute ops require a CUDA (MI355X) device tensor — "
                "there is no CPU compute path")
        if A.is_complex():
            # complex panels: the batched MFMA cgemm with batch=1
            # (ref MatrixMult.py supports complex dtypes throughout)
            _ffi.checked(_ffi.lib().pam_cgemm_batched(
                _stream(A), A.contiguous().data_ptr(),
                B.contiguous().data_ptr(), C.data_ptr(), 1, M, N, K,
                0, 0, 0, 0, 1 if accumulate else 0,
                _ffi.dtype_code(A.dtype)), "cgemm")
        else:
            _ffi.checked(_ffi.lib().pam_gemm(
                _stream(A), A.contiguous().data_ptr(),
                B.contiguous().data_ptr(), C.data_ptr(), M, N, K, K, N, N,
                1 if accumulate else 0, _ffi.dtype_code(A.dtype)), "gemm")
        return C

    def _local_transpose(self, A: torch.Tensor) -> torch.Tensor:
        """A^T for real dtypes, A^H for complex (ref :317-318,416,737
        ``A.T.conj()`` — conj is a no-op on reals)."""
        if A.device.type != "cuda":
            raise RuntimeError(
                "pam: compute ops require a CUDA (MI355X) device tensor — "
                "there is no CPU compute path")
        At = torch.empty((A.shape[1], A.shape[0]), dtype=A.dtype,
                         device=A.device)
        if A.is_complex():
            _ffi.checked(_ffi.lib().pam_ctranspose(
                _stream(A), A.contiguous().data_ptr(), At.data_ptr(),
                A.shape[0], A.shape[1], 1, _ffi.dtype_code(A.dtype)),
                "ctranspose")
        else:
            _ffi.checked(_ffi.lib().pam_transpose(
                _stream(A), A.contiguous().data_ptr(), At.data_ptr(),
                A.shape[0], A.shape[1], _ffi.dtype_code(A.dtype)),
                "transpose")
        return At

    def _AH(self) -> torch.Tensor:
        # ref :416,737: use saved At or compute A.T.conj() on the fly
        return self.At if hasattr(self, "At") else self._local_transpose(self.A)

    @staticmethod
    def _check_scatter(x: DistributedArray):
        if x.partition != Partition.SCATTER:
            raise ValueError(f"x should have partition={Partition.SCATTER} "
                             f"Got {x.partition} instead...")


class _MPIBlockMatrixMult(_MatMultBase):
    """1-D block-row A x column-replicated X (ref :178-427)."""

    def __init__(self, A: torch.Tensor, M: int, saveAt: bool = False,
                 base_comm: Optional[PamComm] = None, dtype="float64"):
        comm = base_comm if base_comm is not None else get_default_comm()
        rank, size = comm.rank, comm.size
        self.base_comm_grid = comm
        P = self._make_grid(comm)
        self._col_id = rank % P
        self._row_id = rank // P
        # ref :305-306
        self._row_comm = comm.split_by([r // P for r in range(size)],
                                       [r % P for r in range(size)])
        self._col_comm = comm.split_by([r % P for r in range(size)],
                                       [r // P for r in range(size)])
        self.A = A.to(as_torch_dtype(np.dtype(dtype)))
        if saveAt:
            self.At = self._local_transpose(self.A)
        # N = sum of A-rows across the row communicator (ref :320)
        all_rows = comm.allgather_obj(int(A.shape[0]))
        row_members = [r for r in range(size) if r // P == self._row_id]
        self._row_nlocs = [all_rows[r] for r in row_members]
        self.N = int(sum(self._row_nlocs))
        self.K = int(A.shape[1])
        self.M = M
        block_cols = int(math.ceil(self.M / P))
        blk_rows = int(math.ceil(self.N / P))
        self._row_start = self._col_id * blk_rows
        self._row_end = min(self.N, self._row_start + blk_rows)
        self._col_start = self._row_id * block_cols
        self._col_end = min(self.M, self._col_start + block_cols)
        self._local_ncols = max(0, self._col_end - self._col_start)
        self._rank_col_lens = comm.allgather_obj(self._local_ncols)
        total_ncols = int(np.sum(self._rank_col_lens))
        dims = (self.K, total_ncols)
        dimsd = (self.N, total_ncols)
        super().__init__(dims=dims, dimsd=dimsd, dtype=np.dtype(dtype),
                         base_comm=comm)

    def _matvec(self, x: DistributedArray) -> DistributedArray:
        # ref :341-377
        self._check_scatter(x)
        y = DistributedArray(
            int(self.N * self.dimsd[1]), x.base_comm, Partition.SCATTER, 0,
            local_shapes=[(int(self.N * c),) for c in self._rank_col_lens],
            dtype=self.dtype)
        my_cols = self._rank_col_lens[self.rank]
        x_arr = x.local_array.reshape(self.dims[0], my_cols).to(self.A.dtype)
        prod = self._local_gemm(self.A, x_arr)          # (N_loc, M_loc)
        tiles = self._row_comm.allgather_tensors(
            prod.reshape(-1),
            [(n, my_cols) for n in self._row_nlocs])    # ref :369-375
        y[:] = torch.vstack(tiles).reshape(-1)
        return y

    def _rmatvec(self, x: DistributedArray) -> DistributedArray:
        # ref :379-427
        self._check_scatter(x)
        y = DistributedArray(
            int(self.K * self.dimsd[1]), x.base_comm, Partition.SCATTER, 0,
            local_shapes=[(int(self.K * c),) for c in self._rank_col_lens],
            dtype=self.dtype)
        x_arr = x.local_array.reshape(self.N, self._local_ncols).to(self.A.dtype)
        X_tile = x_arr[self._row_start: self._row_end, :].contiguous()
        Y_local = self._local_gemm(self._AH(), X_tile)  # (K, M_loc)
        self._row_comm.allreduce_(Y_local, "sum")       # ref :419-425
        y[:] = Y_local.reshape(-1)
        return y


class _MPISummaMatrixMult(_MatMultBase):
    """2-D SUMMA (ref :430-765)."""

    def __init__(self, A: torch.Tensor, M: int, saveAt: bool = False,
                 base_comm: Optional[PamComm] = None, dtype="float64"):
        comm = base_comm if base_comm is not None else get_default_comm()
        rank, size = comm.rank, comm.size
        self.base_comm_grid = comm
        P = self._make_grid(comm)
        self._row_id, self._col_id = divmod(rank, P)
        self._row_comm = comm.split_by([r // P for r in range(size)],
                                       [r % P for r in range(size)])
        self._col_comm = comm.split_by([r % P for r in range(size)],
                                       [r // P for r in range(size)])
        self.A = A.to(as_torch_dtype(np.dtype(dtype)))
        all_rows = comm.allgather_obj(int(A.shape[0]))
        all_cols = comm.allgather_obj(int(A.shape[1]))
        col_members = [r for r in range(size) if r % P == self._col_id]
        row_members = [r for r in range(size) if r // P == self._row_id]
        self.N = int(sum(all_rows[r] for r in col_members))   # ref :585
        self.K = int(sum(all_cols[r] for r in row_members))   # ref :586
        self.M = M
        self._N_padded = math.ceil(self.N / P) * P
        self._K_padded = math.ceil(self.K / P) * P
        self._M_padded = math.ceil(self.M / P) * P
        bn = self._N_padded // P
        bk = self._K_padded // P
        pr = (bn - int(A.shape[0])) if self._row_id == P - 1 else 0
        pc = (bk - int(A.shape[1])) if self._col_id == P - 1 else 0
        if pr > 0 or pc > 0:  # ref :597-601
            Ap = torch.zeros((int(A.shape[0]) + pr, int(A.shape[1]) + pc),
                             dtype=self.A.dtype, device=self.A.device)
            Ap[: A.shape[0], : A.shape[1]] = self.A
            self.A = Ap
        if saveAt:
            self.At = self._local_transpose(self.A)
        super().__init__(dims=(self.K, self.M), dimsd=(self.N, self.M),
                         dtype=np.dtype(dtype), base_comm=comm)

    def _tile_sizes(self):
        P = self._P_prime
        bn = self._N_padded // P
        bk = self._K_padded // P
        bm = self._M_padded // P
        local_n = bn if self._row_id != P - 1 else self.N - (P - 1) * bn
        local_k = bk if self._row_id != P - 1 else self.K - (P - 1) * bk
        local_m = bm if self._col_id != P - 1 else self.M - (P - 1) * bm
        return bn, bk, bm, local_n, local_k, local_m

    def _all_counts(self, kind: str):
        """Deterministic per-rank output element counts (the reference
        allgathers them, ref :630,688)."""
        P = self._P_prime
        bn = self._N_padded // P
        bk = self._K_padded // P
        bm = self._M_padded // P
        out = []
        for q in range(P * P):
            qr, qc = divmod(q, P)
            ln = bn if qr != P - 1 else self.N - (P - 1) * bn
            lk = bk if qr != P - 1 else self.K - (P - 1) * bk
            lm = bm if qc != P - 1 else self.M - (P - 1) * bm
            out.append(int((ln if kind == "n" else lk) * lm))
        return out

    def _pad_block(self, blk: torch.Tensor, rows: int, cols: int):
        if blk.shape[0] == rows and blk.shape[1] == cols:
            return blk.contiguous()
        out = torch.zeros((rows, cols), dtype=blk.dtype, device=blk.device)
        out[: blk.shape[0], : blk.shape[1]] = blk
        return out

    def _matvec(self, x: DistributedArray) -> DistributedArray:
        # ref :610-672
        self._check_scatter(x)
        P = self._P_prime
        bn, bk, bm, local_n, local_k, local_m = self._tile_sizes()
        y = DistributedArray(
            int(self.N * self.M), x.base_comm, Partition.SCATTER, 0,
            local_shapes=[(c,) for c in self._all_counts("n")],
            dtype=self.dtype)
        x_block = self._pad_block(
            x.local_array.reshape(local_k, local_m).to(self.A.dtype), bk, bm)
        Y_local = torch.zeros((self.A.shape[0], bm), dtype=self.A.dtype,
                              device=self.A.device)
        for k in range(P):
            Atemp = self.A.contiguous() if self._col_id == k \
                else torch.empty_like(self.A)
            Xtemp = x_block if self._row_id == k \
                else torch.empty_like(x_block)
            self._row_comm.broadcast_(Atemp, root=k)   # ref :666
            self._col_comm.broadcast_(Xtemp, root=k)   # ref :667
            self._local_gemm(Atemp, Xtemp, Y_local, accumulate=True)
        y[:] = Y_local[:local_n, :local_m].reshape(-1)
        return y

    def _rmatvec(self, x: DistributedArray) -> DistributedArray:
        # ref :674-765
        self._check_scatter(x)
        P = self._P_prime
        bn, bk, bm, local_n, local_k, local_m = self._tile_sizes()
        y = DistributedArray(
            int(self.K * self.M), x.base_comm, Partition.SCATTER, 0,
            local_shapes=[(c,) for c in self._all_counts("k")],
            dtype=self.dtype)
        x_block = self._pad_block(
            x.local_array.reshape(local_n, local_m).to(self.A.dtype), bn, bm)
        A_local = self._AH()                         # (bk, bn)
        Y_local = torch.zeros((self.A.shape[1], bm), dtype=self.A.dtype,
                              device=self.A.device)
        comm = self.base_comm_grid
        me = comm.rank
        for k in range(P):
            Xtemp = x_block if self._row_id == k \
                else torch.empty_like(x_block)
            self._col_comm.broadcast_(Xtemp, root=k)
            # A^T routing (ref :745-760): rank (r,c) consumes the A^T of
            # grid rank (k, r); grid-row-k members send theirs to every
            # member of grid row <their col_id>.
            srcA = k * P + self._row_id
            sends, recvs = [], []
            if self._row_id == k:
                for moving_col in range(P):
                    destA = self._col_id * P + moving_col
                    if destA != me:
                        sends.append((A_local.contiguous(), destA))
            if srcA == me:
                ATtemp = A_local
            else:
                ATtemp = torch.empty_like(A_local)
                recvs.append((ATtemp, srcA))
            comm.exchange(sends, recvs)
            self._local_gemm(ATtemp, Xtemp, Y_local, accumulate=True)
        y[:] = Y_local[:local_k, :local_m].reshape(-1)
        return y


def MPIMatrixMult(A: torch.Tensor, M: int, saveAt: bool = False,
                  base_comm: Optional[PamComm] = None,
                  kind: str = "summa", dtype="float64"):
    """Factory, ref :768-872."""
    if kind == "summa":
        return _MPISummaMatrixMult(A, M, saveAt, base_comm, dtype)
    elif kind == "block":
        return _MPIBlockMatrixMult(A, M, saveAt, base_comm, dtype)
    raise NotImplementedError("kind must be summa or block")
