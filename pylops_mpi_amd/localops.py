"""Local (per-rank, serial) operators used inside MPIBlockDiag and the
serial-operator wrap of MPILinearOperator.

The reference uses pylops.LinearOperator here (ref BlockDiag.py:128-143
duck-types .shape/.matvec/.rmatvec on the rank-local array); our local
operators follow the same protocol on torch device tensors, with the
compute in hand-written HIP kernels (no torch math on the product path).
"""
from typing import Callable, Optional, Tuple

import numpy as np
import torch

from . import _ffi

# per-device GEMV scratch, sized for the largest (nr, nc) seen
_gemv_ws = {}


def _gemv_buffer(device, nr: int, nc: int) -> torch.Tensor:
    need = int(_ffi.lib().pam_gemv_ws_elems(nr, nc))
    key = (device.type, device.index)
    ws = _gemv_ws.get(key)
    if ws is None or ws.numel() < need:
        ws = torch.empty(need, dtype=torch.float64, device=device)
        _gemv_ws[key] = ws
    return ws


class LocalOperator:
    """Protocol: shape (n, m); matvec/rmatvec on 1-D device tensors."""

    shape: Tuple[int, int]
    dtype: np.dtype

    def matvec(self, x: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def rmatvec(self, x: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError


class DenseLocal(LocalOperator):
    """Dense matrix operator (the serial pylops.MatrixMult analog used by
    the reference's BlockDiag examples, ref examples/plot_cgls.py:30-33):
    matvec = A @ x, rmatvec = A^T @ x via pam_gemv (HBM-bound HIP
    kernels; real dtypes — A^H == A^T).

    ``saveAt`` (default True) materializes A^T on the first rmatvec so
    the adjoint runs the row-parallel n-path GEMV (6.4 TB/s) instead of
    the two-stage transpose path (4.7 TB/s, fixed partial-combine
    overhead — r01 measured table).  Costs one extra copy of A in HBM
    (trivial against 288 GB); set saveAt=False to keep single-copy."""

    def __init__(self, A: torch.Tensor, saveAt: bool = True):
        if A.ndim != 2:
            raise ValueError("DenseLocal expects a 2-D matrix")
        self.A = A.contiguous()
        self.shape = (int(A.shape[0]), int(A.shape[1]))
        self.saveAt = bool(saveAt)
        self._At = None
        self.dtype = np.dtype(
            {torch.float64: np.float64, torch.float32: np.float32}[A.dtype])

    def _gemv(self, A: torch.Tensor, x: torch.Tensor, trans: int,
              nout: int) -> torch.Tensor:
        if A.device.type != "cuda":
            raise RuntimeError(
                "pam: compute ops require a CUDA (MI355X) device tensor — "
                "there is no CPU compute path")
        nr, nc = int(A.shape[0]), int(A.shape[1])
        out = torch.empty(nout, dtype=A.dtype, device=A.device)
        ws = _gemv_buffer(A.device, nr, nc)
        stream = torch.cuda.current_stream(A.device).cuda_stream
        _ffi.checked(_ffi.lib().pam_gemv(
            stream, trans, A.data_ptr(), x.contiguous().data_ptr(),
            out.data_ptr(), nr, nc, ws.data_ptr(),
            _ffi.dtype_code(A.dtype)), "gemv")
        return out

    def matvec(self, x: torch.Tensor) -> torch.Tensor:
        return self._gemv(self.A, x.reshape(-1), 0, self.shape[0])

    def rmatvec(self, x: torch.Tensor) -> torch.Tensor:
        if self.saveAt:
            if self._At is None:
                if self.A.device.type != "cuda":
                    raise RuntimeError(
                        "pam: compute ops require a CUDA (MI355X) device "
                        "tensor — there is no CPU compute path")
                At = torch.empty((self.shape[1], self.shape[0]),
                                 dtype=self.A.dtype, device=self.A.device)
                stream = torch.cuda.current_stream(
                    self.A.device).cuda_stream
                _ffi.checked(_ffi.lib().pam_transpose(
                    stream, self.A.data_ptr(), At.data_ptr(),
                    self.shape[0], self.shape[1],
                    _ffi.dtype_code(self.A.dtype)), "transpose")
                self._At = At
            return self._gemv(self._At, x.reshape(-1), 0, self.shape[1])
        return self._gemv(self.A, x.reshape(-1), 1, self.shape[1])


class AdjointLocal(LocalOperator):
    """Adjoint view of a local operator (swap matvec/rmatvec)."""

    def __init__(self, op: LocalOperator):
        self.op = op
        self.shape = (op.shape[1], op.shape[0])
        self.dtype = op.dtype

    def matvec(self, x):
        return self.op.rmatvec(x)

    def rmatvec(self, x):
        return self.op.matvec(x)


class CallableLocal(LocalOperator):
    """Adapter for tests/composition: wrap a (matvec, rmatvec) pair."""

    def __init__(self, shape: Tuple[int, int],
                 matvec: Callable[[torch.Tensor], torch.Tensor],
                 rmatvec: Optional[Callable[[torch.Tensor],
                                            torch.Tensor]] = None,
                 dtype=np.float64):
        self.shape = shape
        self.dtype = np.dtype(dtype)
        self._mv = matvec
        self._rmv = rmatvec

    def matvec(self, x):
        return self._mv(x)

    def rmatvec(self, x):
        if self._rmv is None:
            raise NotImplementedError
        return self._rmv(x)
