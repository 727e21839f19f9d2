"""Serial per-rank derivative local operators (the pylops serial
FirstDerivative/SecondDerivative the reference drops inside MPIBlockDiag
for Gradient/Laplacian axes >= 1, ref basicoperators/Gradient.py:109-116,
Laplacian.py:107-125).  Same stencil tables/edge semantics as the
distributed kernels (csrc fd_serial_kernel), along any axis of a local
block of shape ``dims``."""
import numpy as np
import torch

from . import _ffi
from .derivative import _FD1_OPS, _FD2_OPS
from .localops import LocalOperator


class _FDLocalBase(LocalOperator):

    def __init__(self, dims, axis, sampling, kind, edge, dtype):
        self.dims = (dims,) if isinstance(dims, (int, np.integer)) \
            else tuple(int(v) for v in dims)
        axis = axis if axis >= 0 else len(self.dims) + axis
        if not 0 <= axis < len(self.dims):
            raise ValueError(f"axis {axis} out of range for dims {self.dims}")
        self.axis = axis
        self.sampling = sampling
        self.kind = kind
        self.edge = edge
        self.dtype = np.dtype(dtype)
        n = int(np.prod(self.dims))
        self.shape = (n, n)
        self.batch = int(np.prod(self.dims[:axis], initial=1))
        self.d = self.dims[axis]
        self.m = int(np.prod(self.dims[axis + 1:], initial=1))

    def _run(self, x: torch.Tensor, op: int) -> torch.Tensor:
        if x.device.type != "cuda":
            raise RuntimeError(
                "pam: compute ops require a CUDA (MI355X) device tensor — "
                "there is no CPU compute path")
        flat = x.reshape(-1).contiguous()
        y = torch.empty_like(flat)
        batch, d, m = self.batch, self.d, self.m
        if flat.is_complex():
            rb, rm = batch, 2 * m  # componentwise on the interleaved view
            xr = torch.view_as_real(flat).reshape(-1)
            yr = torch.view_as_real(y).reshape(-1)
            dt = _ffi.dtype_code(xr.dtype)
            ptrs = (xr.data_ptr(), yr.data_ptr())
        else:
            rb, rm = batch, m
            dt = _ffi.dtype_code(flat.dtype)
            ptrs = (flat.data_ptr(), y.data_ptr())
        stream = torch.cuda.current_stream(x.device).cuda_stream
        _ffi.checked(_ffi.lib().pam_fd_serial(
            stream, op, 1 if self.edge else 0, ptrs[0], ptrs[1], rb, d, rm,
            self._coeff(), dt), "fd_serial")
        return y

    def matvec(self, x):
        return self._run(x, self._op_mv)

    def rmatvec(self, x):
        return self._run(x, self._op_rmv)


class FirstDerivativeLocal(_FDLocalBase):
    def __init__(self, dims, axis: int = -1, sampling: float = 1.0,
                 kind: str = "centered", edge: bool = False, order: int = 3,
                 dtype=np.float64):
        super().__init__(dims, axis, sampling, kind, edge, dtype)
        key = (kind, order if kind == "centered" else 0)
        if key not in _FD1_OPS:
            raise NotImplementedError(
                "'kind' must be 'forward', 'centered', or 'backward'")
        self._op_mv, self._op_rmv = _FD1_OPS[key]

    def _coeff(self):
        return 1.0 / self.sampling


class SecondDerivativeLocal(_FDLocalBase):
    def __init__(self, dims, axis: int = -1, sampling: float = 1.0,
                 kind: str = "centered", edge: bool = False,
                 dtype=np.float64):
        super().__init__(dims, axis, sampling, kind, edge, dtype)
        if kind not in _FD2_OPS:
            raise NotImplementedError(
                "'kind' must be 'forward', 'centered' or 'backward'")
        self._op_mv, self._op_rmv = _FD2_OPS[kind]

    def _coeff(self):
        return 1.0 / self.sampling ** 2
