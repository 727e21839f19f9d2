"""Serial (per-rank) FFT and Identity local operators for the MDC chain.

The reference wraps serial pylops FFT/Identity in MPILinearOperator
(ref waveeqprocessing/MDC.py:55-64); pylops itself is not vendored under
/root/reference, so the real-FFT convention is RE-DERIVED here (SURVEY.md
§8c: "pylops' real-FFT sqrt-scaling convention re-derived and locked by
dottest + round-trip tests"):

  forward  : y = rfft(x, n=nt, axis=0, norm="ortho");
             y[k] *= sqrt(2) for the bins with a conjugate twin
             (0 < k < nt - k) — this makes F an isometry onto its range;
  adjoint  : x = irfft(z, n=nt, axis=0, norm="ortho") with the twin bins
             divided by sqrt(2) — exactly F^H, and F^H F = I.

The FFT itself runs on rocFFT via torch.fft ("rocFFT first");
the scaling is elementwise on-device.  Identity follows pylops
Identity(N, M): forward truncates to the first N elements, adjoint
zero-pads back to M (the frequency mask, ref MDC.py:60-64).
"""
import math
from typing import Tuple

import numpy as np
import torch

from .localops import LocalOperator

_C_OF = {np.dtype(np.float64): np.complex128, np.dtype(np.float32): np.complex64}


class FFTLocal(LocalOperator):
    """Real FFT along axis 0 of ``dims``, flattened in/out."""

    def __init__(self, dims: Tuple[int, ...], real: bool = True,
                 ifftshift_before: bool = False, dtype=np.float64):
        if not real:
            raise NotImplementedError("only the real FFT used by MDC")
        self.dims = tuple(int(d) for d in dims)
        self.nt = self.dims[0]
        self.nfft = self.nt // 2 + 1
        rest = int(np.prod(self.dims[1:], initial=1))
        self.rest = rest
        self.rdtype = np.dtype(dtype)
        self.dtype = np.dtype(_C_OF[self.rdtype])  # operator dtype: complex
        self.ifftshift_before = ifftshift_before
        self.shape = (self.nfft * rest, self.nt * rest)
        # bins with a conjugate twin: 0 < k < nt - k
        self._tw0, self._tw1 = 1, (self.nt + 1) // 2

    def matvec(self, x: torch.Tensor) -> torch.Tensor:
        xr = x.reshape((self.nt, self.rest)).real
        if self.ifftshift_before:
            xr = torch.fft.ifftshift(xr, dim=0)
        y = torch.fft.rfft(xr, n=self.nt, dim=0, norm="ortho")
        y[self._tw0: self._tw1] *= math.sqrt(2.0)
        return y.reshape(-1)

    def rmatvec(self, y: torch.Tensor) -> torch.Tensor:
        z = y.reshape((self.nfft, self.rest)).clone()
        z[self._tw0: self._tw1] /= math.sqrt(2.0)
        x = torch.fft.irfft(z, n=self.nt, dim=0, norm="ortho")
        if self.ifftshift_before:
            x = torch.fft.fftshift(x, dim=0)
        return x.reshape(-1)


class IdentityLocal(LocalOperator):
    """pylops.Identity(N, M): forward = first N of M, adjoint = zero-pad."""

    def __init__(self, N: int, M: int, dtype=np.complex64):
        self.shape = (int(N), int(M))
        self.dtype = np.dtype(dtype)

    def matvec(self, x: torch.Tensor) -> torch.Tensor:
        N, M = self.shape
        if N == M:
            return x.clone()
        if N < M:
            return x[:N].clone()
        out = torch.zeros(N, dtype=x.dtype, device=x.device)
        out[:M] = x
        return out

    def rmatvec(self, x: torch.Tensor) -> torch.Tensor:
        N, M = self.shape
        if N == M:
            return x.clone()
        if N < M:
            out = torch.zeros(M, dtype=x.dtype, device=x.device)
            out[:N] = x
            return out
        return x[:M].clone()
