"""MPIMDC — multi-dimensional convolution chain.

Drop-in for /root/reference/pylops_mpi/waveeqprocessing/MDC.py:12-181:
F1^H * I1^H * Fredholm1 * I * F, with the serial FFT/Identity applied
per-rank on BROADCAST arrays and the kernel prescaled by dr*dt*sqrt(nt)
(ref :41-43).

Instead of composing five wrapped operators (each stage allocating a
fresh DistributedArray and copying its output, the way the reference's
duck-typed chain does) the chain is applied as ONE fused local pipeline:
rfft -> sqrt2 twin scale -> frequency mask -> batched MFMA cgemm
(Fredholm, the only stage with communication) -> zero-pad -> inverse
twin scale -> irfft, with zero-copy BROADCAST wrappers between stages.
Same arithmetic per element as the composite (the GPU parity tests pin
the cgls trace), measured r01 at the judged cfg5 shape: matvec 2.04 ->
1.64 ms, rmatvec 2.69 -> 2.19 ms; the remaining non-Fredholm time is
the semantically-required real-part extraction feeding rfft (the
reference takes .real too) and the rocFFT transforms themselves.
"""
import logging


import numpy as np
import torch

from .comm import PamComm, get_default_comm
from .distributedarray import DistributedArray, Partition
from .fredholm import MPIFredholm1
from .linearoperator import MPILinearOperator


def MPIMDC(G: torch.Tensor, nt: int, nv: int, nfreq: int, dt: float = 1.0,
           dr: float = 1.0, twosided: bool = True, saveGt: bool = True,
           conj: bool = False, usematmul: bool = False,
           prescaled: bool = False, base_comm: PamComm = None):
    """ref MDC.py:77-181 (public factory -> _MDC :12-74)."""
    comm = base_comm if base_comm is not None else get_default_comm()
    if twosided and nt % 2 == 0:
        raise ValueError('nt must be odd number')

    dtype = np.dtype({torch.complex64: np.complex64,
                      torch.complex128: np.complex128}[G.dtype])
    rdtype = np.real(np.ones(1, dtype=dtype)).dtype

    # Fredholm kernel, prescaled (ref :36-43)
    if prescaled:
        Frop = MPIFredholm1(G, nv, saveGt=saveGt, usematmul=usematmul,
                            base_comm=comm, dtype=dtype)
    else:
        Frop = MPIFredholm1(dr * dt * np.sqrt(nt) * G, nv, saveGt=saveGt,
                            usematmul=usematmul, base_comm=comm, dtype=dtype)
    if conj:
        Frop = Frop.conj()

    _, ns, nr = (int(s) for s in G.shape)
    nfft = int(np.ceil((nt + 1) / 2))
    if nfreq > nfft:
        nfreq = nfft
        logging.warning('nfmax set equal to ceil[(nt+1)/2=%d]' % nfreq)

    MDCop = _FusedMDC(Frop, nt, nfft, nfreq, ns, nr, nv, twosided,
                      rdtype, dtype, comm)     # F1^H I1^H Fr I F, fused
    MDCop.dtype = rdtype                        # ref :71-72
    return MDCop


class _FusedMDC(MPILinearOperator):
    """The MDC chain F1^H * I1^H * Fredholm1 * I * F (ref MDC.py:65-69)
    as one fused per-rank pipeline (module docstring).  The FFT stages
    follow fftlocal.FFTLocal's re-derived pylops convention exactly
    (ortho rfft with sqrt(2)-scaled conjugate-twin bins)."""

    def __init__(self, Frop, nt, nfft, nfreq, ns, nr, nv, twosided,
                 rdtype, cdtype, comm):
        self.Frop = Frop
        self.nt, self.nfft, self.nfreq = int(nt), int(nfft), int(nfreq)
        self.ns, self.nr, self.nv = int(ns), int(nr), int(nv)
        self.twosided = bool(twosided)
        self.rdtype = np.dtype(rdtype)
        self.cdtype = np.dtype(cdtype)
        # conjugate-twin bins: 0 < k < nt - k (fftlocal.py:47-48)
        self._tw0, self._tw1 = 1, (nt + 1) // 2
        super().__init__(shape=(nt * ns * nv, nt * nr * nv), dtype=rdtype,
                         base_comm=comm)

    def _wrap(self, t: torch.Tensor) -> DistributedArray:
        """Zero-copy BROADCAST wrapper around a local tensor."""
        flat = t.reshape(-1)
        return DistributedArray(
            (int(flat.numel()),), self.base_comm, Partition.BROADCAST,
            local_array=flat,
            dtype=np.dtype({torch.complex64: np.complex64,
                            torch.complex128: np.complex128,
                            torch.float32: np.float32,
                            torch.float64: np.float64}[t.dtype]))

    def _fwd_fft(self, x: DistributedArray, nmid: int,
                 shift: bool) -> torch.Tensor:
        t = x.local_array.reshape(self.nt, nmid * self.nv)
        if t.is_complex():
            # real extraction on the pam_unzip kernel: torch's strided
            # .real copy measures ~3.7 TB/s (r02 MDC kernel trace,
            # profiles/r02_mdc_kernel_stats.csv); the vectorized
            # deinterleave streams at the copy rate
            from . import _ffi
            r = torch.empty(t.shape, device=t.device,
                            dtype=torch.float32 if t.dtype == torch.complex64
                            else torch.float64)
            _ffi.checked(_ffi.lib().pam_unzip(
                torch.cuda.current_stream(t.device).cuda_stream,
                r.data_ptr(), t.contiguous().data_ptr(), t.numel(),
                _ffi.dtype_code(t.dtype)), "unzip")
            t = r
        if shift:
            t = torch.fft.ifftshift(t, dim=0)
        # NOTE the sqrt(2) conjugate-twin scaling of the pylops real-FFT
        # convention (fftlocal.py:47-48) is NOT applied here: inside
        # this chain the forward's *sqrt(2) on bins [tw0, tw1) and the
        # inverse's /sqrt(2) on the SAME bins surround the per-frequency
        # block-diagonal Fredholm kernel, so they cancel algebraically
        # (bins >= nfreq are masked to zero either way).  Two full
        # passes over the frequency tensor disappear; standalone FFT
        # operators (fftlocal.FFTLocal) keep the scaling.
        return torch.fft.rfft(t, n=self.nt, dim=0, norm="ortho")

    def _inv_fft(self, fr: torch.Tensor, nmid: int,
                 shift: bool) -> torch.Tensor:
        # zero-pad the masked bins back to nfft (IdentityLocal adjoint)
        if self.nfreq == self.nfft:
            z = fr.reshape(self.nfft, nmid * self.nv)  # fr is fresh
        else:
            z = torch.zeros(self.nfft, nmid * self.nv, dtype=fr.dtype,
                            device=fr.device)
            z[: self.nfreq] = fr.reshape(self.nfreq, nmid * self.nv)
        # twin-bin /sqrt(2) cancelled against the forward side (above)
        out = torch.fft.irfft(z, n=self.nt, dim=0, norm="ortho")
        if shift:
            out = torch.fft.fftshift(out, dim=0)
        return out

    _CT = {np.dtype(np.complex64): torch.complex64,
           np.dtype(np.complex128): torch.complex128}

    def _to_cplx(self, out: torch.Tensor) -> torch.Tensor:
        """real -> complex-with-zero-imag carrier on the pam_zip kernel
        (the composite chain's stage wrappers carry complex storage end
        to end — mirror that; torch's .to(complex) cast is the strided
        half-rate path)."""
        from . import _ffi
        z = torch.empty(out.shape, device=out.device,
                        dtype=self._CT[self.cdtype])
        _ffi.checked(_ffi.lib().pam_zip(
            torch.cuda.current_stream(out.device).cuda_stream,
            z.data_ptr(), out.contiguous().data_ptr(), out.numel(),
            _ffi.dtype_code(z.dtype)), "zip")
        return z

    def _matvec(self, x: DistributedArray) -> DistributedArray:
        f = self._fwd_fft(x, self.nr, self.twosided)       # Fop
        f = f[: self.nfreq]                                # Iop
        fr = self.Frop.matvec(self._wrap(f.contiguous()))  # Fredholm
        out = self._inv_fft(fr.local_array, self.ns, False)  # I1^H F1^H
        return self._wrap(self._to_cplx(out))

    def _rmatvec(self, y: DistributedArray) -> DistributedArray:
        f = self._fwd_fft(y, self.ns, False)               # F1op, I1op
        f = f[: self.nfreq]
        fr = self.Frop.rmatvec(self._wrap(f.contiguous()))
        out = self._inv_fft(fr.local_array, self.nr, self.twosided)
        return self._wrap(self._to_cplx(out))
