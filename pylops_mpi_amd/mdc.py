"""MPIMDC — multi-dimensional convolution chain.

Drop-in for /root/reference/pylops_mpi/waveeqprocessing/MDC.py:12-181:
F1^H * I1^H * Fredholm1 * I * F, with the serial FFT/Identity applied
per-rank on BROADCAST arrays and the kernel prescaled by dr*dt*sqrt(nt)
(ref :41-43).

Instead of composing five wrapped operators (each stage allocating a
fresh DistributedArray and copying its output, the way the reference's
duck-typed chain does) the chain is applied as ONE fused local pipeline:
real-extract (pam_unzip) -> rfft -> frequency mask -> batched MFMA
cgemm (Fredholm, the only stage with communication) -> zero-pad ->
irfft -> complex carrier (pam_zip), with zero-copy BROADCAST wrappers
between stages.  The pylops real-FFT convention's sqrt(2) twin-bin
scale and its inverse surround the per-frequency block-diagonal
Fredholm kernel, so they cancel algebraically and neither pass is run
(r02; see _fwd_fft).  Same operator as the composite (GPU parity tests
pin the cgls trace).  Measured at the judged cfg5 shape: r01 composite
2.04/2.69 ms (matvec/rmatvec) -> r01 fused 1.64/2.19 -> r02 (twin-scale
cancellation + pam_unzip/zip + the Fredholm output-wrap copy
elimination) — see DESIGN.md §8 for the final figures.
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

    # Fredholm kernel, prescaled (ref :36-43).  The EXTRA 1/nt folds the
    # ortho 1/sqrt(nt) of the chain's forward and inverse FFTs into G
    # (our strided rocFFT transforms are unscaled; a per-frequency
    # scalar commutes with the per-frequency block-diagonal kernel, so
    # the operator is unchanged — see _FusedMDC)
    if prescaled:
        Frop = MPIFredholm1(G * (1.0 / nt), nv, saveGt=saveGt,
                            usematmul=usematmul, base_comm=comm,
                            dtype=dtype)
    else:
        Frop = MPIFredholm1((dr * dt * np.sqrt(nt) / nt) * G, nv,
                            saveGt=saveGt, usematmul=usematmul,
                            base_comm=comm, dtype=dtype)
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
    as one fused per-rank pipeline (module docstring).  The chain AS A
    WHOLE equals the composite built from fftlocal.FFTLocal's pylops
    convention (ortho rfft with sqrt(2)-scaled conjugate-twin bins);
    internally the twin scales cancel and the ortho norms are folded
    into the Fredholm kernel, so the strided rocFFT transforms run
    unscaled (exact operator algebra — parity-tested)."""

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

    @property
    def _rt(self):
        return torch.float32 if self._CT[self.cdtype] == torch.complex64 \
            else torch.float64

    # PAM_MDC_FFT=strided falls back to the strided rocFFT plans (the
    # r02 A/B loser at the cfg5 shape: rocFFT's strided real plans run
    # their own full pack/unpack copy kernels; the contig pipeline
    # replaces them with the transpose-FUSED unzip/zip kernels + one
    # LDS-tiled ctranspose per direction)
    _CONTIG = __import__("os").environ.get("PAM_MDC_FFT", "contig") \
        != "strided"

    def _fwd_fft(self, x: DistributedArray, nmid: int,
                 shift: bool) -> torch.Tensor:
        """real-extract + rfft along the time axis -> (nfft, m) complex.

        The transform is UNSCALED and the pylops real-FFT convention's
        sqrt(2) twin-bin scale is NOT applied: around the per-frequency
        block-diagonal Fredholm kernel the twins cancel and the ortho
        1/sqrt(nt) of both directions folds into G (MPIMDC factory) —
        exact operator algebra, parity-tested.  Standalone FFT
        operators (fftlocal.FFTLocal) keep the full convention."""
        from . import _ffi
        t = x.local_array.reshape(self.nt, nmid * self.nv)
        m = t.shape[1]
        stream = torch.cuda.current_stream(t.device).cuda_stream
        ct = self._CT[self.cdtype]
        rcode = _ffi.dtype_code(self._rt)
        if self._CONTIG:
            # complex (nt, m) -> real (m, nt), transpose fused with the
            # real extraction
            r = torch.empty((m, self.nt), device=t.device, dtype=self._rt)
            if t.is_complex():
                _ffi.checked(_ffi.lib().pam_unzip_t(
                    stream, r.data_ptr(), t.contiguous().data_ptr(),
                    self.nt, m, _ffi.dtype_code(t.dtype)), "unzip_t")
            else:
                _ffi.checked(_ffi.lib().pam_transpose(
                    stream, t.to(self._rt).contiguous().data_ptr(),
                    r.data_ptr(), self.nt, m, rcode), "transpose")
            if shift:
                r = torch.fft.ifftshift(r, dim=1)
            F = torch.empty((m, self.nfft), device=t.device, dtype=ct)
            _ffi.checked(_ffi.lib().pam_rfft_contig(
                stream, r.contiguous().data_ptr(), F.data_ptr(), self.nt,
                m, rcode), "rfft_contig")
            f = torch.empty((self.nfft, m), device=t.device, dtype=ct)
            _ffi.checked(_ffi.lib().pam_ctranspose(
                stream, F.data_ptr(), f.data_ptr(), m, self.nfft, 0,
                _ffi.dtype_code(ct)), "ctranspose")
            return f
        if t.is_complex():
            r = torch.empty(t.shape, device=t.device, dtype=self._rt)
            _ffi.checked(_ffi.lib().pam_unzip(
                stream, r.data_ptr(), t.contiguous().data_ptr(),
                t.numel(), _ffi.dtype_code(t.dtype)), "unzip")
            t = r
        else:
            t = t.to(self._rt)
        if shift:
            t = torch.fft.ifftshift(t, dim=0)
        f = torch.empty((self.nfft, m), device=t.device, dtype=ct)
        _ffi.checked(_ffi.lib().pam_rfft_strided(
            stream, t.contiguous().data_ptr(), f.data_ptr(), self.nt, m,
            rcode), "rfft_strided")
        return f

    def _inv_fft(self, fr: torch.Tensor, nmid: int,
                 shift: bool) -> torch.Tensor:
        """zero-pad + irfft + complex carrier -> (nt, m) complex (the
        composite chain's stage wrappers carry complex storage end to
        end — mirrored)."""
        from . import _ffi
        m = nmid * self.nv
        ct = self._CT[self.cdtype]
        rcode = _ffi.dtype_code(self._rt)
        stream = torch.cuda.current_stream(fr.device).cuda_stream
        # zero-pad the masked bins back to nfft (IdentityLocal adjoint)
        if self.nfreq == self.nfft:
            z = fr.reshape(self.nfft, m)  # fr is fresh (clobberable)
        else:
            z = torch.zeros(self.nfft, m, dtype=fr.dtype,
                            device=fr.device)
            z[: self.nfreq] = fr.reshape(self.nfreq, m)
        if self._CONTIG:
            Zt = torch.empty((m, self.nfft), device=z.device, dtype=ct)
            _ffi.checked(_ffi.lib().pam_ctranspose(
                stream, z.contiguous().data_ptr(), Zt.data_ptr(),
                self.nfft, m, 0, _ffi.dtype_code(ct)), "ctranspose")
            r = torch.empty((m, self.nt), device=z.device, dtype=self._rt)
            _ffi.checked(_ffi.lib().pam_irfft_contig(
                stream, Zt.data_ptr(), r.data_ptr(), self.nt, m, rcode),
                "irfft_contig")
            if shift:
                r = torch.fft.fftshift(r, dim=1)
            out = torch.empty((self.nt, m), device=z.device, dtype=ct)
            _ffi.checked(_ffi.lib().pam_zip_t(
                stream, out.data_ptr(), r.contiguous().data_ptr(),
                self.nt, m, _ffi.dtype_code(ct)), "zip_t")
            return out
        r = torch.empty((self.nt, m), device=z.device, dtype=self._rt)
        _ffi.checked(_ffi.lib().pam_irfft_strided(
            stream, z.contiguous().data_ptr(), r.data_ptr(), self.nt, m,
            rcode), "irfft_strided")
        if shift:
            r = torch.fft.fftshift(r, dim=0)
        out = torch.empty((self.nt, m), device=z.device, dtype=ct)
        _ffi.checked(_ffi.lib().pam_zip(
            stream, out.data_ptr(), r.contiguous().data_ptr(),
            r.numel(), _ffi.dtype_code(ct)), "zip")
        return out

    _CT = {np.dtype(np.complex64): torch.complex64,
           np.dtype(np.complex128): torch.complex128}

    def _matvec(self, x: DistributedArray) -> DistributedArray:
        f = self._fwd_fft(x, self.nr, self.twosided)       # Fop
        f = f[: self.nfreq]                                # Iop
        fr = self.Frop.matvec(self._wrap(f.contiguous()))  # Fredholm
        out = self._inv_fft(fr.local_array, self.ns, False)  # I1^H F1^H
        return self._wrap(out)

    def _rmatvec(self, y: DistributedArray) -> DistributedArray:
        f = self._fwd_fft(y, self.ns, False)               # F1op, I1op
        f = f[: self.nfreq]
        fr = self.Frop.rmatvec(self._wrap(f.contiguous()))
        out = self._inv_fft(fr.local_array, self.nr, self.twosided)
        return self._wrap(out)
