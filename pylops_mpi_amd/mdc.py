"""MPIMDC — multi-dimensional convolution chain.

Drop-in for /root/reference/pylops_mpi/waveeqprocessing/MDC.py:12-181:
F1^H * I1^H * Fredholm1 * I * F, with the serial FFT/Identity wrapped in
MPILinearOperator (BROADCAST arrays) and the kernel prescaled by
dr*dt*sqrt(nt) (ref :41-43).
"""
import logging

import numpy as np
import torch

from .comm import PamComm, get_default_comm
from .fftlocal import FFTLocal, IdentityLocal
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

    Fop = MPILinearOperator(
        Op=FFTLocal((nt, nr, nv), real=True, ifftshift_before=twosided,
                    dtype=rdtype), base_comm=comm)
    F1op = MPILinearOperator(
        Op=FFTLocal((nt, ns, nv), real=True, ifftshift_before=False,
                    dtype=rdtype), base_comm=comm)
    Iop = MPILinearOperator(
        Op=IdentityLocal(nfreq * nr * nv, nfft * nr * nv, dtype=dtype),
        base_comm=comm)
    I1op = MPILinearOperator(
        Op=IdentityLocal(nfreq * ns * nv, nfft * ns * nv, dtype=dtype),
        base_comm=comm)

    MDCop = F1op.H * I1op.H * Frop * Iop * Fop  # ref :65-69
    MDCop.dtype = rdtype                        # ref :71-72
    return MDCop
