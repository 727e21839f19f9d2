"""Rank-simulated MPIFredholm1 + MDC chain — TEST INFRASTRUCTURE ONLY.

Restates /root/reference/pylops_mpi/signalprocessing/Fredholm1.py:78-169
(slice-of-own-rows batched matmul + allgather, BROADCAST in/out) and the
MDC chain /root/reference/pylops_mpi/waveeqprocessing/MDC.py:12-74
(F1^H I1^H Fredholm I F with the dr*dt*sqrt(nt) prescale).

pylops is not vendored, so the serial real-FFT convention is re-derived
(sqrt(2)-scaled twin bins over the ortho rfft => F^H F = I; see
pylops_mpi_amd/fftlocal.py) and locked by the adjoint/round-trip tests in
tests/test_oracle_fredholm.py.
"""
import math
from typing import Sequence

import numpy as np

from .ranksim import Partition, SimArray


class SimFredholm1:
    def __init__(self, G_blocks: Sequence[np.ndarray], nz: int = 1,
                 saveGt: bool = False):
        self.G = [np.asarray(g) for g in G_blocks]
        self.nz = nz
        self.nsls = [g.shape[0] for g in self.G]
        self.nx, self.ny = self.G[0].shape[1], self.G[0].shape[2]
        nslstot = int(sum(self.nsls))
        self.islstart = np.insert(np.cumsum(self.nsls)[:-1], 0, 0)
        self.islend = np.cumsum(self.nsls)
        self.dims = (nslstot, self.ny, self.nz)
        self.dimsd = (nslstot, self.nx, self.nz)
        self.shape = (int(np.prod(self.dimsd)), int(np.prod(self.dims)))
        self.saveGt = saveGt
        if saveGt:
            self.GT = [g.transpose(0, 2, 1).conj() for g in self.G]
        self.dtype = self.G[0].dtype

    def _apply(self, x: SimArray, forward: bool) -> SimArray:
        assert x.partition in (Partition.BROADCAST,
                               Partition.UNSAFE_BROADCAST)
        P = x.size
        dims = self.dims if forward else self.dimsd
        y1s = []
        for r in range(P):
            xl = x.locals[r].reshape(dims)
            xs = xl[int(self.islstart[r]): int(self.islend[r])]
            if forward:
                y1s.append(np.matmul(self.G[r], xs))        # ref :123
            elif self.saveGt:
                y1s.append(np.matmul(self.GT[r], xs))       # ref :150
            else:
                y1s.append(np.matmul(xs.transpose(0, 2, 1).conj(),
                                     self.G[r]).transpose(0, 2, 1).conj())
        full = np.vstack(y1s).ravel()                        # ref :129,167
        return SimArray([full.copy() for _ in range(P)], full.shape,
                        partition=Partition.BROADCAST)

    def matvec(self, x: SimArray) -> SimArray:
        return self._apply(x, True)

    def rmatvec(self, x: SimArray) -> SimArray:
        return self._apply(x, False)

    def dense(self) -> np.ndarray:
        """Explicit block-diagonal-over-slices matrix (independent pin)."""
        nsl = sum(self.nsls)
        A = np.zeros((nsl * self.nx * self.nz, nsl * self.ny * self.nz),
                     dtype=self.G[0].dtype)
        Gall = np.vstack(self.G)
        for isl in range(nsl):
            for z in range(self.nz):
                rows = (np.arange(self.nx) * self.nz + z
                        + isl * self.nx * self.nz)
                cols = (np.arange(self.ny) * self.nz + z
                        + isl * self.ny * self.nz)
                A[np.ix_(rows, cols)] = Gall[isl]
        return A


# --------------------------------------------------------------- serial FFT
def serial_rfft_op(x: np.ndarray, nt: int, ifftshift_before=False):
    """Forward of the re-derived pylops real-FFT convention (axis 0).
    Complex inputs take the real part (real FFT of a real model that may
    be carried in complex storage by the solver)."""
    x = np.asarray(x).real
    if ifftshift_before:
        x = np.fft.ifftshift(x, axes=0)
    y = np.fft.rfft(x, n=nt, axis=0, norm="ortho")
    y[1:(nt + 1) // 2] *= math.sqrt(2.0)
    return y


def serial_rfft_adj(z: np.ndarray, nt: int, ifftshift_before=False):
    z = z.copy()
    z[1:(nt + 1) // 2] /= math.sqrt(2.0)
    x = np.fft.irfft(z, n=nt, axis=0, norm="ortho")
    if ifftshift_before:
        x = np.fft.fftshift(x, axes=0)
    return x


class SimMDC:
    """The MDC chain on rank-simulated BROADCAST arrays (ref MDC.py:12-74)."""

    def __init__(self, G_blocks: Sequence[np.ndarray], nt: int, nv: int,
                 nfreq: int, dt: float = 1.0, dr: float = 1.0,
                 twosided: bool = False, saveGt: bool = True,
                 prescaled: bool = False):
        if twosided and nt % 2 == 0:
            raise ValueError('nt must be odd number')
        self.nt, self.nv = nt, nv
        _, self.ns, self.nr = G_blocks[0].shape
        self.nfft = int(np.ceil((nt + 1) / 2))
        self.nfreq = min(nfreq, self.nfft)
        self.twosided = twosided
        scale = 1.0 if prescaled else dr * dt * np.sqrt(nt)  # ref :36-43
        self.Fr = SimFredholm1([scale * g for g in G_blocks], nv,
                               saveGt=saveGt)
        self.shape = (nt * self.ns * nv, nt * self.nr * nv)

    def matvec(self, x: SimArray) -> SimArray:
        P = x.size
        nt, nr, ns, nv = self.nt, self.nr, self.ns, self.nv
        xg = x.locals[0].reshape(nt, nr, nv)
        f = serial_rfft_op(xg, nt, self.twosided)             # Fop
        f = f.reshape(-1)[: self.nfreq * nr * nv]             # Iop
        fr_in = SimArray([f.copy() for _ in range(P)], f.shape,
                         partition=Partition.BROADCAST)
        fr = self.Fr.matvec(fr_in).locals[0]                  # Frop
        z = np.zeros(self.nfft * ns * nv, dtype=fr.dtype)     # I1op^H
        z[: self.nfreq * ns * nv] = fr
        out = serial_rfft_adj(z.reshape(self.nfft, ns, nv), nt)  # F1op^H
        out = out.reshape(-1)
        return SimArray([out.copy() for _ in range(P)], out.shape,
                        partition=Partition.BROADCAST)

    def rmatvec(self, y: SimArray) -> SimArray:
        P = y.size
        nt, nr, ns, nv = self.nt, self.nr, self.ns, self.nv
        yg = y.locals[0].reshape(nt, ns, nv)
        f = serial_rfft_op(yg, nt, False)                     # F1op
        f = f.reshape(-1)[: self.nfreq * ns * nv]             # I1op
        fr_in = SimArray([f.copy() for _ in range(P)], f.shape,
                         partition=Partition.BROADCAST)
        fr = self.Fr.rmatvec(fr_in).locals[0]                 # Frop^H
        z = np.zeros(self.nfft * nr * nv, dtype=fr.dtype)     # Iop^H
        z[: self.nfreq * nr * nv] = fr
        out = serial_rfft_adj(z.reshape(self.nfft, nr, nv), nt,
                              self.twosided)                  # Fop^H
        out = out.reshape(-1)
        return SimArray([out.copy() for _ in range(P)], out.shape,
                        partition=Partition.BROADCAST)
