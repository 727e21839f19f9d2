"""Serial N-D FFT restatement — TEST INFRASTRUCTURE ONLY.

Restates the serial pylops FFTND convention the reference's distributed
FFTs are tested against (ref tests/test_ffts.py:120-128 compares
MPIFFTND with pylops.signalprocessing.FFTND on rank 0; pylops is not
vendored under /root/reference, so the convention is restated from the
reference's own semantics in signalprocessing/FFTND.py:214-316):

  forward: unnormalized rfftn/fftn over ``axes`` (real transform on
           axes[-1]); real=True scales the conjugate-twin bins
           (1 .. 1+(n-1)//2 along axes[-1]) by sqrt(2);
           norm="1/n" multiplies by 1/prod(nffts);
  adjoint: ifftn (+ irfft on axes[-1] last) with numpy's backward
           normalization (1/N); norm="none" multiplies by prod(nffts)
           (true adjoint of the unnormalized forward); real / float
           dtypes return the real part.

Pinned by the adjoint-identity and round-trip tests in
tests/test_oracle_fftnd.py.
"""
from typing import Sequence

import numpy as np


def _flags(v, n):
    if isinstance(v, (bool, np.bool_)):
        return np.full(n, bool(v))
    return np.asarray([bool(x) for x in v])


def _prep(dims, axes, real, ifftshift_before, fftshift_after):
    nd = len(dims)
    axes = np.asarray([a % nd for a in
                       (axes if isinstance(axes, Sequence) else (axes,))])
    nffts = tuple(int(dims[a]) for a in axes)
    ifb = _flags(ifftshift_before, len(axes))
    fsa = _flags(fftshift_after, len(axes))
    dimsd = list(dims)
    for a, n in zip(axes, nffts):
        dimsd[a] = n
    if real:
        dimsd[axes[-1]] = nffts[-1] // 2 + 1
    return axes, nffts, ifb, fsa, tuple(dimsd)


def serial_fftnd_mv(x: np.ndarray, dims, axes, norm="none", real=False,
                    ifftshift_before=False, fftshift_after=False,
                    clinear=None) -> np.ndarray:
    axes, nffts, ifb, fsa, dimsd = _prep(dims, axes, real,
                                         ifftshift_before, fftshift_after)
    if clinear is None:
        clinear = not (real or np.issubdtype(np.asarray(x).dtype,
                                             np.floating))
    v = np.asarray(x).reshape(dims)
    if ifb.any():
        v = np.fft.ifftshift(v, axes=axes[ifb])
    if not clinear:
        v = v.real
    if real:
        y = np.fft.rfft(v, n=nffts[-1], axis=int(axes[-1]))
        if len(axes) > 1:
            y = np.fft.fftn(y, axes=[int(a) for a in axes[:-1]])
        sl = [slice(None)] * y.ndim
        sl[int(axes[-1])] = slice(1, 1 + (nffts[-1] - 1) // 2)
        y[tuple(sl)] *= np.sqrt(2)
    else:
        y = np.fft.fftn(v, axes=[int(a) for a in axes])
    if norm == "1/n":
        y = y * (1.0 / np.prod(nffts))
    if fsa.any():
        y = np.fft.fftshift(y, axes=axes[fsa])
    return np.asarray(y, dtype=np.complex128).ravel()


def serial_fftnd_rmv(y: np.ndarray, dims, axes, norm="none", real=False,
                     ifftshift_before=False, fftshift_after=False,
                     clinear=None) -> np.ndarray:
    axes, nffts, ifb, fsa, dimsd = _prep(dims, axes, real,
                                         ifftshift_before, fftshift_after)
    if clinear is None:
        clinear = not real
    v = np.asarray(y).reshape(dimsd).astype(np.complex128)
    if fsa.any():
        v = np.fft.ifftshift(v, axes=axes[fsa])
    if real:
        sl = [slice(None)] * v.ndim
        sl[int(axes[-1])] = slice(1, 1 + (nffts[-1] - 1) // 2)
        v = v.copy()
        v[tuple(sl)] /= np.sqrt(2)
        if len(axes) > 1:
            v = np.fft.ifftn(v, axes=[int(a) for a in axes[:-1]])
        z = np.fft.irfft(v, n=nffts[-1], axis=int(axes[-1]))
    else:
        z = np.fft.ifftn(v, axes=[int(a) for a in axes])
    if norm == "none":
        z = z * float(np.prod(nffts))
    if not clinear:
        z = z.real
    if ifb.any():
        z = np.fft.fftshift(z, axes=axes[ifb])
    return z.ravel()
