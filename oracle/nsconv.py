"""Serial non-stationary 1-D convolution restatement — TEST ONLY.

The pylops NonStationaryConvolve1D convention (pylops is not vendored
under /root/reference; re-derived from its published algorithm and locked
by dense-transpose adjoint tests): filters hs[nf][hsize] (odd, centred)
anchored at regularly-sampled ih; the filter at position ix linearly
interpolates the two nearest anchors (clamped outside); forward scatters
x[ix] * h_ix around ix (ref MPI wrapper: signalprocessing/
NonStatConvolve1d.py notes :20-23,63-77).
"""
import numpy as np


def interp_h(hs: np.ndarray, ix: float, oh: float, dh: float) -> np.ndarray:
    q = (ix - oh) / dh
    ic = int(np.floor(q))
    if ic < 0:
        return hs[0]
    if ic >= len(hs) - 1:
        return hs[-1]
    w = q - ic
    return (1 - w) * hs[ic] + w * hs[ic + 1]


def serial_nsconv_mv(x: np.ndarray, dims, hs, ih, axis=-1) -> np.ndarray:
    """Scatter form: y[..., n] += x[..., ix] * h_ix[n - ix + hh]."""
    dims = tuple(dims)
    axis = axis % len(dims)
    oh, dh = float(ih[0]), float(ih[1] - ih[0]) if len(ih) > 1 else 1.0
    xa = np.moveaxis(np.asarray(x).reshape(dims), axis, -1)
    d = xa.shape[-1]
    hsize = hs.shape[1]
    hh = hsize // 2
    y = np.zeros_like(xa)
    for ix in range(d):
        h = interp_h(hs, ix, oh, dh)
        x0, x1 = max(0, ix - hh), min(ix + hh + 1, d)
        h0, h1 = max(0, hh - ix), min(hsize, hh + (d - ix))
        y[..., x0:x1] += xa[..., ix:ix + 1] * h[h0:h1]
    return np.moveaxis(y, -1, axis).ravel()


def serial_nsconv_rmv(x: np.ndarray, dims, hs, ih, axis=-1) -> np.ndarray:
    """Adjoint: z[..., ix] = sum_n x[..., n] * h_ix[n - ix + hh]."""
    dims = tuple(dims)
    axis = axis % len(dims)
    oh, dh = float(ih[0]), float(ih[1] - ih[0]) if len(ih) > 1 else 1.0
    xa = np.moveaxis(np.asarray(x).reshape(dims), axis, -1)
    d = xa.shape[-1]
    hsize = hs.shape[1]
    hh = hsize // 2
    z = np.zeros_like(xa)
    for ix in range(d):
        h = interp_h(hs, ix, oh, dh)
        x0, x1 = max(0, ix - hh), min(ix + hh + 1, d)
        h0, h1 = max(0, hh - ix), min(hsize, hh + (d - ix))
        z[..., ix] = np.sum(xa[..., x0:x1] * h[h0:h1], axis=-1)
    return np.moveaxis(z, -1, axis).ravel()
