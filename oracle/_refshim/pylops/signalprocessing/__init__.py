"""pylops.signalprocessing stub: import-time symbols only (the
reference's MDC/NonStatConv wrap these serially; the ref-parity suite
pins the distributed Fredholm1/derivatives instead)."""


class _ImportOnly:
    _name = "stub"

    def __init__(self, *a, **k):
        raise NotImplementedError(
            f"pylops stub: serial {self._name} is not implemented — the "
            "ref-parity suite does not construct it")


class FFT(_ImportOnly):
    _name = "FFT"


class Fredholm1(_ImportOnly):
    _name = "Fredholm1"


class NonStationaryConvolve1D:
    """Working serial stand-in (the reference wraps it per rank inside
    MPIBlockDiag + MPIHalo, ref signalprocessing/NonStatConvolve1d.py:
    141-190): pylops' published interpolated-filter convolution as
    restated by oracle/nsconv.py (pylops itself is absent — SURVEY
    §8c); the ref-parity suite pins the DISTRIBUTED halo/blockdiag
    composition against the global serial result."""

    def __init__(self, dims, hs, ih, axis=-1, dtype="float64"):
        import numpy as _np
        self.dims = (dims,) if isinstance(dims, int) \
            else tuple(int(d) for d in dims)
        self.hs = _np.asarray(hs)
        self.ih = _np.asarray(ih)
        self.axis = int(axis) % len(self.dims)
        n = int(_np.prod(self.dims))
        self.shape = (n, n)
        self.dtype = _np.dtype(dtype)

    def matvec(self, x):
        from oracle.nsconv import serial_nsconv_mv
        return serial_nsconv_mv(x, self.dims, self.hs, self.ih, self.axis)

    def rmatvec(self, x):
        from oracle.nsconv import serial_nsconv_rmv
        return serial_nsconv_rmv(x, self.dims, self.hs, self.ih,
                                 self.axis)
