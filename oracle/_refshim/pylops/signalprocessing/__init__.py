"""pylops.signalprocessing stub: import-time symbols only (the
reference's MDC/NonStatConv wrap these serially; the ref-parity suite
pins the distributed Fredholm1/derivatives instead)."""


class _ImportOnly:
    _name = "stub"

    def __init__(self, *a, **k):
        raise NotImplementedError(
            f"pylops stub: serial {self._name} is not implemented — the "
            "ref-parity suite does not construct it")


class FFT:
    """Working serial real-FFT stand-in for the reference MDC chain
    (ref waveeqprocessing/MDC.py:56-59): pylops' real-FFT convention
    (ortho + sqrt2-scaled conjugate-twin bins) as re-derived by
    oracle/fredholm.py serial_rfft_op/adj (pylops absent — SURVEY §8c).
    The ref-parity suite pins the DISTRIBUTED chain construction
    (prescale, masks, composite products/adjoints) with the serial FFT
    held common."""

    def __init__(self, dims, axis=0, real=True, ifftshift_before=False,
                 dtype="float64", **kw):
        import numpy as _np
        dims = tuple(int(d) for d in dims)
        if axis != 0 or not real:
            raise NotImplementedError("stub FFT: axis=0 real only")
        self.dims = dims
        self.nt = dims[0]
        self.m = int(_np.prod(dims[1:])) if len(dims) > 1 else 1
        nfft = self.nt // 2 + 1
        self.shape = (nfft * self.m, self.nt * self.m)
        self.shift = bool(ifftshift_before)
        self.dtype = (_np.ones(1, dtype=_np.dtype(dtype))
                      + 1j).dtype  # complex operator dtype (pylops)

    def _matvec(self, x):
        from oracle.fredholm import serial_rfft_op
        import numpy as _np
        y = serial_rfft_op(_np.asarray(x).reshape(self.nt, self.m),
                           self.nt, self.shift)
        return y.ravel()

    def _rmatvec(self, x):
        from oracle.fredholm import serial_rfft_adj
        import numpy as _np
        nfft = self.nt // 2 + 1
        z = _np.asarray(x).reshape(nfft, self.m).astype(self.dtype)
        return serial_rfft_adj(z, self.nt, self.shift).ravel()

    matvec = _matvec
    rmatvec = _rmatvec


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
