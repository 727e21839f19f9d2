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


class NonStationaryConvolve1D(_ImportOnly):
    _name = "NonStationaryConvolve1D"
