"""pylops.signalprocessing._baseffts stub: the _FFTNorms enum the
reference's FFT base classes import (ref signalprocessing/_baseffts.py:7)."""
from enum import Enum, auto


class _FFTNorms(Enum):
    ORTHO = auto()
    NONE = auto()
    ONE_OVER_N = auto()
