"""Minimal pylops stub (TEST INFRASTRUCTURE — oracle side only).

Provides exactly the symbols /root/reference/pylops_mpi imports
(inventoried per-module; see SURVEY.md §8c third-party list).  The
serial operators used in the ref-parity suites (MatrixMult for
BlockDiag blocks) are tiny faithful numpy implementations of pylops'
published semantics; symbols only touched at import time (FFT,
NonStationaryConvolve1D, ...) raise on use.
"""
from .linearoperator import LinearOperator
from .basicoperators import (FirstDerivative, Identity, MatrixMult,
                             SecondDerivative)

__all__ = ["LinearOperator", "Identity", "MatrixMult", "FirstDerivative",
           "SecondDerivative"]
