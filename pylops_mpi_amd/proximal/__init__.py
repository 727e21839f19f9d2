"""Distributed proximal operators and solvers — the MI355X-native
equivalent of the reference subpackage pylops_mpi/proximal/ (ref
proximal/__init__.py:1-24, proximal/proximal/__init__.py,
proximal/optimization/__init__.py:1-21).

The reference wraps pyproximal (an UNPINNED pip dependency, absent from
/root/reference); the separable local operators it relies on (Box, L0,
L1) are restated natively in .operators from pyproximal's published
definitions and anchored on the reference's own call sites and tests
(ref tests/test_prox.py, tests/test_proxsolver.py).
"""
from .operators import (MPIProxOperator, MPIL2,  # noqa: F401
                        ProxOperator, Box, L0, L1)
from . import optimization  # noqa: F401

__all__ = ["MPIProxOperator", "MPIL2", "ProxOperator", "Box", "L0", "L1",
           "optimization"]
