"""In-container mpi4py shim (TEST INFRASTRUCTURE — oracle side only).

Executes the REFERENCE package /root/reference/pylops_mpi in this
container, which has no MPI at all: P ranks run as P
threads of one process, and this module provides the exact mpi4py API
surface the reference uses (inventoried from its sources —
utils/_mpi.py buffered calls, Distributed.py dispatch,
DistributedArray.py:74-100 subcomm_split, MatrixMult.py:61-79
active_grid_comm, benchmark.py:73 Barrier).

Only tests/golden generation may import this (VERDICT r01 item 1:
"pin parity against the reference executed here"); nothing in the
product package touches it.
"""
from . import MPI  # noqa: F401
