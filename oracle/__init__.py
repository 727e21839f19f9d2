"""CPU oracle for the pylops-mpi hot path — TEST INFRASTRUCTURE ONLY.

This package is a pure-NumPy, rank-simulating restatement of the reference
(PyLops/pylops-mpi @ /root/reference) semantics for the hot path named by
BASELINE.json's north_star:

  - ``local_split`` remainder rule        (ref DistributedArray.py:42-71)
  - ``to_dist`` scatter slicing           (ref DistributedArray.py:438-491)
  - ``add_ghost_cells`` halo exchange     (ref DistributedArray.py:955-1032)
  - elementwise math / dot / norm         (ref DistributedArray.py:605-838)
  - the ``reshaped`` rebalance arithmetic (ref utils/decorators.py:44-82)
  - MPIFirstDerivative / MPISecondDerivative stencils
                                          (ref basicoperators/FirstDerivative.py:141-318,
                                               basicoperators/SecondDerivative.py:124-256)
  - CG / CGLS recurrences                 (ref optimization/cls_basic.py:12-531)

P "ranks" are simulated sequentially in one process (the reference needs
mpi4py + pylops, neither of which is installed in this container).

Pinning: the reference itself cannot be imported here, so parity is pinned
the way the reference's own tests pin it (tests/test_derivative.py:197-229):
the rank-simulated operator is compared against an INDEPENDENT serial
restatement of the stencil formulas (oracle/serial.py) at rtol 1e-14, and
every adjoint is additionally checked against the explicit dense transpose
of its forward matrix on small sizes.

IMPORT RESTRICTIONS: only ``tests/``, ``__graft_entry__.smoke()`` and
``bench.py``'s ``cpu_baseline`` leg may import or execute anything in this
package. The product path (pylops_mpi_amd) never touches it.
"""
from .ranksim import (  # noqa: F401
    Partition,
    local_split,
    to_dist,
    add_ghost_cells,
    SimArray,
)
from .serial import (  # noqa: F401
    serial_fd1_matvec,
    serial_fd2_matvec,
    dense_matrix_from_matvec,
)
from .stencils import (  # noqa: F401
    SimFirstDerivative,
    SimSecondDerivative,
)
from .cgls import sim_cgls, sim_cg  # noqa: F401
from .sparsity import (sim_ista, sim_fista,  # noqa: F401
                       sim_power_iteration, powerit_rand)
from .proximal import (sim_proximal_gradient_l2l1,  # noqa: F401
                       sim_admml2_l1, SimScaledOp)
from .blockdiag import (SimBlockDiag, SimStackedBlockDiag,  # noqa: F401
                        SimStackedVStack)
from .fredholm import (SimFredholm1, SimMDC,  # noqa: F401
                       serial_rfft_adj, serial_rfft_op)
from .nsconv import (serial_nsconv_mv,  # noqa: F401
                     serial_nsconv_rmv)
from .proximal import (soft_threshold, hard_threshold,  # noqa: F401
                       half_threshold,
                       SerBox, SerL0, SerL1, SerL2,
                       dense_cg, dense_cgls,
                       ser_proximal_gradient, ser_admml2)
from .fftnd import serial_fftnd_mv, serial_fftnd_rmv  # noqa: F401
