"""pylops_mpi_amd — MI355X-native rebuild of the pylops-mpi hot path.

DistributedArray + MPILinearOperator matvec/rmatvec (stencils, elementwise
math, dot/norm, CGLS) as hand-written HIP/CDNA4 kernels + RCCL over xGMI,
drop-in under the reference's operator API
(reference: PyLops/pylops-mpi @ /root/reference).
"""
from .distributedarray import (DistributedArray, Partition,  # noqa: F401
                               local_split)
from .linearoperator import (MPILinearOperator,  # noqa: F401
                             asmpilinearoperator)
from . import deps  # noqa: F401
from .benchmark import benchmark, mark  # noqa: F401
from .blockdiag import MPIBlockDiag  # noqa: F401
from . import matmult  # noqa: F401
from .matmult import (MPIMatrixMult, active_grid_comm,  # noqa: F401
                      block_gather, local_block_split)
from .fredholm import MPIFredholm1  # noqa: F401
from .halo import MPIHalo  # noqa: F401
from .nonstatconv import (MPINonStationaryConvolve1D,  # noqa: F401
                          NonStationaryConvolve1DLocal, halo_block_split)
from .fdlocal import (FirstDerivativeLocal,  # noqa: F401
                      SecondDerivativeLocal)
from .gradient import MPIGradient, MPILaplacian  # noqa: F401
from .mdc import MPIMDC  # noqa: F401
from .fftlocal import FFTLocal, IdentityLocal  # noqa: F401
from .localops import DenseLocal, CallableLocal, LocalOperator  # noqa: F401
from .derivative import (MPIFirstDerivative,  # noqa: F401
                         MPISecondDerivative)
from .solvers import CG, CGLS, cg, cgls, power_iteration  # noqa: F401
from .sparsity import ISTA, FISTA, ista, fista  # noqa: F401
from .stacked import (StackedDistributedArray,  # noqa: F401
                      MPIStackedLinearOperator)
from .vstack import (MPIVStack, MPIHStack,  # noqa: F401
                     MPIStackedVStack, MPIStackedBlockDiag)
from .localops import AdjointLocal  # noqa: F401
from .dottest import dottest  # noqa: F401
from .comm import (PamComm, get_default_comm,  # noqa: F401
                   init_default_comm)
from .fftnd import MPIFFTND, MPIFFT2D  # noqa: F401
from .fft_helper import fftshift_nd, ifftshift_nd  # noqa: F401
from . import proximal  # noqa: F401  (ref pylops_mpi/proximal/)
from .plotting import (plot_distributed_array,  # noqa: F401
                       plot_local_arrays)

__version__ = "0.1.0"
