"""Full-scale property tests (BASELINE.json-size inputs, tier contract:
size-independent properties where the oracle cannot run).

The workload crosses the 2^31-element boundary (2049*4096*256 > 2^31) to
exercise 64-bit indexing in every kernel.  Properties used:
  * derivative of a constant field is exactly zero (edge=False);
  * linearity: matvec(a*x) == a*matvec(x) to fp64 roundoff;
  * adjoint identity <Op x, y> == <x, Op^H y> at full size;
  * norm/dot consistency: dot(x, x) == norm(x)^2.
"""
import numpy as np
import pytest
import torch

import pylops_mpi_amd as pm

pytestmark = pytest.mark.gpu

DIMS = (2049, 4096, 256)  # 2.148e9 pts = 17.2 GB fp64, > 2^31 elements


@pytest.fixture(scope="module", autouse=True)
def _init():
    from pylops_mpi_amd.comm import init_default_comm
    init_default_comm(torch.device("cuda:0"))
    if torch.cuda.get_device_properties(0).total_memory < 150e9:
        pytest.skip("needs a large-HBM device")


def test_fullsize_properties():
    n = int(np.prod(DIMS))
    op = pm.MPIFirstDerivative(DIMS, kind="centered", order=3)

    # constant field -> exactly zero derivative
    x = pm.DistributedArray((n,))
    x[:] = 3.25
    y = op.matvec(x)
    assert float(y.norm(np.inf)) == 0.0
    del y

    # dot/norm consistency on the constant field (exact value known)
    assert float(x.dot(x)) == 3.25 * 3.25 * n
    np.testing.assert_allclose(float(x.norm()), np.sqrt(3.25 * 3.25 * n),
                               rtol=1e-14)

    # linearity + adjoint identity on random data
    g = torch.Generator(device="cuda").manual_seed(7)
    x[:] = torch.randn(n, generator=g, dtype=torch.float64, device="cuda")
    v = pm.DistributedArray((n,))
    v[:] = torch.randn(n, generator=g, dtype=torch.float64, device="cuda")
    y1 = op.matvec(x)
    y2 = op.matvec(x * 2.5)
    y2.iaxpy_(-2.5, y1)                    # y2 = mv(2.5 x) - 2.5 mv(x)
    assert float(y2.norm(np.inf)) < 1e-12
    del y2
    # adjoint identity
    lhs = float(y1.dot(v))
    rhs = float(x.dot(op.rmatvec(v)))
    np.testing.assert_allclose(lhs, rhs, rtol=1e-12)
    del y1

    # determinism at full size (fixed reduction tree)
    vals = {float(x.dot(v)) for _ in range(3)}
    assert len(vals) == 1


def test_fullsize_cgls_steps():
    n = int(np.prod(DIMS))
    op = pm.MPIFirstDerivative(DIMS, kind="centered", order=3)
    g = torch.Generator(device="cuda").manual_seed(8)
    y = pm.DistributedArray((n,))
    y[:] = torch.randn(n, generator=g, dtype=torch.float64, device="cuda")
    x0 = pm.DistributedArray((n,))
    x0[:] = 0.0
    xs, istop, iters, r1, r2, cost = pm.cgls(op, y, x0, niter=3, damp=0.1,
                                             tol=0.0)
    assert iters == 3 and np.all(np.isfinite(cost))
    assert cost[1] <= cost[0]  # residual decreases


def test_fullsize_fftnd_properties():
    """MPIFFTND at a >2^31-element real workload: linearity and the
    F^H F = N identity (norm='none', real) at sizes the serial oracle
    cannot run — the tier's size-independent-property gate."""
    dims = (1025, 2048, 1024)  # 2.149e9 pts, 17.2 GB fp64
    n = int(np.prod(dims))
    op = pm.MPIFFTND(dims=dims, axes=(0, 1, 2), real=True,
                     dtype=np.float64)
    g = torch.Generator(device="cuda").manual_seed(11)
    x = pm.DistributedArray((n,))
    x[:] = torch.randn(n, generator=g, dtype=torch.float64, device="cuda")
    y = op.matvec(x)
    assert y.global_shape == (int(np.prod(op.dimsd)),)
    z = op.rmatvec(y)
    # F^H F = N * I on real inputs (sqrt2-twin convention)
    z.iaxpy_(-float(np.prod(dims)), x)
    num = float(z.norm())
    den = float(np.prod(dims)) * float(x.norm())
    assert num / den < 1e-13
    del y, z
    # linearity
    y1 = op.matvec(x)
    y2 = op.matvec(x * 2.0)
    # complex arrays: use the real view for the fused axpy
    y2.iaxpy_(-2.0, y1)
    assert float(y2.norm()) / float(y1.norm()) < 1e-13
