"""Product HIP path vs the REFERENCE-generated P=1 fixtures for the
round-3 golden additions: MPIHStack, stacked operators, stacked damped
CGLS, and the proximal subpackage (ProximalGradient/ADMML2).

The fixtures in tests/golden/golden_ref.npz were produced by executing
/root/reference/pylops_mpi in the build container (tests/golden/
refgen.py); this closes HIP == oracle == reference for these paths on
the same inputs (hs_/sbd_/svs_/scgls_/pg_/admm_ keys)."""
import os
import sys

import numpy as np
import pytest
import torch
from numpy.testing import assert_allclose

import pylops_mpi_amd as pm

pytestmark = pytest.mark.gpu

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "golden"))
import refgen  # noqa: E402


@pytest.fixture(scope="module", autouse=True)
def _init():
    from pylops_mpi_amd.comm import init_default_comm
    init_default_comm(torch.device("cuda:0"))


@pytest.fixture(scope="module")
def golden():
    return np.load(refgen.GOLDEN_PATH)


def dev(a):
    return torch.as_tensor(np.ascontiguousarray(a), device="cuda:0")


def host(t):
    return t.cpu().numpy()


def _counts_dist(vec, n):
    d = pm.DistributedArray((n,))
    d[:] = dev(vec)
    return d


def test_hstack_vs_reference_fixtures(golden):
    vmats = refgen.vstack_mats(1)[0]
    hop = pm.MPIHStack([pm.DenseLocal(dev(A.T)) for A in vmats])
    nv_rows = int(sum(A.shape[0] for A in vmats))
    xg = refgen.make_global_x(nv_rows, 1, seed_shift=1)   # refgen vs_y
    yg = refgen.make_global_x(7, 1)                       # refgen vs_x
    y = hop.matvec(pm.DistributedArray.to_dist(dev(xg)))
    assert_allclose(host(y.asarray()), golden["hs_P1_mv"], rtol=1e-12)
    xb = pm.DistributedArray((7,), partition=pm.Partition.BROADCAST)
    xb[:] = dev(yg)
    z = hop.rmatvec(xb)
    assert_allclose(host(z.asarray()), golden["hs_P1_rmv"], rtol=1e-12)


def _spd_ops():
    spd = refgen.spd_mats(1)[0]
    spd2 = refgen.spd2_mats(1)[0]
    bop1 = pm.MPIBlockDiag([pm.DenseLocal(dev(A)) for A in spd])
    bop2 = pm.MPIBlockDiag([pm.DenseLocal(dev(A)) for A in spd2])
    n = int(sum(A.shape[0] for A in spd))
    return bop1, bop2, n


def test_stacked_ops_vs_reference_fixtures(golden):
    bop1, bop2, n = _spd_ops()
    bd = refgen.blockdiag_mats(1)[0]
    bdop = pm.MPIBlockDiag([pm.DenseLocal(dev(A)) for A in bd])
    nrb = int(sum(A.shape[0] for A in bd))
    ncb = int(sum(A.shape[1] for A in bd))
    sbd = pm.MPIStackedBlockDiag([bdop, bop1])
    xst = pm.StackedDistributedArray(
        [pm.DistributedArray.to_dist(dev(refgen.make_global_x(ncb, 1))),
         _counts_dist(refgen.make_global_x(n, 1, seed_shift=6), n)])
    got = np.concatenate([host(d.asarray())
                          for d in sbd.matvec(xst).distarrays])
    assert_allclose(got, golden["sbd_P1_mv"], rtol=1e-12)
    yst = pm.StackedDistributedArray(
        [pm.DistributedArray.to_dist(
            dev(refgen.make_global_x(nrb, 1, seed_shift=1))),
         _counts_dist(refgen.make_global_x(n, 1, seed_shift=7), n)])
    got = np.concatenate([host(d.asarray())
                          for d in sbd.rmatvec(yst).distarrays])
    assert_allclose(got, golden["sbd_P1_rmv"], rtol=1e-12)
    svop = pm.MPIStackedVStack([bop1, bop2])
    xsv = _counts_dist(refgen.make_global_x(n, 1), n)
    got = np.concatenate([host(d.asarray())
                          for d in svop.matvec(xsv).distarrays])
    assert_allclose(got, golden["svs_P1_mv"], rtol=1e-12)
    ysv = pm.StackedDistributedArray(
        [_counts_dist(refgen.make_global_x(n, 1, seed_shift=8), n),
         _counts_dist(refgen.make_global_x(n, 1, seed_shift=9), n)])
    got = host(svop.rmatvec(ysv).asarray())
    assert_allclose(got, golden["svs_P1_rmv"], rtol=1e-12)
    # damped CGLS over the stacked VStack: full trajectory
    xs, *_, cost = pm.cgls(svop, ysv, _counts_dist(np.zeros(n), n),
                           niter=refgen.CGLS_NITER, damp=0.4, tol=0.0)
    assert_allclose(host(xs.asarray()), golden["scgls_P1_x"],
                    rtol=1e-9, atol=1e-11)
    assert_allclose(np.asarray(cost), golden["scgls_P1_cost"], rtol=1e-9)


def test_proximal_vs_reference_fixtures(golden):
    from pylops_mpi_amd.proximal import L1, MPIL2, MPIProxOperator
    from pylops_mpi_amd.proximal.optimization.primal import (ADMML2,
                                                             ProximalGradient)
    bop1, bop2, n = _spd_ops()
    ypg = _counts_dist(refgen.make_global_x(n, 1, seed_shift=10), n)
    l1 = MPIProxOperator(L1(sigma=0.3))
    l2 = MPIL2(Op=bop1, b=ypg, x0=_counts_dist(np.zeros(n), n))
    for an, acc in (("none", None), ("fista", "fista")):
        xpg = ProximalGradient(l2, l1, x0=_counts_dist(np.zeros(n), n),
                               epsg=1.0, tau=0.01, niter=10,
                               acceleration=acc)
        assert_allclose(host(xpg.asarray()), golden[f"pg_P1_{an}_x"],
                        rtol=1e-9, atol=1e-12)
    xa, za = ADMML2(l1, bop1, ypg, bop2, _counts_dist(np.zeros(n), n),
                    tau=0.05, niter=6,
                    kwargs_solver={"niter": 8, "tol": 0.0})
    assert_allclose(host(xa.asarray()), golden["admm_P1_x"],
                    rtol=1e-8, atol=1e-10)
    assert_allclose(host(za.asarray()), golden["admm_P1_z"],
                    rtol=1e-8, atol=1e-10)
