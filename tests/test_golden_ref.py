"""Oracle vs the REFERENCE-generated golden fixtures.

golden_ref.npz was produced by executing /root/reference/pylops_mpi in
the build container (tests/golden/generate_golden_ref.py — P ranks as
threads over the mpi4py/pylops shims).  This test runs EVERYWHERE
(fixtures travel; the reference does not), closing the parity chain

    HIP path  ==  oracle  ==  reference outputs

whose first link is tests/test_gpu_parity.py.  Covers every FD kind/
order/edge at P in {1..4} on 1/2/3-D dims, DistributedArray math and
ghost cells, CGLS traces (damped and undamped), BlockDiag, Fredholm1
(both dtypes, saveGt both ways), MatrixMult block+SUMMA (real and
complex), Gradient/Laplacian, MPIHalo, NonStationaryConvolve1D and the
MDC chain (prescale + masks + composite products/adjoints, serial FFT
convention held common, P=1..4), ISTA/FISTA (soft+hard) and
power_iteration (rank-deterministic init draw), MPIHStack, stacked
operators (MPIStackedBlockDiag/VStack + StackedDistributedArray) and
damped CGLS over a stacked VStack, and the proximal subpackage
(ProximalGradient none/fista + ADMML2 over MPIL2/L1), and masked
sub-communicator reductions (dot/norms per mask group), and the
MatrixMult grid helpers (active_grid_comm / local_block_split /
block_gather, incl. inactive ranks at non-square P, plus the full
non-square-world flow: block+SUMMA MatrixMult on the active 2x2
sub-communicator of an 8-rank world) — 653 pinned arrays.
"""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "golden"))
import refgen  # noqa: E402


@pytest.fixture(scope="module")
def golden():
    assert os.path.exists(refgen.GOLDEN_PATH), \
        "golden_ref.npz missing — run tests/golden/generate_golden_ref.py"
    return np.load(refgen.GOLDEN_PATH)


@pytest.fixture(scope="module")
def oracle_out():
    return refgen.compute_oracle()


def test_same_key_sets(golden, oracle_out):
    assert set(golden.files) == set(oracle_out.keys())


@pytest.mark.parametrize("prefix", [
    "fd1_", "fd2_", "math_", "cgls_", "cg_", "bd_", "fred_",
    "mm_", "vs_", "grad_", "lap_", "halo_", "nsc_", "mdc_",
    "ista_", "fista_", "powit_", "hs_", "sbd_", "svs_", "scgls_", "pg_", "admm_", "mask_", "mmu_", "mmsub_"])
def test_oracle_matches_reference(golden, oracle_out, prefix):
    keys = [k for k in golden.files if k.startswith(prefix)]
    assert keys, f"no golden keys with prefix {prefix}"
    for k in keys:
        a = np.asarray(golden[k]).ravel()
        b = np.asarray(oracle_out[k]).ravel()
        assert a.shape == b.shape, k
        np.testing.assert_allclose(a, b, rtol=1e-13, atol=0, err_msg=k)
