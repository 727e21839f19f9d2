"""Deterministic odd-shape sweep: every op family at awkward sizes
(primes, size-1 trailing dims, non-divisible splits, tiny arrays) vs the
oracle — the fixed parity tests use round sizes; this catches edge-case
indexing bugs (tail columns, guard rows, remainder splits)."""
import numpy as np
import pytest
import torch
from numpy.testing import assert_allclose

import oracle
import pylops_mpi_amd as pm

pytestmark = pytest.mark.gpu


def dev(a):
    return torch.as_tensor(np.ascontiguousarray(a), device="cuda")


def host(t):
    return t.cpu().numpy()


@pytest.fixture(scope="module", autouse=True)
def _init():
    from pylops_mpi_amd.comm import init_default_comm
    init_default_comm(torch.device("cuda:0"))


FD_SHAPES = [(7,), (11, 3), (5, 1), (6, 7, 1), (13, 2, 3), (8, 1, 1),
             (97,), (31, 5), (9, 11)]


@pytest.mark.parametrize("dims", FD_SHAPES)
def test_fuzz_fd_odd_shapes(dims):
    rng = np.random.default_rng(hash(dims) % 2 ** 31)
    n = int(np.prod(dims))
    for kind, order in (("forward", 3), ("backward", 3), ("centered", 3),
                        ("centered", 5)):
        if dims[0] < (5 if order == 5 else 3):
            continue
        for edge in (False, True):
            op = pm.MPIFirstDerivative(dims, kind=kind, order=order,
                                       edge=edge, sampling=1.3)
            sop = oracle.SimFirstDerivative(dims, 1.3, kind, edge, order)
            x = rng.standard_normal(n)
            got = host(op.matvec(
                pm.DistributedArray.to_dist(dev(x))).asarray())
            want = sop.matvec(oracle.to_dist(x, 1)).asarray()
            assert_allclose(got, want, rtol=1e-12, atol=1e-13,
                            err_msg=f"{dims} {kind}{order} edge={edge} mv")
            got = host(op.rmatvec(
                pm.DistributedArray.to_dist(dev(x))).asarray())
            want = sop.rmatvec(oracle.to_dist(x, 1)).asarray()
            assert_allclose(got, want, rtol=1e-12, atol=1e-13,
                            err_msg=f"{dims} {kind}{order} edge={edge} rmv")


@pytest.mark.parametrize("seed", range(6))
def test_fuzz_gemm_odd_shapes(seed):
    rng = np.random.default_rng(100 + seed)
    M, K, N = (int(v) for v in rng.integers(1, 200, 3))
    A = rng.standard_normal((M, K))
    B = rng.standard_normal((K, N))
    from pylops_mpi_amd import _ffi
    a, b = dev(A), dev(B)
    c = torch.empty((M, N), dtype=torch.float64, device="cuda")
    s = torch.cuda.current_stream().cuda_stream
    _ffi.checked(_ffi.lib().pam_gemm(
        s, a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, K, N, N, 0,
        0), "gemm")
    assert_allclose(host(c), A @ B, rtol=1e-11, atol=1e-11,
                    err_msg=f"f64 {M}x{K}x{N}")
    a32, b32 = a.to(torch.float32), b.to(torch.float32)
    c32 = torch.empty((M, N), dtype=torch.float32, device="cuda")
    _ffi.checked(_ffi.lib().pam_gemm(
        s, a32.data_ptr(), b32.data_ptr(), c32.data_ptr(), M, N, K, K, N,
        N, 0, 1), "gemm")
    assert_allclose(host(c32), (A @ B).astype(np.float32), rtol=2e-4,
                    atol=2e-4 * np.sqrt(K), err_msg=f"f32 {M}x{K}x{N}")


@pytest.mark.parametrize("seed", range(6))
def test_fuzz_cgemm_odd_shapes(seed):
    rng = np.random.default_rng(200 + seed)
    nb = int(rng.integers(1, 5))
    M, K, N = (int(v) for v in rng.integers(1, 90, 3))
    G = (rng.standard_normal((nb, M, K))
         + 1j * rng.standard_normal((nb, M, K))).astype(np.complex64)
    X = (rng.standard_normal((nb, K, N))
         + 1j * rng.standard_normal((nb, K, N))).astype(np.complex64)
    from pylops_mpi_amd import _ffi
    g, x = dev(G), dev(X)
    y = torch.empty((nb, M, N), dtype=torch.complex64, device="cuda")
    s = torch.cuda.current_stream().cuda_stream
    _ffi.checked(_ffi.lib().pam_cgemm_batched(
        s, g.data_ptr(), x.data_ptr(), y.data_ptr(), nb, M, N, K,
        M * K, K * N, M * N, 0, 0, _ffi.dtype_code(torch.complex64)),
        "cgemm")
    assert_allclose(host(y), np.matmul(G, X), rtol=1e-3,
                    atol=1e-3 * np.sqrt(K), err_msg=f"b{nb} {M}x{K}x{N}")


@pytest.mark.parametrize("dims,axes,real", [
    ((3, 2), (0, 1), False), ((1, 5), (0, 1), False),
    ((7, 1, 3), (0, 2), True), ((2, 3, 5), (2, 1, 0), True),
    ((13, 7), (1, 0), True), ((4, 4, 1), (0, 1, 2), False),
])
def test_fuzz_fft_odd_shapes(dims, axes, real):
    rng = np.random.default_rng(hash((dims, axes)) % 2 ** 31)
    dt = np.float64 if real else np.complex128
    n = int(np.prod(dims))
    x = rng.standard_normal(n)
    if not real:
        x = x + 1j * rng.standard_normal(n)
    x = x.astype(dt)
    op = pm.MPIFFTND(dims=dims, axes=axes, real=real, dtype=dt)
    got = host(op.matvec(pm.DistributedArray.to_dist(dev(x))).asarray())
    want = oracle.serial_fftnd_mv(x, dims, axes, real=real)
    assert_allclose(got, want, rtol=1e-10, atol=1e-11)
    yv = rng.standard_normal(op.shape[0]) \
        + 1j * rng.standard_normal(op.shape[0])
    got = host(op.rmatvec(pm.DistributedArray.to_dist(dev(yv))).asarray())
    want = oracle.serial_fftnd_rmv(yv, dims, axes, real=real)
    assert_allclose(got, want, rtol=1e-10, atol=1e-11)
