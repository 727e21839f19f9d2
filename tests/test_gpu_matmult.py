"""GPU parity for the MFMA GEMM path: pam_gemm / pam_transpose kernels and
MPIMatrixMult (1x1 grid; multi-rank grids covered by the world-4 gloo
suite + oracle layout pins)."""
import numpy as np
import pytest
import torch
from numpy.testing import assert_allclose

import oracle.matmult as om
import pylops_mpi_amd as pm
from pylops_mpi_amd import _ffi

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _init():
    from pylops_mpi_amd.comm import init_default_comm
    init_default_comm(torch.device("cuda:0"))


def dev(a, dt=None):
    return torch.as_tensor(a, device="cuda:0", dtype=dt)


def run_gemm(A, B, C=None, accumulate=False):
    A, B = A.contiguous(), B.contiguous()
    M, K = A.shape
    N = B.shape[1]
    if C is None:
        C = torch.empty((M, N), dtype=A.dtype, device=A.device)
    s = torch.cuda.current_stream(A.device).cuda_stream
    _ffi.checked(_ffi.lib().pam_gemm(
        s, A.data_ptr(), B.data_ptr(), C.data_ptr(), M, N, K, K, N, N,
        1 if accumulate else 0, _ffi.dtype_code(A.dtype)), "gemm")
    return C


SHAPES = [(128, 128, 128), (256, 160, 512), (100, 37, 53), (64, 64, 1),
          (1, 130, 257), (513, 129, 65)]


@pytest.mark.parametrize("M,K,N", SHAPES)
def test_gemm_f64(M, K, N):
    rng = np.random.default_rng(1)
    # asymmetric operands (transpose-detecting, guide §5.4 rule 16)
    A = rng.standard_normal((M, K))
    B = rng.standard_normal((K, N)) + np.arange(N)[None, :] * 1e-3
    C = run_gemm(dev(A), dev(B))
    assert_allclose(C.cpu().numpy(), A @ B, rtol=1e-12, atol=1e-10)


@pytest.mark.parametrize("M,K,N", SHAPES)
def test_gemm_f32(M, K, N):
    rng = np.random.default_rng(2)
    A = rng.standard_normal((M, K)).astype(np.float32)
    B = (rng.standard_normal((K, N))
         + np.arange(N)[None, :] * 1e-3).astype(np.float32)
    C = run_gemm(dev(A), dev(B))
    ref = A.astype(np.float64) @ B.astype(np.float64)
    err = np.abs(C.cpu().numpy() - ref).max()
    scale = np.abs(ref).max() + 1
    assert err / scale < 5e-6 * max(1, K / 64), (err, scale)


def test_gemm_accumulate():
    rng = np.random.default_rng(3)
    A = rng.standard_normal((96, 70))
    B = rng.standard_normal((70, 110))
    C0 = rng.standard_normal((96, 110))
    C = dev(C0.copy())
    run_gemm(dev(A), dev(B), C, accumulate=True)
    assert_allclose(C.cpu().numpy(), C0 + A @ B, rtol=1e-12, atol=1e-10)


def test_transpose():
    rng = np.random.default_rng(4)
    for nr, nc in [(32, 32), (100, 37), (1, 257), (513, 65)]:
        A = rng.standard_normal((nr, nc))
        At = torch.empty((nc, nr), dtype=torch.float64, device="cuda:0")
        s = torch.cuda.current_stream().cuda_stream
        Ad = dev(A)
        _ffi.checked(_ffi.lib().pam_transpose(
            s, Ad.data_ptr(), At.data_ptr(), nr, nc, 0), "t")
        assert np.array_equal(At.cpu().numpy(), A.T)


@pytest.mark.parametrize("kind", ["block", "summa"])
@pytest.mark.parametrize("dtype", ["float64", "float32"])
def test_matmult_grid1(kind, dtype):
    N, K, M = 67, 45, 33
    rng = np.random.default_rng(5)
    A = rng.standard_normal((N, K)).astype(dtype)
    X = rng.standard_normal((K, M)).astype(dtype)
    Y = rng.standard_normal((N, M)).astype(dtype)
    op = pm.MPIMatrixMult(dev(A), M, kind=kind, dtype=dtype)
    assert op.N == N and op.K == K
    x = pm.DistributedArray((K * M,), dtype=np.dtype(dtype))
    x[:] = dev(X.ravel())
    got = op.matvec(x)
    tol = 1e-12 if dtype == "float64" else 2e-5
    assert_allclose(got.local_array.cpu().numpy(),
                    (A @ X).ravel(), rtol=tol, atol=tol * 10)
    yv = pm.DistributedArray((N * M,), dtype=np.dtype(dtype))
    yv[:] = dev(Y.ravel())
    gotr = op.rmatvec(yv)
    assert_allclose(gotr.local_array.cpu().numpy(),
                    (A.T @ Y).ravel(), rtol=tol, atol=tol * 10)


def test_matmult_saveat_and_dottest():
    N, K, M = 48, 40, 24
    rng = np.random.default_rng(6)
    A = rng.standard_normal((N, K))
    op = pm.MPIMatrixMult(dev(A), M, kind="summa", saveAt=True)
    u = pm.DistributedArray((K * M,))
    u[:] = dev(rng.standard_normal(K * M))
    v = pm.DistributedArray((N * M,))
    v[:] = dev(rng.standard_normal(N * M))
    assert pm.dottest(op, u, v, rtol=1e-10)


@pytest.mark.parametrize("kind", ["block", "summa"])
@pytest.mark.parametrize("dtype", ["complex64", "complex128"])
def test_matmult_grid1_complex(kind, dtype):
    """Complex MatrixMult (r01 advice): batch-1 MFMA cgemm panels +
    conj-transpose materialization, vs dense A@x / A^H y (the reference
    supports complex throughout, ref MatrixMult.py:346-352,416,737)."""
    N, K, M = 67, 45, 33
    rng = np.random.default_rng(7)
    A = (rng.standard_normal((N, K))
         + 1j * rng.standard_normal((N, K))).astype(dtype)
    X = (rng.standard_normal((K, M))
         + 1j * rng.standard_normal((K, M))).astype(dtype)
    Y = (rng.standard_normal((N, M))
         + 1j * rng.standard_normal((N, M))).astype(dtype)
    for saveAt in (False, True):
        op = pm.MPIMatrixMult(dev(A), M, kind=kind, dtype=dtype,
                              saveAt=saveAt)
        assert op.N == N and op.K == K
        x = pm.DistributedArray((K * M,), dtype=np.dtype(dtype))
        x[:] = dev(X.ravel())
        got = op.matvec(x)
        tol = 1e-12 if dtype == "complex128" else 2e-4
        assert_allclose(got.local_array.cpu().numpy(),
                        (A @ X).ravel(), rtol=tol, atol=tol * 10)
        yv = pm.DistributedArray((N * M,), dtype=np.dtype(dtype))
        yv[:] = dev(Y.ravel())
        gotr = op.rmatvec(yv)
        assert_allclose(gotr.local_array.cpu().numpy(),
                        (A.conj().T @ Y).ravel(), rtol=tol, atol=tol * 10)


@pytest.mark.parametrize("dtype", ["complex64", "complex128"])
def test_ctranspose(dtype):
    rng = np.random.default_rng(8)
    for nr, nc in [(32, 32), (33, 65), (1, 7), (100, 3)]:
        A = (rng.standard_normal((nr, nc))
             + 1j * rng.standard_normal((nr, nc))).astype(dtype)
        Ad = dev(A).contiguous()
        At = torch.empty((nc, nr), dtype=Ad.dtype, device="cuda:0")
        s = torch.cuda.current_stream().cuda_stream
        for conj in (0, 1):
            _ffi.checked(_ffi.lib().pam_ctranspose(
                s, Ad.data_ptr(), At.data_ptr(), nr, nc, conj,
                _ffi.dtype_code(Ad.dtype)), "ct")
            want = A.conj().T if conj else A.T
            assert np.array_equal(At.cpu().numpy(), want)


@pytest.mark.parametrize("dtype", ["complex64", "complex128"])
def test_cgemm_accumulate(dtype):
    """C += op(A) @ B (the accumulate flag added for complex SUMMA)."""
    rng = np.random.default_rng(9)
    B_, M, K, N = 3, 64, 48, 40
    A = (rng.standard_normal((B_, M, K))
         + 1j * rng.standard_normal((B_, M, K))).astype(dtype)
    X = (rng.standard_normal((B_, K, N))
         + 1j * rng.standard_normal((B_, K, N))).astype(dtype)
    C0 = (rng.standard_normal((B_, M, N))
          + 1j * rng.standard_normal((B_, M, N))).astype(dtype)
    Cd = dev(C0).contiguous()
    Ad, Xd = dev(A).contiguous(), dev(X).contiguous()
    s = torch.cuda.current_stream().cuda_stream
    _ffi.checked(_ffi.lib().pam_cgemm_batched(
        s, Ad.data_ptr(), Xd.data_ptr(), Cd.data_ptr(), B_, M, N, K,
        M * K, K * N, M * N, 0, 1, _ffi.dtype_code(Ad.dtype)), "cgemm")
    tol = 1e-12 if dtype == "complex128" else 2e-4
    assert_allclose(Cd.cpu().numpy(), C0 + A @ X, rtol=tol, atol=tol * 10)


def test_gemm_kt_glds():
    """pam_gemm_kt (256^2-tile glds pipeline, k-major A): correctness vs
    pam_gemm.  Kept as a measured NEGATIVE for throughput (r02: 125-127
    TF vs 136 at 8192^3 — 1 block/CU loses the 4-WG occupancy the 128^2
    register-pipeline kernel enjoys; DESIGN.md negative-results list)."""
    n = 512
    rng = np.random.default_rng(14)
    A = rng.standard_normal((n, n)).astype(np.float32)
    B = rng.standard_normal((n, n)).astype(np.float32)
    Ad, Bd = dev(A).contiguous(), dev(B).contiguous()
    At = torch.empty((n, n), dtype=torch.float32, device="cuda:0")
    C = torch.empty((n, n), dtype=torch.float32, device="cuda:0")
    s = torch.cuda.current_stream().cuda_stream
    _ffi.checked(_ffi.lib().pam_transpose(
        s, Ad.data_ptr(), At.data_ptr(), n, n, 1), "t")
    _ffi.checked(_ffi.lib().pam_gemm_kt(
        s, At.data_ptr(), Bd.data_ptr(), C.data_ptr(), n, n, n, 0, 1),
        "gemm_kt")
    ref = A.astype(np.float64) @ B.astype(np.float64)
    assert_allclose(C.cpu().numpy(), ref, rtol=2e-4, atol=1e-2)
    # accumulate variant
    C0 = rng.standard_normal((n, n)).astype(np.float32)
    Cd = dev(C0).contiguous()
    _ffi.checked(_ffi.lib().pam_gemm_kt(
        s, At.data_ptr(), Bd.data_ptr(), Cd.data_ptr(), n, n, n, 1, 1),
        "gemm_kt_acc")
    assert_allclose(Cd.cpu().numpy(), C0 + ref, rtol=2e-4, atol=1e-2)
    # fast-path precondition: unaligned sizes are refused, not wrong
    bad = _ffi.lib().pam_gemm_kt(
        s, At.data_ptr(), Bd.data_ptr(), C.data_ptr(), n - 8, n, n, 0, 1)
    assert bad != 0
