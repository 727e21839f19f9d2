"""GPU parity for the complex path: complex DistributedArray math, the
batched complex GEMM kernel, MPIFredholm1 and the MPIMDC chain vs the
oracle (world size 1; multi-rank slicing covered by the oracle's rank
simulation + the Fredholm allgather is the comm-tested pad-to-max
scheme)."""
import numpy as np
import pytest
import torch
from numpy.testing import assert_allclose

import oracle
import pylops_mpi_amd as pm
from pylops_mpi_amd import _ffi

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _init():
    from pylops_mpi_amd.comm import init_default_comm
    init_default_comm(torch.device("cuda:0"))


def dev(a):
    return torch.as_tensor(a, device="cuda:0")


def host(t):
    return t.cpu().numpy()


def crand(rng, shape, dtype=np.complex128):
    return (rng.standard_normal(shape)
            + 1j * rng.standard_normal(shape)).astype(dtype)


# ------------------------------------------------------- complex array math
@pytest.mark.parametrize("dtype,rtol", [(np.complex128, 1e-13),
                                        (np.complex64, 1e-5)])
def test_complex_array_math(dtype, rtol):
    rng = np.random.default_rng(0)
    a, b = crand(rng, 1000, dtype), crand(rng, 1000, dtype)
    da, db = pm.DistributedArray.to_dist(dev(a)), \
        pm.DistributedArray.to_dist(dev(b))
    assert_allclose(host((da + db).asarray()), a + b, rtol=rtol)
    assert_allclose(host((da - db).asarray()), a - b, rtol=rtol)
    assert_allclose(host((da * db).asarray()), a * b, rtol=rtol)
    assert_allclose(host((da * (2 - 3j)).asarray()), a * (2 - 3j), rtol=rtol)
    assert_allclose(host((2.5 * da).asarray()), 2.5 * a, rtol=rtol)
    assert_allclose(host((-da).asarray()), -a, rtol=rtol)
    assert_allclose(host(da.conj().asarray()), a.conj(), rtol=rtol)
    c = da.copy()
    c.iaxpy_(0.5, db)
    assert_allclose(host(c.asarray()), a + 0.5 * b, rtol=rtol)


def test_complex_dot_norm():
    rng = np.random.default_rng(1)
    a, b = crand(rng, 4097), crand(rng, 4097)
    da, db = pm.DistributedArray.to_dist(dev(a)), \
        pm.DistributedArray.to_dist(dev(b))
    assert_allclose(complex(da.dot(db)), np.sum(a * b), rtol=1e-12)
    assert_allclose(complex(da.dot(db, vdot=True)), np.vdot(a, b),
                    rtol=1e-12)
    # ref pattern: r.dot(r.conj()) == sum |r|^2 (real)
    assert_allclose(complex(da.dot(da.conj())).real,
                    np.sum(np.abs(a) ** 2), rtol=1e-12)
    assert_allclose(float(da.norm()), np.linalg.norm(a), rtol=1e-12)
    assert_allclose(float(da.norm(np.inf)), np.abs(a).max(), rtol=1e-13)
    assert_allclose(float(da.norm(1)), np.sum(np.abs(a)), rtol=1e-12)


# --------------------------------------------------------- batched cgemm
@pytest.mark.parametrize("dtype,tol", [(np.complex128, 1e-11),
                                       (np.complex64, 2e-4)])
@pytest.mark.parametrize("B,M,K,N", [(5, 4, 6, 5), (3, 64, 256, 256),
                                     (2, 33, 17, 9), (1, 2, 2, 1)])
def test_cgemm_batched(dtype, tol, B, M, K, N):
    rng = np.random.default_rng(2)
    A = crand(rng, (B, M, K), dtype)
    X = crand(rng, (B, K, N), dtype)
    Y = torch.empty((B, M, N), dtype=dev(A).dtype, device="cuda:0")
    s = torch.cuda.current_stream().cuda_stream
    Ad, Xd = dev(A).contiguous(), dev(X).contiguous()
    _ffi.checked(_ffi.lib().pam_cgemm_batched(
        s, Ad.data_ptr(), Xd.data_ptr(), Y.data_ptr(), B, M, N, K,
        M * K, K * N, M * N, 0, 0, _ffi.dtype_code(Ad.dtype)), "cgemm")
    assert_allclose(host(Y), A @ X, rtol=tol, atol=tol)
    # conj-transpose op: Y2 = A^H @ X2, A stored [K=M_stored rows...]
    X2 = crand(rng, (B, M, N), dtype)
    Y2 = torch.empty((B, K, N), dtype=Ad.dtype, device="cuda:0")
    X2d = dev(X2).contiguous()
    _ffi.checked(_ffi.lib().pam_cgemm_batched(
        s, Ad.data_ptr(), X2d.data_ptr(), Y2.data_ptr(), B, K, N, M,
        M * K, M * N, K * N, 1, 0, _ffi.dtype_code(Ad.dtype)), "cgemm")
    assert_allclose(host(Y2), A.conj().transpose(0, 2, 1) @ X2,
                    rtol=tol, atol=tol)


# ------------------------------------------------------------- Fredholm1
@pytest.mark.parametrize("dtype,tol", [("complex128", 1e-11),
                                       ("complex64", 2e-4),
                                       ("float64", 1e-11),
                                       ("float32", 2e-4)])
@pytest.mark.parametrize("saveGt", [True, False])
def test_fredholm_vs_oracle(dtype, tol, saveGt):
    ndt = np.dtype(dtype)
    # arange-patterned kernel as ref tests/test_fredholm.py:120-123
    nsl, nx, ny = 12, 4, 6
    G = (np.arange(nsl * nx * ny).reshape(nsl, nx, ny)
         .astype(np.float64) / (nsl * nx * ny))
    if ndt.kind == "c":
        G = (G + 1j * G[::-1]).astype(ndt)
    else:
        G = G.astype(ndt)
    blocks = [G.copy()]
    sop = oracle.SimFredholm1(blocks, nz=5, saveGt=saveGt)
    op = pm.MPIFredholm1(dev(G), nz=5, saveGt=saveGt, dtype=dtype)
    assert op.shape == sop.shape
    rng = np.random.default_rng(3)
    x = rng.standard_normal(op.shape[1]).astype(ndt)
    y = rng.standard_normal(op.shape[0]).astype(ndt)
    if ndt.kind == "c":
        x = x + 1j * rng.standard_normal(op.shape[1]).astype(ndt)
        y = y + 1j * rng.standard_normal(op.shape[0]).astype(ndt)
    xd = pm.DistributedArray.to_dist(dev(x),
                                     partition=pm.Partition.BROADCAST)
    yd = pm.DistributedArray.to_dist(dev(y),
                                     partition=pm.Partition.BROADCAST)
    from oracle.ranksim import Partition as SP, SimArray
    bx = SimArray([x.copy()], x.shape, partition=SP.BROADCAST)
    by = SimArray([y.copy()], y.shape, partition=SP.BROADCAST)
    assert_allclose(host(op.matvec(xd).local_array),
                    sop.matvec(bx).locals[0], rtol=tol, atol=tol)
    assert_allclose(host(op.rmatvec(yd).local_array),
                    sop.rmatvec(by).locals[0], rtol=tol, atol=tol)


# ------------------------------------------------------------------ MDC
def _mdc_pair(nt=20, nv=2, nfreq=8, ns=3, nr=4, dtype=np.complex128):
    rng = np.random.default_rng(4)
    nfft = nt // 2 + 1
    G = crand(rng, (nfft, ns, nr), dtype)[:nfreq]
    op = pm.MPIMDC(dev(G), nt, nv, nfreq, dt=0.4, dr=2.0, twosided=False)
    sop = oracle.SimMDC([G.copy()], nt, nv, nfreq, dt=0.4, dr=2.0,
                        twosided=False)
    return op, sop, rng


def test_mdc_vs_oracle():
    op, sop, rng = _mdc_pair()
    from oracle.ranksim import Partition as SP, SimArray
    u = rng.standard_normal(op.shape[1])
    v = rng.standard_normal(op.shape[0])
    ud = pm.DistributedArray.to_dist(dev(u.astype(np.complex128)),
                                     partition=pm.Partition.BROADCAST)
    vd = pm.DistributedArray.to_dist(dev(v.astype(np.complex128)),
                                     partition=pm.Partition.BROADCAST)
    got = host(op.matvec(ud).local_array)
    want = sop.matvec(SimArray([u], u.shape, partition=SP.BROADCAST)).locals[0]
    assert_allclose(got.real, want, rtol=1e-10, atol=1e-12)
    assert np.abs(got.imag).max() < 1e-12
    gotr = host(op.rmatvec(vd).local_array)
    wantr = sop.rmatvec(SimArray([v], v.shape,
                                 partition=SP.BROADCAST)).locals[0]
    assert_allclose(gotr.real, wantr, rtol=1e-10, atol=1e-12)


def test_mdc_dottest():
    # real-valued vectors in complex storage: MDC is real-linear only
    # (the real FFT discards imag; pylops marks it clinear=False)
    op, sop, rng = _mdc_pair()
    u = pm.DistributedArray.to_dist(
        dev(rng.standard_normal(op.shape[1]).astype(np.complex128)),
        partition=pm.Partition.BROADCAST)
    v = pm.DistributedArray.to_dist(
        dev(rng.standard_normal(op.shape[0]).astype(np.complex128)),
        partition=pm.Partition.BROADCAST)
    assert pm.dottest(op, u, v, rtol=1e-10)


def test_mdc_cgls_trace():
    """CGLS driving the MDC chain (the reference's mdd inversion pattern,
    ref tutorials mdd.py) vs the oracle recurrence."""
    op, sop, rng = _mdc_pair()
    n, m = op.shape
    yg = rng.standard_normal(n)
    yd = pm.DistributedArray.to_dist(dev(yg.astype(np.complex128)),
                                     partition=pm.Partition.BROADCAST)
    x0 = pm.DistributedArray((m,), partition=pm.Partition.BROADCAST,
                             dtype=np.complex128)
    x0[:] = 0.0
    xs, _, _, _, _, cost = pm.cgls(op, yd, x0, niter=12, damp=0.3, tol=0.0)
    from oracle.ranksim import Partition as SP, SimArray

    ys = SimArray([yg.astype(np.complex128)], (n,), partition=SP.BROADCAST)
    x0s = SimArray([np.zeros(m, np.complex128)], (m,),
                   partition=SP.BROADCAST)
    xo, cost_ref = oracle.sim_cgls(sop, ys, x0s, niter=12, damp=0.3, tol=0.0)
    assert_allclose(np.asarray(cost), np.asarray(cost_ref), rtol=1e-6,
                    atol=1e-12)
    assert_allclose(host(xs.asarray()), xo.locals[0], rtol=1e-6, atol=1e-9)


def test_cgls_complex_operator():
    """CGLS over a genuinely complex operator (the reference's par*j
    solver cases, ref tests/test_solver.py:40-100 parameter grid) vs the
    dense complex recurrence."""
    import oracle
    rng = np.random.default_rng(17)
    nsl, nx, ny = 6, 8, 10
    G = (rng.standard_normal((nsl, nx, ny))
         + 1j * rng.standard_normal((nsl, nx, ny)))
    op = pm.MPIFredholm1(dev(G.astype(np.complex128)), nz=1, saveGt=True,
                         dtype="complex128")
    A = oracle.SimFredholm1([G], nz=1).dense()
    n_model = op.shape[1]
    xg = rng.standard_normal(n_model) + 1j * rng.standard_normal(n_model)
    x = pm.DistributedArray((n_model,), partition=pm.Partition.BROADCAST,
                            dtype=np.complex128)
    x[:] = dev(xg)
    y = op.matvec(x)
    x0 = pm.DistributedArray((n_model,), partition=pm.Partition.BROADCAST,
                             dtype=np.complex128)
    x0[:] = 0.0
    xinv, istop, iters, r1, r2, cost = pm.cgls(op, y, x0, niter=30,
                                               damp=0.1, tol=0.0)
    yg = A @ xg
    xref = oracle.dense_cgls(A, yg, np.zeros(n_model, np.complex128),
                             niter=30, damp=0.1, tol=0.0)
    # 30 complex CGLS iterations amplify backend fp-order differences;
    # measured divergence ~4e-8 relative
    assert_allclose(host(xinv.asarray()), xref, rtol=1e-6, atol=1e-9)


@pytest.mark.parametrize("dtype,tol", [(np.float64, 1e-12),
                                       (np.float32, 2e-4)])
def test_gemm_batched_real(dtype, tol):
    """pam_gemm_batched: z-batched real MFMA panels (the r02 replacement
    for the per-slice host loop on real Fredholm kernels)."""
    rng = np.random.default_rng(11)
    B_, M, K, N = 5, 48, 33, 40
    A = rng.standard_normal((B_, M, K)).astype(dtype)
    X = rng.standard_normal((B_, K, N)).astype(dtype)
    s = torch.cuda.current_stream().cuda_stream
    Ad, Xd = dev(A).contiguous(), dev(X).contiguous()
    Y = torch.empty((B_, M, N), dtype=Ad.dtype, device="cuda:0")
    _ffi.checked(_ffi.lib().pam_gemm_batched(
        s, Ad.data_ptr(), Xd.data_ptr(), Y.data_ptr(), B_, M, N, K,
        M * K, K * N, M * N, 0, 0, _ffi.dtype_code(Ad.dtype)), "rgb")
    assert_allclose(host(Y), A @ X, rtol=tol, atol=tol)
    # transpose op + accumulate
    X2 = rng.standard_normal((B_, M, N)).astype(dtype)
    C0 = rng.standard_normal((B_, K, N)).astype(dtype)
    X2d, Cd = dev(X2).contiguous(), dev(C0).contiguous()
    _ffi.checked(_ffi.lib().pam_gemm_batched(
        s, Ad.data_ptr(), X2d.data_ptr(), Cd.data_ptr(), B_, K, N, M,
        M * K, M * N, K * N, 1, 1, _ffi.dtype_code(Ad.dtype)), "rgb")
    assert_allclose(host(Cd), C0 + A.transpose(0, 2, 1) @ X2,
                    rtol=tol, atol=tol * 10)


def test_fredholm_real_dtypes_vs_oracle():
    """Real-G Fredholm (ref test_fredholm.py float32 params) through the
    z-batched path."""
    import oracle
    from oracle.ranksim import Partition as SP, SimArray
    rng = np.random.default_rng(12)
    nsl, nx, ny, nz = 21, 4, 6, 5
    for dt, tol in ((np.float64, 1e-12), (np.float32, 2e-4)):
        G = rng.standard_normal((nsl, nx, ny)).astype(dt)
        for sg in (False, True):
            op = pm.MPIFredholm1(dev(G), nz, saveGt=sg, dtype=dt)
            sop = oracle.SimFredholm1([G.copy()], nz=nz, saveGt=sg)
            x = rng.standard_normal(op.shape[1]).astype(dt)
            y = rng.standard_normal(op.shape[0]).astype(dt)
            xd = pm.DistributedArray.to_dist(
                dev(x), partition=pm.Partition.BROADCAST)
            yd = pm.DistributedArray.to_dist(
                dev(y), partition=pm.Partition.BROADCAST)
            got = host(op.matvec(xd).local_array)
            want = sop.matvec(
                SimArray([x], x.shape, partition=SP.BROADCAST)).locals[0]
            assert_allclose(got, want, rtol=tol, atol=tol)
            gotr = host(op.rmatvec(yd).local_array)
            wantr = sop.rmatvec(
                SimArray([y], y.shape, partition=SP.BROADCAST)).locals[0]
            assert_allclose(gotr, wantr, rtol=tol, atol=tol)


@pytest.mark.parametrize("dtype,tol", [(np.complex128, 0),
                                       (np.complex64, 0)])
def test_zip_unzip(dtype, tol):
    """pam_unzip/pam_zip: complex<->real (de)interleave (MDC chain)."""
    rng = np.random.default_rng(13)
    for n in (1024, 999, 7, 65536):
        a = crand(rng, n, dtype)
        ad = dev(a).contiguous()
        rt = torch.float32 if dtype == np.complex64 else torch.float64
        r = torch.empty(n, dtype=rt, device="cuda:0")
        s = torch.cuda.current_stream().cuda_stream
        _ffi.checked(_ffi.lib().pam_unzip(
            s, r.data_ptr(), ad.data_ptr(), n, _ffi.dtype_code(ad.dtype)),
            "unzip")
        assert np.array_equal(host(r), a.real)
        z = torch.empty(n, dtype=ad.dtype, device="cuda:0")
        _ffi.checked(_ffi.lib().pam_zip(
            s, z.data_ptr(), r.data_ptr(), n, _ffi.dtype_code(ad.dtype)),
            "zip")
        assert np.array_equal(host(z), a.real.astype(dtype))
