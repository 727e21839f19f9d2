"""Pin the oracle (rank-simulated restatement) against independent serial
restatements + dense adjoints, mirroring the reference's own test scheme
(/root/reference/tests/test_derivative.py:197-229: seed-42 normal(rank,10)
inputs, rank-0 comparison vs the serial operator at rtol 1e-14, plus an
adjoint dottest)."""
import numpy as np
import pytest
from numpy.testing import assert_allclose

from oracle import (Partition, SimArray, SimFirstDerivative,
                    SimSecondDerivative, dense_matrix_from_matvec,
                    local_split, serial_fd1_matvec, serial_fd2_matvec,
                    sim_cg, sim_cgls, to_dist)
from oracle.ranksim import add_ghost_cells, split_bounds

RANKS = [1, 2, 3, 4]
DIMS = [(32,), (17, 5), (16, 4, 3)]


def make_x(dims, P, seed=42, flat=True):
    """Per-rank seed-42 normal(rank, 10) input, ref test_derivative.py:25,207.

    When ``flat``, ranks hold the element-balanced split of the flattened
    global vector (the shape a solver vector has, exercising the
    ``reshaped`` rebalance)."""
    n = int(np.prod(dims))
    locals_ = []
    for r in range(P):
        np.random.seed(seed)
        shape = local_split((n,), P, r) if flat else local_split(dims, P, r)
        locals_.append(np.random.normal(r, 10, shape))
    gshape = (n,) if flat else dims
    return SimArray(locals_, gshape)


# ---------------------------------------------------------------- local_split
def test_local_split_remainder_rule():
    # ref DistributedArray.py:67-70: first N%P ranks get one extra
    assert [local_split((10,), 4, r)[0] for r in range(4)] == [3, 3, 2, 2]
    assert [local_split((8,), 4, r)[0] for r in range(4)] == [2, 2, 2, 2]
    assert sum(local_split((1023, 7), 8, r, axis=0)[0] for r in range(8)) == 1023
    assert local_split((5, 7), 2, 0, Partition.BROADCAST) == (5, 7)


def test_to_dist_roundtrip():
    rng = np.random.default_rng(42)
    x = rng.standard_normal((13, 4))
    for P in RANKS:
        d = to_dist(x, P)
        assert_allclose(d.asarray(), x, rtol=0)
        bounds = split_bounds(13, P)
        for r in range(P):
            assert d.locals[r].shape[0] == bounds[r][1] - bounds[r][0]


def test_ghost_cells_values():
    # 2 ranks, axis 0: front ghost of rank1 = last rows of rank0
    a = np.arange(12.0).reshape(6, 2)
    d = to_dist(a, 2)
    g = add_ghost_cells(d.locals, cells_front=[1, 1], cells_back=[1, 1])
    assert_allclose(g[0], np.vstack([a[:3], a[3:4]]))
    assert_allclose(g[1], np.vstack([a[2:3], a[3:]]))
    # oversize request raises like ref :996-1002
    with pytest.raises(ValueError):
        add_ghost_cells(d.locals, cells_front=[4, 4], cells_back=None)


# ---------------------------------------------------------- derivative parity
@pytest.mark.parametrize("P", RANKS)
@pytest.mark.parametrize("dims", DIMS)
@pytest.mark.parametrize("kind,order,edge", [
    ("forward", 3, False), ("backward", 3, False),
    ("centered", 3, False), ("centered", 3, True),
    ("centered", 5, False), ("centered", 5, True),
])
def test_fd1_vs_serial(P, dims, kind, order, edge):
    op = SimFirstDerivative(dims, sampling=1.5, kind=kind, edge=edge,
                            order=order)
    x = make_x(dims, P)
    y = op.matvec(x)
    xg = x.asarray().reshape(dims)
    assert_allclose(y.asarray().reshape(dims),
                    serial_fd1_matvec(xg, 1.5, kind, edge, order), rtol=1e-14)


@pytest.mark.parametrize("P", RANKS)
@pytest.mark.parametrize("dims", DIMS)
@pytest.mark.parametrize("kind,edge", [
    ("forward", False), ("backward", False),
    ("centered", False), ("centered", True),
])
def test_fd2_vs_serial(P, dims, kind, edge):
    op = SimSecondDerivative(dims, sampling=1.2, kind=kind, edge=edge)
    x = make_x(dims, P)
    y = op.matvec(x)
    xg = x.asarray().reshape(dims)
    assert_allclose(y.asarray().reshape(dims),
                    serial_fd2_matvec(xg, 1.2, kind, edge), rtol=1e-14)


@pytest.mark.parametrize("P", [1, 3])
@pytest.mark.parametrize("op_factory", [
    lambda: SimFirstDerivative((24,), 0.7, "forward"),
    lambda: SimFirstDerivative((24,), 0.7, "backward"),
    lambda: SimFirstDerivative((24,), 0.7, "centered", order=3),
    lambda: SimFirstDerivative((24,), 0.7, "centered", edge=True, order=3),
    lambda: SimFirstDerivative((24,), 0.7, "centered", order=5),
    lambda: SimFirstDerivative((24,), 0.7, "centered", edge=True, order=5),
    lambda: SimSecondDerivative((24,), 0.7, "forward"),
    lambda: SimSecondDerivative((24,), 0.7, "backward"),
    lambda: SimSecondDerivative((24,), 0.7, "centered"),
    lambda: SimSecondDerivative((24,), 0.7, "centered", edge=True),
])
def test_adjoint_vs_dense_transpose(P, op_factory):
    """rmatvec must equal A.T @ x where A is the explicit forward matrix —
    the strongest adjoint pin (ref dottest only checks one random pair)."""
    op = op_factory()
    n = op.shape[0]

    def serial_mv(v):
        return op.matvec(to_dist(v, P)).asarray()

    A = dense_matrix_from_matvec(serial_mv, n)
    rng = np.random.default_rng(7)
    v = rng.standard_normal(n)
    r = op.rmatvec(to_dist(v, P)).asarray()
    assert_allclose(r, A.T @ v, rtol=1e-12, atol=1e-13)


@pytest.mark.parametrize("P", RANKS)
def test_dottest(P):
    # ref utils/dottest.py:11-107 at rtol 1e-6 (we hold 1e-10 in fp64)
    op = SimFirstDerivative((16, 6), kind="centered", order=5)
    rng = np.random.default_rng(3)
    u = to_dist(rng.standard_normal(96), P)
    v = to_dist(rng.standard_normal(96), P)
    y = op.matvec(u)
    x = op.rmatvec(v)
    yy = np.vdot(y.asarray(), v.asarray())
    xx = np.vdot(u.asarray(), x.asarray())
    assert np.isclose(xx, yy, rtol=1e-10)


# ----------------------------------------------------------------- math/norm
@pytest.mark.parametrize("P", RANKS)
def test_array_math_vs_numpy(P):
    # ref tests/test_distributedarray.py:177-222 tolerances
    rng = np.random.default_rng(0)
    a, b = rng.standard_normal(100), rng.standard_normal(100)
    da, db = to_dist(a, P), to_dist(b, P)
    assert_allclose((da + db).asarray(), a + b, rtol=1e-14)
    assert_allclose((da - db).asarray(), a - b, rtol=1e-14)
    assert_allclose((da * db).asarray(), a * b, rtol=1e-14)
    assert_allclose((2.5 * da).asarray(), 2.5 * a, rtol=1e-14)
    assert_allclose((-da).asarray(), -a, rtol=1e-14)
    assert_allclose(da.dot(db), np.dot(a, b), rtol=1e-14)
    assert_allclose(da.norm(), np.linalg.norm(a), rtol=1e-13)
    assert_allclose(da.norm(1), np.linalg.norm(a, 1), rtol=1e-14)
    assert_allclose(da.norm(np.inf), np.linalg.norm(a, np.inf), rtol=1e-14)
    assert_allclose(da.norm(-np.inf), np.linalg.norm(a, -np.inf), rtol=1e-14)
    assert_allclose(da.norm(0), np.count_nonzero(a), rtol=0)


# -------------------------------------------------------------------- solver
@pytest.mark.parametrize("P", [1, 2, 4])
def test_cgls_solves_lsq(P):
    """CGLS on FD1 + damping converges to the damped least-squares solution
    (independent pin: dense normal equations)."""
    dims, damp, niter = (40,), 0.5, 120
    op = SimFirstDerivative(dims, kind="centered", order=3)
    n = op.shape[0]
    A = dense_matrix_from_matvec(lambda v: op.matvec(to_dist(v, 1)).asarray(), n)
    rng = np.random.default_rng(11)
    yg = rng.standard_normal(n)
    x_ref = np.linalg.solve(A.T @ A + damp ** 2 * np.eye(n), A.T @ yg)
    x, cost = sim_cgls(op, to_dist(yg, P), to_dist(np.zeros(n), P),
                       niter=niter, damp=damp, tol=1e-30)
    assert_allclose(x.asarray(), x_ref, rtol=1e-8, atol=1e-9)


def test_cgls_trace_rank_invariant():
    """cost history identical (to fp64 reduction-order noise) across P."""
    dims = (24, 5)
    n = int(np.prod(dims))
    rng = np.random.default_rng(5)
    yg = rng.standard_normal(n)
    op = SimFirstDerivative(dims, kind="centered", order=3)
    traces = []
    for P in (1, 2, 4):
        _, cost = sim_cgls(op, to_dist(yg, P), to_dist(np.zeros(n), P),
                           niter=30, damp=1e-1, tol=1e-30)
        traces.append(np.asarray(cost))
    assert_allclose(traces[1], traces[0], rtol=1e-9)
    assert_allclose(traces[2], traces[0], rtol=1e-9)


@pytest.mark.parametrize("P", [1, 3])
def test_cg_solves_spd(P):
    """CG on the SPD normal-matrix operator A^T A + eps I."""
    dims = (30,)
    op = SimFirstDerivative(dims, kind="centered", order=3)
    n = op.shape[0]

    class NormalOp:
        def matvec(self, x):
            return op.rmatvec(op.matvec(x)) + 0.1 * x

    A = dense_matrix_from_matvec(
        lambda v: NormalOp().matvec(to_dist(v, 1)).asarray(), n)
    rng = np.random.default_rng(13)
    yg = rng.standard_normal(n)
    x, cost = sim_cg(NormalOp(), to_dist(yg, P), to_dist(np.zeros(n), P),
                     niter=200, tol=1e-30)
    assert_allclose(x.asarray(), np.linalg.solve(A, yg), rtol=1e-7, atol=1e-8)


# ----------------------------------------------------------------- blockdiag
@pytest.mark.parametrize("P", RANKS)
def test_blockdiag_vs_dense(P):
    """SimBlockDiag vs its explicit global block-diagonal matrix, with
    uneven per-rank blocks (ref BlockDiag.py:100-144 semantics)."""
    from oracle import SimBlockDiag
    rng = np.random.default_rng(21)
    mats = []
    for r in range(P):
        ms = [rng.standard_normal((3 + r, 4)), rng.standard_normal((2, 2 + r))]
        mats.append(ms)
    op = SimBlockDiag(mats)
    A = op.dense()
    n, m = op.shape
    x = to_dist(rng.standard_normal(m), P)
    y = to_dist(rng.standard_normal(n), P)
    assert_allclose(op.matvec(x).asarray(), A @ x.asarray(), rtol=1e-13)
    assert_allclose(op.rmatvec(y).asarray(), A.T @ y.asarray(), rtol=1e-13)


@pytest.mark.parametrize("P", [1, 2, 4])
def test_cgls_blockdiag(P):
    """CGLS over a block-diagonal dense system (the reference's
    examples/plot_cgls.py:30-43 recipe) vs a direct solve."""
    from oracle import SimBlockDiag
    rng = np.random.default_rng(30)
    mats = [[rng.standard_normal((6, 6)) + 6 * np.eye(6)] for _ in range(P)]
    op = SimBlockDiag(mats)
    A = op.dense()
    n = op.shape[0]
    yg = rng.standard_normal(n)
    x, cost = sim_cgls(op, to_dist(yg, P), to_dist(np.zeros(n), P),
                       niter=80, damp=0.0, tol=1e-30)
    assert_allclose(x.asarray(), np.linalg.solve(A, yg), rtol=1e-8, atol=1e-9)


@pytest.mark.parametrize("P", [1, 3])
def test_fd1_complex_vs_serial(P):
    # ref tests/test_derivative.py includes complex128 parameter sets
    dims = (17, 5)
    n = int(np.prod(dims))
    rng = np.random.default_rng(42)
    xg = rng.standard_normal(n) + 1j * rng.standard_normal(n)
    op = SimFirstDerivative(dims, 1.5, "centered", True, 5,
                            dtype=np.complex128)
    y = op.matvec(to_dist(xg, P))
    want = serial_fd1_matvec(xg.reshape(dims), 1.5, "centered", True, 5)
    assert_allclose(y.asarray().reshape(dims), want, rtol=1e-14)


def test_nsconv_adjoint_vs_dense():
    """Non-stationary convolution restatement: rmatvec == A.T @ x where
    A is the explicit forward matrix (locks the re-derived pylops
    interpolation convention)."""
    from oracle import serial_nsconv_mv, serial_nsconv_rmv
    rng = np.random.default_rng(15)
    dims, hsize = (24,), 7
    ih = np.array([3, 9, 15, 21])
    hs = rng.standard_normal((len(ih), hsize))
    A = dense_matrix_from_matvec(
        lambda v: serial_nsconv_mv(v, dims, hs, ih), 24)
    v = rng.standard_normal(24)
    assert_allclose(serial_nsconv_rmv(v, dims, hs, ih), A.T @ v,
                    rtol=1e-12, atol=1e-13)
    # batched axis=-1 of a 2-D block
    dims2 = (3, 24)
    A2 = dense_matrix_from_matvec(
        lambda v: serial_nsconv_mv(v, dims2, hs, ih, axis=-1), 72)
    v2 = rng.standard_normal(72)
    assert_allclose(serial_nsconv_rmv(v2, dims2, hs, ih, axis=-1),
                    A2.T @ v2, rtol=1e-12, atol=1e-13)
