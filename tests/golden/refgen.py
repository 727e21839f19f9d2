"""Shared case table for the REFERENCE-executed golden fixtures.

Two symmetric computations over the SAME deterministic inputs:

  * :func:`compute_reference` — runs /root/reference/pylops_mpi in this
    container (P ranks as threads, oracle/_refshim mpi4py + pylops
    stubs) and collects the distributed outputs;
  * :func:`compute_oracle` — runs the repo's rank-simulating oracle on
    the identical cases.

``tests/golden/generate_golden_ref.py`` saves the reference outputs to
``golden_ref.npz`` (committed — the GPU box has no /root/reference);
``tests/test_golden_ref.py`` asserts oracle == fixtures everywhere, and
``tests/test_ref_parity.py`` asserts fixtures == a live reference run
when /root/reference is present (fixture staleness guard).

Inputs follow the reference's own test recipe (seed-42
``normal(rank, 10, local_shape)`` per rank,
ref tests/test_derivative.py:25,207).
"""
import os
import sys

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
_ROOT = os.path.dirname(os.path.dirname(_HERE))
if _ROOT not in sys.path:
    sys.path.insert(0, _ROOT)

import oracle  # noqa: E402
from oracle import matmult as om  # noqa: E402

GOLDEN_PATH = os.path.join(_HERE, "golden_ref.npz")

PS = (1, 2, 3, 4, 8)
DIMS = [(32,), (17, 5), (16, 4, 3)]
# P=8 splits the small DIMS to 2 rows/rank, below the 5-point stencils'
# ghost width — the REFERENCE itself raises there — so P=8 uses taller
# row counts (>= 4 rows per rank)
DIMS_P8 = [(40,), (33, 5), (32, 4, 3)]
SAMPLING = 1.5
FD1_CASES = [("forward", 3, False), ("backward", 3, False),
             ("centered", 3, False), ("centered", 3, True),
             ("centered", 5, False), ("centered", 5, True)]
FD2_CASES = [("forward", False), ("backward", False),
             ("centered", False), ("centered", True)]
MATH_N = (13, 4)
MATH_N8 = (33, 4)   # P=8: >= 4 rows/rank so width-2 ghosts exist
NORM_ORDS = [("2", None), ("0", 0), ("inf", np.inf), ("ninf", -np.inf),
             ("1p5", 1.5)]
CGLS_DIMS = (17, 5)
CGLS_NITER = 10
CGLS_DAMPS = [("d0", 0.0), ("d05", 0.5)]
CG_NITER = 12
ISTA_NITER = 10
ISTA_EPS = 0.2
ISTA_ALPHA = 0.002   # < 1/lambda_max(B^T B) for the SPD BlockDiag
POWIT_NITER = 12
GRAD_DIMS = (10, 6, 4)
GRAD_SAMP = (1.5, 2.0, 0.5)
LAP_AXES = (0, 1, 2)
LAP_W = (1.0, 2.0, 0.5)
LAP_SAMP = (1.0, 1.5, 2.0)
MM_SHAPES = (7, 5, 9)       # N, K, M (uneven vs any P in PS)
MM_DTYPES = ["float64", "complex128"]
FRED_SHAPE = (21, 4, 6, 5)  # nsl, nx, ny, nz (the reference test's)
FRED_DTYPES = ["float64", "complex128"]


def tagd(dims):
    return "x".join(map(str, dims))


def make_global_x(n, P, seed_shift=0):
    """seed-42 normal(rank,10) per rank, concatenated (ref test recipe)."""
    parts = []
    for r in range(P):
        np.random.seed(42 + seed_shift)
        parts.append(np.random.normal(
            r, 10, oracle.local_split((n,), P, r)))
    return np.concatenate(parts)


def blockdiag_mats(P):
    """Per-rank dense serial blocks (examples/plot_cgls.py:30-33 style)."""
    rng = np.random.default_rng(31)
    shapes = [[(4, 6)], [(3, 2), (5, 5)], [(2, 3)], [(4, 4), (1, 2)]]
    return [[rng.standard_normal(s) for s in shapes[r % 4]]
            for r in range(P)]


def spd_mats(P):
    """Per-rank SPD blocks for CG (B = M M^T + n I)."""
    rng = np.random.default_rng(41)
    sizes = [4, 6, 3, 5]
    out = []
    for r in range(P):
        n = sizes[r % 4]
        M = rng.standard_normal((n, n))
        out.append([M @ M.T + n * np.eye(n)])
    return out


def spd2_mats(P):
    """Second SPD family (same sizes, fresh seed) for the stacked
    VStack/CGLS cases."""
    rng = np.random.default_rng(43)
    sizes = [4, 6, 3, 5]
    out = []
    for r in range(P):
        n = sizes[r % 4]
        M = rng.standard_normal((n, n))
        out.append([M @ M.T + n * np.eye(n)])
    return out


def vstack_mats(P):
    """Per-rank blocks sharing 7 columns (MPIVStack requirement)."""
    rng = np.random.default_rng(51)
    shapes = [[(4, 7)], [(3, 7), (5, 7)], [(2, 7)], [(4, 7), (1, 7)]]
    return [[rng.standard_normal(s) for s in shapes[r % 4]]
            for r in range(P)]


def fred_G(dtype):
    nsl, nx, ny, _ = FRED_SHAPE
    G = (np.outer(np.ones(nsl), np.arange(nx * ny))
         .reshape(nsl, nx, ny).astype(dtype))
    if np.dtype(dtype).kind == "c":
        G = G + 1j * (G[:, ::-1, ::-1] / 7.0)
    return G


def fred_x(dtype, forward):
    nsl, nx, ny, nz = FRED_SHAPE
    rng = np.random.default_rng(17)
    shape = (nsl, ny, nz) if forward else (nsl, nx, nz)
    x = rng.standard_normal(shape).astype(dtype)
    if np.dtype(dtype).kind == "c":
        x = x + 1j * rng.standard_normal(shape).astype(dtype)
    return x


def fred_split(P):
    nsl = FRED_SHAPE[0]
    return [oracle.local_split((nsl,), P, r)[0] for r in range(P)]


def _serial_axis_fd(xg, dims, axis, sampling, edge, fwd, second=False):
    """Serial derivative along ``axis`` of the global array (the exact
    map the reference composes for Gradient/Laplacian — distribution-
    independent dense expectation)."""
    from oracle.ranksim import to_dist
    from oracle.stencils import SimFirstDerivative, SimSecondDerivative
    arr = np.moveaxis(xg.reshape(dims), axis, 0)
    moved = arr.shape
    sim = (SimSecondDerivative(moved, sampling, "centered", edge) if second
           else SimFirstDerivative(moved, sampling, "centered", edge, 3))
    d = to_dist(arr.ravel(), 1)
    res = (sim.matvec(d) if fwd else sim.rmatvec(d)).asarray()
    return np.ascontiguousarray(
        np.moveaxis(res.reshape(moved), 0, axis)).ravel()


def plane_counts(dims, P):
    return [int(np.prod(oracle.local_split(dims, P, r)))
            for r in range(P)]


def _oracle_dist_planes(xg, dims, P):
    return _oracle_dist_counts(xg, plane_counts(dims, P))


def _oracle_dist_counts(xg, counts):
    from oracle.ranksim import Partition, SimArray
    offs = np.cumsum([0] + list(counts))
    return SimArray([xg[offs[r]: offs[r + 1]] for r in range(len(counts))],
                    (int(xg.size),), 0, Partition.SCATTER)


# ===================================================== oracle side
def compute_oracle():
    out = {}
    for P in PS:
        # FD operators
        for dims in (DIMS_P8 if P >= 8 else DIMS):
            n = int(np.prod(dims))
            xg = make_global_x(n, P)
            yg = make_global_x(n, P, seed_shift=1)
            for kind, order, edge in FD1_CASES:
                key = f"fd1_P{P}_{kind}{order}{'e' if edge else 'n'}_" \
                      f"{tagd(dims)}"
                op = oracle.SimFirstDerivative(dims, SAMPLING, kind, edge,
                                               order)
                out[key + "_mv"] = op.matvec(oracle.to_dist(xg, P)).asarray()
                out[key + "_rmv"] = op.rmatvec(oracle.to_dist(yg, P)).asarray()
            for kind, edge in FD2_CASES:
                key = f"fd2_P{P}_{kind}{'e' if edge else 'n'}_{tagd(dims)}"
                op = oracle.SimSecondDerivative(dims, SAMPLING, kind, edge)
                out[key + "_mv"] = op.matvec(oracle.to_dist(xg, P)).asarray()
                out[key + "_rmv"] = op.rmatvec(oracle.to_dist(yg, P)).asarray()
        # array math
        mn = MATH_N8 if P >= 8 else MATH_N
        n = int(np.prod(mn))
        xg = make_global_x(n, P)
        yg = make_global_x(n, P, seed_shift=1)
        dx = oracle.to_dist(xg, P)
        dy = oracle.to_dist(yg, P)
        out[f"math_P{P}_dot"] = np.asarray(dx.dot(dy))
        dxa = oracle.to_dist(np.abs(xg), P)
        for name, o in NORM_ORDS:
            # |x| input for the fractional norm: the reference's
            # float_power(negative, 1.5) is NaN (ref :786) — pin values
            d = dxa if name == "1p5" else dx
            out[f"math_P{P}_norm{name}"] = np.asarray(d.norm(o))
        out[f"math_P{P}_addmul"] = ((dx + dy) * dx).asarray()
        d2 = oracle.to_dist(xg.reshape(mn), P)
        for w in (1, 2):
            gh = d2.add_ghost_cells(cells_front=w, cells_back=w)
            out[f"math_P{P}_ghost{w}"] = np.concatenate(
                [g.ravel() for g in gh])
        # redistribute axis 0 -> axis 1 (ref :493-552): the result is the
        # BALANCED column split of the global array — dense expectation
        G2 = xg.reshape(mn)
        cols = [oracle.local_split((mn[1],), P, r)[0] for r in range(P)]
        co = np.cumsum([0] + cols)
        out[f"math_P{P}_redist"] = np.concatenate(
            [G2[:, co[r]: co[r + 1]].ravel() for r in range(P)])
        # masked reductions (ref DistributedArray.py:74-99, :715,
        # :745-788): each rank reduces over its mask group only, in
        # rank order within the group
        if P >= 4:
            mk = [r // 2 for r in range(P)]
            cnts = [oracle.local_split((n,), P, r)[0] for r in range(P)]
            moff = np.cumsum([0] + cnts)
            chx = [xg[moff[r]: moff[r + 1]] for r in range(P)]
            chy = [yg[moff[r]: moff[r + 1]] for r in range(P)]
            dots, n2s, nis = [], [], []
            for r in range(P):
                grp = [s_ for s_ in range(P) if mk[s_] == mk[r]]
                dots.append(sum(np.dot(chx[s_], chy[s_]) for s_ in grp))
                n2s.append(np.power(
                    sum(np.sum(np.abs(np.float_power(chx[s_], 2)))
                        for s_ in grp), 0.5))
                nis.append(max(np.max(np.abs(chx[s_])) for s_ in grp))
            out[f"mask_P{P}_dot"] = np.asarray(dots)
            out[f"mask_P{P}_norm2"] = np.asarray(n2s)
            out[f"mask_P{P}_norminf"] = np.asarray(nis)
            out[f"mask_P{P}_asarray"] = np.concatenate(
                [np.concatenate([chx[s_] for s_ in range(P)
                                 if mk[s_] == mk[r]])
                 for r in range(P)])
        # CGLS on FD1 centered3.  x0 must be plane-aligned: the
        # reference's reshaped wrapper leaves operator OUTPUTS on the
        # plane split (ref decorators.py:79-82), and CGLS subtracts
        # damped_x (x0's split) from rmatvec outputs (ref
        # cls_basic.py:348) — a default-split x0 raises "Local Array
        # Shape Mismatch" in the reference itself when dims[0] % P != 0.
        nc = int(np.prod(CGLS_DIMS))
        xg = make_global_x(nc, P)
        op = oracle.SimFirstDerivative(CGLS_DIMS, SAMPLING, "centered",
                                       False, 3)
        y = op.matvec(oracle.to_dist(xg, P))
        x0 = _oracle_dist_planes(np.zeros(nc), CGLS_DIMS, P)
        for dn, damp in CGLS_DAMPS:
            xs, cost = oracle.sim_cgls(op, y, x0, CGLS_NITER, damp=damp,
                                       tol=0.0)
            out[f"cgls_P{P}_{dn}_x"] = xs.asarray()
            out[f"cgls_P{P}_{dn}_cost"] = np.asarray(cost)
        # BlockDiag
        mats = blockdiag_mats(P)
        bop = oracle.SimBlockDiag(mats)
        nr, nc2 = bop.shape
        xg = make_global_x(nc2, P)
        yg = make_global_x(nr, P, seed_shift=1)
        out[f"bd_P{P}_mv"] = bop.matvec(oracle.to_dist(xg, P)).asarray()
        out[f"bd_P{P}_rmv"] = bop.rmatvec(oracle.to_dist(yg, P)).asarray()
        # CG on an SPD BlockDiag (ref optimization/basic.py:13 cg)
        smats = spd_mats(P)
        sop2 = oracle.SimBlockDiag(smats)
        nspd = sop2.shape[0]
        yg2 = make_global_x(nspd, P)
        spd_counts = [int(sum(A.shape[0] for A in ms)) for ms in smats]
        xs2, cost2 = oracle.sim_cg(
            sop2, _oracle_dist_counts(yg2, spd_counts),
            _oracle_dist_counts(np.zeros(nspd), spd_counts),
            CG_NITER, tol=0.0)
        out[f"cg_P{P}_x"] = xs2.asarray()
        out[f"cg_P{P}_cost"] = np.asarray(cost2)
        # ISTA/FISTA on the same SPD BlockDiag (ref optimization/
        # sparsity.py:11,136 -> cls_sparsity.py ISTA/FISTA)
        ysp = _oracle_dist_counts(yg2, spd_counts)
        x0sp = _oracle_dist_counts(np.zeros(nspd), spd_counts)
        for tk in ("soft", "hard"):
            xi, _, ci = oracle.sim_ista(sop2, ysp, x0sp, ISTA_NITER,
                                        ISTA_EPS, ISTA_ALPHA,
                                        threshkind=tk, tol=0.0)
            out[f"ista_P{P}_{tk}_x"] = xi.asarray()
            out[f"ista_P{P}_{tk}_cost"] = np.asarray(ci)
        xf, _, cf = oracle.sim_fista(sop2, ysp, x0sp, ISTA_NITER,
                                     ISTA_EPS, ISTA_ALPHA, tol=0.0)
        out[f"fista_P{P}_x"] = xf.asarray()
        out[f"fista_P{P}_cost"] = np.asarray(cf)
        # power_iteration (ref optimization/eigs.py:10-102; the rand
        # init is the shared rank-deterministic powerit_rand draw)
        me, bk, nit = oracle.sim_power_iteration(sop2, spd_counts,
                                                 niter=POWIT_NITER,
                                                 tol=1e-5)
        out[f"powit_P{P}_eig"] = np.array([me])
        out[f"powit_P{P}_vec"] = bk.asarray()
        out[f"powit_P{P}_n"] = np.array([float(nit)])
        # VStack vs dense (ref basicoperators/VStack.py:121-150; the
        # oracle side IS the dense algebra — reference == dense pins it)
        vmats = vstack_mats(P)
        flat = [A for ms in vmats for A in ms]
        nv_rows = int(sum(A.shape[0] for A in flat))
        xv = make_global_x(7, P)
        yv = make_global_x(nv_rows, P, seed_shift=1)
        out[f"vs_P{P}_mv"] = np.concatenate([A @ xv for A in flat])
        offs = np.cumsum([0] + [A.shape[0] for A in flat])
        # fold in the reference's exact order: per-rank vstack-sum, then
        # rank-ordered allreduce (bitwise at any P)
        per_rank, idx = [], 0
        for ms in vmats:
            segs = []
            for A in ms:
                segs.append(A.T @ yv[offs[idx]: offs[idx + 1]])
                idx += 1
            per_rank.append(np.sum(np.vstack(segs), axis=0))
        acc = per_rank[0]
        for v in per_rank[1:]:
            acc = acc + v
        out[f"vs_P{P}_rmv"] = acc
        # HStack dense (ref HStack.py:90-107: VStack-of-adjoints fold —
        # matvec is the vs_rmv-style rank-ordered reduce of A.T @ seg)
        hsegs = [yv[offs[i]: offs[i + 1]] for i in range(len(flat))]
        hper, idx = [], 0
        for ms in vmats:
            ss = []
            for A in ms:
                ss.append(A.T @ hsegs[idx])
                idx += 1
            hper.append(np.sum(np.vstack(ss), axis=0))
        hacc = hper[0]
        for v in hper[1:]:
            hacc = hacc + v
        out[f"hs_P{P}_mv"] = hacc
        out[f"hs_P{P}_rmv"] = np.concatenate([A @ xv for A in flat])
        # Stacked operators + stacked arrays (ref BlockDiag.py:147-189,
        # VStack.py:153-203; SimStackedArray ranksim mirror of
        # DistributedArray.py:1041-1300)
        from oracle.ranksim import SimStackedArray
        sbd_o = oracle.SimStackedBlockDiag([bop, sop2])
        xst = SimStackedArray([
            oracle.to_dist(make_global_x(nc2, P), P),
            _oracle_dist_counts(make_global_x(nspd, P, seed_shift=6),
                                spd_counts)])
        out[f"sbd_P{P}_mv"] = sbd_o.matvec(xst).asarray()
        yst = SimStackedArray([
            oracle.to_dist(make_global_x(nr, P, seed_shift=1), P),
            _oracle_dist_counts(make_global_x(nspd, P, seed_shift=7),
                                spd_counts)])
        out[f"sbd_P{P}_rmv"] = sbd_o.rmatvec(yst).asarray()
        svop_o = oracle.SimStackedVStack(
            [sop2, oracle.SimBlockDiag(spd2_mats(P))])
        xsv = _oracle_dist_counts(yg2, spd_counts)
        out[f"svs_P{P}_mv"] = svop_o.matvec(xsv).asarray()
        ysv = SimStackedArray([
            _oracle_dist_counts(make_global_x(nspd, P, seed_shift=8),
                                spd_counts),
            _oracle_dist_counts(make_global_x(nspd, P, seed_shift=9),
                                spd_counts)])
        out[f"svs_P{P}_rmv"] = svop_o.rmatvec(ysv).asarray()
        # CGLS over the stacked VStack (damped, stacked data vector)
        xsc, costsc = oracle.sim_cgls(
            svop_o, ysv, _oracle_dist_counts(np.zeros(nspd), spd_counts),
            CGLS_NITER, damp=0.4, tol=0.0)
        out[f"scgls_P{P}_x"] = xsc.asarray()
        out[f"scgls_P{P}_cost"] = np.asarray(costsc)
        # proximal subpackage (rank-sims of ProximalGradient / ADMML2
        # with MPIL2 grad + per-rank pyproximal-L1 prox)
        ypg_o = _oracle_dist_counts(make_global_x(nspd, P, seed_shift=10),
                                    spd_counts)
        for an, acc in (("none", None), ("fista", "fista")):
            xpg = oracle.sim_proximal_gradient_l2l1(
                sop2, ypg_o, _oracle_dist_counts(np.zeros(nspd),
                                                 spd_counts),
                0.01, 0.3, 10, acceleration=acc)
            out[f"pg_P{P}_{an}_x"] = xpg.asarray()
        xa, za = oracle.sim_admml2_l1(
            sop2, ypg_o, svop_o.ops[1],
            _oracle_dist_counts(np.zeros(nspd), spd_counts),
            0.05, 0.3, 6, 8, 0.0)
        out[f"admm_P{P}_x"] = xa.asarray()
        out[f"admm_P{P}_z"] = za.asarray()
        # Gradient / Laplacian (composed operators; dense expectations
        # are distribution-independent).  P <= 4 only: at P=8 the
        # 10-row dims leave the reference's reshaped rebalance with
        # ghost moves larger than a rank's block — the REFERENCE itself
        # raises there.
        if P <= 4:
            out.update(_grad_lap_oracle(P))
        # MPIHalo (Cartesian ghost exchange, ref Halo.py) vs the
        # restated window semantics
        for ci, (hdims, grid, hspec) in enumerate(_halo_cases(P)):
            rngh = np.random.default_rng(90 + ci)
            G = rngh.standard_normal(hdims)
            out[f"halo_P{P}_c{ci}_mv"] = np.concatenate(
                [_halo_window(G, grid, q, hspec).ravel()
                 for q in range(P)])
            out[f"halo_P{P}_c{ci}_rmv"] = np.concatenate(
                [b.ravel() for b in _halo_blocks(G, grid)])
        # NonStationaryConvolve1D: the distributed halo/blockdiag
        # composition equals the GLOBAL serial convolution at P <= 2
        # (both ranks take the edge branches).  At P >= 3 the middle-
        # rank branch's sliced anchor set (ihidx+-1, ref
        # NonStatConvolve1d.py:178-185) clamps filter interpolation in
        # the outer halo margin where the global operator still
        # interpolates — a boundary-localized difference inherent to
        # the reference's slicing, so those P are not fixtured.
        if P <= 2:
            from oracle.nsconv import serial_nsconv_mv, serial_nsconv_rmv
            ndims, hs, ih = _nsc_setup()
            nn = int(np.prod(ndims))
            xn = make_global_x(nn, P)
            yn = make_global_x(nn, P, seed_shift=1)
            out[f"nsc_P{P}_mv"] = serial_nsconv_mv(xn, ndims, hs, ih, 0)
            out[f"nsc_P{P}_rmv"] = serial_nsconv_rmv(yn, ndims, hs, ih, 0)
        # MDC chain (composite F1^H I1^H Fr I F with the serial FFT
        # convention held common — pins the reference's chain
        # construction: prescale, masks, product/adjoint composites);
        # P<=4: at P=8 the nfreq=8 split leaves 1 freq/rank and the
        # REFERENCE itself raises ("at least 2 or more elements in the
        # first dimension")
        if P <= 4:
            nt_, ns_, nr_, nv_, nfreq_, Gm = _mdc_setup()
            mop = oracle.SimMDC(
                [Gm[off: off + c] for off, c in _mdc_slices(P)],
                nt_, nv_, nfreq_, dt=0.4, dr=2.0, twosided=False)
            xm = make_global_x(nt_ * nr_ * nv_, P)
            ymv = make_global_x(nt_ * ns_ * nv_, P, seed_shift=1)
            from oracle.ranksim import Partition as _SP, SimArray as _SA
            out[f"mdc_P{P}_mv"] = mop.matvec(
                _SA([xm.copy() for _ in range(P)], xm.shape,
                    partition=_SP.BROADCAST)).locals[0]
            # the reference records max|imag| of its (complex-typed)
            # output; the chain is exactly real, so pin it to 0
            out[f"mdc_P{P}_mv_imagmax"] = np.array([0.0])
            out[f"mdc_P{P}_rmv"] = mop.rmatvec(
                _SA([ymv.copy() for _ in range(P)], ymv.shape,
                    partition=_SP.BROADCAST)).locals[0]
        # Fredholm1
        for dt in FRED_DTYPES:
            G = fred_G(dt)
            nsls = fred_split(P)
            blocks, off = [], 0
            for r in range(P):
                blocks.append(G[off: off + nsls[r]])
                off += nsls[r]
            for sg in (False, True):
                sop = oracle.SimFredholm1(blocks, nz=FRED_SHAPE[3],
                                          saveGt=sg)
                key = f"fred_P{P}_{np.dtype(dt).char}_{'s' if sg else 'n'}"
                xb = fred_x(dt, True).ravel()
                yb = fred_x(dt, False).ravel()
                from oracle.ranksim import Partition, SimArray
                xd = SimArray([xb.copy() for _ in range(P)], xb.shape,
                              partition=Partition.BROADCAST)
                yd = SimArray([yb.copy() for _ in range(P)], yb.shape,
                              partition=Partition.BROADCAST)
                out[key + "_mv"] = sop.matvec(xd).asarray()
                out[key + "_rmv"] = sop.rmatvec(yd).asarray()
    # MatrixMult (square P only, like the reference)
    N, K, M = MM_SHAPES
    for P in (1, 4):
        rng = np.random.default_rng(77)
        for dt in MM_DTYPES:
            A = rng.standard_normal((N, K)).astype(dt)
            X = rng.standard_normal((K, M)).astype(dt)
            Y = rng.standard_normal((N, M)).astype(dt)
            if np.dtype(dt).kind == "c":
                A = A + 1j * rng.standard_normal((N, K))
                X = X + 1j * rng.standard_normal((K, M))
                Y = Y + 1j * rng.standard_normal((N, M))
            for kind in ("block", "summa"):
                key = f"mm_{kind}_P{P}_{np.dtype(dt).char}"
                if kind == "block":
                    mv = om.block_expected_mv(A, X, P)
                    rmv = om.block_expected_rmv(A, Y, P)
                else:
                    mv = om.summa_expected_mv(A, X, P)
                    rmv = om.summa_expected_rmv(A, Y, P)
                out[key + "_mv"] = np.concatenate(
                    [v.ravel() for v in mv])
                out[key + "_rmv"] = np.concatenate(
                    [v.ravel() for v in rmv])
    # grid helpers (ref MatrixMult.py:24-171): active-grid selection,
    # ceil-division block slices, allgather reassembly — dense
    # restatement of the published arithmetic, incl. inactive ranks
    import math as _math
    for P in PS:
        for (N, M) in ((7, 9), (2, 9)):
            pp = _math.isqrt(P)
            ad = min(N, M, pp)
            active = [r for r in range(P)
                      if (r // pp) < ad and (r % pp) < ad]
            ppn = _math.isqrt(len(active))
            grid, blocks = [], []
            for r in range(P):
                row, col = divmod(r, pp)
                act = row < ad and col < ad
                if act:
                    nr_ = active.index(r)
                    nrow, ncol = divmod(nr_, ppn)
                    grid.append([nr_, nrow, ncol, 1.0])
                    new_r = -(-N // ppn) * ppn
                    new_c = -(-M // ppn) * ppn
                    blkr, blkc = new_r // ppn, new_c // ppn
                    rs, cs = nrow * blkr, ncol * blkc
                    re_, ce = min(rs + blkr, N), min(cs + blkc, M)
                    blocks.append([rs, re_, cs, ce])
                else:
                    grid.append([r, row, col, 0.0])
                    blocks.append([-1.0, -1.0, -1.0, -1.0])
            out[f"mmu_P{P}_{N}x{M}_grid"] = np.asarray(
                grid, dtype=float).ravel()
            out[f"mmu_P{P}_{N}x{M}_block"] = np.asarray(
                blocks, dtype=float).ravel()
            # block_gather reassembles the blocks exactly into G
            rng = np.random.default_rng(55)
            out[f"mmu_P{P}_{N}x{M}_gather"] = rng.standard_normal((N, M))
    # the non-square-world flow at P=8: the operator runs on the 2x2
    # ACTIVE sub-grid, so expectations are the P=4 dense folds
    rng = np.random.default_rng(77)
    A = rng.standard_normal(MM_SHAPES[:2])
    X = rng.standard_normal(MM_SHAPES[1:])
    out["mmsub_P8_block_mv"] = np.concatenate(
        [v.ravel() for v in om.block_expected_mv(A, X, 4)])
    out["mmsub_P8_summa_mv"] = np.concatenate(
        [v.ravel() for v in om.summa_expected_mv(A, X, 4)])
    return out


# ================================================== reference side
def compute_reference():
    from oracle.refrun import run_reference

    out = {}
    from oracle.sparsity import powerit_rand

    def _rank_rand(*shape):
        # rank-deterministic stand-in for the eigs.py:72-75 init draw
        # (threads share the global RNG, so the real np.random.rand
        # would race AND be order-dependent)
        from mpi4py import MPI
        assert len(shape) == 1
        return powerit_rand(MPI.COMM_WORLD.Get_rank(), int(shape[0]))

    for P in PS:
        saved_rand = np.random.rand
        np.random.rand = _rank_rand
        try:
            outs = run_reference(P, _ref_rank_fn(P))
        finally:
            np.random.rand = saved_rand
        for key, val in outs[0].items():
            if key.startswith("__perrank__"):
                # rank-resolved pieces (ghost cells) come back per rank
                real = key[len("__perrank__"):]
                out[real] = np.concatenate(
                    [np.asarray(o[key]).ravel() for o in outs])
            else:
                out[key] = np.asarray(val)
    for P in (1, 4):
        outs = run_reference(P, _ref_mm_fn(P))
        for key, val in outs[0].items():
            out[key] = np.asarray(val)
    for P in PS:
        outs = run_reference(P, _ref_mmutil_fn(P))
        for key, val in outs[0].items():
            if key.startswith("__perrank__"):
                real = key[len("__perrank__"):]
                out[real] = np.concatenate(
                    [np.asarray(o[key]).ravel() for o in outs])
            else:
                out[key] = np.asarray(val)
    return out


def _ref_rank_fn(P):
    # precompute EVERY input in the spawning thread: make_global_x uses
    # the global numpy RNG (the reference test recipe's np.random.seed),
    # which is not thread-safe across the P rank threads
    fd_inputs = {}
    for dims in (DIMS_P8 if P >= 8 else DIMS):
        n = int(np.prod(dims))
        fd_inputs[dims] = (make_global_x(n, P),
                           make_global_x(n, P, seed_shift=1))
    mn = MATH_N8 if P >= 8 else MATH_N
    nmath = int(np.prod(mn))
    math_x = make_global_x(nmath, P)
    math_y = make_global_x(nmath, P, seed_shift=1)
    ncgls = int(np.prod(CGLS_DIMS))
    cgls_x = make_global_x(ncgls, P)
    bd_mats = blockdiag_mats(P)
    nr_bd = int(sum(sum(A.shape[0] for A in ms) for ms in bd_mats))
    nc_bd = int(sum(sum(A.shape[1] for A in ms) for ms in bd_mats))
    bd_x = make_global_x(nc_bd, P)
    bd_y = make_global_x(nr_bd, P, seed_shift=1)
    spd = spd_mats(P)
    n_spd = int(sum(sum(A.shape[0] for A in ms) for ms in spd))
    cg_y = make_global_x(n_spd, P)
    vmats = vstack_mats(P)
    nv_rows = int(sum(A.shape[0] for ms in vmats for A in ms))
    vs_x = make_global_x(7, P)
    vs_y = make_global_x(nv_rows, P, seed_shift=1)
    ng = int(np.prod(GRAD_DIMS))
    grad_x = make_global_x(ng, P)
    grad_ys = [make_global_x(ng, P, seed_shift=2 + i) for i in range(3)]
    lap_y = make_global_x(ng, P, seed_shift=5)
    spd2 = spd2_mats(P)
    sbd_x2 = make_global_x(n_spd, P, seed_shift=6)
    sbd_y2 = make_global_x(n_spd, P, seed_shift=7)
    scgls_y1 = make_global_x(n_spd, P, seed_shift=8)
    scgls_y2 = make_global_x(n_spd, P, seed_shift=9)
    pg_y = make_global_x(n_spd, P, seed_shift=10)
    nsc_ndims, nsc_hs, nsc_ih = _nsc_setup()
    nn_nsc = int(np.prod(nsc_ndims))
    nsc_x = make_global_x(nn_nsc, P)
    nsc_y = make_global_x(nn_nsc, P, seed_shift=1)
    mdc_nt, mdc_ns, mdc_nr, mdc_nv, mdc_nfreq, mdc_G = _mdc_setup()
    mdc_x = make_global_x(mdc_nt * mdc_nr * mdc_nv, P)
    mdc_y = make_global_x(mdc_nt * mdc_ns * mdc_nv, P, seed_shift=1)

    def fn(rank):
        from pylops_mpi import (DistributedArray, MPIBlockDiag,
                                MPIFirstDerivative, MPISecondDerivative,
                                Partition, cgls)
        from pylops_mpi.signalprocessing import MPIFredholm1
        import pylops

        def dist_from_global(xg, dims=None):
            n = int(xg.size)
            d = DistributedArray(global_shape=n, dtype=xg.dtype)
            counts = [oracle.local_split((n,), P, r)[0] for r in range(P)]
            off = int(np.sum(counts[:rank], initial=0))
            d[:] = xg[off: off + d.local_shape[0]]
            return d

        res = {}
        for dims in (DIMS_P8 if P >= 8 else DIMS):
            n = int(np.prod(dims))
            xg, yg = fd_inputs[dims]
            for kind, order, edge in FD1_CASES:
                key = f"fd1_P{P}_{kind}{order}{'e' if edge else 'n'}_" \
                      f"{tagd(dims)}"
                op = MPIFirstDerivative(dims, sampling=SAMPLING, kind=kind,
                                        edge=edge, order=order)
                res[key + "_mv"] = op.matvec(dist_from_global(xg)).asarray()
                res[key + "_rmv"] = op.rmatvec(
                    dist_from_global(yg)).asarray()
            for kind, edge in FD2_CASES:
                key = f"fd2_P{P}_{kind}{'e' if edge else 'n'}_{tagd(dims)}"
                op = MPISecondDerivative(dims, sampling=SAMPLING, kind=kind,
                                         edge=edge)
                res[key + "_mv"] = op.matvec(dist_from_global(xg)).asarray()
                res[key + "_rmv"] = op.rmatvec(
                    dist_from_global(yg)).asarray()
        n = nmath
        xg, yg = math_x, math_y
        dx, dy = dist_from_global(xg), dist_from_global(yg)
        counts0 = [oracle.local_split((n,), P, r)[0] for r in range(P)]
        res[f"math_P{P}_dot"] = np.asarray(dx.dot(dy))
        dxa = dist_from_global(np.abs(xg))
        for name, o in NORM_ORDS:
            d = dxa if name == "1p5" else dx
            res[f"math_P{P}_norm{name}"] = np.asarray(
                d.norm() if o is None else d.norm(o))
        res[f"math_P{P}_addmul"] = ((dx + dy) * dx).asarray()
        d2 = DistributedArray(global_shape=mn, dtype=np.float64)
        counts = [oracle.local_split(mn, P, r)[0] for r in range(P)]
        off = int(np.sum(counts[:rank], initial=0))
        d2[:] = xg.reshape(mn)[off: off + d2.local_shape[0]]
        for w in (1, 2):
            res[f"__perrank__math_P{P}_ghost{w}"] = d2.add_ghost_cells(
                cells_front=w, cells_back=w)
        rd = d2.redistribute(axis=1)
        res[f"__perrank__math_P{P}_redist"] = np.asarray(rd.local_array)
        # masked (sub-communicator) reductions: dot and norms reduce
        # over the mask group only (ref DistributedArray.py:74-99,
        # :194-195, :715, :745-788) — results are rank-dependent
        if P >= 4:
            mk = [r // 2 for r in range(P)]
            md = DistributedArray(global_shape=n, mask=mk,
                                  dtype=np.float64)
            nd = DistributedArray(global_shape=n, mask=mk,
                                  dtype=np.float64)
            offm = int(np.sum(counts0[:rank], initial=0))
            md[:] = xg[offm: offm + md.local_shape[0]]
            nd[:] = yg[offm: offm + nd.local_shape[0]]
            res[f"__perrank__mask_P{P}_dot"] = np.atleast_1d(md.dot(nd))
            res[f"__perrank__mask_P{P}_norm2"] = np.atleast_1d(md.norm())
            res[f"__perrank__mask_P{P}_norminf"] = np.atleast_1d(
                md.norm(np.inf))
            # masked asarray gathers over the sub-communicator only
            # (ref DistributedArray.py:401-436) — collective on every
            # rank, result differs per mask group
            res[f"__perrank__mask_P{P}_asarray"] = md.asarray(
                masked=True)
        # CGLS
        nc = ncgls
        xg = cgls_x
        op = MPIFirstDerivative(CGLS_DIMS, sampling=SAMPLING,
                                kind="centered", edge=False, order=3)
        y = op.matvec(dist_from_global(xg))
        pcounts = plane_counts(CGLS_DIMS, P)
        for dn, damp in CGLS_DAMPS:
            # plane-aligned x0 (see the oracle-side comment)
            x0 = DistributedArray(
                global_shape=nc, local_shapes=[(c,) for c in pcounts],
                dtype=np.float64)
            x0[:] = 0.0
            xs, istop, iit, r1, r2, cost = cgls(
                op, y, x0, niter=CGLS_NITER, damp=damp, tol=0.0,
                show=False)
            res[f"cgls_P{P}_{dn}_x"] = xs.asarray()
            res[f"cgls_P{P}_{dn}_cost"] = np.asarray(cost)
        # BlockDiag with serial dense blocks (pylops stub MatrixMult)
        ops = [pylops.MatrixMult(A) for A in bd_mats[rank]]
        bop = MPIBlockDiag(ops=ops)
        assert bop.shape == (nr_bd, nc_bd)
        xg, yg = bd_x, bd_y
        res[f"bd_P{P}_mv"] = bop.matvec(dist_from_global(xg)).asarray()
        res[f"bd_P{P}_rmv"] = bop.rmatvec(dist_from_global(yg)).asarray()
        # CG on an SPD BlockDiag
        from pylops_mpi import cg
        sops = [pylops.MatrixMult(A) for A in spd[rank]]
        sbop = MPIBlockDiag(ops=sops)
        spd_counts = [int(sum(A.shape[0] for A in ms)) for ms in spd]

        def dist_from_counts(vec, counts):
            d = DistributedArray(
                global_shape=int(np.sum(counts)),
                local_shapes=[(int(c),) for c in counts],
                dtype=np.float64)
            off = int(np.sum(counts[:rank], initial=0))
            d[:] = vec[off: off + counts[rank]]
            return d

        ycg = dist_from_counts(cg_y, spd_counts)
        x0cg = dist_from_counts(np.zeros(n_spd), spd_counts)
        xs2, iit2, cost2 = cg(sbop, ycg, x0cg, niter=CG_NITER, tol=0.0,
                              show=False)
        res[f"cg_P{P}_x"] = xs2.asarray()
        # ISTA/FISTA on the same SPD BlockDiag (ref optimization/
        # sparsity.py:11,136)
        from pylops_mpi.optimization.sparsity import ista, fista
        ysp = dist_from_counts(cg_y, spd_counts)
        x0sp = dist_from_counts(np.zeros(n_spd), spd_counts)
        for tk in ("soft", "hard"):
            xi, _, ci = ista(sbop, ysp, x0sp, niter=ISTA_NITER,
                             eps=ISTA_EPS, alpha=ISTA_ALPHA,
                             threshkind=tk, tol=0.0, show=False)
            res[f"ista_P{P}_{tk}_x"] = xi.asarray()
            res[f"ista_P{P}_{tk}_cost"] = np.asarray(ci)
        xf, _, cf = fista(sbop, ysp, x0sp, niter=ISTA_NITER,
                          eps=ISTA_EPS, alpha=ISTA_ALPHA, tol=0.0,
                          show=False)
        res[f"fista_P{P}_x"] = xf.asarray()
        res[f"fista_P{P}_cost"] = np.asarray(cf)
        # power_iteration (ref optimization/eigs.py:10-102; np.random
        # .rand is patched rank-deterministically by compute_reference)
        from pylops_mpi.optimization.eigs import power_iteration
        b0 = dist_from_counts(np.zeros(n_spd), spd_counts)
        me, bk, nit = power_iteration(sbop, b0, niter=POWIT_NITER,
                                      tol=1e-5)
        res[f"powit_P{P}_eig"] = np.array([float(me)])
        res[f"powit_P{P}_vec"] = bk.asarray()
        res[f"powit_P{P}_n"] = np.array([float(nit)])
        res[f"cg_P{P}_cost"] = np.asarray(cost2)
        # VStack (matvec: BROADCAST in, SCATTER out; rmatvec: allreduce)
        from pylops_mpi import MPIVStack
        vops = [pylops.MatrixMult(A) for A in vmats[rank]]
        vop = MPIVStack(ops=vops)
        xvd = DistributedArray(global_shape=7,
                               partition=Partition.BROADCAST,
                               dtype=np.float64)
        xvd[:] = vs_x
        res[f"vs_P{P}_mv"] = vop.matvec(xvd).asarray()
        yvd = dist_from_global(vs_y)
        res[f"vs_P{P}_rmv"] = vop.rmatvec(yvd).asarray()
        # HStack (= VStack-of-adjoints, adjointed: ref HStack.py:90-107)
        from pylops_mpi import MPIHStack
        hop = MPIHStack(ops=[pylops.MatrixMult(A.T) for A in vmats[rank]])
        res[f"hs_P{P}_mv"] = hop.matvec(dist_from_global(vs_y)).asarray()
        yhd = DistributedArray(global_shape=7,
                               partition=Partition.BROADCAST,
                               dtype=np.float64)
        yhd[:] = vs_x
        res[f"hs_P{P}_rmv"] = hop.rmatvec(yhd).asarray()
        # Stacked operators + stacked arrays (ref BlockDiag.py:147-189,
        # VStack.py:153-203, DistributedArray.py:1041-1300)
        from pylops_mpi import StackedDistributedArray
        from pylops_mpi.basicoperators import (MPIStackedBlockDiag,
                                               MPIStackedVStack)
        sbop_st = MPIStackedBlockDiag(ops=[bop, sbop])
        xst = StackedDistributedArray(
            [dist_from_global(bd_x), dist_from_counts(sbd_x2, spd_counts)])
        res[f"sbd_P{P}_mv"] = np.concatenate(
            [d.asarray() for d in sbop_st.matvec(xst).distarrays])
        yst = StackedDistributedArray(
            [dist_from_global(bd_y), dist_from_counts(sbd_y2, spd_counts)])
        res[f"sbd_P{P}_rmv"] = np.concatenate(
            [d.asarray() for d in sbop_st.rmatvec(yst).distarrays])
        sops2 = [pylops.MatrixMult(A) for A in spd2[rank]]
        sbop2 = MPIBlockDiag(ops=sops2)
        svop = MPIStackedVStack(ops=[sbop, sbop2])
        xsv = dist_from_counts(cg_y, spd_counts)
        res[f"svs_P{P}_mv"] = np.concatenate(
            [d.asarray() for d in svop.matvec(xsv).distarrays])
        ysv = StackedDistributedArray(
            [dist_from_counts(scgls_y1, spd_counts),
             dist_from_counts(scgls_y2, spd_counts)])
        res[f"svs_P{P}_rmv"] = svop.rmatvec(ysv).asarray()
        # CGLS over the stacked VStack (damped): stacked data vector
        x0sv = dist_from_counts(np.zeros(n_spd), spd_counts)
        xsc, _, _, _, _, costsc = cgls(svop, ysv, x0sv, niter=CGLS_NITER,
                                       damp=0.4, tol=0.0, show=False)
        res[f"scgls_P{P}_x"] = xsc.asarray()
        res[f"scgls_P{P}_cost"] = np.asarray(costsc)
        # proximal subpackage (ref proximal/ProxOperator.py:113-121,
        # proximal/proximal/L2.py:180-189, proximal/optimization/
        # primal.py:135-168 and :306-340)
        from pyproximal import L1
        from pylops_mpi.proximal import MPIL2, MPIProxOperator
        from pylops_mpi.proximal.optimization.primal import (ADMML2,
                                                             ProximalGradient)
        l1 = MPIProxOperator(L1(sigma=0.3))
        ypg = dist_from_counts(pg_y, spd_counts)
        l2 = MPIL2(Op=sbop, b=ypg,
                   x0=dist_from_counts(np.zeros(n_spd), spd_counts))
        for an, acc in (("none", None), ("fista", "fista")):
            xpg = ProximalGradient(
                l2, l1, x0=dist_from_counts(np.zeros(n_spd), spd_counts),
                epsg=1.0, tau=0.01, niter=10, acceleration=acc)
            res[f"pg_P{P}_{an}_x"] = xpg.asarray()
        xa, za = ADMML2(l1, sbop, ypg, sbop2,
                        dist_from_counts(np.zeros(n_spd), spd_counts),
                        tau=0.05, niter=6,
                        kwargs_solver={"niter": 8, "tol": 0.0})
        res[f"admm_P{P}_x"] = xa.asarray()
        res[f"admm_P{P}_z"] = za.asarray()
        # Gradient (StackedVStack composition) + Laplacian (scaled-sum
        # composite algebra), serial axis>=1 blocks via the pylops stub.
        # P <= 4 only (the reference's own rebalance cannot ghost the
        # 10-row dims at P=8 — see the oracle-side note).
        if P <= 4:
            from pylops_mpi import MPIGradient, MPILaplacian
            gop = MPIGradient(dims=GRAD_DIMS, sampling=GRAD_SAMP, edge=False,
                              kind="centered")
            gx = dist_from_global(grad_x)
            ym = gop.matvec(gx)
            for i in range(3):
                res[f"grad_P{P}_mv{i}"] = ym.distarrays[i].asarray()
            # fill the stacked input with seeded globals (component splits
            # come from the matvec output structure)
            for i, di in enumerate(ym.distarrays):
                counts = [int(np.prod(sh)) for sh in di.local_shapes]
                off = int(np.sum(counts[:rank], initial=0))
                di[:] = grad_ys[i][off: off + counts[rank]].reshape(
                    di.local_shape)
            res[f"grad_P{P}_rmv"] = gop.rmatvec(ym).asarray()
            lop = MPILaplacian(dims=GRAD_DIMS, axes=LAP_AXES, weights=LAP_W,
                               sampling=LAP_SAMP, edge=False, kind="centered")
            res[f"lap_P{P}_mv"] = lop.matvec(gx).asarray()
            res[f"lap_P{P}_rmv"] = lop.rmatvec(
                dist_from_global(lap_y)).asarray()
        # MPIHalo
        if P <= 4:
            from pylops_mpi import MPIHalo
            for ci, (hdims, grid, hspec) in enumerate(_halo_cases(P)):
                rngh = np.random.default_rng(90 + ci)
                G = rngh.standard_normal(hdims)
                hop = MPIHalo(hdims, hspec, proc_grid_shape=grid)
                blocks = _halo_blocks(G, grid)
                counts = [b.size for b in blocks]
                hx = DistributedArray(
                    global_shape=int(np.prod(hdims)),
                    local_shapes=[(int(v),) for v in counts],
                    dtype=np.float64)
                hx[:] = blocks[rank].ravel()
                hy = hop.matvec(hx)
                res[f"halo_P{P}_c{ci}_mv"] = hy.asarray()
                res[f"halo_P{P}_c{ci}_rmv"] = hop.rmatvec(hy).asarray()
        # NonStationaryConvolve1D composition (P <= 2 — see the
        # oracle-side note)
        if P <= 2:
            from pylops_mpi.signalprocessing import \
                MPINonStationaryConvolve1D
            nop = MPINonStationaryConvolve1D(nsc_ndims, nsc_hs, nsc_ih)
            xn = dist_from_global(nsc_x)
            yn = dist_from_global(nsc_y)
            res[f"nsc_P{P}_mv"] = nop.matvec(xn).asarray()
            res[f"nsc_P{P}_rmv"] = nop.rmatvec(yn).asarray()
        # MDC chain (serial FFT/Identity via the pylops stubs)
        # (P<=4 — the reference needs >=2 freqs/rank, see oracle side)
        if P <= 4:
            from pylops_mpi.waveeqprocessing import MPIMDC
            goff, gcnt = _mdc_slices(P)[rank]
            mop = MPIMDC(mdc_G[goff: goff + gcnt], nt=mdc_nt, nv=mdc_nv,
                         nfreq=mdc_nfreq, dt=0.4, dr=2.0, twosided=False)
            xm = mdc_x
            ymv = mdc_y
            xd = DistributedArray(global_shape=xm.size,
                                  partition=Partition.BROADCAST,
                                  dtype=np.float64)
            xd[:] = xm
            got = mop.matvec(xd).asarray()
            res[f"mdc_P{P}_mv"] = np.real(got)
            res[f"mdc_P{P}_mv_imagmax"] = np.array(
                [float(np.max(np.abs(np.imag(got))))])
            yd = DistributedArray(global_shape=ymv.size,
                                  partition=Partition.BROADCAST,
                                  dtype=np.float64)
            yd[:] = ymv
            gotr = mop.rmatvec(yd).asarray()
            res[f"mdc_P{P}_rmv"] = np.real(gotr)
        # Fredholm1
        for dt in FRED_DTYPES:
            G = fred_G(dt)
            nsls = fred_split(P)
            off = int(np.sum(nsls[:rank], initial=0))
            Gl = G[off: off + nsls[rank]]
            for sg in (False, True):
                fop = MPIFredholm1(Gl, nz=FRED_SHAPE[3], saveGt=sg,
                                   usematmul=True, dtype=dt)
                key = f"fred_P{P}_{np.dtype(dt).char}_{'s' if sg else 'n'}"
                xb = fred_x(dt, True).ravel()
                yb = fred_x(dt, False).ravel()
                xd = DistributedArray(global_shape=xb.size,
                                      partition=Partition.BROADCAST,
                                      dtype=dt)
                xd[:] = xb
                yd = DistributedArray(global_shape=yb.size,
                                      partition=Partition.BROADCAST,
                                      dtype=dt)
                yd[:] = yb
                res[key + "_mv"] = fop.matvec(xd).asarray()
                res[key + "_rmv"] = fop.rmatvec(yd).asarray()
        return res
    return fn


def _ref_mm_fn(P):
    def fn(rank):
        from pylops_mpi import DistributedArray, Partition
        from pylops_mpi.basicoperators.MatrixMult import MPIMatrixMult

        N, K, M = MM_SHAPES
        rng = np.random.default_rng(77)
        res = {}
        for dt in MM_DTYPES:
            A = rng.standard_normal((N, K)).astype(dt)
            X = rng.standard_normal((K, M)).astype(dt)
            Y = rng.standard_normal((N, M)).astype(dt)
            if np.dtype(dt).kind == "c":
                A = A + 1j * rng.standard_normal((N, K))
                X = X + 1j * rng.standard_normal((K, M))
                Y = Y + 1j * rng.standard_normal((N, M))
            for kind in ("block", "summa"):
                key = f"mm_{kind}_P{P}_{np.dtype(dt).char}"
                if kind == "block":
                    pp = om._isqrt(P)
                    inputs = om.block_inputs(A, X, P)
                    ylocals = [Y[:, om.split_slice(M, pp, q // pp)].ravel()
                               for q in range(P)]
                else:
                    inputs = om.summa_inputs(A, X, P)
                    ylocals = [om.summa_tile(Y, P, q).ravel()
                               for q in range(P)]
                op = MPIMatrixMult(inputs[rank][0], M, kind=kind, dtype=dt)
                counts = [v.size for _, v in inputs]
                xd = DistributedArray(
                    global_shape=int(sum(counts)),
                    local_shapes=[(int(v),) for v in counts], dtype=dt)
                xd[:] = inputs[rank][1].ravel()
                res[key + "_mv"] = op.matvec(xd).asarray()
                ycounts = [v.size for v in ylocals]
                yd = DistributedArray(
                    global_shape=int(sum(ycounts)),
                    local_shapes=[(int(v),) for v in ycounts], dtype=dt)
                yd[:] = ylocals[rank]
                res[key + "_rmv"] = op.rmatvec(yd).asarray()
        return res
    return fn


def _ref_mmutil_fn(P):
    """active_grid_comm / local_block_split / block_gather pins (ref
    MatrixMult.py:24-171) — pure grid arithmetic plus the allgather
    reassembly, incl. inactive ranks at non-square P."""
    def fn(rank):
        from mpi4py import MPI
        from pylops_mpi import DistributedArray
        from pylops_mpi.basicoperators.MatrixMult import (
            active_grid_comm, block_gather, local_block_split)
        res = {}
        for (N, M) in ((7, 9), (2, 9)):
            comm, nr, row, col, act = active_grid_comm(
                MPI.COMM_WORLD, N, M)
            res[f"__perrank__mmu_P{P}_{N}x{M}_grid"] = np.array(
                [nr, row, col, int(act)], dtype=float)
            if act:
                rs, cs = local_block_split((N, M), nr, comm)
                res[f"__perrank__mmu_P{P}_{N}x{M}_block"] = np.array(
                    [rs.start, rs.stop, cs.start, cs.stop], dtype=float)
                rng = np.random.default_rng(55)
                G = rng.standard_normal((N, M))
                pa = comm.Get_size()
                blocks = []
                for q in range(pa):
                    qrs, qcs = local_block_split((N, M), q, comm)
                    blocks.append(G[qrs, qcs])
                xd = DistributedArray(
                    global_shape=int(sum(b.size for b in blocks)),
                    base_comm=comm,
                    local_shapes=[(int(b.size),) for b in blocks],
                    dtype=np.float64)
                xd[:] = blocks[nr].ravel()
                C = block_gather(xd, (N, M), comm)   # collective
                if rank == 0:
                    res[f"mmu_P{P}_{N}x{M}_gather"] = C
            else:
                res[f"__perrank__mmu_P{P}_{N}x{M}_block"] = np.full(
                    4, -1.0)
        # the documented non-square-world flow (ref MatrixMult.py
        # docstrings): build MPIMatrixMult on the ACTIVE sub-comm of a
        # non-square world; inactive ranks stay idle
        if P == 8:
            from pylops_mpi.basicoperators.MatrixMult import MPIMatrixMult
            Nm, Km, Mm = MM_SHAPES
            rng = np.random.default_rng(77)
            A = rng.standard_normal((Nm, Km))
            X = rng.standard_normal((Km, Mm))
            comm, nr, _, _, act = active_grid_comm(MPI.COMM_WORLD, Nm, Mm)
            if act:
                pa = comm.Get_size()
                for kind in ("block", "summa"):
                    if kind == "block":
                        inputs = om.block_inputs(A, X, pa)
                    else:
                        inputs = om.summa_inputs(A, X, pa)
                    op = MPIMatrixMult(inputs[nr][0], Mm, kind=kind,
                                       base_comm=comm, dtype="float64")
                    counts = [v.size for _, v in inputs]
                    xd = DistributedArray(
                        global_shape=int(sum(counts)), base_comm=comm,
                        local_shapes=[(int(v),) for v in counts],
                        dtype=np.float64)
                    xd[:] = inputs[nr][1].ravel()
                    yfull = op.matvec(xd).asarray()   # collective
                    if rank == 0:
                        res[f"mmsub_P{P}_{kind}_mv"] = yfull
        return res
    return fn


def _halo_cases(P):
    """(dims, grid, halo_spec) per P — Cartesian grids tiling P ranks."""
    if P == 1:
        return [((9, 8), (1, 1), 1), ((9, 8), (1, 1), (2, 2, 1, 1))]
    if P == 2:
        return [((9, 8), (1, 2), 1), ((9, 8), (2, 1), (2, 2, 1, 1))]
    if P == 4:
        return [((9, 8), (1, 4), 1), ((9, 8), (2, 2), (2, 2, 1, 1))]
    return []


def _halo_blocks(G, grid):
    import math as _m
    blocks = []
    for q in range(int(np.prod(grid))):
        coords = np.unravel_index(q, grid)
        sl = []
        for gdim, cc, pp in zip(G.shape, coords, grid):
            blk = _m.ceil(gdim / pp)
            sl.append(slice(cc * blk, min(cc * blk + blk, gdim)))
        blocks.append(G[tuple(sl)].copy())
    return blocks


def _halo_window(G, grid, rank, halo_spec):
    """The reference's matvec semantics (ref Halo.py:197-227,362-398):
    scalar halo widths trim to 0 at global borders; tuple widths keep
    their size with zero padding beyond the domain."""
    import math as _m
    nd = G.ndim
    coords = np.unravel_index(rank, grid)
    if isinstance(halo_spec, (int, np.integer)):
        halo = [halo_spec] * (2 * nd)
        for a in range(nd):
            if coords[a] == 0:
                halo[2 * a] = 0
            if coords[a] == grid[a] - 1:
                halo[2 * a + 1] = 0
    else:
        h = tuple(halo_spec)
        halo = list(h if len(h) == 2 * nd else
                    sum(((d, d) for d in h), ()))
    starts, ends = [], []
    for gdim, c, pp in zip(G.shape, coords, grid):
        blk = _m.ceil(gdim / pp)
        starts.append(c * blk)
        ends.append(min(c * blk + blk, gdim))
    ext = tuple((ends[a] - starts[a]) + halo[2 * a] + halo[2 * a + 1]
                for a in range(nd))
    out = np.zeros(ext, dtype=G.dtype)
    src, dst = [], []
    for a in range(nd):
        lo = starts[a] - halo[2 * a]
        hi = ends[a] + halo[2 * a + 1]
        s0, s1 = max(lo, 0), min(hi, G.shape[a])
        src.append(slice(s0, s1))
        dst.append(slice(s0 - lo, (s0 - lo) + (s1 - s0)))
    out[tuple(dst)] = G[tuple(src)]
    return out


def _mdc_slices(P):
    nfreq = 8  # the Fredholm kernel carries the MASKED frequencies only
    counts = [oracle.local_split((nfreq,), P, r)[0] for r in range(P)]
    offs = np.cumsum([0] + counts)
    return [(int(offs[r]), int(counts[r])) for r in range(P)]


def _mdc_setup():
    nt, ns, nr, nv, nfreq = 20, 3, 4, 2, 8
    nfft = (nt + 1 + 1) // 2  # ceil((nt+1)/2)
    rng = np.random.default_rng(66)
    G = (rng.standard_normal((nfft, ns, nr))
         + 1j * rng.standard_normal((nfft, ns, nr)))[:nfreq]
    return nt, ns, nr, nv, nfreq, G


def _nsc_setup():
    # 48-sample domain, anchors every 6 (halo = spacing + hsize//2 + 1
    # must fit the 12-sample P=4 blocks, ref NonStatConvolve1d.py:120-136)
    nf, hsize = 8, 5
    rng = np.random.default_rng(77)
    hs = rng.standard_normal((nf, hsize))
    ih = np.arange(3, 48, 6)
    return (48,), hs, ih


def _grad_lap_oracle(P):
    out = {}
    ng = int(np.prod(GRAD_DIMS))
    xg = make_global_x(ng, P)
    for i in range(3):
        out[f"grad_P{P}_mv{i}"] = _serial_axis_fd(
            xg, GRAD_DIMS, i, GRAD_SAMP[i], False, True)
    ys = [make_global_x(ng, P, seed_shift=2 + i) for i in range(3)]
    out[f"grad_P{P}_rmv"] = sum(
        _serial_axis_fd(ys[i], GRAD_DIMS, i, GRAD_SAMP[i], False,
                        False) for i in range(3))
    out[f"lap_P{P}_mv"] = sum(
        LAP_W[i] * _serial_axis_fd(xg, GRAD_DIMS, LAP_AXES[i],
                                   LAP_SAMP[i], False, True,
                                   second=True) for i in range(3))
    yl = make_global_x(ng, P, seed_shift=5)
    out[f"lap_P{P}_rmv"] = sum(
        LAP_W[i] * _serial_axis_fd(yl, GRAD_DIMS, LAP_AXES[i],
                                   LAP_SAMP[i], False, False,
                                   second=True) for i in range(3))
    return out
