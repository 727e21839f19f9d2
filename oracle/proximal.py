"""Serial restatements of the proximal subpackage — TEST INFRASTRUCTURE
ONLY.

Restates, in plain NumPy, ref pylops_mpi/proximal/: the separable local
operators the reference takes from pyproximal (an UNPINNED dependency
absent from /root/reference — Box/L0/L1 restated from pyproximal's
published definitions, anchored on ref tests/test_prox.py:64-228), the
MPIL2 operator (ref proximal/proximal/L2.py:15-192) and the
ProximalGradient/ADMML2 solvers (ref proximal/optimization/
primal.py:22-347).  Serial == world-1 distributed, the comparison the
reference's own tests make on rank 0 (ref tests/test_proxsolver.py:
143-161).
"""
from math import sqrt
from typing import Optional

import numpy as np


# ------------------------------------------------------- local operators
def soft_threshold(x: np.ndarray, thresh: float) -> np.ndarray:
    """pylops/pyproximal soft rule (matches pam_thresh kind=0)."""
    if np.iscomplexobj(x):
        a = np.abs(x)
        with np.errstate(divide="ignore", invalid="ignore"):
            s = np.where(a > 0, np.maximum(a - thresh, 0.0) / a, 0.0)
        return x * s
    return np.sign(x) * np.maximum(np.abs(x) - thresh, 0.0)


def hard_threshold(x: np.ndarray, thresh: float) -> np.ndarray:
    """pylops/pyproximal hard rule (matches pam_thresh kind=1)."""
    return np.where(np.abs(x) >= np.sqrt(2.0 * thresh), x, 0.0)


class SerBox:
    hasgrad = False

    def __init__(self, lower=-np.inf, upper=np.inf):
        self.lower, self.upper = lower, upper

    def __call__(self, x):
        return bool(np.all((x >= self.lower) & (x <= self.upper)))

    def prox(self, x, tau):
        return np.clip(x, self.lower, self.upper)


class SerL0:
    hasgrad = False

    def __init__(self, sigma=1.0):
        self.sigma = sigma

    def __call__(self, x):
        return self.sigma * float(np.count_nonzero(x))

    def prox(self, x, tau):
        return hard_threshold(x, tau * self.sigma)


class SerL1:
    hasgrad = False

    def __init__(self, sigma=1.0):
        self.sigma = sigma

    def __call__(self, x):
        return self.sigma * float(np.sum(np.abs(x)))

    def prox(self, x, tau):
        return soft_threshold(x, tau * self.sigma)


# ------------------------------------------------------------ dense CGLS
def dense_cg(A, y, x0, niter=10, tol=1e-4):
    """The CG recurrence (ref cls_basic.py:86-141) on dense arrays.
    ``A`` may be a matrix or a callable matvec (to mirror op-by-op
    application of composite operators, as pylops/pyproximal do)."""
    mv = A if callable(A) else (lambda v: A @ v)
    x = x0.copy()
    r = y - mv(x)
    c = r.copy()
    kold = float(np.abs(np.dot(r.conj(), r)))
    iiter = 0
    while iiter < niter and kold > tol:
        Ac = mv(c)
        cAc = np.abs(np.dot(c.conj(), Ac))
        a = float(kold / cAc)
        x = x + a * c
        r = r - a * Ac
        k = float(np.abs(np.dot(r.conj(), r)))
        b = float(k / kold)
        c = r + b * c
        kold = k
        iiter += 1
    return x


def dense_cgls(A, y, x0, niter=10, damp=0.0, tol=1e-4):
    """The CGLS recurrence (ref cls_basic.py:308-404) on dense arrays.
    ``A`` may be a matrix or an ``(mv, rmv)`` callable pair (to mirror
    op-by-op application of composite/stacked operators)."""
    if isinstance(A, tuple):
        mv, rmv = A
    else:
        mv = lambda v: A @ v           # noqa: E731
        rmv = lambda s: A.conj().T @ s  # noqa: E731
    damp2 = damp ** 2
    x = x0.copy()
    s = y - mv(x)
    r = rmv(s) - damp * x
    c = r.copy()
    q = mv(c)
    kold = float(np.abs(np.dot(r.conj(), r)))
    iiter = 0
    while iiter < niter and kold > tol:
        a = float(np.abs(kold / (np.dot(q.conj(), q)
                                 + damp2 * np.dot(c.conj(), c))))
        x = x + a * c
        s = s - a * q
        r = rmv(s) - damp2 * x
        k = float(np.abs(np.dot(r.conj(), r)))
        b = float(k / kold)
        c = r + b * c
        q = mv(c)
        kold = k
        iiter += 1
    return x


# ------------------------------------------------------------------- L2
class SerL2:
    """ref proximal/proximal/L2.py:15-192 with a dense Op."""

    hasgrad = True

    def __init__(self, Op: Optional[np.ndarray] = None, b=None, q=None,
                 sigma=1.0, alpha=1.0, qgrad=True, niter=10, x0=None,
                 warm=True, solver="cgls", kwargs_solver=None):
        if Op is not None and x0 is None:
            raise ValueError("x0 must be passed when Op is not None")
        self.Op, self.b, self.q = Op, b, q
        self.sigma, self.alpha, self.qgrad = sigma, alpha, qgrad
        self.niter, self.x0, self.warm = niter, x0, warm
        self.normaleqs = solver == "cg"
        self.kwargs_solver = {} if kwargs_solver is None else kwargs_solver
        self.count = 0
        if Op is not None and b is not None and self.normaleqs:
            self.OpTb = sigma * (Op.conj().T @ b)

    def __call__(self, x):
        if self.Op is not None and self.b is not None:
            f = (self.sigma / 2.0) * float(
                np.linalg.norm(self.Op @ x - self.b)) ** 2
        elif self.b is not None:
            f = (self.sigma / 2.0) * float(np.linalg.norm(x - self.b)) ** 2
        else:
            f = (self.sigma / 2.0) * float(np.linalg.norm(x)) ** 2
        if self.q is not None:
            f += self.alpha * float(np.dot(self.q, x))
        return float(f)

    def prox(self, x, tau):
        self.count += 1
        niter = self.niter if isinstance(self.niter, int) \
            else self.niter(self.count)
        if self.Op is not None and self.b is not None:
            n = x.shape[0]
            if self.normaleqs:
                y = x + tau * self.OpTb
                if self.q is not None:
                    y = y - (tau * self.alpha) * self.q
                # functional normal-equations op, mirroring the composite
                # Iop + tau*sigma*(Op.H @ Op) applied op-by-op (ref
                # L2.py:149-155, pylops composite semantics)
                A, ts = self.Op, float(tau * self.sigma)
                mv = lambda v: v.copy() + ts * (   # noqa: E731
                    A.conj().T @ (A @ v))
                x = dense_cg(mv, y, self.x0, niter=niter,
                             **self.kwargs_solver)
            else:
                y = x
                if self.q is not None:
                    y = y - (tau * self.alpha) * self.q
                # functional stacked op, mirroring MPIStackedVStack
                # [sqrt(tau*sigma)*Op, Iop] (ref L2.py:156-170)
                A, c0 = self.Op, sqrt(tau * self.sigma)
                ny = A.shape[0]
                mv = lambda v: np.concatenate(    # noqa: E731
                    [c0 * (A @ v), v.copy()])
                rmv = lambda s: c0 * (            # noqa: E731
                    A.conj().T @ s[:ny]) + s[ny:]
                breg = np.concatenate([c0 * self.b, y])
                x = dense_cgls((mv, rmv), breg, self.x0, niter=niter,
                               **self.kwargs_solver)
            if self.warm:
                self.x0 = x
        elif self.b is not None:
            num = x + (tau * self.sigma) * self.b
            if self.q is not None:
                num = num - (tau * self.alpha) * self.q
            x = (1.0 / (1.0 + tau * self.sigma)) * num
        else:
            num = x
            if self.q is not None:
                num = num - (tau * self.alpha) * self.q
            x = (1.0 / (1.0 + tau * self.sigma)) * num
        return x

    def grad(self, x):
        if self.Op is not None and self.b is not None:
            g = self.sigma * (self.Op.conj().T @ (self.Op @ x - self.b))
        elif self.b is not None:
            g = self.sigma * (x - self.b)
        else:
            g = self.sigma * x
        if self.q is not None and self.qgrad:
            g = g + self.alpha * self.q
        return g


# ---------------------------------------------------------------- solvers
def ser_proximal_gradient(proxf, proxg, x0, epsg=1.0, tau=None, eta=1.0,
                          niter=10, acceleration=None, tol=None):
    """ref primal.py:22-201, serial."""
    epsg = np.asarray(epsg, dtype=float)
    if epsg.size == 1:
        epsg = epsg * np.ones(niter)
    t = 1.0
    x = x0.copy()
    y = x.copy()
    pfg = np.inf
    for iiter in range(niter):
        xold = x.copy()
        if eta == 1.0:
            x = proxg.prox(y - tau * proxf.grad(y), epsg[iiter] * tau)
        else:
            x = x + eta * (
                proxg.prox(x - tau * proxf.grad(x), epsg[iiter] * tau) - x)
        if acceleration == "vandenberghe":
            omega = iiter / (iiter + 3)
        elif acceleration == "fista":
            told = t
            t = (1.0 + np.sqrt(1.0 + 4.0 * t ** 2)) / 2.0
            omega = (told - 1.0) / t
        else:
            omega = 0
        y = x + omega * (x - xold)
        if tol is not None:
            pfgold = pfg
            pfg = proxf(x) + np.sum(epsg[iiter] * proxg(x))
            if np.abs(1.0 - pfg / pfgold) < tol:
                break
    return x


def ser_admml2(proxg, Op, b, A, x0, tau, niter=10, z0=None, gfirst=False,
               kwargs_solver=None):
    """ref primal.py:209-347, serial (dense Op and A)."""
    kwargs_solver = {} if kwargs_solver is None else dict(kwargs_solver)
    x = x0.copy()
    z = z0.copy() if z0 is not None else A @ x
    u = np.zeros_like(z)
    sqrttau = 1.0 / sqrt(tau)
    ny = Op.shape[0]
    # functional stacked op, mirroring MPIStackedVStack [Op, sqrttau*A]
    mv = lambda v: np.concatenate(          # noqa: E731
        [Op @ v, sqrttau * (A @ v)])
    rmv = lambda s: Op.conj().T @ s[:ny] + sqrttau * (  # noqa: E731
        A.conj().T @ s[ny:])
    for _ in range(niter):
        if gfirst:
            Ax = A @ x
            z = proxg.prox(Ax + u, tau)
            breg = np.concatenate([b, sqrttau * (z - u)])
            x = dense_cgls((mv, rmv), breg, x, **kwargs_solver)
        else:
            breg = np.concatenate([b, sqrttau * (z - u)])
            x = dense_cgls((mv, rmv), breg, x, **kwargs_solver)
            Ax = A @ x
            z = proxg.prox(Ax + u, tau)
        u = u + Ax - z
    return x, z


def half_threshold(x: np.ndarray, thresh: float) -> np.ndarray:
    """L1/2 ('half') threshold — the published Xu et al. (2012) closed
    form pylops' _halfthreshold implements: the EXACT prox of
    (thresh/2)*|v|^(1/2) (verified to ~1e-12 against a dense grid scan;
    pylops is absent from /root/reference, so the formula is pinned by
    the prox-optimality property test in tests/test_oracle_proximal.py).
    Complex: magnitude rule."""
    a = np.abs(x)
    cut = (54.0 ** (1.0 / 3.0) / 4.0) * thresh ** (2.0 / 3.0)
    with np.errstate(divide="ignore", invalid="ignore"):
        phi = np.arccos(np.clip((thresh / 8.0) * (a / 3.0) ** (-1.5),
                                -1.0, 1.0))
        fac = (2.0 / 3.0) * (1.0 + np.cos(2.0 * np.pi / 3.0
                                          - (2.0 / 3.0) * phi))
    return np.where(a > cut, x * fac, 0.0)


class SimScaledOp:
    """scalar * MPILinearOperator wrap (pylops _ScaledLinearOperator
    semantics the reference reaches via ``sqrttau * A``)."""

    def __init__(self, op, s):
        self.op = op
        self.s = float(s)
        self.shape = op.shape

    def matvec(self, x):
        return self.s * self.op.matvec(x)

    def rmatvec(self, x):
        return self.s * self.op.rmatvec(x)


def sim_proximal_gradient_l2l1(Op, b, x0, tau, sigma_l1, niter,
                               acceleration=None, epsg=1.0):
    """Rank-sim of the reference's ProximalGradient loop (ref proximal/
    optimization/primal.py:135-168) with proxf = MPIL2(Op, b)
    (grad = Op^H(Op x - b), ref proximal/proximal/L2.py:180-189) and
    proxg = MPIProxOperator(pyproximal L1) (per-rank soft threshold,
    ref proximal/ProxOperator.py:113-121)."""
    epsg = np.asarray(epsg, dtype=float)
    if epsg.size == 1:
        epsg = epsg * np.ones(niter)
    t = 1.0
    x = x0.copy()
    y = x.copy()
    for iiter in range(niter):
        xold = x.copy()
        g = Op.rmatvec(Op.matvec(y) - b)
        v = y - tau * g
        v.locals = [soft_threshold(a, epsg[iiter] * tau * sigma_l1)
                    for a in v.locals]
        x = v
        if acceleration == "vandenberghe":
            omega = iiter / (iiter + 3)
        elif acceleration == "fista":
            told = t
            t = (1.0 + np.sqrt(1.0 + 4.0 * t ** 2)) / 2.0
            omega = (told - 1.0) / t
        else:
            omega = 0
        y = x + omega * (x - xold)
    return x


def sim_admml2_l1(Op, b, A, x0, tau, sigma_l1, niter, solver_niter,
                  solver_tol=0.0, gfirst=False):
    """Rank-sim of the reference's ADMML2 (ref proximal/optimization/
    primal.py:306-340): augmented CGLS over MPIStackedVStack
    [Op, (1/sqrt(tau)) A], per-rank L1 prox on z, running dual u."""
    from .blockdiag import SimStackedVStack
    from .cgls import sim_cgls
    from .ranksim import SimStackedArray

    x = x0.copy()
    z = A.matvec(x)
    u = z.zeros_like()
    sqrttau = 1.0 / sqrt(tau)
    Opreg = SimStackedVStack([Op, SimScaledOp(A, sqrttau)])

    def _prox(v):
        v.locals = [soft_threshold(a, tau * sigma_l1) for a in v.locals]
        return v

    for _ in range(niter):
        if gfirst:
            Ax = A.matvec(x)
            z = _prox(Ax + u)
            breg = SimStackedArray([b, sqrttau * (z - u)])
            x, _ = sim_cgls(Opreg, breg, x, solver_niter, tol=solver_tol)
        else:
            breg = SimStackedArray([b, sqrttau * (z - u)])
            x, _ = sim_cgls(Opreg, breg, x, solver_niter, tol=solver_tol)
            Ax = A.matvec(x)
            z = _prox(Ax + u)
        u = u + Ax - z
    return x, z
