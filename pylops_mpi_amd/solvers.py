"""CG / CGLS solvers — same recurrences as the reference
(/root/reference/pylops_mpi/optimization/cls_basic.py and basic.py), driving
the operator + DistributedArray surface unchanged ("cgls drives
the loop unchanged").

MI355X-first differences (both bit-identical to the reference recurrence):

* the three per-iteration vector updates use the fused one-pass HIP
  axpy/xpby kernels (DistributedArray.iaxpy_/xpby_) instead of the
  reference's two fresh temporaries per update (ref cls_basic.py:390-391,
  396 via DistributedArray.py:618-683) — same arithmetic per element
  (mul-round, add-round; -ffp-contract=off), one HBM pass instead of three;

* the recurrence scalars (a, b, kold) live in DEVICE memory: the dots
  reduce into 1-element slots (DistributedArray.dot_into, batched into one
  allreduce per group), tiny pam_scalar_alpha/div kernels form a and b on
  device, and the axpy/xpby kernels read them from HBM (pam_axpy_d /
  pam_xpby_d).  The whole iteration is launched back-to-back with a SINGLE
  host synchronization — the k readback the reference's stop test
  (ref cls_basic.py:433) requires — instead of one blocking readback per
  dot/norm (5 per CGLS iteration).  Measured r01 (scripts/
  gpu_cgls_probe.py): at the bench config the iteration is GPU-bound
  (~17 ms of kernels after the round-final launch-shape retunes) and
  the paths are within ~5% (17.9 vs 18.7 ms/iter); at latency-bound
  sizes (256x256x64 and below)
  the device path is ~35% faster/iter (0.23 vs 0.31 ms), and at N>1 it
  keeps collective latency off the host critical path.  The device path
  covers real SCATTER mask-free CUDA arrays (the north
  star); anything else (complex, BROADCAST, masked, stacked, CPU/gloo
  tests) takes the host-scalar path.  PAM_DISABLE_DEVSCALARS=1 forces the
  host path; the two produce bit-identical iterates and cost traces
  (|x/y| == |x|/|y| in IEEE covers the reference's np.abs placement).
"""
from typing import Optional, Tuple

import numpy as np

from . import _ffi, deps
from .distributedarray import DistributedArray, Partition


def _dev_capable(*arrs) -> bool:
    for d in arrs:
        if not isinstance(d, DistributedArray) or d._is_cplx():
            return False
        if not d.local_array.is_cuda or d.mask is not None:
            return False
        if d.partition != Partition.SCATTER:
            return False
    return True


class CG:
    """ref cls_basic.py:12-249."""

    def __init__(self, Op):
        self.Op = Op

    def setup(self, y: DistributedArray, x0: DistributedArray,
              niter: Optional[int] = None, tol: float = 1e-4,
              show: bool = False) -> DistributedArray:
        # ref :57-108
        self.y = y
        self.niter = niter
        self.tol = tol
        x = x0.copy()
        self.r = self.y - self.Op.matvec(x)
        self.rank = x.rank
        self.c = self.r.copy()
        self.kold = float(np.abs(self.r.dot(self.r.conj())))
        self.cost = [float(np.sqrt(self.kold))]
        self.iiter = 0
        self._dev = (deps.devscalars_enabled
                     and not deps.env_flag("PAM_DISABLE_DEVSCALARS")
                     and _dev_capable(x, self.r, self.c))
        if self._dev:
            import torch
            self._B = torch.zeros(8, dtype=torch.float64,
                                  device=x.local_array.device)
            self._B[7] = self.kold
        return x

    def step(self, x: DistributedArray, show: bool = False):
        # ref :110-141
        if self._dev and _dev_capable(self.c):
            return self._step_dev(x)
        Opc = self.Op.matvec(self.c)
        cOpc = np.abs(self.c.dot(Opc.conj()))
        a = float(self.kold / cOpc)
        x.iaxpy_(a, self.c)          # x += a * c
        self.r.iaxpy_(-a, Opc)       # r -= a * Opc
        k = float(np.abs(self.r.dot(self.r.conj())))
        b = float(k / self.kold)
        self.c.xpby_(self.r, b)      # c = r + b * c
        self.kold = k
        self.iiter += 1
        self.cost.append(float(np.sqrt(self.kold)))
        return x

    def _step_dev(self, x: DistributedArray):
        """ref :110-141 with device-resident scalars (module docstring);
        one host sync per iteration (the k readback for the stop test)."""
        B, lib, s = self._B, _ffi.lib(), self.c._stream()
        Opc = self.Op.matvec(self.c)
        self.c.dot_into(Opc, B[0:1])
        x._sub_comm.allreduce_(B[0:1], "sum")
        _ffi.checked(lib.pam_scalar_alpha(          # a = |kold / cOpc|
            s, B[2:3].data_ptr(), B[7:8].data_ptr(), B[0:2].data_ptr(), 0.0),
            "scalar_alpha")
        x.iaxpy_dev_(B[2:3], self.c)                # x += a * c
        self.r.iaxpy_dev_(B[2:3], Opc, -1.0)        # r -= a * Opc
        self.r.dot_into(self.r, B[4:5])
        x._sub_comm.allreduce_(B[4:5], "sum")
        _ffi.checked(lib.pam_scalar_div(            # b = k / kold
            s, B[3:4].data_ptr(), B[4:5].data_ptr(), B[7:8].data_ptr()),
            "scalar_div")
        B[7:8].copy_(B[4:5])                        # kold <- k (device)
        self.c.xpby_dev_(self.r, B[3:4])            # c = r + b * c
        self.kold = abs(float(B[4].item()))         # the ONE sync: stop test
        self.iiter += 1
        self.cost.append(float(np.sqrt(self.kold)))
        return x

    def run(self, x, niter=None, show=False, itershow=(10, 10, 10)):
        niter = self.niter if niter is None else niter
        if niter is None:
            raise ValueError("niter must not be None")
        while self.iiter < niter and self.kold > self.tol:
            x = self.step(x, show)
        return x

    def finalize(self, show: bool = False):
        self.cost = np.array(self.cost)

    def solve(self, y, x0, niter: int = 10, tol: float = 1e-4,
              show: bool = False, itershow=(10, 10, 10)):
        x = self.setup(y=y, x0=x0, niter=niter, tol=tol, show=show)
        x = self.run(x, niter, show=show, itershow=itershow)
        self.finalize(show)
        return x, self.iiter, self.cost


class CGLS:
    """ref cls_basic.py:252-531."""

    def __init__(self, Op):
        self.Op = Op

    def setup(self, y: DistributedArray, x0: DistributedArray,
              niter: Optional[int] = None, damp: float = 0.0,
              tol: float = 1e-4, show: bool = False) -> DistributedArray:
        # ref :308-368
        self.y = y
        self.damp = damp ** 2
        self.tol = tol
        self.niter = niter
        x = x0.copy()
        self.s = self.y - self.Op.matvec(x)
        r = self.Op.rmatvec(self.s)
        if damp != 0.0:
            r.iaxpy_(-damp, x)  # r -= damp * x, fused (ref :347-348)
        self.rank = x.rank
        self.c = r.copy()
        self.q = self.Op.matvec(self.c)
        self.kold = float(np.abs(r.dot(r.conj())))
        self.cost = [float(self.s.norm())]
        # NB: the reference uses the raw damp here and damp**2 in step
        # (ref :358 vs :401) — reproduced faithfully.
        self.cost1 = [float(np.sqrt(self.cost[0] ** 2
                                    + damp * np.abs(x.dot(x.conj()))))]
        self.iiter = 0
        self._dev = (deps.devscalars_enabled
                     and not deps.env_flag("PAM_DISABLE_DEVSCALARS")
                     and _dev_capable(x, self.s, self.c, self.q))
        if self._dev:
            import torch
            self._B = torch.zeros(8, dtype=torch.float64,
                                  device=x.local_array.device)
            self._B[7] = self.kold
        return x

    def step(self, x: DistributedArray, show: bool = False):
        # ref :370-404; vector updates fused (module docstring)
        if self._dev and _dev_capable(self.c, self.q):
            return self._step_dev(x)
        a = float(np.abs(self.kold / (self.q.dot(self.q.conj())
                                      + self.damp * self.c.dot(self.c.conj()))))
        x.iaxpy_(a, self.c)          # x += a * c        (ref :390)
        self.s.iaxpy_(-a, self.q)    # s -= a * q        (ref :391)
        r = self.Op.rmatvec(self.s)
        if self.damp != 0.0:
            r.iaxpy_(-self.damp, x)  # r -= damp^2 * x, fused (ref :392-393)
        k = float(np.abs(r.dot(r.conj())))
        b = float(k / self.kold)
        self.c.xpby_(r, b)           # c = r + b * c     (ref :396)
        self.q = self.Op.matvec(self.c)
        self.kold = k
        self.iiter += 1
        self.cost.append(float(self.s.norm()))
        self.cost1.append(float(np.sqrt(self.cost[self.iiter] ** 2
                                        + self.damp
                                        * np.abs(x.dot(x.conj())))))
        return x

    def _step_dev(self, x: DistributedArray):
        """ref :370-404 with device-resident scalars (module docstring);
        one host sync per iteration (the k/ss/xx readback: stop test +
        cost/cost1 bookkeeping, ref :398-403)."""
        B, lib, s = self._B, _ffi.lib(), self.c._stream()
        self.q.dot_into(self.q, B[0:1])             # q.q
        self.c.dot_into(self.c, B[1:2])             # c.c
        x._sub_comm.allreduce_(B[0:2], "sum")       # one collective for both
        _ffi.checked(lib.pam_scalar_alpha(          # a = |kold/(qq+damp^2 cc)|
            s, B[2:3].data_ptr(), B[7:8].data_ptr(), B[0:2].data_ptr(),
            self.damp), "scalar_alpha")
        x.iaxpy_dev_(B[2:3], self.c)                # x += a * c   (ref :390)
        self.s.iaxpy_dev_(B[2:3], self.q, -1.0)     # s -= a * q   (ref :391)
        r = self.Op.rmatvec(self.s)
        if self.damp != 0.0:
            r.iaxpy_(-self.damp, x)                 # r -= damp^2 * x (host a)
        r.dot_into(r, B[4:5])                       # k
        self.s.dot_into(self.s, B[5:6])             # |s|^2   (cost)
        x.dot_into(x, B[6:7])                       # |x|^2   (cost1)
        x._sub_comm.allreduce_(B[4:7], "sum")       # one collective for all 3
        _ffi.checked(lib.pam_scalar_div(            # b = k / kold
            s, B[3:4].data_ptr(), B[4:5].data_ptr(), B[7:8].data_ptr()),
            "scalar_div")
        B[7:8].copy_(B[4:5])                        # kold <- k (device)
        self.c.xpby_dev_(r, B[3:4])                 # c = r + b * c (ref :396)
        self.q = self.Op.matvec(self.c)
        k, ss, xx = B[4:7].tolist()                 # the ONE sync
        self.kold = abs(k)
        self.iiter += 1
        self.cost.append(float(ss ** (1.0 / 2)))
        self.cost1.append(float(np.sqrt(self.cost[self.iiter] ** 2
                                        + self.damp * abs(xx))))
        return x

    def run(self, x, niter=None, show=False, itershow=(10, 10, 10)):
        # ref :406-449
        niter = self.niter if niter is None else niter
        if niter is None:
            raise ValueError("niter must not be None")
        while self.iiter < niter and self.kold > self.tol:
            x = self.step(x, show)
        return x

    def finalize(self, show: bool = False):
        # ref :451-469
        self.istop = 1 if self.kold < self.tol else 2
        self.r1norm = self.kold
        self.r2norm = self.cost1[self.iiter]
        self.cost = np.array(self.cost)

    def solve(self, y, x0, niter: int = 10, damp: float = 0.0,
              tol: float = 1e-4, show: bool = False,
              itershow=(10, 10, 10)):
        # ref :471-531
        x = self.setup(y=y, x0=x0, niter=niter, damp=damp, tol=tol,
                       show=show)
        x = self.run(x, niter, show=show, itershow=itershow)
        self.finalize(show)
        return x, self.istop, self.iiter, self.r1norm, self.r2norm, self.cost


def cg(Op, y, x0, niter: int = 10, tol: float = 1e-4, show: bool = False,
       itershow: Tuple[int, int, int] = (10, 10, 10)):
    """ref optimization/basic.py:13-70."""
    cgsolve = CG(Op)
    x, iiter, cost = cgsolve.solve(y=y, x0=x0, niter=niter, tol=tol,
                                   show=show, itershow=itershow)
    return x, iiter, cost


def cgls(Op, y, x0, niter: int = 10, damp: float = 0.0, tol: float = 1e-4,
         show: bool = False, itershow: Tuple[int, int, int] = (10, 10, 10)):
    """ref optimization/basic.py:73-148."""
    cgsolve = CGLS(Op)
    x, istop, iiter, r1norm, r2norm, cost = cgsolve.solve(
        y=y, x0=x0, niter=niter, damp=damp, tol=tol, show=show,
        itershow=itershow)
    return x, istop, iiter, r1norm, r2norm, cost


def power_iteration(Op, b_k, niter: int = 10, tol: float = 1e-5,
                    dtype="float64"):
    """Largest-eigenpair power iteration,
    ref optimization/eigs.py:10-100 (random re-init of b_k, vdot Rayleigh
    quotient, renormalization via the HIP scale kernel)."""
    import torch

    from .stacked import StackedDistributedArray

    cmpx = np.issubdtype(np.dtype(dtype), np.complexfloating)

    def _randomize(d):
        t = torch.rand(d.local_shape, dtype=torch.float64,
                       device=d.local_array.device)
        if cmpx:
            t = t + 1j * torch.rand(d.local_shape, dtype=torch.float64,
                                    device=d.local_array.device)
        d[:] = t.to(d.local_array.dtype)

    if isinstance(b_k, StackedDistributedArray):
        for d in b_k.distarrays:
            _randomize(d)
    else:
        _randomize(b_k)
    b_k = b_k * (1.0 / float(b_k.norm()))
    maxeig_old = 0.0
    iiter = 0
    for iiter in range(niter):
        b1_k = Op.matvec(b_k)
        maxeig = b_k.dot(b1_k, vdot=True)
        maxeig = complex(maxeig) if cmpx else float(maxeig)
        b1_k_norm = float(b1_k.norm())
        b_k = b1_k * (1.0 / b1_k_norm)
        if abs(maxeig - maxeig_old) < tol * abs(maxeig):
            break
        maxeig_old = maxeig
    return maxeig, b_k, iiter + 1
