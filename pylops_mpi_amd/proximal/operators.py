"""MPI-enabled proximal operators.

MPIProxOperator restates ref proximal/ProxOperator.py:17-140 (separable
prox applied per-rank + the name-keyed reduction of functional values);
MPIL2 restates ref proximal/proximal/L2.py:15-192 on our operator stack.

The reference delegates the LOCAL prox math to pyproximal — an unpinned
dependency absent from /root/reference — so the separable operators it
supports (ref ProxOperator.py:10-14 `_call_reduce_op`: Box, L0, L1) are
restated here from pyproximal's published definitions:

  Box(lower, upper): call = all(lower <= x <= upper); prox = clip
  L0(sigma):         call = sigma * count_nonzero;    prox = hard thresh
  L1(sigma):         call = sigma * sum|x|;           prox = soft thresh

with the soft/hard threshold rules already used by ISTA/FISTA
(include/pam.h pam_thresh; soft: sign(x)*max(|x|-t,0), hard: x kept iff
|x| >= sqrt(2 t)).  Parity is anchored on the reference's call sites and
tests (ref tests/test_prox.py:64-228, tests/test_proxsolver.py:103-300)
and on oracle/proximal.py.

Thresholded proxes (L0/L1) run through the HIP pam_thresh kernel and
require CUDA tensors (fail-loud, no CPU fallback); the scalar functional
evaluations and Box's clip are torch device ops (plumbing).
"""
import math
from typing import Any, Callable, Optional

import numpy as np
import torch

from .. import _ffi
from ..distributedarray import DistributedArray, Partition
from ..linearoperator import MPILinearOperator
from ..blockdiag import MPIBlockDiag
from ..fftlocal import IdentityLocal
from ..solvers import cg, cgls
from ..stacked import StackedDistributedArray
from ..vstack import MPIStackedVStack


def _thresh_tensor(x: torch.Tensor, kind: int, thresh: float) -> torch.Tensor:
    if not x.is_cuda:
        raise RuntimeError(
            "pylops_mpi_amd.proximal thresholds run the HIP pam_thresh "
            "kernel and need a CUDA tensor (no CPU fallback)")
    out = torch.empty_like(x)
    stream = torch.cuda.current_stream(x.device).cuda_stream
    _ffi.checked(_ffi.lib().pam_thresh(
        stream, out.data_ptr(), x.data_ptr(), x.numel(), kind,
        float(thresh), _ffi.dtype_code(x.dtype)), "thresh")
    return out


class ProxOperator:
    """Local separable proximal operator on torch device tensors
    (restates the pyproximal.ProxOperator interface the reference wraps,
    ref ProxOperator.py:36-48)."""

    hasgrad = False

    def __call__(self, x: torch.Tensor):
        raise NotImplementedError

    def prox(self, x: torch.Tensor, tau: float) -> torch.Tensor:
        raise NotImplementedError

    def proxdual(self, x: torch.Tensor, tau: float) -> torch.Tensor:
        # Moreau decomposition (pyproximal ProxOperator.proxdual)
        return x - tau * self.prox(x / tau, 1.0 / tau)

    def grad(self, x: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError


class Box(ProxOperator):
    def __init__(self, lower: float = -np.inf, upper: float = np.inf):
        self.lower = float(lower)
        self.upper = float(upper)

    def __call__(self, x: torch.Tensor) -> bool:
        return bool(torch.all((x >= self.lower) & (x <= self.upper)).item())

    def prox(self, x: torch.Tensor, tau: float) -> torch.Tensor:
        return torch.clamp(x, self.lower, self.upper)


class L0(ProxOperator):
    def __init__(self, sigma: float = 1.0):
        self.sigma = float(sigma)

    def __call__(self, x: torch.Tensor) -> float:
        return self.sigma * float(torch.count_nonzero(x).item())

    def prox(self, x: torch.Tensor, tau: float) -> torch.Tensor:
        return _thresh_tensor(x, 1, tau * self.sigma)


class L1(ProxOperator):
    def __init__(self, sigma: float = 1.0):
        self.sigma = float(sigma)

    def __call__(self, x: torch.Tensor) -> float:
        return self.sigma * float(torch.sum(torch.abs(x)).item())

    def prox(self, x: torch.Tensor, tau: float) -> torch.Tensor:
        return _thresh_tensor(x, 0, tau * self.sigma)


# functional-value reduction per separable operator
# (ref ProxOperator.py:10-14: Box -> MPI.LAND/all, L0/L1 -> MPI.SUM/sum)
_call_reduce_op = dict(
    Box=("min", all),
    L0=("sum", sum),
    L1=("sum", sum),
)


class MPIProxOperator:
    """ref proximal/ProxOperator.py:17-140."""

    def __init__(self, prox: ProxOperator) -> None:
        prox_name = str(type(prox).__name__)
        if prox_name not in _call_reduce_op:
            raise NotImplementedError(
                f"{prox_name} is not a separable proximal "
                "operator, must be implemented directly...")
        self.proxop = prox
        self.hasgrad = prox.hasgrad

    def __repr__(self) -> str:
        if hasattr(self, "proxop"):
            return f"<{type(self).__name__} ({type(self.proxop).__name__})>"
        return f"<{type(self).__name__}>"

    def __call__(self, x):
        # ref :56-110 (LAND becomes a min over {0,1} on the RCCL plane)
        if isinstance(x, DistributedArray):
            f = self.proxop(x.local_array)
            if x.partition == Partition.SCATTER:
                redop = _call_reduce_op[type(self.proxop).__name__][0]
                t = torch.tensor([float(f)], dtype=torch.float64,
                                 device=x.local_array.device)
                x._sub_comm.allreduce_(t, redop)
                v = float(t.item())
                return bool(v != 0.0) if isinstance(f, bool) else v
            return f
        # StackedDistributedArray (ref :106-110)
        red = _call_reduce_op[type(self.proxop).__name__][1]
        return red([self(x[iarr]) for iarr in range(x.narrays)])

    def prox(self, x, tau: float, **kwargs: Any):
        # ref :112-121
        y = x.empty_like()
        if isinstance(x, DistributedArray):
            y[:] = self.proxop.prox(x.local_array, tau)
        else:
            for iarr in range(x.narrays):
                y[iarr][:] = self.proxop.prox(x[iarr].local_array, tau)
        return y

    def proxdual(self, x, tau: float, **kwargs: Any):
        # ref :123-140
        y = x.empty_like()
        if isinstance(x, DistributedArray):
            y[:] = self.proxop.proxdual(x.local_array, tau)
        else:
            for iarr in range(x.narrays):
                y[iarr][:] = self.proxop.proxdual(x[iarr].local_array, tau)
        return y


def _identity_op(x: DistributedArray, dtype):
    """The reference's per-partition identity wrap
    (ref L2.py:149-152,160-163)."""
    n = int(np.prod(x.local_shape))
    if x.partition == Partition.SCATTER:
        return MPIBlockDiag([IdentityLocal(n, n, dtype=dtype)])
    return MPILinearOperator(Op=IdentityLocal(n, n, dtype=dtype),
                             shape=(n, n), dtype=dtype)


class MPIL2(MPIProxOperator):
    """ref proximal/proximal/L2.py:15-192."""

    def __init__(
        self,
        Op: Optional[MPILinearOperator] = None,
        b: Optional[DistributedArray] = None,
        q: Optional[DistributedArray] = None,
        sigma: float = 1.0,
        alpha: float = 1.0,
        qgrad: bool = True,
        niter=10,
        x0: Optional[DistributedArray] = None,
        warm: bool = True,
        solver: Optional[str] = "cgls",
        kwargs_solver: Optional[dict] = None,
    ) -> None:
        # ref :73-109
        if Op is not None and x0 is None:
            raise ValueError("x0 must be passed when Op is not None")
        self.Op = Op
        self.hasgrad = True
        self.b = b
        self.q = q
        self.sigma = sigma
        self.alpha = alpha
        self.qgrad = qgrad
        self.niter = niter
        self.x0 = x0
        self.warm = warm
        self.solver = solver
        self.count = 0
        self.kwargs_solver = {} if kwargs_solver is None else kwargs_solver
        if self.solver == "cg":
            self.normaleqs = True
        elif self.solver == "cgls":
            self.normaleqs = False
        else:
            raise ValueError(f"Provided solver={self.solver}. "
                             "Available options are 'cg' or 'cgls'.")
        if self.Op is not None and self.b is not None and self.normaleqs:
            self.OpTb = self.sigma * (self.Op.H @ self.b)

    def __call__(self, x: DistributedArray) -> float:
        # ref :111-120
        if self.Op is not None and self.b is not None:
            f = (self.sigma / 2.0) * float((self.Op @ x - self.b).norm()) ** 2
        elif self.b is not None:
            f = (self.sigma / 2.0) * float((x - self.b).norm()) ** 2
        else:
            f = (self.sigma / 2.0) * float(x.norm()) ** 2
        if self.q is not None:
            f += self.alpha * float(self.q.dot(x))
        return float(f)

    def prox(self, x: DistributedArray, tau: float,
             **kwargs: Any) -> DistributedArray:
        # ref :131-181 (with the _increment_count wrapper inlined)
        self.count += 1
        niter = self.niter if isinstance(self.niter, int) \
            else self.niter(self.count)
        if self.Op is not None and self.b is not None:
            if self.normaleqs:
                y = x + tau * self.OpTb
                if self.q is not None:
                    y = y - (tau * self.alpha) * self.q
                Iop = _identity_op(x, self.Op.dtype)
                Op1 = Iop + float(tau * self.sigma) * (self.Op.H @ self.Op)
                x = cg(Op1, y, self.x0, niter=niter,
                       **self.kwargs_solver)[0]
            else:
                y = x
                if self.q is not None:
                    y = y - (tau * self.alpha) * self.q
                Iop = _identity_op(x, self.Op.dtype)
                Opreg = MPIStackedVStack([
                    math.sqrt(tau * self.sigma) * self.Op,
                    Iop,
                ])
                breg = StackedDistributedArray(
                    [math.sqrt(tau * self.sigma) * self.b, y])
                x = cgls(Opreg, breg, self.x0, niter=niter,
                         **self.kwargs_solver)[0]
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

    def grad(self, x: DistributedArray) -> DistributedArray:
        # ref :183-192
        if self.Op is not None and self.b is not None:
            g = self.sigma * (self.Op.H @ (self.Op @ x - self.b))
        elif self.b is not None:
            g = self.sigma * (x - self.b)
        else:
            g = self.sigma * x
        if self.q is not None and self.qgrad:
            g = g + self.alpha * self.q
        return g
