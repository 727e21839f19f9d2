"""Rank-simulated MPIFirstDerivative / MPISecondDerivative — TEST ONLY.

Faithful restatement of the reference's distributed stencil slice algebra,
rank by rank.  Citations:
  FirstDerivative  -> /root/reference/pylops_mpi/basicoperators/FirstDerivative.py
  SecondDerivative -> /root/reference/pylops_mpi/basicoperators/SecondDerivative.py
"""
from typing import Tuple

import numpy as np

from .ranksim import (Partition, SimArray, add_ghost_cells, reshaped_apply,
                      to_dist)


def _z(k: int, rest: Tuple[int, ...], dtype) -> np.ndarray:
    return np.zeros((k,) + tuple(rest), dtype=dtype)


class _SimDerivativeBase:
    def __init__(self, dims, sampling=1.0, kind="centered", edge=False,
                 dtype=np.float64):
        self.dims = (dims,) if isinstance(dims, int) else tuple(dims)
        self.sampling = sampling
        self.kind = kind
        self.edge = edge
        self.dtype = dtype
        n = int(np.prod(self.dims))
        self.shape = (n, n)

    def _wrap(self, x: SimArray, body) -> SimArray:
        # BROADCAST -> SCATTER conversion, ref FirstDerivative.py:128-138
        if x.partition is Partition.BROADCAST:
            x = to_dist(x.locals[0], x.size)
        return reshaped_apply(body, self.dims, x)

    def matvec(self, x: SimArray) -> SimArray:
        return self._wrap(x, self._body_matvec)

    def rmatvec(self, x: SimArray) -> SimArray:
        return self._wrap(x, self._body_rmatvec)


class SimFirstDerivative(_SimDerivativeBase):
    """ref FirstDerivative.py:84-318."""

    def __init__(self, dims, sampling=1.0, kind="centered", edge=False,
                 order=3, dtype=np.float64):
        super().__init__(dims, sampling, kind, edge, dtype)
        self.order = order
        key = (kind, order if kind == "centered" else 0)
        table = {
            ("forward", 0): (self._mv_forward, self._rmv_forward),
            ("backward", 0): (self._mv_backward, self._rmv_backward),
            ("centered", 3): (self._mv_centered3, self._rmv_centered3),
            ("centered", 5): (self._mv_centered5, self._rmv_centered5),
        }
        if key not in table:
            raise NotImplementedError(f"kind={kind} order={order}")
        self._body_matvec, self._body_rmatvec = table[key]

    # ---- forward, ref :141-168
    def _mv_forward(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        gx = add_ghost_cells(xs, None, [1] * P)
        ys = []
        for r in range(P):
            g = gx[r]
            yf = g[1:] - g[:-1]
            if r == P - 1:
                yf = np.append(yf, _z(1, rest, dt), axis=0)
            ys.append(yf / self.sampling)
        return ys

    def _rmv_forward(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        gx = add_ghost_cells(xs, [1] * P, None)
        ys = []
        for r in range(P):
            x = xs[r]
            y = np.zeros_like(x)
            if r == P - 1:
                y[:-1] -= x[:-1]
            else:
                y[:] -= x[:]
            yf = gx[r][:-1]
            if r == 0:
                yf = np.append(_z(1, rest, dt), yf, axis=0)
            y[:] += yf
            ys.append(y / self.sampling)
        return ys

    # ---- backward, ref :171-198
    def _mv_backward(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        gx = add_ghost_cells(xs, [1] * P, None)
        ys = []
        for r in range(P):
            g = gx[r]
            yb = g[1:] - g[:-1]
            if r == 0:
                yb = np.append(_z(1, rest, dt), yb, axis=0)
            ys.append(yb / self.sampling)
        return ys

    def _rmv_backward(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        gx = add_ghost_cells(xs, None, [1] * P)
        ys = []
        for r in range(P):
            x = xs[r]
            y = np.zeros_like(x)
            yb = gx[r][1:]
            if r == P - 1:
                yb = np.append(yb, _z(1, rest, dt), axis=0)
            y[:] -= yb
            if r == 0:
                y[1:] += x[1:]
            else:
                y[:] += x[:]
            ys.append(y / self.sampling)
        return ys

    # ---- centered3, ref :201-246
    def _mv_centered3(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        N = self.dims[0]
        gx = add_ghost_cells(xs, [1] * P, [1] * P)
        ys = []
        for r in range(P):
            g = gx[r]
            yc = 0.5 * (g[2:] - g[:-2])
            if r == 0:
                yc = np.append(_z(1, rest, dt), yc, axis=0)
            if r == P - 1:
                yc = np.append(yc, _z(min(N - 1, 1), rest, dt), axis=0)
            y = yc.copy()
            if self.edge:
                x = xs[r]
                if r == 0:
                    y[0] = x[1] - x[0]
                if r == P - 1:
                    y[-1] = x[-1] - x[-2]
            ys.append(y / self.sampling)
        return ys

    def _rmv_centered3(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        N = self.dims[0]
        ys = [np.zeros_like(x) for x in xs]
        gx = add_ghost_cells(xs, None, [2] * P)
        for r in range(P):
            yc = 0.5 * gx[r][1:-1]
            if r == P - 1:
                yc = np.append(yc, _z(min(N, 2), rest, dt), axis=0)
            ys[r][:] -= yc
        gx = add_ghost_cells(xs, [2] * P, None)
        for r in range(P):
            yc = 0.5 * gx[r][1:-1]
            if r == 0:
                yc = np.append(_z(min(N, 2), rest, dt), yc, axis=0)
            ys[r][:] += yc
        for r in range(P):
            if self.edge:
                x = xs[r]
                if r == 0:
                    ys[r][0] -= x[0]
                    ys[r][1] += x[0]
                if r == P - 1:
                    ys[r][-2] -= x[-1]
                    ys[r][-1] += x[-1]
            ys[r][:] /= self.sampling
        return ys

    # ---- centered5, ref :249-318
    def _mv_centered5(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        N = self.dims[0]
        gx = add_ghost_cells(xs, [2] * P, [2] * P)
        ys = []
        for r in range(P):
            g = gx[r]
            yc = (g[:-4] / 12.0 - 2 * g[1:-3] / 3.0
                  + 2 * g[3:-1] / 3.0 - g[4:] / 12.0)
            if r == 0:
                yc = np.append(_z(min(N, 2), rest, dt), yc, axis=0)
            if r == P - 1:
                yc = np.append(yc, _z(min(N - 2, 2), rest, dt), axis=0)
            y = yc.copy()
            if self.edge:
                x = xs[r]
                if r == 0:
                    y[0] = x[1] - x[0]
                    y[1] = 0.5 * (x[2] - x[0])
                if r == P - 1:
                    y[-1] = x[-1] - x[-2]
                    y[-2] = 0.5 * (x[-1] - x[-3])
            ys.append(y / self.sampling)
        return ys

    def _rmv_centered5(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        N = self.dims[0]
        ys = [np.zeros_like(x) for x in xs]
        gx = add_ghost_cells(xs, None, [4] * P)
        for r in range(P):
            yc = gx[r][2:-2] / 12.0
            if r == P - 1:
                yc = np.append(yc, _z(min(N, 4), rest, dt), axis=0)
            ys[r][:] += yc
        gx = add_ghost_cells(xs, [1] * P, [3] * P)
        for r in range(P):
            yc = 2.0 * gx[r][2:-2] / 3.0
            if r == 0:
                yc = np.append(_z(1, rest, dt), yc, axis=0)
            if r == P - 1:
                yc = np.append(yc, _z(min(N - 1, 3), rest, dt), axis=0)
            ys[r][:] -= yc
        gx = add_ghost_cells(xs, [3] * P, [1] * P)
        for r in range(P):
            yc = 2.0 * gx[r][2:-2] / 3.0
            if r == 0:
                yc = np.append(_z(min(N, 3), rest, dt), yc, axis=0)
            if r == P - 1:
                yc = np.append(yc, _z(min(N - 3, 1), rest, dt), axis=0)
            ys[r][:] += yc
        gx = add_ghost_cells(xs, [4] * P, None)
        for r in range(P):
            yc = gx[r][2:-2] / 12.0
            if r == 0:
                yc = np.append(_z(min(N, 4), rest, dt), yc, axis=0)
            ys[r][:] -= yc
        for r in range(P):
            if self.edge:
                x = xs[r]
                if r == 0:
                    ys[r][0] -= x[0] + 0.5 * x[1]
                    ys[r][1] += x[0]
                    ys[r][2] += 0.5 * x[1]
                if r == P - 1:
                    ys[r][-3] -= 0.5 * x[-2]
                    ys[r][-2] -= x[-1]
                    ys[r][-1] += 0.5 * x[-2] + x[-1]
            ys[r][:] /= self.sampling
        return ys


class SimSecondDerivative(_SimDerivativeBase):
    """ref SecondDerivative.py:84-256."""

    def __init__(self, dims, sampling=1.0, kind="centered", edge=False,
                 dtype=np.float64):
        super().__init__(dims, sampling, kind, edge, dtype)
        table = {
            "forward": (self._mv_forward, self._rmv_forward),
            "backward": (self._mv_backward, self._rmv_backward),
            "centered": (self._mv_centered, self._rmv_centered),
        }
        if kind not in table:
            raise NotImplementedError(f"kind={kind}")
        self._body_matvec, self._body_rmatvec = table[kind]

    # ---- forward, ref :124-160
    def _mv_forward(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        N = self.dims[0]
        gx = add_ghost_cells(xs, None, [2] * P)
        ys = []
        for r in range(P):
            g = gx[r]
            yf = g[2:] - 2 * g[1:-1] + g[:-2]
            if r == P - 1:
                yf = np.append(yf, _z(min(N, 2), rest, dt), axis=0)
            ys.append(yf / self.sampling ** 2)
        return ys

    def _rmv_forward(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        N = self.dims[0]
        ys = []
        for r in range(P):
            x = xs[r]
            y = np.zeros_like(x)
            if r == P - 1:
                y[:-2] += x[:-2]
            else:
                y[:] += x[:]
            ys.append(y)
        gx = add_ghost_cells(xs, [1] * P, [1] * P)
        for r in range(P):
            yf = gx[r][:-2]
            if r == 0:
                yf = np.append(_z(1, rest, dt), yf, axis=0)
            if r == P - 1:
                yf = np.append(yf, _z(min(1, N - 1), rest, dt), axis=0)
            ys[r][:] -= 2 * yf
        gx = add_ghost_cells(xs, [2] * P, None)
        for r in range(P):
            yf = gx[r][:-2]
            if r == 0:
                yf = np.append(_z(min(N, 2), rest, dt), yf, axis=0)
            ys[r][:] += yf
            ys[r][:] /= self.sampling ** 2
        return ys

    # ---- backward, ref :162-199
    def _mv_backward(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        N = self.dims[0]
        gx = add_ghost_cells(xs, [2] * P, None)
        ys = []
        for r in range(P):
            g = gx[r]
            yb = g[2:] - 2 * g[1:-1] + g[:-2]
            if r == 0:
                yb = np.append(_z(min(N, 2), rest, dt), yb, axis=0)
            ys.append(yb / self.sampling ** 2)
        return ys

    def _rmv_backward(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        N = self.dims[0]
        ys = [np.zeros_like(x) for x in xs]
        gx = add_ghost_cells(xs, None, [2] * P)
        for r in range(P):
            yb = gx[r][2:]
            if r == P - 1:
                yb = np.append(yb, _z(min(2, N), rest, dt), axis=0)
            ys[r][:] += yb
        gx = add_ghost_cells(xs, [1] * P, [1] * P)
        for r in range(P):
            yb = 2 * gx[r][2:]
            if r == 0:
                yb = np.append(_z(1, rest, dt), yb, axis=0)
            if r == P - 1:
                yb = np.append(yb, _z(min(1, N - 1), rest, dt), axis=0)
            ys[r][:] -= yb
        for r in range(P):
            x = xs[r]
            if r == 0:
                ys[r][2:] += x[2:]
            else:
                ys[r][:] += x[:]
            ys[r][:] /= self.sampling ** 2
        return ys

    # ---- centered, ref :201-256
    def _mv_centered(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        N = self.dims[0]
        gx = add_ghost_cells(xs, [1] * P, [1] * P)
        ys = []
        for r in range(P):
            g = gx[r]
            yc = g[2:] - 2 * g[1:-1] + g[:-2]
            if r == 0:
                yc = np.append(_z(1, rest, dt), yc, axis=0)
            if r == P - 1:
                yc = np.append(yc, _z(min(1, N - 1), rest, dt), axis=0)
            y = yc.copy()
            if self.edge:
                x = xs[r]
                if r == 0:
                    y[0] = x[0] - 2 * x[1] + x[2]
                if r == P - 1:
                    y[-1] = x[-3] - 2 * x[-2] + x[-1]
            ys.append(y / self.sampling ** 2)
        return ys

    def _rmv_centered(self, xs):
        P, rest, dt = len(xs), self.dims[1:], self.dtype
        N = self.dims[0]
        ys = [np.zeros_like(x) for x in xs]
        gx = add_ghost_cells(xs, None, [2] * P)
        for r in range(P):
            yc = gx[r][1:-1]
            if r == P - 1:
                yc = np.append(yc, _z(min(2, N), rest, dt), axis=0)
            ys[r][:] += yc
        gx = add_ghost_cells(xs, [1] * P, [1] * P)
        for r in range(P):
            yc = 2 * gx[r][1:-1]
            if r == 0:
                yc = np.append(_z(1, rest, dt), yc, axis=0)
            if r == P - 1:
                yc = np.append(yc, _z(min(1, N - 1), rest, dt), axis=0)
            ys[r][:] -= yc
        gx = add_ghost_cells(xs, [2] * P, None)
        for r in range(P):
            yc = gx[r][1:-1]
            if r == 0:
                yc = np.append(_z(min(N, 2), rest, dt), yc, axis=0)
            ys[r][:] += yc
        for r in range(P):
            if self.edge:
                x = xs[r]
                if r == 0:
                    ys[r][0] += x[0]
                    ys[r][1] -= 2 * x[0]
                    ys[r][2] += x[0]
                if r == P - 1:
                    ys[r][-3] += x[-1]
                    ys[r][-2] -= 2 * x[-1]
                    ys[r][-1] += x[-1]
            ys[r][:] /= self.sampling ** 2
        return ys
