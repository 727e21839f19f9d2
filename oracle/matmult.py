"""Expected-layout builders for MPIMatrixMult — TEST INFRASTRUCTURE ONLY.

The reference's own tests pin the distributed matmul against the global
product ``A_glob @ X_glob`` (ref tests/test_matrixmult.py:37-45,146-161 —
SURVEY.md §8c names this the sanctioned recipe).  These helpers build the
per-rank inputs each rank passes to the operator and the per-rank outputs
the operator must return, from the global arrays, restating the layout
arithmetic of /root/reference/pylops_mpi/basicoperators/MatrixMult.py:
  block kind layout -> ref :320-338 (blk_rows by col_id, block_cols by
                       row_id)
  summa tiles       -> ref :82-129 local_block_split (ceil blocks)
"""
import math
from typing import List

import numpy as np


def _isqrt(P: int) -> int:
    p = math.isqrt(P)
    assert p * p == P, "P must be a perfect square"
    return p


def split_slice(n: int, nblk: int, idx: int) -> slice:
    blk = math.ceil(n / nblk)
    s = idx * blk
    return slice(s, min(n, s + blk))


# --------------------------------------------------------------- block kind
def block_inputs(A: np.ndarray, X: np.ndarray, P: int):
    """Per-rank (A_block, x_local_flat) the user hands the operator.
    ref :320-338: rank (row r, col c) holds A rows by c, X cols by r."""
    Pp = _isqrt(P)
    N, K = A.shape
    M = X.shape[1]
    out = []
    for q in range(P):
        r, c = divmod(q, Pp)
        rs = split_slice(N, Pp, c)
        cs = split_slice(M, Pp, r)
        out.append((A[rs].copy(), X[:, cs].copy().ravel()))
    return out


def block_col_lens(N: int, M: int, P: int) -> List[int]:
    Pp = _isqrt(P)
    out = []
    for q in range(P):
        r, _ = divmod(q, Pp)
        cs = split_slice(M, Pp, r)
        out.append(cs.stop - cs.start)
    return out


def block_expected_mv(A: np.ndarray, X: np.ndarray, P: int):
    """Per-rank flattened (N, M_loc) outputs of the forward."""
    Pp = _isqrt(P)
    Y = A @ X
    M = X.shape[1]
    return [Y[:, split_slice(M, Pp, q // Pp)].ravel() for q in range(P)]


def block_expected_rmv(A: np.ndarray, Y: np.ndarray, P: int):
    """Per-rank flattened (K, M_loc) outputs of the adjoint."""
    Pp = _isqrt(P)
    Z = A.conj().T @ Y
    M = Y.shape[1]
    return [Z[:, split_slice(M, Pp, q // Pp)].ravel() for q in range(P)]


# -------------------------------------------------------------------- summa
def summa_tile(G: np.ndarray, P: int, q: int) -> np.ndarray:
    Pp = _isqrt(P)
    r, c = divmod(q, Pp)
    return G[split_slice(G.shape[0], Pp, r),
             split_slice(G.shape[1], Pp, c)].copy()


def summa_inputs(A: np.ndarray, X: np.ndarray, P: int):
    """Per-rank (A_tile, x_tile_flat): A 2-D tiled over (N,K), X over
    (K,M) (ref :483-497)."""
    return [(summa_tile(A, P, q), summa_tile(X, P, q).ravel())
            for q in range(P)]


def summa_expected_mv(A: np.ndarray, X: np.ndarray, P: int):
    Y = A @ X
    return [summa_tile(Y, P, q).ravel() for q in range(P)]


def summa_expected_rmv(A: np.ndarray, Y: np.ndarray, P: int):
    Z = A.conj().T @ Y
    return [summa_tile(Z, P, q).ravel() for q in range(P)]
