"""Generate golden parity fixtures for the hot path.

Inputs follow the reference's own test recipe (seed-42 ``normal(rank, 10)``
per simulated rank, /root/reference/tests/test_derivative.py:25,207); outputs
are produced by the INDEPENDENT serial restatement (oracle/serial.py) plus
the dense-transpose adjoint, so the fixtures pin both the oracle and the HIP
path without circularity.

Run from the repo root:  python tests/golden/generate_golden.py
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from oracle import (dense_matrix_from_matvec, local_split,  # noqa: E402
                    serial_fd1_matvec, serial_fd2_matvec)

OUT = os.path.join(os.path.dirname(__file__), "golden_fd.npz")

CASES_FD1 = [
    ("forward", 3, False), ("backward", 3, False),
    ("centered", 3, False), ("centered", 3, True),
    ("centered", 5, False), ("centered", 5, True),
]
CASES_FD2 = [("forward", False), ("backward", False),
             ("centered", False), ("centered", True)]
DIMS = [(32,), (17, 5), (16, 4, 3)]
P = 4  # simulated ranks used to build the input
SAMPLING = 1.5


def make_global_x(dims):
    n = int(np.prod(dims))
    parts = []
    for r in range(P):
        np.random.seed(42)
        parts.append(np.random.normal(r, 10, local_split((n,), P, r)))
    return np.concatenate(parts)


def main():
    data = {}
    for dims in DIMS:
        tag = "x".join(map(str, dims))
        xg = make_global_x(dims)
        data[f"x_{tag}"] = xg
        n = xg.size
        for kind, order, edge in CASES_FD1:
            name = f"fd1_{kind}{order}_{'e' if edge else 'n'}_{tag}"
            mv = lambda v: serial_fd1_matvec(  # noqa: E731
                v.reshape(dims), SAMPLING, kind, edge, order).ravel()
            data[f"{name}_mv"] = mv(xg)
            A = dense_matrix_from_matvec(mv, n)
            data[f"{name}_rmv"] = A.T @ xg
        for kind, edge in CASES_FD2:
            name = f"fd2_{kind}_{'e' if edge else 'n'}_{tag}"
            mv = lambda v: serial_fd2_matvec(  # noqa: E731
                v.reshape(dims), SAMPLING, kind, edge).ravel()
            data[f"{name}_mv"] = mv(xg)
            A = dense_matrix_from_matvec(mv, n)
            data[f"{name}_rmv"] = A.T @ xg
    np.savez_compressed(OUT, **data)
    print(f"wrote {OUT}: {len(data)} arrays")


if __name__ == "__main__":
    main()
