"""pylops.optimization.cls_sparsity stub: the threshold helpers the
reference's ISTA/FISTA import (pylops' published closed forms; the
distributed sparsity parity is pinned on our own oracle, not here)."""
import numpy as np


def _softthreshold(x, thresh):
    return np.maximum(np.abs(x) - thresh, 0.0) * np.sign(x)


def _hardthreshold(x, thresh):
    return x * (np.abs(x) >= np.sqrt(2 * thresh))


def _halfthreshold(x, thresh):
    phi = np.arccos(np.clip((thresh / 8.0)
                            * (np.abs(x) / 3.0) ** (-1.5), -1.0, 1.0))
    out = (2.0 / 3.0) * x * (1 + np.cos((2.0 * np.pi - 2.0 * phi) / 3.0))
    out[np.abs(x) <= (54 ** (1.0 / 3.0) / 4.0) * thresh ** (2.0 / 3.0)] = 0
    return out
