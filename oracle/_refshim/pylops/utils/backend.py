"""pylops.utils.backend stub: numpy-only (no cupy in this container)."""
import numpy as np


def get_module(backend="numpy"):
    if backend != "numpy":
        raise ModuleNotFoundError(
            f"pylops stub: only the numpy engine exists here ({backend!r})")
    return np


def get_array_module(x):
    return np


def get_module_name(mod):
    return "numpy"


def to_numpy(x):
    return np.asarray(x)


def get_normalize_axis_index():
    def normalize_axis_index(axis, ndim):
        if not -ndim <= axis < ndim:
            raise np.exceptions.AxisError(axis, ndim)
        return axis % ndim
    return normalize_axis_index
