"""pylops.utils stub: the names the reference imports from here."""
from typing import Any, Sequence, Union

import numpy as np

from . import deps  # noqa: F401
from .backend import (get_array_module, get_module,  # noqa: F401
                      get_module_name, to_numpy)

NDArray = np.ndarray
DTypeLike = Any
ShapeLike = Sequence[int]
InputDimsLike = Union[int, Sequence[int]]


def get_normalize_axis_index():
    def normalize_axis_index(axis, ndim):
        if not -ndim <= axis < ndim:
            raise np.exceptions.AxisError(axis, ndim)
        return axis % ndim
    return normalize_axis_index


def get_real_dtype(dtype):
    return np.real(np.ones(1, dtype=np.dtype(dtype))).dtype


def get_complex_dtype(dtype):
    return (np.ones(1, dtype=np.dtype(dtype)) + 1j).dtype
