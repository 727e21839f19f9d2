"""pylops.utils._internal stub."""
import numpy as np


def _value_or_sized_to_tuple(value, repeat=1):
    try:
        iter(value)
    except TypeError:
        return tuple([value] * repeat)
    return tuple(value)


def _value_or_sized_to_array(value, repeat=1):
    return np.asarray(_value_or_sized_to_tuple(value, repeat))


def _raise_on_wrong_dtype(arr, dtype, name):
    if np.dtype(dtype) != arr.dtype:
        raise TypeError(f"Wrong input type for `{name}`.")
