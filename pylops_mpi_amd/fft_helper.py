"""Distributed fftshift/ifftshift (ref utils/fft_helper.py:11-107).

Shifts along local axes are a torch.roll on the local block; a shift
along the DISTRIBUTED axis first realigns the distribution to another
axis (one batched all-to-all, DistributedArray.redistribute) so the roll
is local, and leaves the array in the redistributed state — exactly the
reference's behavior.
"""
from typing import Optional, Sequence

import numpy as np
import torch

from .distributedarray import DistributedArray


def _shift_nd(x: DistributedArray, axes: Optional[Sequence[int]],
              sign: int, name: str) -> DistributedArray:
    if x.ndim < 2:
        raise ValueError(
            f"{name} requires a 2D or higher array, but got ndim={x.ndim}. ")
    if axes is None:
        axes = tuple(range(x.ndim))
    elif np.isscalar(axes):
        axes = (int(axes),)
    axes = [int(a) for a in axes]
    local_axes = [ax for ax in axes if ax != x.axis]
    dist_axes = [ax for ax in axes if ax == x.axis]
    if local_axes:
        shifts = [sign * (x.global_shape[ax] // 2) for ax in local_axes]
        x[:] = torch.roll(x.local_array, shifts=shifts, dims=local_axes)
    if dist_axes:
        new_axis = 1 if x.axis == 0 else 0
        x = x.redistribute(axis=new_axis)
        shifts = [sign * (x.global_shape[ax] // 2) for ax in dist_axes]
        x[:] = torch.roll(x.local_array, shifts=shifts, dims=dist_axes)
    return x


def fftshift_nd(x: DistributedArray,
                axes: Optional[Sequence[int]] = None) -> DistributedArray:
    """ref fft_helper.py:11-57."""
    return _shift_nd(x, axes, +1, "fftshift_nd")


def ifftshift_nd(x: DistributedArray,
                 axes: Optional[Sequence[int]] = None) -> DistributedArray:
    """ref fft_helper.py:59-107."""
    return _shift_nd(x, axes, -1, "ifftshift_nd")
