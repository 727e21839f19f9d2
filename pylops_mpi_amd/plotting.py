"""Plotting helpers for DistributedArray (ref plotting/plotting.py:13-75).

Pure visualization — the tensors are gathered and moved to host NumPy
before matplotlib sees them; nothing here touches the compute path.
"""
from typing import Any, Optional

import numpy as np
import torch

from .distributedarray import DistributedArray, Partition


def plot_distributed_array(arr: DistributedArray) -> None:
    """ref plotting.py:13-43 — color the global array by owning rank."""
    from matplotlib import pyplot as plt

    if not isinstance(arr, DistributedArray):
        raise TypeError("Not a DistributedArray")
    if arr.partition is Partition.BROADCAST:
        raise NotImplementedError("Use Scatter for plot")
    dist_array = DistributedArray(global_shape=arr.global_shape,
                                  base_comm=arr.base_comm,
                                  partition=arr.partition, axis=arr.axis,
                                  local_shapes=arr.local_shapes,
                                  dtype=arr.dtype)
    dist_array[:] = torch.full(dist_array.local_shape, float(arr.rank),
                               dtype=dist_array.local_array.dtype,
                               device=dist_array.local_array.device)
    full_dist_arr = dist_array.asarray().cpu().numpy()
    full_arr = arr.asarray().cpu().numpy()
    if arr.rank == 0:
        figure, (ax1, ax2) = plt.subplots(nrows=1, ncols=2, figsize=(18, 5))
        ax1.matshow(np.real(full_arr), cmap="rainbow")
        ax1.set_title("Original Array")
        im2 = ax2.matshow(np.real(full_dist_arr), cmap="rainbow")
        ax2.set_title(f"Distributed over axis {arr.axis}")
        cbar = figure.colorbar(im2)
        cbar.set_ticks(np.arange(arr.size))
        cbar.set_label("Ranks")
        plt.tight_layout()


def plot_local_arrays(arr: DistributedArray, title: Optional[str] = None,
                      vmin: Optional[Any] = None,
                      vmax: Optional[Any] = None) -> None:
    """ref plotting.py:47-75 — one panel per rank's local block."""
    from matplotlib import pyplot as plt

    shapes = [tuple(s) for s in arr.local_shapes]
    gathered = arr.base_comm.allgather_tensors(
        arr.local_array.contiguous(), shapes)
    if arr.rank == 0:
        figure, ax = plt.subplots(nrows=1, ncols=arr.size, figsize=(18, 5))
        ax = [ax] if arr.size == 1 else ax
        for i in range(arr.size):
            loc = gathered[i].reshape(shapes[i]).cpu().numpy()
            ax[i].imshow(np.real(loc), cmap="rainbow", vmin=vmin, vmax=vmax)
            ax[i].set_xticks(np.arange(loc.shape[1]))
            ax[i].set_yticks(np.arange(loc.shape[0]))
            ax[i].set_title(f"Rank-{i}")
        plt.suptitle(title)
        plt.tight_layout()
