"""The ``@reshaped`` input rebalance (ref utils/decorators.py:44-82).

Moves a 1-D SCATTER DistributedArray to an arbitrary per-rank element
split via the reference's cumulative-imbalance ghost arithmetic
(ref decorators.py:66-77 + DistributedArray.py:976-1031), but with all
ranks' counts known locally (no object allgathers) and a zero-copy fast
path when the split already matches.
"""
from typing import List

import numpy as np
import torch


def rebalance_1d(x, counts: List[int]) -> torch.Tensor:
    """Return this rank's slice of the re-split flat vector (a view when
    no movement is needed)."""
    P, r = x.size, x.rank
    x_counts = [int(np.prod(s)) for s in x.local_shapes]
    if list(x_counts) == list(counts):
        return x.local_array.reshape(-1)
    dif = np.cumsum(np.asarray(counts) - np.asarray(x_counts))
    cf = [int(abs(min(0, dif[q - 1]))) for q in range(P)]
    cb = [int(max(0, dif[q])) for q in range(P)]
    t = x.local_array.reshape(-1)
    # rank r sends its last cf[r+1] elements to r+1 and its first
    # cb[r-1] elements to r-1
    if r < P - 1 and cf[r + 1] > t.numel():
        raise ValueError(
            f"Local Shape at rank={r} along axis=0 should be > {cf[r + 1]}")
    if r > 0 and cb[r - 1] > t.numel():
        raise ValueError(
            f"Local Shape at rank={r} along axis=0 should be > {cb[r - 1]}")
    send_next = t[-cf[r + 1]:].contiguous() \
        if r < P - 1 and cf[r + 1] > 0 else None
    send_prev = t[: cb[r - 1]].contiguous() \
        if r > 0 and cb[r - 1] > 0 else None
    recv_front = torch.empty(cf[r], dtype=t.dtype, device=t.device) \
        if r > 0 and cf[r] > 0 else None
    recv_back = torch.empty(cb[r], dtype=t.dtype, device=t.device) \
        if r < P - 1 and cb[r] > 0 else None
    x.base_comm.sendrecv_neighbors(send_prev, send_next,
                                   recv_front, recv_back)
    parts = [p for p in (recv_front, t, recv_back) if p is not None]
    ghosted = torch.cat(parts) if len(parts) > 1 else t
    index = int(max(0, dif[r - 1]))
    return ghosted[index: index + counts[r]]
