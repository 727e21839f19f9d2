"""Emulate the N>1 halo-overlap launch split on one GPU: the stencil at
the per-rank block runs as interior rows + two boundary bands
(derivative.py overlap path).  Times the 3-launch split vs the single
launch at the (512, 4096, 256) fp64 per-rank shape so the first real
8-GPU run executes a pre-tuned path."""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pylops_mpi_amd import _ffi  # noqa: E402
from pylops_mpi_amd.comm import init_default_comm  # noqa: E402


def main():
    init_default_comm(torch.device("cuda:0"))
    dims = tuple(int(v) for v in
                 os.environ.get("DIMS", "512x4096x256").split("x"))
    nloc, m = dims[0], int(np.prod(dims[1:]))
    g = torch.Generator(device="cuda").manual_seed(3)
    x = torch.randn((nloc, m), generator=g, dtype=torch.float64,
                    device="cuda")
    gf = torch.randn((1, m), generator=g, dtype=torch.float64,
                     device="cuda")
    gb = torch.randn((1, m), generator=g, dtype=torch.float64,
                     device="cuda")
    y = torch.empty_like(x)
    s = torch.cuda.current_stream().cuda_stream
    op, w = 4, 1  # centered3 matvec

    def apply_range(r0, r1, use_halo):
        _ffi.checked(_ffi.lib().pam_fd_apply(
            s, op, 0, x.data_ptr(),
            gf.data_ptr() if use_halo else None,
            gb.data_ptr() if use_halo else None,
            y.data_ptr(), nloc, m, nloc, 8 * nloc, r0, r1, 1.0, 0),
            "fd")

    def run_single():
        apply_range(0, nloc, True)

    def run_split():
        apply_range(w, nloc - w, False)   # interior
        apply_range(0, w, True)           # boundary bands
        apply_range(nloc - w, nloc, True)

    for fn, name in ((run_single, "single"), (run_split, "split3")):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t = time.perf_counter()
        for _ in range(20):
            fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t) / 20
        print(f"{name}: {dt*1e3:7.3f} ms  "
              f"{16 * nloc * m / dt / 1e12:5.2f} TB/s")


if __name__ == "__main__":
    main()
