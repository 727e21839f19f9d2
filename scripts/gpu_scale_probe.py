"""Judged-config-scale single-GPU measurements: Fredholm/MDC (cfg5),
SUMMA panel at large size (cfg4 local panel), BlockDiag GEMV (cfg3).
Prints one JSON line per measurement."""
import json
import os
import sys
import time

import numpy as np
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import pylops_mpi_amd as pm  # noqa: E402
from pylops_mpi_amd.comm import init_default_comm  # noqa: E402


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    init_default_comm(torch.device("cuda:0"))
    g = torch.Generator(device="cuda").manual_seed(42)

    # ---- cfg5: Fredholm1 at judged scale (1-GPU share of nf=513)
    nf, ns, nr, nv = 513, 64, 256, 256
    G = (torch.rand((nf, ns, nr), generator=g, device="cuda") - 0.5
         + 1j * (torch.rand((nf, ns, nr), generator=g, device="cuda") - 0.5)
         ).to(torch.complex64)
    Fr = pm.MPIFredholm1(G, nz=nv, saveGt=True, dtype="complex64")
    x = pm.DistributedArray((Fr.shape[1],), partition=pm.Partition.BROADCAST,
                            dtype=np.complex64)
    x[:] = (torch.rand(Fr.shape[1], generator=g, device="cuda") - 0.5
            ).to(torch.complex64)
    sec = timeit(lambda: Fr.matvec(x), iters=10)
    flops = 8.0 * nf * ns * nr * nv  # complex MAC = 8 real flops
    print(json.dumps({"probe": "fredholm_cfg5_matvec", "ms": sec * 1e3,
                      "TF_real": flops / sec / 1e12}), flush=True)
    y = Fr.matvec(x)
    sec = timeit(lambda: Fr.rmatvec(y), iters=10)
    print(json.dumps({"probe": "fredholm_cfg5_rmatvec", "ms": sec * 1e3,
                      "TF_real": flops / sec / 1e12}), flush=True)

    # ---- cfg5: full MDC chain
    nt, nfreq = 1024, nf
    mdc = pm.MPIMDC(G, nt, nv, nfreq, dt=0.004, dr=1.0, twosided=False)
    xm = pm.DistributedArray((mdc.shape[1],),
                             partition=pm.Partition.BROADCAST,
                             dtype=np.complex64)
    xm[:] = (torch.rand(mdc.shape[1], generator=g, device="cuda") - 0.5
             ).to(torch.complex64)
    sec = timeit(lambda: mdc.matvec(xm), iters=5)
    print(json.dumps({"probe": "mdc_cfg5_matvec", "ms": sec * 1e3}),
          flush=True)

    # ---- cfg4 local panel: SUMMA at P=1, 16384^2 fp32
    n = 16384
    A = (torch.rand((n, n), generator=g, device="cuda") * 2 - 1)
    X = (torch.rand((n, n), generator=g, device="cuda") * 2 - 1)
    op = pm.MPIMatrixMult(A, n, kind="summa", dtype="float32")
    xd = pm.DistributedArray((n * n,), dtype=np.float32)
    xd[:] = X.reshape(-1)
    sec = timeit(lambda: op.matvec(xd), iters=3, warmup=1)
    print(json.dumps({"probe": "summa_p1_16384_f32_matvec", "ms": sec * 1e3,
                      "TF": 2.0 * n ** 3 / sec / 1e12}), flush=True)

    # ---- cfg3: BlockDiag dense 4096^2 fp64 matvec+rmatvec pair
    Ad = torch.rand((4096, 4096), generator=g, dtype=torch.float64,
                    device="cuda")
    bd = pm.MPIBlockDiag([pm.DenseLocal(Ad)])
    xb = pm.DistributedArray((4096,))
    xb[:] = torch.rand(4096, generator=g, dtype=torch.float64, device="cuda")
    sec = timeit(lambda: bd.rmatvec(bd.matvec(xb)), iters=20)
    gbs = 2 * 4096 * 4096 * 8 / sec / 1e9
    print(json.dumps({"probe": "blockdiag_cfg3_pair", "ms": sec * 1e3,
                      "GB/s": gbs}), flush=True)


if __name__ == "__main__" and not os.environ.get("PAM_PROBE_EXTRA"):
    main()


def extra():
    """Mix-ceiling references + the full cfg2 cgls(50)."""
    from pylops_mpi_amd import _ffi
    init_default_comm(torch.device("cuda:0"))
    n = 1 << 29  # 4 GiB fp64 working set
    x = torch.rand(n, dtype=torch.float64, device="cuda")
    y = torch.rand(n, dtype=torch.float64, device="cuda")
    s = torch.cuda.current_stream().cuda_stream

    def axpy():
        _ffi.checked(_ffi.lib().pam_axpy(s, y.data_ptr(), x.data_ptr(),
                                         0.5, n, 0), "axpy")
    sec = timeit(axpy, iters=10)
    print(json.dumps({"probe": "axpy_f64_2R1W", "ms": sec * 1e3,
                      "GB/s": 24.0 * n / sec / 1e9}), flush=True)

    import pylops_mpi_amd as pm
    ws = torch.empty(2048, dtype=torch.float64, device="cuda")
    out = torch.empty(2, dtype=torch.float64, device="cuda")

    def dot():
        _ffi.checked(_ffi.lib().pam_dot(s, x.data_ptr(), y.data_ptr(), n,
                                        ws.data_ptr(), out.data_ptr(), 0),
                     "dot")
    sec = timeit(dot, iters=10)
    print(json.dumps({"probe": "dot_f64_2R", "ms": sec * 1e3,
                      "GB/s": 16.0 * n / sec / 1e9}), flush=True)
    del x, y
    torch.cuda.empty_cache()

    # cfg2: MPIFirstDerivative + cgls(niter=50) end to end (1 GPU)
    import numpy as np
    dims = (2048, 2048, 128)
    ntot = int(np.prod(dims))
    op = pm.MPIFirstDerivative(dims, kind="centered", order=3)
    g = torch.Generator(device="cuda").manual_seed(42)
    yd = pm.DistributedArray((ntot,))
    yd[:] = torch.randn(ntot, generator=g, dtype=torch.float64,
                        device="cuda")
    x0 = pm.DistributedArray((ntot,))
    x0[:] = 0.0
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    xs, istop, iit, r1, r2, cost = pm.cgls(op, yd, x0, niter=50, damp=0.1,
                                           tol=0.0)
    torch.cuda.synchronize()
    wall = time.perf_counter() - t0
    print(json.dumps({"probe": "cfg2_cgls50", "s": wall,
                      "iters": int(iit), "s_per_iter": wall / max(iit, 1)}),
          flush=True)


if __name__ == "__main__" and os.environ.get("PAM_PROBE_EXTRA"):
    extra()
