"""Kernel perf probe (run on the GPU box): GEMM TF/s, GEMV GB/s, stencil
variants.  Prints one JSON line per measurement."""
import json
import os
import sys
import time

import numpy as np
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

from pylops_mpi_amd import _ffi  # noqa: E402
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


def bench_gemm(M, K, N, tdt, label):
    # random data (guide §5.4 rule 25: never zero-filled operands)
    A = torch.rand((M, K), dtype=tdt, device="cuda") * 2 - 1
    B = torch.rand((K, N), dtype=tdt, device="cuda") * 2 - 1
    C = torch.empty((M, N), dtype=tdt, device="cuda")
    s = torch.cuda.current_stream().cuda_stream
    dt = _ffi.dtype_code(tdt)

    def run():
        _ffi.checked(_ffi.lib().pam_gemm(
            s, A.data_ptr(), B.data_ptr(), C.data_ptr(), M, N, K, K, N, N,
            0, dt), "gemm")

    sec = timeit(run)
    tf = 2.0 * M * N * K / sec / 1e12
    print(json.dumps({"probe": label, "M": M, "K": K, "N": N,
                      "ms": sec * 1e3, "TF": tf}), flush=True)


def bench_gemv(n, label):
    A = torch.rand((n, n), dtype=torch.float64, device="cuda")
    x = torch.rand(n, dtype=torch.float64, device="cuda")
    y = torch.empty(n, dtype=torch.float64, device="cuda")
    ws = torch.empty(int(_ffi.lib().pam_gemv_ws_elems(n, n)),
                     dtype=torch.float64, device="cuda")
    s = torch.cuda.current_stream().cuda_stream
    for trans in (0, 1):
        def run(tr=trans):
            _ffi.checked(_ffi.lib().pam_gemv(
                s, tr, A.data_ptr(), x.data_ptr(), y.data_ptr(), n, n,
                ws.data_ptr(), 0), "gemv")
        sec = timeit(run, iters=20)
        gbs = 8.0 * n * n / sec / 1e9
        print(json.dumps({"probe": f"{label}_t{trans}", "n": n,
                          "ms": sec * 1e3, "GB/s": gbs}), flush=True)


def main():
    init_default_comm(torch.device("cuda:0"))
    bench_gemm(4096, 4096, 4096, torch.float32, "gemm_f32")
    bench_gemm(8192, 8192, 8192, torch.float32, "gemm_f32")
    bench_gemm(4096, 4096, 4096, torch.float64, "gemm_f64")
    bench_gemm(8192, 8192, 8192, torch.float64, "gemm_f64")
    bench_gemv(4096, "gemv_f64")
    bench_gemv(8192, "gemv_f64")


if __name__ == "__main__":
    main()
