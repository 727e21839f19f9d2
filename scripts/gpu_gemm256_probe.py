"""A/B: pam_gemm_kt (glds 256^2-tile, k-major A) vs pam_gemm (f32)."""
import os, sys, time
import numpy as np, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pylops_mpi_amd import _ffi
from pylops_mpi_amd.comm import init_default_comm

def timeit(fn, iters=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / iters

def main():
    init_default_comm(torch.device("cuda:0"))
    g = torch.Generator(device="cuda").manual_seed(9)
    s = torch.cuda.current_stream().cuda_stream
    for n in (4096, 8192, 16384):
        A = torch.rand((n, n), generator=g, dtype=torch.float32,
                       device="cuda") * 2 - 1
        B = torch.rand((n, n), generator=g, dtype=torch.float32,
                       device="cuda") * 2 - 1
        C = torch.empty((n, n), dtype=torch.float32, device="cuda")
        At = torch.empty((n, n), dtype=torch.float32, device="cuda")
        _ffi.checked(_ffi.lib().pam_transpose(
            s, A.data_ptr(), At.data_ptr(), n, n, 1), "t")
        fl = 2.0 * n * n * n

        def run_old():
            _ffi.checked(_ffi.lib().pam_gemm(
                s, A.data_ptr(), B.data_ptr(), C.data_ptr(), n, n, n,
                n, n, n, 0, 1), "gemm")

        def run_kt():
            _ffi.checked(_ffi.lib().pam_gemm_kt(
                s, At.data_ptr(), B.data_ptr(), C.data_ptr(), n, n, n,
                0, 1), "gemm_kt")

        t_old = timeit(run_old)
        # correctness: compare kt vs old on a checksum
        run_old()
        torch.cuda.synchronize()
        ref = C[::1037, ::911].clone()
        run_kt()
        torch.cuda.synchronize()
        got = C[::1037, ::911].clone()
        ok = torch.allclose(ref, got, rtol=2e-4, atol=1e-2)
        t_kt = timeit(run_kt)
        print(f"n={n}: pam_gemm {fl/t_old/1e12:6.1f} TF ({t_old*1e3:7.2f} ms)"
              f"  gemm_kt {fl/t_kt/1e12:6.1f} TF ({t_kt*1e3:7.2f} ms)"
              f"  match={ok}")

if __name__ == "__main__":
    main()
