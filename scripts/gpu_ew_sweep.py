"""A/B the 1-D grid cap for axpy/dot at the bench size."""
import os, sys, time
import numpy as np, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pylops_mpi_amd as pm
from pylops_mpi_amd import _ffi
from pylops_mpi_amd.comm import init_default_comm

def main():
    init_default_comm(torch.device("cuda:0"))
    n = 2048 * 2048 * 128
    g = torch.Generator(device="cuda").manual_seed(0)
    x = torch.randn(n, generator=g, dtype=torch.float64, device="cuda")
    y = torch.randn(n, generator=g, dtype=torch.float64, device="cuda")
    ws = torch.empty(2048, dtype=torch.float64, device="cuda")
    out = torch.empty(2, dtype=torch.float64, device="cuda")
    s = torch.cuda.current_stream().cuda_stream
    def axpy():
        _ffi.checked(_ffi.lib().pam_axpy(s, y.data_ptr(), x.data_ptr(), 0.5, n, 0), "a")
    def dot():
        _ffi.checked(_ffi.lib().pam_dot(s, x.data_ptr(), y.data_ptr(), n, ws.data_ptr(), out.data_ptr(), 0), "d")
    for name, fn, byt in (("axpy", axpy, 24 * n), ("dot", dot, 16 * n)):
        for _ in range(5): fn()
        torch.cuda.synchronize()
        t = time.perf_counter()
        for _ in range(30): fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t) / 30
        print(f"{name}: {dt*1e3:7.3f} ms  {byt/dt/1e12:6.3f} TB/s")

main()
