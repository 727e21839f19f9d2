"""ctypes binding of the pam C-ABI (include/pam.h -> libpam.so).

The product compute path goes EXCLUSIVELY through this library; there is no
CPU fallback.  If the extension is missing or fails to load, every compute
op raises loudly.
"""
import ctypes
import os

_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                         "libpam.so")
_lib = None
_load_error = None

F64, F32, C128, C64 = 0, 1, 2, 3  # PAM_* dtype codes

_SIGS = {
    "pam_version": ([], ctypes.c_int64),
    "pam_reduce_ws_elems": ([], ctypes.c_int64),
    "pam_fill": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
                  ctypes.c_double, ctypes.c_int], ctypes.c_int),
    "pam_neg": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                 ctypes.c_int64, ctypes.c_int], ctypes.c_int),
    "pam_add": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                 ctypes.c_void_p, ctypes.c_int64, ctypes.c_int], ctypes.c_int),
    "pam_sub": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                 ctypes.c_void_p, ctypes.c_int64, ctypes.c_int], ctypes.c_int),
    "pam_mul": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                 ctypes.c_void_p, ctypes.c_int64, ctypes.c_int], ctypes.c_int),
    "pam_scale": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                   ctypes.c_double, ctypes.c_int64, ctypes.c_int],
                  ctypes.c_int),
    "pam_axpy": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                  ctypes.c_double, ctypes.c_int64, ctypes.c_int],
                 ctypes.c_int),
    "pam_xpby": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                  ctypes.c_double, ctypes.c_int64, ctypes.c_int],
                 ctypes.c_int),
    "pam_axpy_d": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                    ctypes.c_void_p, ctypes.c_double, ctypes.c_int64,
                    ctypes.c_int],
                   ctypes.c_int),
    "pam_xpby_d": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                    ctypes.c_void_p, ctypes.c_double, ctypes.c_int64,
                    ctypes.c_int],
                   ctypes.c_int),
    "pam_scalar_alpha": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                          ctypes.c_void_p, ctypes.c_double],
                         ctypes.c_int),
    "pam_scalar_div": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                        ctypes.c_void_p],
                       ctypes.c_int),
    "pam_dot": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                 ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
                 ctypes.c_int], ctypes.c_int),
    "pam_cmul": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                  ctypes.c_void_p, ctypes.c_int64, ctypes.c_int],
                 ctypes.c_int),
    "pam_cscale": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                    ctypes.c_double, ctypes.c_double, ctypes.c_int64,
                    ctypes.c_int], ctypes.c_int),
    "pam_conj": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                  ctypes.c_int64, ctypes.c_int], ctypes.c_int),
    "pam_thresh": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                    ctypes.c_int64, ctypes.c_int, ctypes.c_double,
                    ctypes.c_int], ctypes.c_int),
    "pam_cdot": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                  ctypes.c_int64, ctypes.c_int, ctypes.c_void_p,
                  ctypes.c_void_p, ctypes.c_int], ctypes.c_int),
    "pam_cgemm_batched": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                           ctypes.c_void_p, ctypes.c_int64, ctypes.c_int64,
                           ctypes.c_int64, ctypes.c_int64, ctypes.c_int64,
                           ctypes.c_int64, ctypes.c_int64, ctypes.c_int,
                           ctypes.c_int, ctypes.c_int], ctypes.c_int),
    "pam_ctranspose": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                        ctypes.c_int64, ctypes.c_int64, ctypes.c_int,
                        ctypes.c_int], ctypes.c_int),
    "pam_rfft_strided": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                          ctypes.c_int64, ctypes.c_int64, ctypes.c_int],
                         ctypes.c_int),
    "pam_irfft_strided": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                           ctypes.c_int64, ctypes.c_int64, ctypes.c_int],
                          ctypes.c_int),
    "pam_rfft_contig": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                         ctypes.c_int64, ctypes.c_int64, ctypes.c_int],
                        ctypes.c_int),
    "pam_irfft_contig": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                          ctypes.c_int64, ctypes.c_int64, ctypes.c_int],
                         ctypes.c_int),
    "pam_unzip_t": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                     ctypes.c_int64, ctypes.c_int64, ctypes.c_int],
                    ctypes.c_int),
    "pam_zip_t": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                   ctypes.c_int64, ctypes.c_int64, ctypes.c_int],
                  ctypes.c_int),
    "pam_unzip": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                   ctypes.c_int64, ctypes.c_int], ctypes.c_int),
    "pam_zip": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                 ctypes.c_int64, ctypes.c_int], ctypes.c_int),
    "pam_gemm_batched": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                          ctypes.c_void_p, ctypes.c_int64, ctypes.c_int64,
                          ctypes.c_int64, ctypes.c_int64, ctypes.c_int64,
                          ctypes.c_int64, ctypes.c_int64, ctypes.c_int,
                          ctypes.c_int, ctypes.c_int], ctypes.c_int),
    "pam_norm_local": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
                        ctypes.c_int, ctypes.c_double, ctypes.c_void_p,
                        ctypes.c_void_p, ctypes.c_int], ctypes.c_int),
    "pam_gemv_ws_elems": ([ctypes.c_int64, ctypes.c_int64], ctypes.c_int64),
    "pam_gemv": ([ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
                  ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
                  ctypes.c_int64, ctypes.c_void_p, ctypes.c_int],
                 ctypes.c_int),
    "pam_gemm": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                  ctypes.c_void_p, ctypes.c_int64, ctypes.c_int64,
                  ctypes.c_int64, ctypes.c_int64, ctypes.c_int64,
                  ctypes.c_int64, ctypes.c_int, ctypes.c_int], ctypes.c_int),
    "pam_gemm_kt": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                     ctypes.c_void_p, ctypes.c_int64, ctypes.c_int64,
                     ctypes.c_int64, ctypes.c_int, ctypes.c_int],
                    ctypes.c_int),
    "pam_transpose": ([ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                       ctypes.c_int64, ctypes.c_int64, ctypes.c_int],
                      ctypes.c_int),
    "pam_fd_halo_width": ([ctypes.c_int], ctypes.c_int64),
    "pam_nsconv": ([ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
                    ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
                    ctypes.c_int64, ctypes.c_int64, ctypes.c_int64,
                    ctypes.c_int64, ctypes.c_double, ctypes.c_double,
                    ctypes.c_int], ctypes.c_int),
    "pam_fd_serial": ([ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
                       ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
                       ctypes.c_int64, ctypes.c_int64, ctypes.c_double,
                       ctypes.c_int], ctypes.c_int),
    "pam_fd_apply": ([ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
                      ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                      ctypes.c_void_p, ctypes.c_int64, ctypes.c_int64,
                      ctypes.c_int64, ctypes.c_int64, ctypes.c_int64,
                      ctypes.c_int64, ctypes.c_double, ctypes.c_int],
                     ctypes.c_int),
}


def _load():
    global _lib, _load_error
    if _lib is not None:
        return _lib
    if _load_error is not None:
        raise _load_error
    try:
        lib = ctypes.CDLL(_LIB_PATH)
        for name, (argtypes, restype) in _SIGS.items():
            fn = getattr(lib, name)
            fn.argtypes = argtypes
            fn.restype = restype
        ver = lib.pam_version()
        if ver != 1:
            raise RuntimeError(f"pam ABI version mismatch: {ver}")
    except OSError as e:
        _load_error = ImportError(
            f"pam HIP extension not found/loadable at {_LIB_PATH} "
            f"(build it with __graft_entry__.build()): {e}")
        raise _load_error
    _lib = lib
    return lib


def lib():
    """The loaded C-ABI; raises ImportError if libpam.so is missing."""
    return _load()


def available() -> bool:
    try:
        _load()
        return True
    except ImportError:
        return False


def checked(rc: int, what: str) -> None:
    if rc != 0:
        raise RuntimeError(f"pam: {what} failed with code {rc}")


def dtype_code(torch_dtype) -> int:
    import torch
    if torch_dtype == torch.float64:
        return F64
    if torch_dtype == torch.float32:
        return F32
    if torch_dtype == torch.complex128:
        return C128
    if torch_dtype == torch.complex64:
        return C64
    raise TypeError(f"pam: unsupported dtype {torch_dtype}")
