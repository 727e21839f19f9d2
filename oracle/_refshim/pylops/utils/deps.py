"""pylops.utils.deps stub: cupy is absent, so cupy_import returns a
message (the reference gates its NCCL import on it being None,
ref Distributed.py:10-13, DistributedArray.py:14-17)."""
cupy_enabled = False


def cupy_import(message=None):
    return ("cupy not installed (pylops stub: CPU-only reference "
            "execution in-container)")
