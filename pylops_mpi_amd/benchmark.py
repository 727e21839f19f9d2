"""benchmark/mark timing utility — drop-in for the reference's
pylops_mpi/utils/benchmark.py:25-173: a decorator that brackets the
wrapped function (and any user mark() points inside it) with a global
device sync + barrier and prints an indented call tree from rank 0.

Sync here = torch.cuda.synchronize (HIP) + an RCCL/gloo barrier; the gate
env var is BENCH_PYLOPS_AMD (reference: BENCH_PYLOPS_MPI, ref :25).
"""
__all__ = ["benchmark", "mark"]

import functools
import logging
import os
import time
from typing import Callable, List, Optional

ENABLE_BENCHMARK = int(os.getenv("BENCH_PYLOPS_AMD", 1)) == 1

_mark_func_stack: List[Callable] = []
_markers: List = []


def _parse_output_tree(markers):
    """ref :32-68 — nested calls indented one level per decorator depth."""
    global _markers
    output = []
    stack = []
    i = 0
    while i < len(markers):
        label, t, level = markers[i]
        if label.startswith("[decorator]"):
            indent = "\t" * (level - 1)
            output.append(f"{indent}{label}: total runtime: {t:6f} s\n")
        else:
            if stack:
                prev_label, prev_time, prev_level = stack[-1]
                if prev_level == level:
                    indent = "\t" * level
                    output.append(
                        f"{indent}{prev_label}-->{label}: "
                        f"{t - prev_time:6f} s\n")
                    stack.pop()
            if i + 1 <= len(markers) - 1:
                _, _, next_level = markers[i + 1]
                if next_level >= level:
                    stack.append(markers[i])
        i += 1
    _markers = []
    return output


def _sync():
    """Device sync + global barrier (ref :70-73)."""
    import torch

    from .comm import get_default_comm
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    get_default_comm().barrier()


def mark(label: str):
    """ref :76-89 — end the previous region, begin a new one."""
    if not ENABLE_BENCHMARK:
        return
    if not _mark_func_stack:
        raise RuntimeError("mark() called outside of a benchmarked region")
    _mark_func_stack[-1](label)


def benchmark(func: Optional[Callable] = None,
              description: Optional[str] = "",
              logger: Optional[logging.Logger] = None):
    """ref :92-173."""

    def noop_decorator(f):
        @functools.wraps(f)
        def wrapped(*args, **kwargs):
            return f(*args, **kwargs)
        return wrapped

    def decorator(f):
        @functools.wraps(f)
        def wrapper(*args, **kwargs):
            from .comm import get_default_comm
            rank = get_default_comm().rank
            level = len(_mark_func_stack) + 1
            _markers.append(
                (f"[decorator]{description or f.__name__}", None, level))
            header_index = len(_markers) - 1

            def local_mark(label):
                _sync()
                _markers.append((label, time.perf_counter(), level))

            _mark_func_stack.append(local_mark)
            _sync()
            start_time = time.perf_counter()
            result = f(*args, **kwargs)
            _sync()
            elapsed = time.perf_counter() - start_time
            _markers[header_index] = (
                f"[decorator]{description or f.__name__}", elapsed, level)
            _mark_func_stack.pop()
            if not _mark_func_stack:
                if rank == 0:
                    output = _parse_output_tree(_markers)
                    if logger:
                        logger.info("".join(output))
                    else:
                        print("".join(output))
            return result
        return wrapper

    if not ENABLE_BENCHMARK:
        return noop_decorator if func is None else noop_decorator(func)
    return decorator if func is None else decorator(func)
