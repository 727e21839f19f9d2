"""Execute the REFERENCE package (/root/reference/pylops_mpi) in this
container: P ranks as P threads over the mpi4py/pylops shims in
oracle/_refshim (TEST INFRASTRUCTURE — VERDICT r01 item 1).

Usage (golden generation and the ref-parity tests only):

    from oracle.refrun import reference_available, run_reference
    outs = run_reference(2, fn)     # fn(rank) -> value, per rank

``fn`` runs with ``pylops_mpi`` importable and ``MPI.COMM_WORLD``
resolving to the P-thread world.  Nothing in the product package
imports this module; /root/reference does not exist on the GPU box, so
GPU-side tests use the committed fixtures in tests/golden instead.
"""
import os
import sys
import threading

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_SHIM = os.path.join(_ROOT, "oracle", "_refshim")
REFERENCE_PATH = os.environ.get("PAM_REFERENCE_PATH", "/root/reference")


def reference_available() -> bool:
    return os.path.isdir(os.path.join(REFERENCE_PATH, "pylops_mpi"))


def _ensure_paths():
    for p in (_SHIM, REFERENCE_PATH):
        if p not in sys.path:
            sys.path.insert(0, p)
    # the reference targets python >= 3.11 (typing.Self); this image is
    # 3.10 — backfill from typing_extensions (runtime-only name)
    import typing
    if not hasattr(typing, "Self"):
        import typing_extensions
        typing.Self = typing_extensions.Self


_import_lock = threading.Lock()


def import_reference():
    """Import the reference pylops_mpi over the shims (idempotent)."""
    if not reference_available():
        raise RuntimeError(
            f"reference not available at {REFERENCE_PATH} (GPU boxes do "
            "not carry /root/reference — use the committed goldens)")
    _ensure_paths()
    with _import_lock:
        import mpi4py  # noqa: F401  (must resolve to the shim)
        assert "_refshim" in mpi4py.__file__, \
            f"unexpected real mpi4py at {mpi4py.__file__}"
        import pylops_mpi
        assert pylops_mpi.__file__.startswith(REFERENCE_PATH), \
            f"unexpected pylops_mpi at {pylops_mpi.__file__}"
        return pylops_mpi


def run_reference(P, fn):
    """Run ``fn(rank) -> value`` on P simulated ranks (threads); returns
    the list of per-rank values.  The first rank exception is re-raised
    (with every thread joined first)."""
    import_reference()
    from mpi4py import MPI as shim_mpi

    world = shim_mpi._World(P)
    outs = [None] * P
    errs = [None] * P

    def tmain(rank):
        shim_mpi._register_thread(world, rank)
        try:
            outs[rank] = fn(rank)
        except BaseException as e:  # noqa: BLE001 — reported to caller
            errs[rank] = e
        finally:
            shim_mpi._unregister_thread()

    threads = [threading.Thread(target=tmain, args=(r,), daemon=True)
               for r in range(P)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=300.0)
    first_err = next((e for e in errs if e is not None), None)
    for t in threads:
        if t.is_alive():
            # a raised rank leaves the others at a barrier — surface the
            # root cause, not the hang
            if first_err is not None:
                raise RuntimeError(
                    f"reference run hung after a rank raised: "
                    f"{first_err!r}") from first_err
            raise RuntimeError("reference run deadlocked (thread alive "
                               "after 300 s)")
    if first_err is not None:
        raise first_err
    return outs
