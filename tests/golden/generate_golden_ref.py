"""Generate golden fixtures FROM THE REFERENCE ITSELF.

Runs /root/reference/pylops_mpi in this container (P ranks as threads
over the oracle/_refshim mpi4py + pylops stubs) across the case table
in refgen.py and commits the outputs as tests/golden/golden_ref.npz.
This pins parity at the reference boundary: the
oracle is checked against these fixtures everywhere (including the GPU
box, which has no /root/reference), and tests/test_ref_parity.py
re-derives them live wherever the reference is present.

Run from the repo root:  python tests/golden/generate_golden_ref.py
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import refgen  # noqa: E402


def main():
    data = refgen.compute_reference()
    np.savez_compressed(refgen.GOLDEN_PATH, **data)
    sz = os.path.getsize(refgen.GOLDEN_PATH)
    print(f"wrote {refgen.GOLDEN_PATH}: {len(data)} arrays, {sz} bytes")


if __name__ == "__main__":
    main()
