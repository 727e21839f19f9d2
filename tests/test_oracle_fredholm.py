"""Oracle pins for Fredholm1 / the MDC chain: dense-matrix adjoints, the
re-derived real-FFT convention (isometry + adjoint), and rank invariance.
Inputs follow the reference's test recipe (ref tests/test_fredholm.py:
36-95,120-123: arange-patterned G, float32/complex64 paths)."""
import numpy as np
import pytest
from numpy.testing import assert_allclose

from oracle import to_dist
from oracle.fredholm import (SimFredholm1, SimMDC, serial_rfft_adj,
                             serial_rfft_op)
from oracle.ranksim import Partition, SimArray

NSL, NX, NY, NZ = 12, 4, 6, 5


def bcast(v, P):
    return SimArray([v.copy() for _ in range(P)], v.shape,
                    partition=Partition.BROADCAST)


def make_G(P, dtype=np.complex64, nsl=NSL, nx=NX, ny=NY):
    # arange-patterned kernel as ref tests/test_fredholm.py:120-123
    G = (np.arange(nsl * nx * ny).reshape(nsl, nx, ny)
         .astype(np.float64) / (nsl * nx * ny))
    if np.issubdtype(dtype, np.complexfloating):
        G = (G + 1j * G[::-1]).astype(dtype)
    else:
        G = G.astype(dtype)
    splits = np.array_split(np.arange(nsl), P)
    return G, [G[s].copy() for s in splits]


@pytest.mark.parametrize("P", [1, 2, 3])
@pytest.mark.parametrize("dtype", [np.float64, np.complex128])
@pytest.mark.parametrize("saveGt", [True, False])
def test_fredholm_vs_dense(P, dtype, saveGt):
    G, blocks = make_G(P, dtype)
    op = SimFredholm1(blocks, nz=NZ, saveGt=saveGt)
    A = op.dense()
    rng = np.random.default_rng(8)
    x = rng.standard_normal(op.shape[1]).astype(dtype)
    y = rng.standard_normal(op.shape[0]).astype(dtype)
    if np.issubdtype(dtype, np.complexfloating):
        x = x + 1j * rng.standard_normal(op.shape[1]).astype(dtype)
        y = y + 1j * rng.standard_normal(op.shape[0]).astype(dtype)
    got = op.matvec(bcast(x, P)).locals[0]
    assert_allclose(got, A @ x, rtol=1e-11, atol=1e-12)
    gotr = op.rmatvec(bcast(y, P)).locals[0]
    assert_allclose(gotr, A.conj().T @ y, rtol=1e-11, atol=1e-12)


@pytest.mark.parametrize("nt", [16, 17, 1024])
@pytest.mark.parametrize("shift", [False, True])
def test_fft_convention(nt, shift):
    """The re-derived pylops real-FFT scaling: F^H F = I (isometry) and
    <Fx, z> == <x, F^H z> (adjoint)."""
    rng = np.random.default_rng(9)
    x = rng.standard_normal((nt, 3))
    y = serial_rfft_op(x, nt, shift)
    back = serial_rfft_adj(y, nt, shift)
    assert_allclose(back, x, rtol=1e-12, atol=1e-13)  # isometry round trip
    nfft = nt // 2 + 1
    z = (rng.standard_normal((nfft, 3))
         + 1j * rng.standard_normal((nfft, 3)))
    lhs = np.vdot(serial_rfft_op(x, nt, shift), z)
    rhs = np.vdot(x, serial_rfft_adj(z, nt, shift))
    assert_allclose(lhs.real, rhs.real, rtol=1e-11)


@pytest.mark.parametrize("P", [1, 2, 4])
def test_mdc_dottest_and_rank_invariance(P):
    nt, nv, nfreq = 20, 2, 8
    ns, nr = 3, 4
    nfft = nt // 2 + 1
    rng = np.random.default_rng(10)
    G = (rng.standard_normal((nfft, ns, nr))
         + 1j * rng.standard_normal((nfft, ns, nr))).astype(np.complex128)
    splits = np.array_split(np.arange(nfreq), P)
    blocks = [G[:nfreq][s].copy() for s in splits]
    op = SimMDC(blocks, nt, nv, nfreq, dt=0.4, dr=2.0, twosided=False)
    u = rng.standard_normal(op.shape[1])
    v = rng.standard_normal(op.shape[0])
    yy = np.vdot(op.matvec(bcast(u, P)).locals[0], v)
    xx = np.vdot(u, op.rmatvec(bcast(v, P)).locals[0])
    assert_allclose(yy.real, xx.real, rtol=1e-10)
    # rank invariance vs P=1
    op1 = SimMDC([G[:nfreq].copy()], nt, nv, nfreq, dt=0.4, dr=2.0,
                 twosided=False)
    assert_allclose(op.matvec(bcast(u, P)).locals[0],
                    op1.matvec(bcast(u, 1)).locals[0], rtol=1e-12)
    assert_allclose(op.rmatvec(bcast(v, P)).locals[0],
                    op1.rmatvec(bcast(v, 1)).locals[0], rtol=1e-12)
