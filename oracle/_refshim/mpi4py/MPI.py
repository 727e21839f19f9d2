"""Threaded MPI rendezvous: the mpi4py surface the reference package
uses, with P ranks as P threads of this process (TEST INFRASTRUCTURE).

Semantics restated from mpi4py's documented behaviour at the
reference's call sites only:
  * object collectives (``allgather``/``allreduce``/``bcast``/``gather``)
    pickle — emulated with ``copy.deepcopy`` at deposit;
  * buffered collectives (``Allgather``/``Allgatherv``/``Allreduce``/
    ``Bcast``) operate on numpy buffers (the reference always passes
    contiguous arrays or ``[buf, count, datatype]`` specs,
    ref utils/_mpi.py:21-276);
  * ``Send``/``Recv``/``Sendrecv`` are tag-matched FIFO queues per
    (src, dst, tag) — buffered-send semantics (never blocks), which is
    what mpi4py gives for the small messages the reference ships;
  * ``Split`` is comm-collective (ref DistributedArray.py:100,
    MatrixMult.py:305-306); ``Create_group`` is group-members-only
    collective (ref MatrixMult.py:73-74 — inactive ranks return before
    calling it);
  * reductions fold in rank order (deterministic; within the 1e-14
    comparison tolerances any MPI reduce-order difference is
    invisible).
"""
import copy
import queue
import threading

import numpy as np

_RECV_TIMEOUT = 120.0  # seconds: a hang in the shim fails the test


# --------------------------------------------------------------- tokens
class Op:
    def __init__(self, name):
        self.name = name

    def __repr__(self):  # pragma: no cover
        return f"MPI.{self.name}"


SUM = Op("SUM")
PROD = Op("PROD")
MAX = Op("MAX")
MIN = Op("MIN")
LAND = Op("LAND")
BOR = Op("BOR")

_BINOPS = {
    "SUM": lambda a, b: a + b,
    "PROD": lambda a, b: a * b,
    "MAX": lambda a, b: np.maximum(a, b) if isinstance(
        a, np.ndarray) else max(a, b),
    "MIN": lambda a, b: np.minimum(a, b) if isinstance(
        a, np.ndarray) else min(a, b),
    "LAND": lambda a, b: bool(a) and bool(b),
    "BOR": lambda a, b: a | b,
}


class Datatype:
    def __init__(self, char):
        self.char = char


class _TypeDict(dict):
    def __missing__(self, key):
        d = Datatype(key)
        self[key] = d
        return d


_typedict = _TypeDict()

PROC_NULL = -2
UNDEFINED = -3
IN_PLACE = object()
COMM_TYPE_SHARED = object()
COMM_NULL = None  # reference never touches a null comm's methods


class Group:
    def __init__(self, world_ranks):
        self._ranks = list(world_ranks)

    def Incl(self, ranks):
        return Group([self._ranks[r] for r in ranks])

    def Get_size(self):
        return len(self._ranks)

    @staticmethod
    def Translate_ranks(g1, ranks, g2):
        return [g2._ranks.index(g1._ranks[r]) for r in ranks]


# --------------------------------------------------------------- world
class _World:
    """One running reference 'job': size threads, ident -> world rank."""

    def __init__(self, size):
        self.size = size
        self.ident2rank = {}
        self.lock = threading.Lock()
        self.group_registry = {}   # members-only rendezvous state
        self.comm = Comm(self, list(range(size)))


_worlds = {}          # thread ident -> _World
_worlds_lock = threading.Lock()


def _register_thread(world, rank):
    ident = threading.get_ident()
    with _worlds_lock:
        _worlds[ident] = world
    world.ident2rank[ident] = rank


def _unregister_thread():
    ident = threading.get_ident()
    with _worlds_lock:
        _worlds.pop(ident, None)


def _current_world():
    try:
        return _worlds[threading.get_ident()]
    except KeyError:
        raise RuntimeError(
            "mpi4py shim: calling thread is not part of a reference run "
            "(use oracle.refrun.run_reference)") from None


def _buf_of(spec):
    """mpi4py buffer spec -> ndarray (the reference passes either a bare
    array or [buf, count, datatype], ref utils/_mpi.py:184,233)."""
    if isinstance(spec, (list, tuple)):
        return np.asarray(spec[0])
    return np.asarray(spec)


def _mem_flat(a):
    """MEMORY-order 1-D view of a contiguous buffer.

    Real mpi4py transmits the raw bytes of the (single-segment) buffer,
    not its logical C-order: the reference's SUMMA adjoint sends
    ``A.T.conj()`` — an F-contiguous array — into an ``empty_like``
    (also F-ordered) receive buffer (ref MatrixMult.py:738,756-760),
    and the two layout flips cancel.  Emulating logical order instead
    silently transposes the tiles (r02 shim bug, caught by the SUMMA
    adjoint parity case)."""
    a = np.asarray(a)
    if a.flags.c_contiguous or a.flags.f_contiguous:
        return a.reshape(-1, order="A")
    raise BufferError("mpi4py shim: buffer is not contiguous "
                      "(real mpi4py would refuse it too)")


# --------------------------------------------------------------- comm
class Comm:
    def __init__(self, world, members):
        self._world = world
        self._members = list(members)           # world ranks, comm order
        self._local_of = {wr: i for i, wr in enumerate(members)}
        n = len(members)
        self._bar = threading.Barrier(n)
        self._slots = [None] * n
        self._qs = {}
        self._qlock = threading.Lock()

    # ------------------------------------------------------------ ranks
    def _r(self):
        return self._local_of[self._world.ident2rank[threading.get_ident()]]

    def Get_rank(self):
        return self._r()

    def Get_size(self):
        return len(self._members)

    def Barrier(self):
        self._bar.wait()

    barrier = Barrier

    def Get_group(self):
        return Group(list(self._members))

    # ------------------------------------------------- collective core
    def _xchg(self, value):
        """Deposit value (already a private copy), barrier, read all,
        barrier (slot reuse safety).  All ranks call collectives in the
        same program order — an MPI requirement the reference upholds."""
        r = self._r()
        self._slots[r] = value
        self._bar.wait()
        vals = list(self._slots)
        self._bar.wait()
        return vals

    # --------------------------------------------------- object layer
    def allgather(self, sendobj):
        return self._xchg(copy.deepcopy(sendobj))

    def gather(self, sendobj, root=0):
        vals = self._xchg(copy.deepcopy(sendobj))
        return vals if self._r() == root else None

    def allreduce(self, sendobj, op=SUM):
        vals = self._xchg(copy.deepcopy(sendobj))
        f = _BINOPS[op.name]
        out = vals[0]
        for v in vals[1:]:
            out = f(out, v)
        return out

    def bcast(self, obj, root=0):
        vals = self._xchg(copy.deepcopy(obj) if self._r() == root else None)
        return vals[root] if self._r() != root else obj

    # -------------------------------------------------- buffered layer
    def Allgather(self, sendbuf, recvbuf):
        send = _buf_of(sendbuf)
        vals = self._xchg(_mem_flat(send).copy())
        recv = _buf_of(recvbuf)
        flat = np.concatenate(vals)
        _mem_flat(recv)[: flat.size] = flat

    def Allgatherv(self, sendbuf, recvspec):
        send = _buf_of(sendbuf)
        vals = self._xchg(_mem_flat(send).copy())
        recv = np.asarray(recvspec[0])
        counts = list(recvspec[1])
        displs = list(recvspec[2]) if recvspec[2] is not None else None
        if displs is None:
            displs = list(np.cumsum([0] + counts[:-1]))
        r = _mem_flat(recv)
        for i, v in enumerate(vals):
            r[displs[i]: displs[i] + counts[i]] = v[: counts[i]]

    def Allreduce(self, sendbuf, recvbuf, op=SUM):
        send = _buf_of(sendbuf)
        vals = self._xchg(_mem_flat(send).copy())
        f = _BINOPS[op.name]
        out = vals[0]
        for v in vals[1:]:
            out = f(out, v)
        recv = _buf_of(recvbuf)
        _mem_flat(recv)[:] = np.asarray(out)

    def Bcast(self, buf, root=0):
        b = _buf_of(buf)
        vals = self._xchg(_mem_flat(b).copy() if self._r() == root
                          else None)
        if self._r() != root:
            _mem_flat(b)[:] = vals[root]

    # ------------------------------------------------------------- p2p
    def _q(self, src, dst, tag):
        key = (src, dst, tag)
        with self._qlock:
            q = self._qs.get(key)
            if q is None:
                q = self._qs[key] = queue.Queue()
            return q

    def Send(self, sendspec, dest=0, tag=0):
        if dest == PROC_NULL:
            return  # MPI: communication with PROC_NULL is a no-op
        send = _buf_of(sendspec)
        self._q(self._r(), dest, tag).put(_mem_flat(send).copy())

    send = Send  # object send at the reference's sites is also an array

    def Recv(self, recvspec, source=0, tag=0):
        if source == PROC_NULL:
            return  # no-op; recv buffer untouched
        data = self._q(source, self._r(), tag).get(timeout=_RECV_TIMEOUT)
        recv = _buf_of(recvspec)
        _mem_flat(recv)[: data.size] = data.reshape(-1)

    def recv(self, source=0, tag=0):
        if source == PROC_NULL:
            return None
        return self._q(source, self._r(), tag).get(timeout=_RECV_TIMEOUT)

    def Sendrecv(self, sendbuf=None, dest=0, sendtag=0, recvbuf=None,
                 source=0, recvtag=0):
        if dest != PROC_NULL:
            send = _buf_of(sendbuf)
            self._q(self._r(), dest, sendtag).put(_mem_flat(send).copy())
        if source != PROC_NULL:
            data = self._q(source, self._r(), recvtag).get(
                timeout=_RECV_TIMEOUT)
            recv = _buf_of(recvbuf)
            _mem_flat(recv)[: data.size] = data.reshape(-1)

    def sendrecv(self, sendobj=None, dest=0, sendtag=0, source=0,
                 recvtag=0):
        if dest != PROC_NULL:
            self._q(self._r(), dest, sendtag).put(copy.deepcopy(sendobj))
        if source == PROC_NULL:
            return None  # MPI: PROC_NULL recv completes with no data
        return self._q(source, self._r(), recvtag).get(
            timeout=_RECV_TIMEOUT)

    # ------------------------------------------------- comm management
    def Split(self, color=0, key=0):
        r = self._r()
        vals = self._xchg((color, key, r))
        mine = sorted((k, lr) for (c, k, lr) in vals if c == color)
        members_local = [lr for _, lr in mine]
        members_world = [self._members[lr] for lr in members_local]
        leader = members_local[0]
        newcomm = Comm(self._world, members_world) if r == leader else None
        slots = self._xchg(newcomm)
        return slots[leader]

    def Create_group(self, group, tag=0):
        """Members-only collective (MPI_Comm_create_group): the
        reference's inactive ranks return before calling this
        (ref MatrixMult.py:67-74)."""
        wranks = tuple(group._ranks)
        me = self._world.ident2rank[threading.get_ident()]
        if me not in wranks:
            return COMM_NULL
        w = self._world
        with w.lock:
            st = w.group_registry.get(wranks)
            if st is None:
                # one comm per member set, reused on repeat calls
                # (threading.Barrier resets after each full trip)
                st = {"bar": threading.Barrier(len(wranks)),
                      "comm": Comm(w, list(wranks))}
                w.group_registry[wranks] = st
        st["bar"].wait()
        return st["comm"]

    def Split_type(self, split_type, key=0):  # COMM_TYPE_SHARED: one node
        return self.Split(color=0, key=key)

    def Create_cart(self, dims, periods=None, reorder=False):
        """Cartesian topology (ref basicoperators/Halo.py:53,231-241):
        identity rank order (a valid 'reorder' outcome), row-major
        coords, PROC_NULL beyond non-periodic edges."""
        r = self._r()
        newc = CartComm(self, dims, periods) if r == 0 else None
        slots = self._xchg(newc)
        return slots[0]

    def Free(self):
        pass

    def Dup(self):
        return self


class CartComm(Comm):
    def __init__(self, parent, dims, periods):
        super().__init__(parent._world, list(parent._members))
        self._dims = [int(d) for d in dims]
        self._periods = list(periods) if periods is not None \
            else [False] * len(self._dims)

    def Get_coords(self, rank):
        coords = []
        for d in reversed(self._dims):
            coords.append(rank % d)
            rank //= d
        return list(reversed(coords))

    def _ravel(self, coords):
        r = 0
        for d, c in zip(self._dims, coords):
            r = r * d + c
        return r

    def Shift(self, direction, disp=1):
        me = self._r()
        coords = self.Get_coords(me)

        def nbr(sign):
            c = list(coords)
            c[direction] += sign * disp
            if 0 <= c[direction] < self._dims[direction]:
                return self._ravel(c)
            if self._periods[direction]:
                c[direction] %= self._dims[direction]
                return self._ravel(c)
            return PROC_NULL
        return nbr(-1), nbr(+1)


# ----------------------------------------------------------- COMM_WORLD
class _CommWorld(Comm):
    """Proxy: resolves to the calling thread's world communicator."""

    def __init__(self):
        pass

    def _real(self):
        return _current_world().comm

    def __getattribute__(self, name):
        if name in ("_real", "__class__", "__repr__"):
            return object.__getattribute__(self, name)
        return getattr(object.__getattribute__(self, "_real")(), name)

    def __repr__(self):  # pragma: no cover
        return "<shim MPI.COMM_WORLD>"


COMM_WORLD = _CommWorld()
