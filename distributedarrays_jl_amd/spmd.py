"""SPMD primitives — the MI355X-native analog of the reference's SPMD
submodule (/root/reference/src/spmd.jl: sendto/recvfrom/barrier/bcast/
scatter/gather over per-worker RemoteChannels).  Here the payloads are
device buffers and the transport is RCCL over xGMI; the `*_host`
convenience forms stage numpy arrays through HBM.

Context semantics (spmd.jl:16-60; concurrent runs pinned at
test/spmd.jl:108-195): the reference keys per-worker RemoteChannels by
context so concurrent @async SPMD runs cannot cross-match messages.
Here a rank is an OS process issuing RCCL ops in program order on one
communicator, and RCCL matches point-to-point ops per peer in issue
order — so concurrent contexts are supported under the SPMD ordering
contract: every rank must issue the interleaved contexts' operations
in the SAME global order (which any deterministic SPMD program does,
including the reference's own concurrent-runs test, whose runs are
spawned in a fixed order).  Context objects carry the reference's
context-local storage (test/spmd.jl:154-195) and an epoch guard that
raises on out-of-order reuse instead of deadlocking."""
import ctypes
import itertools

import numpy as np

from . import comm
from ._ffi import check, lib, DArrayError
from .ops import _Buf

_ctx_counter = itertools.count()
_ctx_last_finished = [-1]


class Context:
    """An SPMD run context — context_local_storage() analog
    (spmd.jl:16-60).  Creation order defines the serialization domain:
    finish contexts (close_ctx) in creation order on every rank."""

    def __init__(self):
        self.id = next(_ctx_counter)
        self.storage = {}
        self.closed = False

    def context_local_storage(self):
        """Per-context dict (the reference returns a per-(context,
        worker) Dict; here per-(context, rank))."""
        if self.closed:
            raise DArrayError("spmd context %d already closed" % self.id)
        return self.storage

    def close(self):
        """close_ctx analog: frees storage and enforces creation-order
        completion (the deterministic-order contract above)."""
        if self.closed:
            return
        if self.id != _ctx_last_finished[0] + 1:
            raise DArrayError(
                "spmd contexts must complete in creation order "
                "(closing %d after %d)" % (self.id,
                                           _ctx_last_finished[0]))
        _ctx_last_finished[0] = self.id
        self.storage = {}
        self.closed = True


def context():
    """spmd_context() analog."""
    return Context()


def _auto():
    if not comm.initialized():
        comm.init()


def barrier():
    """spmd.jl barrier (:145-160)."""
    _auto()
    check(lib.da_barrier())


def sendto(peer, dev_ptr, nbytes):
    """spmd.jl sendto (:117-124) — device buffer to peer."""
    _auto()
    check(lib.da_send(dev_ptr, nbytes, peer))


def recvfrom(peer, dev_ptr, nbytes):
    """spmd.jl recvfrom (:126-136)."""
    _auto()
    check(lib.da_recv(dev_ptr, nbytes, peer))


def sendrecv(sptr, peer_s, rptr, peer_r, nbytes):
    _auto()
    check(lib.da_sendrecv(sptr, peer_s, rptr, peer_r, nbytes))


def bcast(dev_ptr, nbytes, root=0):
    """spmd.jl bcast (:162-178) — in-place device broadcast."""
    _auto()
    check(lib.da_bcast(dev_ptr, nbytes, root))


def bcast_host(arr, root=0):
    """Broadcast a numpy array from root to every rank (staged via HBM)."""
    _auto()
    arr = np.ascontiguousarray(arr)
    buf = _Buf(arr.nbytes)
    if comm.rank() == root:
        check(lib.da_h2d(buf.p, arr.ctypes.data_as(ctypes.c_void_p),
                         arr.nbytes))
    check(lib.da_bcast(buf.p, arr.nbytes, root))
    out = np.empty_like(arr)
    check(lib.da_d2h(buf.p, out.ctypes.data_as(ctypes.c_void_p),
                     arr.nbytes))
    buf.free()
    return out


def scatter_host(parts, root=0):
    """spmd.jl scatter (:180-200): root holds a list of per-rank numpy
    arrays (equal shape); each rank returns its part."""
    _auto()
    r, n = comm.rank_info()
    if r == root:
        if len(parts) != n:
            raise DArrayError("scatter: need %d parts" % n)
        shapes = [np.ascontiguousarray(p) for p in parts]
        mine = shapes[root]
        if n > 1:
            bufs = []
            check(lib.da_group_start())
            for peer in range(n):
                if peer == root:
                    continue
                pb = _Buf(shapes[peer].nbytes)
                check(lib.da_h2d(pb.p, shapes[peer].ctypes.data_as(
                    ctypes.c_void_p), shapes[peer].nbytes))
                check(lib.da_send(pb.p, shapes[peer].nbytes, peer))
                bufs.append(pb)
            check(lib.da_group_end())
            check(lib.da_synchronize())
            for pb in bufs:
                pb.free()
        return mine.copy()
    proto = np.ascontiguousarray(parts[0]) if parts else None
    if proto is None:
        raise DArrayError("scatter: non-root needs a shape prototype part")
    buf = _Buf(proto.nbytes)
    check(lib.da_recv(buf.p, proto.nbytes, root))
    out = np.empty_like(proto)
    check(lib.da_d2h(buf.p, out.ctypes.data_as(ctypes.c_void_p),
                     proto.nbytes))
    buf.free()
    return out


def gather_host(arr, root=0):
    """spmd.jl gather (:202-231): every rank contributes an equal-shape
    numpy array; root returns the list, others None."""
    _auto()
    r, n = comm.rank_info()
    arr = np.ascontiguousarray(arr)
    if n == 1:
        return [arr.copy()] if r == root else None
    if r == root:
        bufs = {}
        check(lib.da_group_start())
        for peer in range(n):
            if peer == root:
                continue
            pb = _Buf(arr.nbytes)
            check(lib.da_recv(pb.p, arr.nbytes, peer))
            bufs[peer] = pb
        check(lib.da_group_end())
        out = []
        for peer in range(n):
            if peer == root:
                out.append(arr.copy())
            else:
                o = np.empty_like(arr)
                check(lib.da_d2h(bufs[peer].p,
                                 o.ctypes.data_as(ctypes.c_void_p),
                                 arr.nbytes))
                bufs[peer].free()
                out.append(o)
        return out
    buf = _Buf(arr.nbytes)
    check(lib.da_h2d(buf.p, arr.ctypes.data_as(ctypes.c_void_p),
                     arr.nbytes))
    check(lib.da_send(buf.p, arr.nbytes, root))
    check(lib.da_synchronize())
    buf.free()
    return None
