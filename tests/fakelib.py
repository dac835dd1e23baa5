"""TEST-ONLY numpy + gloo implementation of the darray_hip C ABI.

Purpose: the product's multi-rank code paths (ops.dmatmul exchanges,
mismatched-cuts gather routing, dreduce_dims slab combine, dsort
all-to-all, spmd veneer, collect) are RCCL-over-xGMI on hardware, but
RCCL refuses two ranks on one GPU, and the CI container has no GPU at
all.  This module lets the REAL package code (ops.py / darray.py /
spmd.py, unmodified) execute at world_size 2..8 on CPU: chunk memory is
numpy, kernels are the oracle's numpy restatements, and da_send/recv/
bcast/allreduce move bytes with torch.distributed (gloo) using the same
pairing semantics as grouped RCCL (FIFO per (src,dst) pair; a group
posts all sends, then all recvs, then waits).

This is test infrastructure in the same category as oracle/ (which it
imports): ONLY tests install it, by explicit monkeypatching via
install().  The product never selects it — there is no env-var hook, no
fallback; on a GPU box the HIP library is the only compute path.
Numerics here are numpy's, NOT the HIP kernels': these tests validate
orchestration (schedules, pairing, buffer management, routing
decisions), while kernel numerics are covered by the -m gpu parity
suite against the same oracle.
"""
import ctypes
import time

import numpy as np

import oracle.ops as oops
import oracle.philox as ophilox
from distributedarrays_jl_amd._opcodes import MAP_OPS, MAP2_OPS

_NPDT = {0: np.dtype("float64"), 1: np.dtype("float32"),
         2: np.dtype("int64")}
_REDOP_NAMES = {0: "add", 1: "mul", 2: "min", 3: "max"}
_REDF_NAMES = {0: "identity", 1: "abs", 2: "abs2", 3: "isnan",
               4: "isfinite", 5: "nonzero"}


def _addr(x):
    if x is None:
        return 0
    if isinstance(x, int):
        return x
    v = getattr(x, "value", None)
    if isinstance(v, int):
        return v
    return ctypes.cast(x, ctypes.c_void_p).value or 0  # byref/arrays


def _u8(addr, nbytes):
    """uint8 view of raw memory at addr (which the caller keeps alive)."""
    p = ctypes.cast(ctypes.c_void_p(_addr(addr)),
                    ctypes.POINTER(ctypes.c_uint8))
    return np.ctypeslib.as_array(p, shape=(int(nbytes),))


def _tv(addr, n, dt):
    """n-element typed view at addr."""
    return _u8(addr, int(n) * dt.itemsize).view(dt)


class FakeLib:
    def __init__(self):
        self.bufs = {}        # base addr -> numpy backing array
        self.user_bytes = 0
        self.rank = 0
        self.nranks = 1
        self.group = None     # None or list of pending p2p ops
        self.events = {}
        self._next_ev = 1

    # ---- torch.distributed helpers ----
    def _td(self):
        import torch.distributed as td
        return td

    # ---- lifecycle ----
    def da_init(self, device, rank, nranks, uid_path):
        self.rank, self.nranks = int(rank), int(nranks)
        return 0

    def da_shutdown(self):
        return 0

    def da_rank(self):
        return self.rank

    def da_nranks(self):
        return self.nranks

    # ---- memory ----
    def da_alloc(self, nbytes, dtype, out):
        arr = np.zeros(max(int(nbytes), 1), dtype=np.uint8)
        self.bufs[arr.ctypes.data] = arr
        self.user_bytes += arr.nbytes
        ctypes.cast(out, ctypes.POINTER(ctypes.c_void_p))[0] = \
            arr.ctypes.data
        return 0

    def da_free(self, chunk):
        a = _addr(chunk)
        arr = self.bufs.pop(a, None)
        if arr is None:
            raise AssertionError("da_free of unknown pointer 0x%x" % a)
        self.user_bytes -= arr.nbytes
        return 0

    def da_pool_trim(self):
        return 0

    def da_pool_bytes(self):
        return 0

    def da_bytes_in_use(self):
        return self.user_bytes

    def da_h2d(self, chunk, host, nbytes):
        ctypes.memmove(_addr(chunk), _addr(host), int(nbytes))
        return 0

    def da_d2h(self, chunk, host, nbytes):
        ctypes.memmove(_addr(host), _addr(chunk), int(nbytes))
        return 0

    def da_d2d(self, dst, src, nbytes):
        ctypes.memmove(_addr(dst), _addr(src), int(nbytes))
        return 0

    def da_copy2d(self, dst, dpitch, src, spitch, width, height):
        d, s = _addr(dst), _addr(src)
        for h in range(int(height)):
            ctypes.memmove(d + h * int(dpitch), s + h * int(spitch),
                           int(width))
        return 0

    # ---- constructors ----
    def da_fill(self, chunk, v, n, dtype):
        dt = _NPDT[int(dtype)]
        _tv(chunk, n, dt)[:] = dt.type(v)
        return 0

    def da_rand(self, chunk, n, dtype, seed, kind, offset):
        dt = _NPDT[int(dtype)]
        n, seed, offset = int(n), int(seed), int(offset)
        if int(kind) == 0:
            gen = {0: ophilox.fill_uniform_f64,
                   1: ophilox.fill_uniform_f32,
                   2: ophilox.fill_int64}[int(dtype)]
        else:
            gen = {0: ophilox.fill_normal_f64,
                   1: ophilox.fill_normal_f32}[int(dtype)]
        _tv(chunk, n, dt)[:] = gen(n, seed, offset)
        return 0

    # ---- elementwise ----
    def da_map(self, opcode, dst, src, n, dtype):
        dt = _NPDT[int(dtype)]
        s = _tv(src, n, dt)
        _tv(dst, n, dt)[:] = oops.MAP_OPS[MAP_OPS[int(opcode)]](s)
        return 0

    def da_map2(self, opcode, dst, a, b, n, dtype):
        dt = _NPDT[int(dtype)]
        av, bv = _tv(a, n, dt), _tv(b, n, dt)
        _tv(dst, n, dt)[:] = oops.MAP2_OPS[MAP2_OPS[int(opcode)]](av, bv)
        return 0

    def da_map2_scalar(self, opcode, dst, src, c, rev, n, dtype):
        dt = _NPDT[int(dtype)]
        s = _tv(src, n, dt)
        cc = dt.type(c)
        f = oops.MAP2_OPS[MAP2_OPS[int(opcode)]]
        _tv(dst, n, dt)[:] = f(cc, s) if int(rev) else f(s, cc)
        return 0

    def da_bcast_fma(self, d, a, b, c, n, dtype):
        dt = _NPDT[int(dtype)]
        av, bv = _tv(a, n, dt), _tv(b, n, dt)
        _tv(d, n, dt)[:] = oops.oracle_bcast_fma(av, bv, c)
        return 0

    def da_expr(self, prog, plen, dst, dst_dims, nd, srcs, src_strides,
                nsrcs, consts, nconsts, n, dtype):
        import oracle.expr as oexpr
        dt = _NPDT[int(dtype)]
        plen, nd = int(plen), int(nd)
        nsrcs, nconsts, n = int(nsrcs), int(nconsts), int(n)
        pv = ctypes.cast(prog, ctypes.POINTER(ctypes.c_int32))
        program = [int(pv[i]) for i in range(plen)]
        cv = ctypes.cast(consts, ctypes.POINTER(ctypes.c_double))
        cc = [float(cv[i]) for i in range(nconsts)]
        sv = ctypes.cast(srcs, ctypes.POINTER(ctypes.c_void_p))
        if _addr(src_strides) == 0:
            args = [_tv(sv[i], n, dt) for i in range(nsrcs)]
            out = oexpr.evaluate(program, args, cc, dt)
            _tv(dst, n, dt)[:] = out.ravel()
            return 0
        dims = ctypes.cast(dst_dims, ctypes.POINTER(ctypes.c_uint64))
        shape = tuple(int(dims[d]) for d in range(nd))
        stv = ctypes.cast(src_strides, ctypes.POINTER(ctypes.c_uint64))
        args = []
        for i in range(nsrcs):
            ss = [int(stv[i * nd + d]) for d in range(nd)]
            base = _tv(sv[i], 1, dt)   # element 0; as_strided walks on
            args.append(np.lib.stride_tricks.as_strided(
                base, shape=shape,
                strides=tuple(s * dt.itemsize for s in ss)))
        out = oexpr.evaluate(program, args, cc, dt)
        _tv(dst, n, dt)[:] = np.asfortranarray(out).ravel(order="F")
        return 0

    def da_cast(self, dst, dst_dtype, src, src_dtype, n):
        dd, sd = _NPDT[int(dst_dtype)], _NPDT[int(src_dtype)]
        sv = _tv(src, n, sd)
        if dd.kind == "i" and sd.kind == "f":
            _tv(dst, n, dd)[:] = np.rint(sv).astype(dd)
        else:
            _tv(dst, n, dd)[:] = sv.astype(dd)
        return 0

    def da_expr_jit_state(self):
        return 0      # no JIT in the fake (numpy evaluator only)

    def da_expr_jit_errstr(self):
        return b""

    def da_axpby(self, y, x, alpha, beta, n, dtype):
        dt = _NPDT[int(dtype)]
        yv, xv = _tv(y, n, dt), _tv(x, n, dt)
        yv[:] = dt.type(alpha) * xv + dt.type(beta) * yv
        return 0

    def da_add(self, dest, src, scale, n, dtype):
        dt = _NPDT[int(dtype)]
        dv, sv = _tv(dest, n, dt), _tv(src, n, dt)
        dv[:] = oops.oracle_add(dv, sv, scale)
        return 0

    def da_scale(self, a, s, n, dtype):
        dt = _NPDT[int(dtype)]
        av = _tv(a, n, dt)
        av[:] = oops.oracle_scale(av, s)
        return 0

    # ---- reductions ----
    def da_reduce(self, mapop, redop, src, n, dtype, out):
        dt = _NPDT[int(dtype)]
        chunk = _tv(src, n, dt)
        val = oops.oracle_chunk_reduce(_REDF_NAMES[int(mapop)],
                                       _REDOP_NAMES[int(redop)], chunk)
        ct = {0: ctypes.c_double, 1: ctypes.c_float, 2: ctypes.c_int64}[
            int(dtype)]
        ctypes.cast(out, ctypes.POINTER(ct))[0] = \
            float(val) if int(dtype) != 2 else int(val)
        return 0

    def da_reduce_dims(self, mapop, redop, src, inner, axis, outer,
                       dtype, dst):
        dt = _NPDT[int(dtype)]
        inner, axis, outer = int(inner), int(axis), int(outer)
        f = oops.MAPRED_FS[_REDF_NAMES[int(mapop)]]
        cube = f(_tv(src, inner * axis * outer, dt)
                 .reshape((inner, axis, outer), order="F"))
        red = _REDOP_NAMES[int(redop)]
        if red == "add":
            r = cube.sum(axis=1, dtype=dt)
        elif red == "mul":
            r = cube.prod(axis=1, dtype=dt)
        elif red == "min":
            r = cube.min(axis=1)
        else:
            r = cube.max(axis=1)
        _tv(dst, inner * outer, dt)[:] = np.asfortranarray(r).ravel(
            order="F")
        return 0

    def da_allreduce(self, inout, count, dtype, redop):
        if self.nranks == 1:
            return 0
        import torch
        td = self._td()
        dt = _NPDT[int(dtype)]
        host = _tv(inout, count, dt)
        t = torch.from_numpy(host.copy())
        op = {0: td.ReduceOp.SUM, 1: td.ReduceOp.PRODUCT,
              2: td.ReduceOp.MIN, 3: td.ReduceOp.MAX}[int(redop)]
        td.all_reduce(t, op=op)
        host[:] = t.numpy()
        return 0

    # ---- gemm ----
    def _gemm(self, C, A, B, m, n, k, lda, ldb, ldc, alpha, beta, dt):
        m, n, k = int(m), int(n), int(k)
        lda, ldb, ldc = int(lda), int(ldb), int(ldc)
        Av = _tv(A, lda * k, dt).reshape((lda, k), order="F")[:m, :]
        Bv = _tv(B, ldb * n, dt).reshape((ldb, n), order="F")[:k, :]
        Cv = _tv(C, ldc * n, dt).reshape((ldc, n), order="F")
        acc = Av @ Bv
        res = dt.type(alpha) * acc if alpha != 1.0 else acc
        if beta == 0.0:
            Cv[:m, :] = res
        else:
            Cv[:m, :] = dt.type(beta) * Cv[:m, :] + res
        return 0

    def da_gemm_f64(self, C, A, B, m, n, k, lda, ldb, ldc, alpha, beta):
        return self._gemm(C, A, B, m, n, k, lda, ldb, ldc, alpha, beta,
                          np.dtype("float64"))

    def da_gemm_f32(self, C, A, B, m, n, k, lda, ldb, ldc, alpha, beta):
        return self._gemm(C, A, B, m, n, k, lda, ldb, ldc, alpha, beta,
                          np.dtype("float32"))

    def da_gemm_i64(self, C, A, B, m, n, k, lda, ldb, ldc, alpha, beta):
        return self._gemm(C, A, B, m, n, k, lda, ldb, ldc, alpha, beta,
                          np.dtype("int64"))

    # ---- transpose / diag ----
    def da_transpose(self, dst, src, m, n, dtype):
        dt = _NPDT[int(dtype)]
        m, n = int(m), int(n)
        s = _tv(src, m * n, dt).reshape((m, n), order="F")
        _tv(dst, m * n, dt)[:] = np.asfortranarray(s.T).ravel(order="F")
        return 0

    def da_diag_scale(self, a, m, n, diag, side, dtype):
        dt = _NPDT[int(dtype)]
        m, n = int(m), int(n)
        av = _tv(a, m * n, dt).reshape((m, n), order="F")
        dv = _tv(diag, m if int(side) == 0 else n, dt)
        if int(side) == 0:
            av *= dv[:, None]
        else:
            av *= dv[None, :]
        return 0

    # ---- sort ----
    def da_sort(self, chunk, n, dtype):
        dt = _NPDT[int(dtype)]
        v = _tv(chunk, n, dt)
        v.sort(kind="stable")
        return 0

    def da_sort_out(self, src, dst, n, dtype):
        dt = _NPDT[int(dtype)]
        _tv(dst, n, dt)[:] = np.sort(_tv(src, n, dt), kind="stable")
        return 0

    def da_lower_bound(self, sorted_, n, dtype, splitters, k, out):
        dt = _NPDT[int(dtype)]
        s = _tv(sorted_, n, dt)
        spl = _tv(splitters, k, dt)
        res = np.searchsorted(s, spl, side="left")
        ov = ctypes.cast(out, ctypes.POINTER(ctypes.c_uint64))
        for i in range(int(k)):
            ov[i] = int(res[i])
        return 0

    # ---- point-to-point over gloo ----
    # Semantics mirror grouped RCCL: inside a group, ops are deferred;
    # group_end posts every isend, then every irecv, then waits (FIFO
    # matching per (src,dst) pair, like NCCL's in-group ordering).
    def da_group_start(self):
        assert self.group is None, "nested da_group_start"
        self.group = []
        return 0

    def da_group_end(self):
        import torch
        td = self._td()
        ops, self.group = self.group, None
        works = []
        for kind, addr, nbytes, peer in ops:
            if kind == "send":
                t = torch.from_numpy(_u8(addr, nbytes).copy())
                works.append((td.isend(t, peer, tag=0), None, None, t))
        for kind, addr, nbytes, peer in ops:
            if kind == "recv":
                t = torch.zeros(int(nbytes), dtype=torch.uint8)
                works.append((td.irecv(t, peer, tag=0), addr, nbytes, t))
        for w, addr, nbytes, t in works:
            w.wait()
            if addr is not None:
                _u8(addr, nbytes)[:] = t.numpy()
        return 0

    def _p2p(self, kind, buf, nbytes, peer):
        if self.group is not None:
            self.group.append((kind, _addr(buf), int(nbytes), int(peer)))
            return 0
        import torch
        td = self._td()
        if kind == "send":
            t = torch.from_numpy(_u8(buf, nbytes).copy())
            td.send(t, int(peer), tag=0)
        else:
            t = torch.zeros(int(nbytes), dtype=torch.uint8)
            td.recv(t, int(peer), tag=0)
            _u8(buf, nbytes)[:] = t.numpy()
        return 0

    def da_send(self, buf, nbytes, peer):
        return self._p2p("send", buf, nbytes, peer)

    def da_recv(self, buf, nbytes, peer):
        return self._p2p("recv", buf, nbytes, peer)

    def da_sendrecv(self, sbuf, peer_s, rbuf, peer_r, nbytes):
        self.da_group_start()
        self.da_send(sbuf, nbytes, peer_s)
        self.da_recv(rbuf, nbytes, peer_r)
        return self.da_group_end()

    def da_bcast(self, buf, nbytes, root):
        if self.nranks == 1:
            return 0
        import torch
        td = self._td()
        t = torch.from_numpy(_u8(buf, nbytes).copy())
        td.broadcast(t, src=int(root))
        _u8(buf, nbytes)[:] = t.numpy()
        return 0

    def da_barrier(self):
        if self.nranks > 1:
            self._td().barrier()
        return 0

    # ---- streams / events (host-synchronous here) ----
    def da_p2p_stream(self, use_comm):
        return 0

    def da_comm_after_compute(self):
        return 0

    def da_main_after_comm(self):
        return 0

    def da_comm_sync(self):
        return 0

    def da_synchronize(self):
        return 0

    def da_event_create(self, out):
        ev = self._next_ev
        self._next_ev += 1
        self.events[ev] = 0.0
        ctypes.cast(out, ctypes.POINTER(ctypes.c_void_p))[0] = ev
        return 0

    def da_event_record(self, ev):
        self.events[_addr(ev)] = time.perf_counter()
        return 0

    def da_event_elapsed(self, e0, e1, out):
        ms = (self.events[_addr(e1)] - self.events[_addr(e0)]) * 1e3
        ctypes.cast(out, ctypes.POINTER(ctypes.c_float))[0] = ms
        return 0

    def da_event_destroy(self, ev):
        self.events.pop(_addr(ev), None)
        return 0

    # ---- introspection ----
    def da_errstr(self, code):
        return b"fakelib error"

    def da_device_props(self, name, name_len, hbm):
        return 0


def install():
    """Swap the package's ABI binding for a FakeLib — explicit,
    test-side-only.  Returns the instance."""
    from distributedarrays_jl_amd import _ffi, ops, darray, spmd, expr
    fake = FakeLib()
    _ffi.lib = fake
    for mod in (ops, darray, spmd, expr):
        mod.lib = fake
    return fake
