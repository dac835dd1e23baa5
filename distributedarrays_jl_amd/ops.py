"""Hot-path operations on DArrays — the API mirror of
src/mapreduce.jl, src/broadcast.jl and src/linalg.jl, executing HIP
kernels on local chunks and RCCL collectives across ranks.

The reference fans a closure out to each worker and runs Base on
localparts; here every rank IS a worker (SPMD), the closure set is the
fixed opcode table (include/darray_hip.h), and the caller-side fold
becomes an RCCL allreduce (SURVEY.md §3.1 build mapping)."""
import ctypes
import numpy as np

from . import geometry, plan
from ._ffi import check, lib, DArrayError
from ._opcodes import (DTYPES, DTYPE_SIZE, NUMPY_DTYPES, MAP_OP, MAP2_OP,
                       RED_OPS, RED_FS, I64_MAP_OPS, I64_MAP2_OPS)
from .darray import DArray


def _same_layout(d0, d):
    """True when the two DArrays have identical chunk layout: same global
    dims/dtype, the same cut boxes AND the same chunk->rank owners.
    Comparing d.idxs (not d.dist) matters: a ragged DVector from
    DArray.from_chunk_sizes has dist=(nr,) like an evenly-cut one, but
    per-rank lnumel differs — treating those as aligned would read past
    the smaller chunk."""
    return (d.dims == d0.dims and d.dtype == d0.dtype
            and list(d.idxs) == list(d0.idxs) and d.ranks == d0.ranks)


def _aligned(*ds):
    d0 = ds[0]
    for d in ds[1:]:
        if not _same_layout(d0, d):
            raise DArrayError(
                "operands must share dims/dtype/cuts/owners (aligned "
                "chunks); mismatched-cuts operands route through the "
                "makelocal gather (map_general/map2_general/"
                "broadcast_fma_general)")
    return d0


def _check_i64_scalar(dtype, c):
    """The ABI carries scalars as double; |i64| > 2^53 would round
    silently (the fill_ guard, generalized — ADVICE r1)."""
    if dtype == "i64":
        ci = int(c)
        if abs(ci) > (1 << 53) or ci != c:
            raise DArrayError(
                "i64 scalar %r not exactly representable through the "
                "double ABI parameter" % (c,))


# ------------------------------------------------------- map / broadcast
def map_(op, dest, src):
    """map!(f, dest, src) — mapreduce.jl:5-12.  Mismatched cuts route
    transparently through the makelocal gather (mapreduce.jl:8), like
    the reference."""
    if dest.dtype == "i64" and op not in I64_MAP_OPS:
        raise DArrayError("op %r invalid for i64" % op)
    if not _same_layout(dest, src):
        return map_general(op, dest, src)
    if dest.lnumel:
        check(lib.da_map(MAP_OP[op], dest._ptr(), src._ptr(),
                         dest.lnumel, DTYPES[dest.dtype]))
    return dest


def dmap(op, src):
    """map(f, d) — mapreduce.jl:3 (out-of-place)."""
    return map_(op, src.similar(), src)


def map2_(op, dest, a, b):
    """dest .= f.(a, b) for the binary table (mapreduce.jl:180-189).
    Mismatched cuts localize per-operand (bclocal, broadcast.jl:140-152)."""
    if dest.dtype == "i64" and op not in I64_MAP2_OPS:
        raise DArrayError("op %r invalid for i64" % op)
    if not (_same_layout(dest, a) and _same_layout(dest, b)):
        return map2_general(op, dest, a, b)
    if dest.lnumel:
        check(lib.da_map2(MAP2_OP[op], dest._ptr(), a._ptr(), b._ptr(),
                          dest.lnumel, DTYPES[dest.dtype]))
    return dest


_PROMOTE_RANK = {"i64": 0, "f32": 1, "f64": 2}


def elementwise(op, a, b):
    """C = f.(a, b) out of place, with Julia's type promotion for mixed
    eltypes (promote_type: Int64 < Float32 < Float64).  In-place map2_
    keeps strict same-dtype semantics (the destination fixes T)."""
    if a.dtype != b.dtype:
        target = (a.dtype if _PROMOTE_RANK[a.dtype] >=
                  _PROMOTE_RANK[b.dtype] else b.dtype)
        ca = a if a.dtype == target else dcast(a, target)
        cb = b if b.dtype == target else dcast(b, target)
        try:
            return map2_(op, ca.similar(), ca, cb)
        finally:
            if ca is not a:
                ca.close()
            if cb is not b:
                cb.close()
    return map2_(op, a.similar(), a, b)


def map2_scalar_(op, dest, src, c, reverse=False):
    """dest .= f.(src, c) (or f.(c, src)) — scalar broadcast argument
    (broadcast.jl:124-133: singletons are not distributed).  Covers the
    cfg-1 plumbing op D .+ 1."""
    if dest.dtype == "i64" and op not in I64_MAP2_OPS:
        raise DArrayError("op %r invalid for i64" % op)
    _check_i64_scalar(dest.dtype, c)
    if not _same_layout(dest, src):
        buf, _ = gather_box(src, _dest_boxes(dest))
        if buf is not None and dest.lnumel:
            check(lib.da_map2_scalar(MAP2_OP[op], dest._ptr(), buf.p,
                                     float(c), 1 if reverse else 0,
                                     dest.lnumel, DTYPES[dest.dtype]))
            check(lib.da_synchronize())
        if buf is not None:
            buf.free()
        return dest
    if dest.lnumel:
        check(lib.da_map2_scalar(MAP2_OP[op], dest._ptr(), src._ptr(),
                                 float(c), 1 if reverse else 0,
                                 dest.lnumel, DTYPES[dest.dtype]))
    return dest


def elementwise_scalar(op, a, c, reverse=False):
    return map2_scalar_(op, a.similar(), a, c, reverse)


def broadcast_fma(dest, a, b, c):
    """D .= A .* B .+ c — the fused cfg-3 broadcast (broadcast.jl:65-85;
    aligned same-cuts args need zero communication).
    Mismatched cuts route through the bclocal/makelocal gather."""
    if not (_same_layout(dest, a) and _same_layout(dest, b)):
        return broadcast_fma_general(dest, a, b, c)
    if dest.lnumel:
        check(lib.da_bcast_fma(dest._ptr(), a._ptr(), b._ptr(), float(c),
                               dest.lnumel, DTYPES[dest.dtype]))
    return dest


def axpy_(alpha, x, y):
    """axpy! — linalg.jl:24-34."""
    _aligned(x, y)
    if y.lnumel:
        check(lib.da_axpby(y._ptr(), x._ptr(), float(alpha), 1.0,
                           y.lnumel, DTYPES[y.dtype]))
    return y


def add_(dest, src, scale=1.0):
    """add! — linalg.jl:62-76.  Mismatched cuts gather src first
    (the fetch of linalg.jl:243-251 partials is the aligned case)."""
    _check_i64_scalar(dest.dtype, scale)
    if not _same_layout(dest, src):
        if dest.dims != src.dims or dest.dtype != src.dtype:
            raise DArrayError("add_: dims/dtype mismatch")
        buf, _ = gather_box(src, _dest_boxes(dest))
        if buf is not None and dest.lnumel:
            check(lib.da_add(dest._ptr(), buf.p, float(scale),
                             dest.lnumel, DTYPES[dest.dtype]))
            check(lib.da_synchronize())
        if buf is not None:
            buf.free()
        return dest
    if dest.lnumel:
        check(lib.da_add(dest._ptr(), src._ptr(), float(scale),
                         dest.lnumel, DTYPES[dest.dtype]))
    return dest


def scale_(a, s):
    """rmul! — linalg.jl:54-59."""
    _check_i64_scalar(a.dtype, s)
    if a.lnumel:
        check(lib.da_scale(a._ptr(), float(s), a.lnumel, DTYPES[a.dtype]))
    return a


def dcast(d, dtype):
    """DArray{T2}(D) — elementwise dtype conversion on device.
    float -> i64 rounds half-even (round(Int, x)); the strict
    InexactError convert is a host-side concern."""
    if dtype == d.dtype:
        return d.copy()
    if dtype not in DTYPES:
        raise DArrayError("dcast: bad dtype %r" % dtype)
    out = d.similar(dtype)
    if out.lnumel:
        check(lib.da_cast(out._ptr(), DTYPES[dtype], d._ptr(),
                          DTYPES[d.dtype], out.lnumel))
    return out


def map_localparts(f, *ds):
    """map_localparts(f, d...) — mapreduce.jl:137-169: apply an
    ARBITRARY host function to whole localparts (numpy in, numpy out;
    the reference runs arbitrary Julia closures on its CPU localparts
    the same way).  This is a host-boundary escape hatch: the chunk
    round-trips HBM -> host -> HBM.  Compositions of the op tables
    should use expr.materialize_ instead (fused, device-resident)."""
    res = f(*[d.localpart() for d in ds])
    out = ds[0].similar()
    out.set_localpart(np.asfortranarray(
        np.asarray(res, dtype=np.dtype(NUMPY_DTYPES[out.dtype]))))
    return out


def map_localparts_(f, d):
    """map_localparts!(f, d) — in-place variant (mapreduce.jl:156-162)."""
    d.set_localpart(np.asfortranarray(
        np.asarray(f(d.localpart()),
                   dtype=np.dtype(NUMPY_DTYPES[d.dtype]))))
    return d


# ------------------------------------------------------------ reductions
_CTYPE = {"f64": ctypes.c_double, "f32": ctypes.c_float,
          "i64": ctypes.c_int64}


def _local_reduce(f, op, d):
    out = _CTYPE[d.dtype]()
    check(lib.da_reduce(RED_FS[f], RED_OPS[op], d._ptr(), d.lnumel,
                        DTYPES[d.dtype], ctypes.byref(out)))
    return out


def mapreduce(f, op, d):
    """mapreduce(f, op, d) — mapreduce.jl:29-35: per-chunk kernel
    reduction, then the cross-worker fold as one scalar RCCL allreduce
    (tree; partials are 8 bytes — latency-bound)."""
    out = _local_reduce(f, op, d)
    if d.nranks > 1:
        check(lib.da_allreduce(ctypes.byref(out), 1, DTYPES[d.dtype],
                               RED_OPS[op]))
    return out.value


def dsum(d):
    return mapreduce("identity", "add", d)


def dprod(d):
    return mapreduce("identity", "mul", d)


def dmaximum(d):
    if d.size == 0:
        raise DArrayError("maximum over empty DArray")  # Julia throws too
    return mapreduce("identity", "max", d)


def dminimum(d):
    if d.size == 0:
        raise DArrayError("minimum over empty DArray")
    return mapreduce("identity", "min", d)


def dextrema(d):
    """extrema — mapreduce.jl:124-131."""
    return (dminimum(d), dmaximum(d))


def dmean(d):
    """mean rides the sum path (ext/StatisticsExt.jl:6)."""
    return dsum(d) / d.size


def dcount(pred, d):
    """count(f, A) — mapreduce.jl:115-122 (pred in {nonzero, isnan,
    isfinite})."""
    return int(mapreduce(pred, "add", d))


def dall(pred, d):
    """all(f, A) — mapreduce.jl:97-104."""
    if d.size == 0:
        return True
    return mapreduce(pred, "min", d) == 1


def dany(pred, d):
    """any(f, A) — mapreduce.jl:106-113."""
    if d.size == 0:
        return False
    return mapreduce(pred, "max", d) == 1


def ddot(x, y):
    """dot — linalg.jl:36-45: aligned cuts multiply locally; mismatched
    cuts localize y via makelocal (linalg.jl:42) inside map2_."""
    if x.dims != y.dims or x.dtype != y.dtype:
        raise DArrayError("ddot: dims/dtype mismatch")
    tmp = elementwise("mul", x, y)
    try:
        return mapreduce("identity", "add", tmp)
    finally:
        tmp.close()


def dnorm(x, p=2):
    """norm(x, p) for any real p — linalg.jl:47-52 (the reference runs
    Base.norm on each localpart then norms the partials; here the
    closed forms: p-power sum via one fused |x|^p kernel)."""
    if p == 2:
        return float(np.sqrt(mapreduce("abs2", "add", x)))
    if p == 1:
        return mapreduce("abs", "add", x)
    if p == float("inf"):
        return mapreduce("abs", "max", x)
    if p == float("-inf"):
        return mapreduce("abs", "min", x)
    if p == 0:
        return float(dcount("nonzero", x))   # Julia norm(x, 0)
    from . import expr as E
    tmp = x.similar()
    E.materialize_(tmp, E.abs(E.ref(x)) ** float(p))
    try:
        return float(mapreduce("identity", "add", tmp)) ** (1.0 / p)
    finally:
        tmp.close()


# --------------------------------------------------------------- matmul
class _Buf:
    """Raw device staging buffer."""

    def __init__(self, nbytes):
        self.p = ctypes.c_void_p()
        check(lib.da_alloc(max(int(nbytes), 1), 0, ctypes.byref(self.p)))

    def at(self, byte_off):
        return ctypes.c_void_p(self.p.value + byte_off)

    def free(self):
        if self.p.value:
            check(lib.da_free(self.p))
            self.p = ctypes.c_void_p()


def _copy2d(dst, dpitch, src, spitch, width, height):
    if width and height:
        check(lib.da_copy2d(dst, dpitch, src, spitch, width, height))


def dmul_(C, A, B, alpha=1.0, beta=0.0):
    """mul!(C, A, B, alpha, beta) — full LinearAlgebra.mul! semantics
    (linalg.jl:255): beta-scale C (fill 0 when beta==0,
    linalg.jl:232-240) then accumulate alpha*A*B into it."""
    R = dmatmul(A, B, alpha=alpha)
    if beta == 0.0:
        C.fill_(0.0)
    elif beta != 1.0:
        scale_(C, beta)
    add_(C, R, 1.0)
    check(lib.da_synchronize())
    R.close()
    return C


def dmatmul(A, B, alpha=1.0):
    """C = alpha * A * B — the `*` wrapper (linalg.jl:266-273) plus
    _matmatmul! (linalg.jl:190-253), re-expressed per plan.py.  f64 only
    (the cfg-4 metric path); local GEMM is the MFMA kernel."""
    if A.dtype not in ("f64", "f32", "i64") or B.dtype != A.dtype:
        raise DArrayError("dmatmul: f64/f32/i64 only (matching dtypes)")
    if A.ndims != 2 or B.ndims != 2 or A.dims[1] != B.dims[0]:
        raise DArrayError("dmatmul: shape mismatch %r x %r"
                          % (A.dims, B.dims))
    if (A.ranks != list(range(A.nchunks))
            or B.ranks != list(range(B.nchunks))):
        # the slab/partial plans treat chunk indices as rank ids;
        # non-identity owners (e.g. a dims-reduction result) would
        # silently drop the off-grid rank's data
        raise DArrayError(
            "dmatmul: operands must have identity chunk->rank mapping "
            "(copy() a reduction result first)")
    if A.dtype == "f64":
        gemm_fn = lib.da_gemm_f64
    elif A.dtype == "f32":
        gemm_fn = lib.da_gemm_f32
    else:
        def gemm_fn(C_, A_, B_, m_, n_, k_, la, lb, lc, al, be):
            return lib.da_gemm_i64(C_, A_, B_, m_, n_, k_, la, lb, lc,
                                   int(al), int(be))
    m, kk = A.dims
    n = B.dims[1]
    I, J = A.dist
    K = plan.c_grid(A.dist, B.dist)[1]
    C = DArray((m, n), A.dtype, (I, K))
    C.fill_(0.0)
    r = A.rank
    pos = plan.a_rank_pos(r, A.dist)
    esz = DTYPE_SIZE[A.dtype]

    pieces = plan.bslab_plan(A.dist, A.cuts[1], B.dims, B.dist, B.idxs)
    my_sends = [p for p in pieces if p[0] == r and p[1] != r]
    my_recvs = [p for p in pieces if p[1] == r and p[0] != r]
    my_local = [p for p in pieces if p[0] == r and p[1] == r]

    # B-slab exchange: ANY rank owning B pieces packs+sends (a B owner
    # outside A's process grid — the arbitrary-AbstractMatrix-B case of
    # linalg.jl:211-226 — has pos None and only sends); A-grid ranks
    # receive/unpack into their row slab.
    slab = None
    srows = rlo = 0
    sendbufs, recvbufs = [], []
    if pos is not None:
        i, j = pos
        rlo, rhi = plan.slab_rows(A.cuts[1], j)
        srows = rhi - rlo
        slab = _Buf(max(srows * n, 1) * esz)
    # pack my outgoing pieces from my B block (da_copy2d: column-major
    # sub-block -> contiguous)
    Brows = B.lshape[0] if B.lnumel else 0
    for (src, dst, rows, cols) in my_sends:
        prows = rows[1] - rows[0]
        pcols = cols[1] - cols[0]
        buf = _Buf(prows * pcols * esz)
        off = ((rows[0] - B.lidx[0][0])
               + (cols[0] - B.lidx[1][0]) * Brows) * esz
        _copy2d(buf.p, prows * esz,
                ctypes.c_void_p(B._ptr().value + off), Brows * esz,
                prows * esz, pcols)
        sendbufs.append(((src, dst, rows, cols), buf))
    for (src, dst, rows, cols) in my_recvs:
        prows = rows[1] - rows[0]
        pcols = cols[1] - cols[0]
        recvbufs.append(((src, dst, rows, cols),
                         _Buf(prows * pcols * esz)))
    # exchange (grouped so send/recv pairs match globally)
    if my_sends or my_recvs:
        check(lib.da_group_start())
        for (src, dst, rows, cols), buf in sendbufs:
            nb = (rows[1] - rows[0]) * (cols[1] - cols[0]) * esz
            check(lib.da_send(buf.p, nb, dst))
        for (src, dst, rows, cols), buf in recvbufs:
            nb = (rows[1] - rows[0]) * (cols[1] - cols[0]) * esz
            check(lib.da_recv(buf.p, nb, src))
        check(lib.da_group_end())
    if pos is not None:
        # unpack into the slab (stream-ordered after the group)
        for (src, dst, rows, cols) in my_local:
            prows = rows[1] - rows[0]
            pcols = cols[1] - cols[0]
            off = ((rows[0] - B.lidx[0][0])
                   + (cols[0] - B.lidx[1][0]) * (B.lshape[0])) * esz
            doff = ((rows[0] - rlo) + cols[0] * srows) * esz
            _copy2d(slab.at(doff), srows * esz,
                    ctypes.c_void_p(B._ptr().value + off),
                    B.lshape[0] * esz, prows * esz, pcols)
        for (src, dst, rows, cols), buf in recvbufs:
            prows = rows[1] - rows[0]
            pcols = cols[1] - cols[0]
            doff = ((rows[0] - rlo) + cols[0] * srows) * esz
            _copy2d(slab.at(doff), srows * esz, buf.p, prows * esz,
                    prows * esz, pcols)

    # local partial GEMMs fused with the per-k partial exchange
    # (linalg.jl:218-251).  With DA_MM_OVERLAP (default on) each k's
    # grouped send/recv rides the comm stream as soon as its GEMM is
    # done, overlapping the xGMI transfer with GEMM k+1; every A-grid
    # rank issues group k in ascending-k order, so the pairing argument
    # of the single-group schedule carries over unchanged.
    import os as _os
    ccols = geometry.ranges1d(C.cuts[1])
    moves = plan.partial_plan(A.dist, K)
    my_psends = [mv for mv in moves if mv[0] == r]
    my_precvs = [mv for mv in moves if mv[1] == r]
    overlap = (_os.environ.get("DA_MM_OVERLAP", "1") == "1"
               and (my_psends or my_precvs))
    precv = {}
    for (src, dst, k) in my_precvs:
        precv[(src, k)] = _Buf(max(C.lshape[0] * C.lshape[1], 1) * esz)
    partials = []
    if overlap:
        check(lib.da_p2p_stream(1))
    for k in range(K):
        if pos is not None:
            # a partial per k even when the local A chunk is empty
            # (zero-size k-cut): its send/recv peers index partials[k],
            # and an empty k-sum contributes zeros
            mloc = A.lshape[0]
            kloc = A.lshape[1]
            clo, chi = ccols[k]
            nk = chi - clo
            pk = _Buf(max(mloc * nk, 1) * esz)
            if A.lnumel:
                check(gemm_fn(pk.p, A._ptr(), slab.at(clo * kloc * esz),
                              mloc, nk, kloc, mloc, kloc, mloc, 1.0, 0.0))
            elif mloc * nk:
                check(lib.da_fill(pk.p, 0.0, mloc * nk, DTYPES[A.dtype]))
            partials.append(pk)
        if overlap:
            check(lib.da_comm_after_compute())
            # zero-size partials are skipped consistently on both sides:
            # sender nb == A.lshape[0]*width == C-owner's C.lnumel
            sends_k = [mv for mv in my_psends if mv[2] == k
                       and A.lshape[0] * (ccols[k][1] - ccols[k][0])]
            recvs_k = [mv for mv in my_precvs if mv[2] == k
                       and C.lnumel]
            if sends_k or recvs_k:
                check(lib.da_group_start())
                for (src, dst, kk) in sends_k:
                    nb = A.lshape[0] * (ccols[kk][1] - ccols[kk][0]) * esz
                    check(lib.da_send(partials[kk].p, nb, dst))
                for (src, dst, kk) in recvs_k:
                    nb = C.lshape[0] * C.lshape[1] * esz
                    check(lib.da_recv(precv[(src, kk)].p, nb, src))
                check(lib.da_group_end())
    if overlap:
        check(lib.da_main_after_comm())
        check(lib.da_p2p_stream(0))
    else:
        psends = [mv for mv in my_psends
                  if A.lshape[0] * (ccols[mv[2]][1] - ccols[mv[2]][0])]
        precvs = [mv for mv in my_precvs if C.lnumel]
        if psends or precvs:
            check(lib.da_group_start())
            for (src, dst, k) in psends:
                nb = A.lshape[0] * (ccols[k][1] - ccols[k][0]) * esz
                check(lib.da_send(partials[k].p, nb, dst))
            for (src, dst, k) in precvs:
                buf = precv[(src, k)]
                nb = C.lshape[0] * C.lshape[1] * esz
                check(lib.da_recv(buf.p, nb, src))
            check(lib.da_group_end())
    if r < I * K and C.lnumel:
        i, myk = r % I, r // I
        for j in plan.accumulate_order(J):
            srcrank = i + I * j
            if srcrank == r:
                check(lib.da_add(C._ptr(), partials[myk].p, float(alpha),
                                 C.lnumel, DTYPES[A.dtype]))
            else:
                buf = precv[(srcrank, myk)]
                check(lib.da_add(C._ptr(), buf.p, float(alpha),
                                 C.lnumel, DTYPES[A.dtype]))

    check(lib.da_synchronize())
    for _, b in sendbufs:
        b.free()
    for _, b in recvbufs:
        b.free()
    for b in precv.values():
        b.free()
    for b in partials:
        b.free()
    if slab is not None:
        slab.free()
    return C


# ------------------------------------------------------- dims-reductions
_RED2MAP2 = {"add": "add", "mul": "mul", "min": "min2", "max": "max2"}


def dreduce_dims(f, op, d, dims):
    """mapreduce(f, op, D; dims=dims) — src/mapreduce.jl:42-94:
    per-chunk kernel reduction (mapreducedim_within), then cross-chunk
    combine onto the lowest-coordinate slab owners
    (reducedim_initarray stores R on A.pids[region -> 1:1],
    mapreducedim_between! pulls co-slabs and folds; here: grouped
    ncclSend/Recv of partial slabs + ascending-source-rank elementwise
    combine).  Result keeps reduced dims as size 1 (Julia keepdims)."""
    if isinstance(dims, int):
        dims = (dims,)
    dims = tuple(sorted(set(int(a) for a in dims)))
    nd = d.ndims
    if any(a < 0 or a >= nd for a in dims):
        raise DArrayError("dreduce_dims: bad dims %r" % (dims,))
    if d.ranks != list(range(d.nchunks)):
        raise DArrayError("dreduce_dims: source must have identity ranks")
    dt = DTYPES[d.dtype]
    esz = DTYPE_SIZE[d.dtype]

    rdims = tuple(1 if a in dims else d.dims[a] for a in range(nd))
    rdist = tuple(1 if a in dims else d.dist[a] for a in range(nd))
    # owner of R chunk c = D-grid rank with reduced coords 0
    nr_chunks = 1
    for c in rdist:
        nr_chunks *= c
    owners = []
    for lin in range(nr_chunks):
        sub = list(geometry.grid_pos(lin, rdist))
        owners.append(geometry.grid_rank(sub, d.dist))
    R = DArray(rdims, d.dtype, rdist, ranks=owners)

    me = d.rank
    partial = None
    pshape = None
    if d.lchunk is not None:
        # sequential per-axis reduction; mapf only on the first pass
        shape = list(d.lshape)
        src_ptr = d._ptr()
        cur = None
        first = True
        for a in dims:
            inner = 1
            for x in shape[:a]:
                inner *= x
            outer = 1
            for x in shape[a + 1:]:
                outer *= x
            axis = shape[a]
            nxt = _Buf(max(inner * outer, 1) * esz)
            check(lib.da_reduce_dims(RED_FS[f if first else "identity"],
                                     RED_OPS[op], src_ptr, inner, axis,
                                     outer, dt, nxt.p))
            if cur is not None:
                cur.free()
            cur = nxt
            src_ptr = cur.p
            shape[a] = 1
            first = False
        partial = cur
        pshape = tuple(shape)

    # exchange: group D-ranks by owner (ascending rank == ascending
    # column-major reduced coords, the co-slab concatenation order of
    # mapreducedim_between!, mapreduce.jl:71-81)
    groups = {}
    for src in range(d.nchunks):
        sub = list(geometry.grid_pos(src, d.dist))
        for a in dims:
            sub[a] = 0
        groups.setdefault(geometry.grid_rank(sub, d.dist), []).append(src)

    my_owner_group = groups.get(me) if R.lchunk is not None else None
    sends = []
    recvs = {}
    if d.lchunk is not None:
        my_group_owner = None
        sub = list(geometry.grid_pos(me, d.dist))
        for a in dims:
            sub[a] = 0
        my_group_owner = geometry.grid_rank(sub, d.dist)
        if my_group_owner != me:
            sends.append((my_group_owner, partial))
    if my_owner_group:
        for src in my_owner_group:
            if src != me:
                recvs[src] = _Buf(max(R.lnumel, 1) * esz)
    if sends or recvs:
        check(lib.da_group_start())
        for (dst, buf) in sends:
            check(lib.da_send(buf.p, max(d2_numel(pshape), 1) * esz, dst))
        for src, buf in recvs.items():
            check(lib.da_recv(buf.p, max(R.lnumel, 1) * esz, src))
        check(lib.da_group_end())

    if my_owner_group:
        # fold ascending source rank; owner (reduced coords 0) is first
        first_src = my_owner_group[0]
        if first_src == me:
            if R.lnumel:
                check(lib.da_d2d(R._ptr(), partial.p, R.lnumel * esz))
        else:
            if R.lnumel:
                check(lib.da_d2d(R._ptr(), recvs[first_src].p,
                                 R.lnumel * esz))
        for src in my_owner_group[1:]:
            buf = partial if src == me else recvs[src]
            if R.lnumel:
                check(lib.da_map2(MAP2_OP[_RED2MAP2[op]], R._ptr(),
                                  R._ptr(), buf.p, R.lnumel, dt))
    check(lib.da_synchronize())
    if partial is not None:
        partial.free()
    for buf in recvs.values():
        buf.free()
    return R


def d2_numel(shape):
    n = 1
    for x in shape:
        n *= x
    return n


def dsum_dims(d, dims):
    return dreduce_dims("identity", "add", d, dims)


def dprod_dims(d, dims):
    return dreduce_dims("identity", "mul", d, dims)


def dmaximum_dims(d, dims):
    return dreduce_dims("identity", "max", d, dims)


def dminimum_dims(d, dims):
    return dreduce_dims("identity", "min", d, dims)


def dmean_dims(d, dims):
    """mean(D; dims) rides the sum path (ext/StatisticsExt.jl:6)."""
    if isinstance(dims, int):
        dims = (dims,)
    dims = tuple(sorted(set(int(a) for a in dims)))  # match dreduce_dims
    R = dsum_dims(d, dims)
    nred = 1
    for a in dims:
        nred *= d.dims[a]
    scale_(R, 1.0 / nred)
    return R


# ---------------------------------------------------------------- matvec
def dmatvec(A, x, alpha=1.0):
    """y = alpha * A * x — mul!(y, A, x) (linalg.jl:78-122): each rank
    (i,j) multiplies its block by the x-slice of its column cut
    (linalg.jl:91: xj shipped per tile), partial vectors travel to the
    y owner (rank i = procs(A)[i,1]) and accumulate ascending j."""
    import numpy as np
    if A.dtype not in ("f64", "f32") or A.ndims != 2:
        raise DArrayError("dmatvec: 2-D f64/f32 only")
    if A.ranks != list(range(A.nchunks)):
        raise DArrayError(
            "dmatvec: A must have identity chunk->rank mapping "
            "(copy() a reduction result first)")
    gemm_fn = lib.da_gemm_f64 if A.dtype == "f64" else lib.da_gemm_f32
    x = np.ascontiguousarray(np.asarray(x,
                             dtype=NUMPY_DTYPES[A.dtype]))
    if x.shape != (A.dims[1],):
        raise DArrayError("dmatvec: x length %d != %d"
                          % (x.shape[0], A.dims[1]))
    m = A.dims[0]
    I, J = A.dist
    y = DArray((m,), A.dtype, (I,))
    y.fill_(0.0)
    r = A.rank
    esz = DTYPE_SIZE[A.dtype]
    partial = None
    if A.lchunk is not None and A.lnumel:
        i, j = r % I, r // I
        jlo, jhi = A.lidx[1]
        xj = x[jlo:jhi]
        xbuf = _Buf(max(xj.size, 1) * esz)
        check(lib.da_h2d(xbuf.p, xj.ctypes.data_as(ctypes.c_void_p),
                         xj.size * esz))
        mloc, kloc = A.lshape
        partial = _Buf(mloc * esz)
        check(gemm_fn(partial.p, A._ptr(), xbuf.p,
                      mloc, 1, kloc, mloc, kloc, mloc, 1.0, 0.0))
        xbuf.free()

    sends, recvs = [], {}
    if A.lchunk is not None and A.lnumel:
        i, j = r % I, r // I
        if j != 0:
            sends.append((i, partial))
    if r < I and A.nchunks > I:
        for j in range(1, J):
            src = r + I * j
            recvs[src] = _Buf(max(y.lnumel, 1) * esz)
    if sends or recvs:
        check(lib.da_group_start())
        for dst, buf in sends:
            check(lib.da_send(buf.p, A.lshape[0] * esz, dst))
        for src, buf in recvs.items():
            check(lib.da_recv(buf.p, y.lnumel * esz, src))
        check(lib.da_group_end())
    if r < I and y.lnumel:
        for j in range(J):
            src = r + I * j
            buf = partial if src == r else recvs[src]
            check(lib.da_add(y._ptr(), buf.p, float(alpha), y.lnumel,
                             DTYPES[A.dtype]))
    check(lib.da_synchronize())
    if partial is not None:
        partial.free()
    for buf in recvs.values():
        buf.free()
    return y


def dmatvec_adj(A, x, alpha=1.0):
    """y = alpha * A' * x — mul!(y, adjoint(A), x) (linalg.jl:124-167):
    rank (i,j) multiplies its block TRANSPOSED by the x-slice of its ROW
    cut (xj = x[A.cuts[1][j]-range], linalg.jl:139), partial vectors of
    y-block j travel to rank j (y.cuts == A.cuts[2], linalg.jl:134) and
    accumulate ascending i.  Real dtypes: adjoint == transpose, so this
    also serves mul!(y, transpose(A), x) (linalg.jl:169 analog).

    The local A_loc'*xj is one da_gemm_f64 with the vector as the 1-row
    left operand: C(1 x kloc) = x_row(1 x mloc) * A_loc(mloc x kloc)."""
    import numpy as np
    if A.dtype not in ("f64", "f32") or A.ndims != 2:
        raise DArrayError("dmatvec_adj: 2-D f64/f32 only")
    if A.ranks != list(range(A.nchunks)):
        raise DArrayError(
            "dmatvec_adj: A must have identity chunk->rank mapping")
    gemm_fn = lib.da_gemm_f64 if A.dtype == "f64" else lib.da_gemm_f32
    x = np.ascontiguousarray(np.asarray(x,
                             dtype=NUMPY_DTYPES[A.dtype]))
    if x.shape != (A.dims[0],):
        raise DArrayError("dmatvec_adj: x length %d != %d"
                          % (x.shape[0], A.dims[0]))
    I, J = A.dist
    y = DArray((A.dims[1],), A.dtype, (J,))
    y.fill_(0.0)
    r = A.rank
    esz = DTYPE_SIZE[A.dtype]
    partial = None
    kloc = 0
    if A.lchunk is not None and A.lnumel:
        ilo, ihi = A.lidx[0]
        xi = x[ilo:ihi]
        xbuf = _Buf(max(xi.size, 1) * esz)
        check(lib.da_h2d(xbuf.p, xi.ctypes.data_as(ctypes.c_void_p),
                         xi.size * esz))
        mloc, kloc = A.lshape
        partial = _Buf(max(kloc, 1) * esz)
        check(gemm_fn(partial.p, xbuf.p, A._ptr(),
                      1, kloc, mloc, 1, mloc, 1, 1.0, 0.0))
        xbuf.free()

    sends, recvs = [], {}
    if A.lchunk is not None and A.lnumel:
        i, j = r % I, r // I
        if j != r and kloc:          # my partial's owner is rank j
            sends.append((j, partial))
    if r < J and y.lnumel:
        for i in range(I):
            src = i + I * r
            if src != r:
                recvs[src] = _Buf(max(y.lnumel, 1) * esz)
    if sends or recvs:
        check(lib.da_group_start())
        for dst, buf in sends:
            check(lib.da_send(buf.p, kloc * esz, dst))
        for src, buf in recvs.items():
            check(lib.da_recv(buf.p, y.lnumel * esz, src))
        check(lib.da_group_end())
    if r < J and y.lnumel:
        for i in range(I):
            src = i + I * r
            buf = partial if src == r else recvs[src]
            check(lib.da_add(y._ptr(), buf.p, float(alpha), y.lnumel,
                             DTYPES[A.dtype]))
    check(lib.da_synchronize())
    if partial is not None:
        partial.free()
    for buf in recvs.values():
        buf.free()
    return y


# ------------------------------------------------- makelocal halo gather
def gather_box(A, boxes_all):
    """Collective makelocal (darray.jl:351-368): every rank passes the
    SAME boxes_all list (boxes_all[r] = the box rank r requests, derived
    from shared metadata, or None) and receives its own box gathered
    into a device buffer (column-major).  Returns (_Buf, shape) or
    (None, None) when this rank requested nothing.

    Fully-local requests degrade to on-device strided copies (the
    reference's zero-copy view case); remote pieces move as grouped
    ncclSend/Recv of packed sub-blocks over xGMI.  N-D: dims >= 3
    iterate per 2-D slice."""
    nd = A.ndims
    esz = DTYPE_SIZE[A.dtype]
    mybox = boxes_all[A.rank] if A.rank < len(boxes_all) else None
    pieces = plan.halo_plan(A.idxs, A.ranks, boxes_all)
    me = A.rank
    my_sends = [p for p in pieces if p[0] == me and p[1] != me]
    my_recvs = [p for p in pieces if p[1] == me and p[0] != me]
    my_local = [p for p in pieces if p[0] == me and p[1] == me]

    out = None
    oshape = None
    if mybox is not None:
        oshape = tuple(hi - lo for lo, hi in mybox)
        out = _Buf(max(geometry.nelems(mybox), 1) * esz)

    def nelems(box):
        n = 1
        for lo, hi in box:
            n *= hi - lo
        return n

    def copy_box(dst_base, dst_shape, dst_org, src_base, src_shape,
                 src_org, box):
        """Copy an N-D box between two column-major buffers; dst/src_org
        are the global coordinates of each buffer's origin.  Dims >= 3
        iterate in Python (one hipMemcpy2DAsync per 2-D slice)."""
        rows = box[0][1] - box[0][0]
        cols = (box[1][1] - box[1][0]) if nd >= 2 else 1
        def org_off(shape, org, outer_idx):
            off = box[0][0] - org[0]
            mul = shape[0]
            if nd >= 2:
                off += (box[1][0] - org[1]) * mul
                mul *= shape[1]
            for d in range(2, nd):
                off += (box[d][0] + outer_idx[d - 2] - org[d]) * mul
                mul *= shape[d]
            return off
        outer_dims = [box[d][1] - box[d][0] for d in range(2, nd)]
        idx = [0] * len(outer_dims)
        while True:
            soff = org_off(src_shape, src_org, idx) * esz
            doff = org_off(dst_shape, dst_org, idx) * esz
            _copy2d(ctypes.c_void_p(dst_base.value + doff),
                    dst_shape[0] * esz,
                    ctypes.c_void_p(src_base.value + soff),
                    src_shape[0] * esz, rows * esz, cols)
            k = 0
            while k < len(outer_dims):
                idx[k] += 1
                if idx[k] < outer_dims[k]:
                    break
                idx[k] = 0
                k += 1
            if k == len(outer_dims):
                break

    lorg = tuple(lo for lo, _ in A.lidx)

    sendbufs = []
    for (src, dst, box) in my_sends:
        buf = _Buf(nelems(box) * esz)
        bshape = tuple(hi - lo for lo, hi in box)
        borg = tuple(lo for lo, _ in box)
        copy_box(buf.p, bshape, borg, A._ptr(), A.lshape, lorg, box)
        sendbufs.append((box, buf, dst))
    recvbufs = []
    for (src, dst, box) in my_recvs:
        recvbufs.append((box, _Buf(nelems(box) * esz), src))
    if my_sends or my_recvs:
        check(lib.da_group_start())
        for box, buf, dst in sendbufs:
            check(lib.da_send(buf.p, nelems(box) * esz, dst))
        for box, buf, src in recvbufs:
            check(lib.da_recv(buf.p, nelems(box) * esz, src))
        check(lib.da_group_end())
    if mybox is not None:
        morg = tuple(lo for lo, _ in mybox)
        for (src, dst, box) in my_local:
            copy_box(out.p, oshape, morg, A._ptr(), A.lshape, lorg, box)
        for box, buf, src in recvbufs:
            bshape = tuple(hi - lo for lo, hi in box)
            borg = tuple(lo for lo, _ in box)
            copy_box(out.p, oshape, morg, buf.p, bshape, borg, box)
    check(lib.da_synchronize())
    for _, buf, _ in sendbufs:
        buf.free()
    for _, buf, _ in recvbufs:
        buf.free()
    return out, oshape


def _dest_boxes(dest):
    """boxes_all for 'each dest owner requests its own index box' — the
    makelocal(src, localindices(dest)) pattern (mapreduce.jl:8,
    broadcast.jl:79, linalg.jl:42)."""
    boxes = [None] * dest.nranks
    for c, r in enumerate(dest.ranks):
        boxes[r] = dest.idxs[c]
    return boxes


def dreshape(A, dims):
    """reshape(A::DVector, dims) — darray.jl:612-636: each chunk of the
    result pulls the source linear ranges backing its columns (the
    reference fetches A[a:a+nr-1] per column).  Here: ONE collective
    gather of each rank's bounding linear interval over xGMI, then a
    strided on-device unpack (da_copy2d per dim-1 slab) — at most a
    rows-span over-fetch for row-split grids, exact for column splits."""
    import numpy as np
    if A.ndims != 1:
        raise DArrayError("dreshape: 1-D source only (as the reference)")
    dims = tuple(int(x) for x in dims)
    total = 1
    for x in dims:
        total *= x
    if total != A.dims[0]:
        raise DArrayError("dreshape: dimensions must be consistent "
                          "with array size")
    out = DArray(dims, A.dtype)
    esz = DTYPE_SIZE[A.dtype]
    nd = len(dims)

    def lin(point):
        a, mul = 0, 1
        for d in range(nd):
            a += point[d] * mul
            mul *= dims[d]
        return a

    boxes = [None] * A.nranks
    spans = {}
    for c, r in enumerate(out.ranks):
        box = out.idxs[c]
        if geometry.nelems(box) == 0:
            continue
        lo = lin([b[0] for b in box])
        hi = lin([b[1] - 1 for b in box]) + 1
        boxes[r] = ((lo, hi),)
        spans[r] = (lo, hi)
    buf, _ = gather_box(A, boxes)
    if buf is not None and out.lnumel:
        g0 = spans[out.rank][0]
        (r0, r1) = out.lidx[0]
        nr = r1 - r0
        outer = [out.lidx[d] for d in range(2, nd)]
        o_sizes = [hi - lo for lo, hi in outer]
        idx = [0] * len(o_sizes)
        lshape = out.lshape
        while True:
            point = [r0, out.lidx[1][0] if nd > 1 else 0] + \
                [outer[k][0] + idx[k] for k in range(len(o_sizes))]
            src_off = lin(point[:nd]) - g0
            dst_off = 0
            mul = 1
            for d in range(2, nd):
                dst_off += idx[d - 2] * mul * lshape[0] * lshape[1]
                mul *= lshape[d]
            ncols = lshape[1] if nd > 1 else 1
            _copy2d(out.at_byte(dst_off * esz), lshape[0] * esz,
                    ctypes.c_void_p(buf.p.value + src_off * esz),
                    dims[0] * esz, nr * esz, ncols)
            k = 0
            while k < len(o_sizes):
                idx[k] += 1
                if idx[k] < o_sizes[k]:
                    break
                idx[k] = 0
                k += 1
            if not o_sizes or k == len(o_sizes):
                break
        check(lib.da_synchronize())
    if buf is not None:
        buf.free()
    return out


def redistribute(D, dist):
    """A new DArray with the same global content on a different chunk
    grid — the reference's re-distribution constructor pattern
    (mapslices builds one at mapreduce.jl:196-202 via the init
    constructor pulling D[I...]).  Device-resident: each new-chunk
    owner gathers its box over xGMI (gather_box); collective."""
    dist = tuple(int(c) for c in dist)
    out = DArray(D.dims, D.dtype, dist)
    boxes = [None] * D.nranks
    for c, r in enumerate(out.ranks):
        boxes[r] = out.idxs[c]
    buf, shape = gather_box(D, boxes)
    if buf is not None and out.lnumel:
        check(lib.da_d2d(out._ptr(), buf.p,
                         out.lnumel * DTYPE_SIZE[out.dtype]))
        check(lib.da_synchronize())
    if buf is not None:
        buf.free()
    return out


def dmapslices(f, D, dims):
    """mapslices(f, D; dims) — mapreduce.jl:191-208: if any sliced dim
    is distributed, redistribute so slices are whole per chunk (the
    reference builds a DD with p[dims]=1), then apply numpy's
    apply-over-slices on each localpart (host boundary, like
    map_localparts) and assemble the result DArray on the non-sliced
    chunk grid."""
    import numpy as np
    if isinstance(dims, int):
        dims = (dims,)
    dims = tuple(sorted(set(int(a) for a in dims)))
    nd = D.ndims
    if any(a < 0 or a >= nd for a in dims):
        raise DArrayError("dmapslices: bad dims %r" % (dims,))
    if any(D.dist[a] != 1 for a in dims):
        nondims = [a for a in range(nd) if a not in dims]
        p = [1] * nd
        sub = geometry.defaultdist([D.dims[a] for a in nondims],
                                   D.nranks)
        for a, c in zip(nondims, sub):
            p[a] = c
        DD = redistribute(D, p)
        try:
            return dmapslices(f, DD, dims)
        finally:
            DD.close()

    lp = D.localpart()

    def apply_local(arr):
        # evaluate f on every dims-slice of arr, stack the results in
        # place of the sliced dims (Julia mapslices semantics for
        # same-shape outputs; general reshaping follows Base)
        nondims = [a for a in range(nd) if a not in dims]
        it_shape = [arr.shape[a] for a in nondims]
        probe_idx = [slice(None)] * nd
        for a in nondims:
            probe_idx[a] = 0
        r1 = np.asarray(f(np.asarray(arr[tuple(probe_idx)])))
        out_shape = list(arr.shape)
        rs = list(r1.shape) + [1] * (len(dims) - r1.ndim)
        for a, s in zip(dims, rs):
            out_shape[a] = s
        out = np.empty(out_shape, dtype=r1.dtype, order="F")
        for pos in np.ndindex(*it_shape):
            idx = [slice(None)] * nd
            for a, v in zip(nondims, pos):
                idx[a] = v
            out[tuple(idx)] = np.asarray(
                f(np.asarray(arr[tuple(idx)]))).reshape(
                [out_shape[a] for a in dims], order="A")
        return out

    local = (apply_local(lp) if D.lnumel else
             np.empty([0] * nd, order="F"))
    # result dims: sliced dims take f's output size (agreed across
    # ranks via the shared metadata — f must be shape-uniform, as in
    # the reference), non-sliced dims keep D's
    if D.lnumel:
        out_sizes = list(local.shape)
    else:
        out_sizes = [0] * nd
    import torch.distributed as td
    if D.nranks > 1 and td.is_initialized():
        lst = [None] * D.nranks
        td.all_gather_object(lst, out_sizes if D.lnumel else None)
        out_sizes = next(s for s in lst if s is not None)
    rdims = tuple(out_sizes[a] if a in dims else D.dims[a]
                  for a in range(nd))
    R = DArray(rdims, D.dtype, D.dist, ranks=list(D.ranks))
    if R.lnumel:
        R.set_localpart(np.asfortranarray(
            np.asarray(local, dtype=np.dtype(NUMPY_DTYPES[D.dtype]))))
    return R


def dppeval(f, *Ds, dim=None):
    """ppeval(f, D...; dim) — mapreduce.jl:258-323: evaluate f on the
    dim-slices of each argument (default: last dim for DArrays, whole
    array broadcast for numpy arguments), stacking results along a new
    last dimension distributed like the first DArray's sliced dim.
    Host boundary (arbitrary f), like the reference's per-worker
    _ppeval on localparts."""
    import numpy as np
    if not Ds or not isinstance(Ds[0], DArray):
        raise DArrayError("dppeval: first argument must be a DArray "
                          "(procs(D[1]) in the reference)")
    # dim entries are 1-based like the reference; <= 0 means the
    # argument is broadcast whole to every evaluation (mapreduce.jl
    # ppeval docstring)
    if dim is None:
        dax = [a.ndims - 1 if isinstance(a, DArray) else -1 for a in Ds]
        sliced = [isinstance(a, DArray) for a in Ds]
    else:
        dax = [d - 1 for d in dim]
        sliced = [isinstance(a, DArray) and d > 0
                  for a, d in zip(Ds, dim)]
    if not sliced[0]:
        raise DArrayError("dppeval: the first DArray must be sliced")
    first = Ds[0]
    for a, d, sl in zip(Ds, dax, sliced):
        if isinstance(a, DArray):
            for ax in range(a.ndims):
                if (ax != d or not sl) and a.dist[ax] != 1:
                    raise DArrayError(
                        "dppeval: dimension %d is distributed; must be "
                        "whole per chunk" % ax)
    locals_ = [a.localpart() if isinstance(a, DArray) else np.asarray(a)
               for a in Ds]

    def slice_at(arr, d, i, is_sliced):
        if not is_sliced:
            return arr
        idx = [slice(None)] * arr.ndim
        idx[d] = i
        return arr[tuple(idx)]

    n_loc = locals_[0].shape[dax[0]]
    outs = []
    for i in range(n_loc):
        args = [slice_at(lp, d, i, sl)
                for lp, d, sl in zip(locals_, dax, sliced)]
        outs.append(np.asarray(f(*args)))
    if outs:
        local = np.stack(outs, axis=-1)
    else:
        local = np.empty((0,), order="F")
    # result: (fshape..., global dimlen), distributed along the last
    # dim with the first DArray's cuts on its sliced dim
    fshape = list(outs[0].shape) if outs else []
    import torch.distributed as td
    if first.nranks > 1 and td.is_initialized():
        lst = [None] * first.nranks
        td.all_gather_object(lst, fshape if outs else None)
        fshape = next(s for s in lst if s is not None)
    gdim = first.dims[dax[0]]
    nchunks = first.dist[dax[0]]
    rdims = tuple(fshape) + (gdim,)
    rdist = tuple([1] * len(fshape)) + (nchunks,)
    R = DArray(rdims, first.dtype, rdist, ranks=list(first.ranks))
    if R.lnumel:
        R.set_localpart(np.asfortranarray(
            np.asarray(local, dtype=np.dtype(NUMPY_DTYPES[R.dtype]))
            .reshape(R.lshape, order="A")))
    return R


def dgetindex(A, *ranges):
    """D[I...] for range indexing -> numpy array on every rank
    (Array(view(A, I...)), the makelocal contract darray.jl:346-368;
    collective: rank 0's box request is the union, gathered via
    gather_box then broadcast on the control plane)."""
    import numpy as np
    box = tuple((lo, hi) for lo, hi in ranges)
    boxes = [None] * A.nranks
    boxes[A.rank] = box          # every rank requests the same box
    for rr in range(A.nranks):
        boxes[rr] = box
    buf, shape = gather_box(A, boxes)
    out = np.empty(shape, dtype=np.dtype(NUMPY_DTYPES[A.dtype]),
                   order="F")
    if buf is not None and out.size:
        check(lib.da_d2h(buf.p, out.ctypes.data_as(ctypes.c_void_p),
                         out.size * DTYPE_SIZE[A.dtype]))
    if buf is not None:
        buf.free()
    return out


def map_general(op, dest, src):
    """map!(f, dest, src) for MISMATCHED cuts (mapreduce.jl:5-12 with
    makelocal): gather src's piece of dest's index box, then map."""
    if dest.dims != src.dims or dest.dtype != src.dtype:
        raise DArrayError("map_general: dims/dtype mismatch")
    if _same_layout(dest, src):
        return map_(op, dest, src)
    buf, shape = gather_box(src, _dest_boxes(dest))
    if buf is not None:
        check(lib.da_map(MAP_OP[op], dest._ptr(), buf.p, dest.lnumel,
                         DTYPES[dest.dtype]))
        check(lib.da_synchronize())
        buf.free()
    return dest


def map2_general(op, dest, a, b):
    """dest .= f.(a, b) with arbitrary (mismatched) cuts — per-operand
    bclocal/makelocal localisation (broadcast.jl:140-152): only the
    operands whose layout differs from dest's are gathered."""
    if (dest.dims != a.dims or dest.dims != b.dims
            or dest.dtype != a.dtype or dest.dtype != b.dtype):
        raise DArrayError("map2_general: dims/dtype mismatch")
    if _same_layout(dest, a) and _same_layout(dest, b):
        return map2_(op, dest, a, b)
    boxes = _dest_boxes(dest)
    abuf = bbuf = None
    if not _same_layout(dest, a):
        abuf, _ = gather_box(a, boxes)
    if not _same_layout(dest, b):
        bbuf, _ = gather_box(b, boxes)
    if dest.lnumel:
        ap = abuf.p if abuf is not None else a._ptr()
        bp = bbuf.p if bbuf is not None else b._ptr()
        check(lib.da_map2(MAP2_OP[op], dest._ptr(), ap, bp,
                          dest.lnumel, DTYPES[dest.dtype]))
    check(lib.da_synchronize())
    if abuf is not None:
        abuf.free()
    if bbuf is not None:
        bbuf.free()
    return dest


def broadcast_fma_general(dest, a, b, c):
    """D .= A .* B .+ c with arbitrary (mismatched) cuts — the
    bclocal/makelocal localisation of broadcast.jl:65-85."""
    if _same_layout(dest, a) and _same_layout(dest, b):
        return broadcast_fma(dest, a, b, c)
    boxes = _dest_boxes(dest)
    abuf = bbuf = None
    if not _same_layout(dest, a):
        abuf, _ = gather_box(a, boxes)
    if not _same_layout(dest, b):
        bbuf, _ = gather_box(b, boxes)
    if dest.lnumel:
        ap = abuf.p if abuf is not None else a._ptr()
        bp = bbuf.p if bbuf is not None else b._ptr()
        check(lib.da_bcast_fma(dest._ptr(), ap, bp, float(c),
                               dest.lnumel, DTYPES[dest.dtype]))
    check(lib.da_synchronize())
    if abuf is not None:
        abuf.free()
    if bbuf is not None:
        bbuf.free()
    return dest


# ------------------------------------------------------------ samplesort
def dsort(d, samples_per_rank=64):
    """sort(::DVector) — the distributed samplesort of src/sort.jl:103-170,
    MI355X-native: per-chunk rocPRIM radix sort, evenly-spaced samples
    gathered to agree on P-1 splitters (control plane; the reference
    gathers samples to the caller the same way), device binary-search
    boundaries, all-to-all of the contiguous sorted segments over xGMI,
    and a local re-sort of the received runs.  Returns a new DVector
    whose (possibly ragged) distribution follows the splitters."""
    import numpy as np
    if d.ndims != 1:
        raise DArrayError("dsort: DVector only")
    esz = DTYPE_SIZE[d.dtype]
    npdt = np.dtype(NUMPY_DTYPES[d.dtype])
    P = d.nranks
    if P == 1:
        out = d.similar()
        check(lib.da_sort_out(d._ptr(), out._ptr(), d.lnumel,
                              DTYPES[d.dtype]))
        check(lib.da_synchronize())
        return out
    out = d.copy()
    if out.lnumel > 1:
        check(lib.da_sort(out._ptr(), out.lnumel, DTYPES[d.dtype]))

    import torch.distributed as td
    if not td.is_initialized():
        raise DArrayError("dsort with nranks>1 needs gloo control plane")

    # evenly spaced samples from the sorted chunk (sort.jl sampling)
    n_loc = out.lnumel
    s = min(samples_per_rank, max(n_loc, 0))
    samples = np.empty(s, dtype=npdt)
    if s:
        idx = ((np.arange(s) + 0.5) * n_loc / s).astype(np.int64)
        # strided device->host via copy2d into a packed staging buffer
        stage = _Buf(max(s, 1) * esz)
        for t, i in enumerate(idx):   # s is small (<=64): cheap copies
            check(lib.da_d2d(stage.at(t * esz),
                             ctypes.c_void_p(out._ptr().value
                                             + int(i) * esz), esz))
        check(lib.da_d2h(stage.p, samples.ctypes.data_as(ctypes.c_void_p),
                         s * esz))
        stage.free()
    gathered = [None] * P
    td.all_gather_object(gathered, samples)
    allsamp = np.sort(np.concatenate([g for g in gathered if g is not None]))
    if allsamp.size < P:
        splitters = allsamp[:max(P - 1, 0)]
    else:
        splitters = allsamp[[(i + 1) * allsamp.size // P
                             for i in range(P - 1)]]
    if splitters.size < P - 1:   # tiny inputs: trailing segments empty
        pad = (np.inf if npdt.kind == "f" else np.iinfo(npdt).max)
        splitters = np.concatenate(
            [splitters, np.full(P - 1 - splitters.size, pad, npdt)])

    # boundaries in my sorted chunk
    bounds = (ctypes.c_uint64 * max(P - 1, 1))()
    if n_loc and P > 1:
        spl = np.ascontiguousarray(splitters, dtype=npdt)
        check(lib.da_lower_bound(out._ptr(), n_loc, DTYPES[d.dtype],
                                 spl.ctypes.data_as(ctypes.c_void_p),
                                 P - 1, bounds))
    edges = [0] + [int(bounds[i]) for i in range(P - 1)] + [n_loc]

    # segment sizes: seg[j] goes to rank j; exchange counts via gloo
    segs = [edges[j + 1] - edges[j] for j in range(P)]
    allsegs = [None] * P
    td.all_gather_object(allsegs, segs)
    recv_sizes = [allsegs[src][d.rank] for src in range(P)]
    my_total = sum(recv_sizes)

    res = DArray.from_chunk_sizes(
        [sum(allsegs[src][dst] for src in range(P)) for dst in range(P)],
        d.dtype)
    assert res.lnumel == my_total
    # grouped all-to-all of contiguous sorted segments
    recvbufs = {}
    check(lib.da_group_start())
    for dst in range(P):
        if dst != d.rank and segs[dst]:
            check(lib.da_send(
                ctypes.c_void_p(out._ptr().value + edges[dst] * esz),
                segs[dst] * esz, dst))
    for src in range(P):
        if src != d.rank and recv_sizes[src]:
            buf = _Buf(recv_sizes[src] * esz)
            recvbufs[src] = buf
            check(lib.da_recv(buf.p, recv_sizes[src] * esz, src))
    check(lib.da_group_end())
    # concatenate runs (own + received) then radix re-sort
    off = 0
    for src in range(P):
        nsz = recv_sizes[src]
        if not nsz:
            continue
        if src == d.rank:
            check(lib.da_d2d(res.at_byte(off),
                             ctypes.c_void_p(out._ptr().value
                                             + edges[d.rank] * esz),
                             nsz * esz))
        else:
            check(lib.da_d2d(res.at_byte(off), recvbufs[src].p, nsz * esz))
        off += nsz * esz
    if res.lnumel > 1:
        check(lib.da_sort(res._ptr(), res.lnumel, DTYPES[d.dtype]))
    check(lib.da_synchronize())
    for buf in recvbufs.values():
        buf.free()
    out.close()
    return res


# ------------------------------------------- transpose / Diagonal scaling
def dtranspose(D):
    """copy(transpose(D)) — linalg.jl:10-17: the result DArray's chunk
    (I) gathers D[reverse(I)] (remote gather -> gather_box over xGMI)
    and transposes locally (LDS-tiled kernel)."""
    import numpy as np
    if D.ndims != 2:
        raise DArrayError("dtranspose: 2-D only")
    out = DArray((D.dims[1], D.dims[0]), D.dtype)
    boxes = [None] * D.nranks
    for c, r in enumerate(out.ranks):
        (rlo, rhi), (clo, chi) = out.idxs[c]
        boxes[r] = ((clo, chi), (rlo, rhi))   # reversed into D's coords
    buf, shape = gather_box(D, boxes)
    if buf is not None and out.lnumel:
        check(lib.da_transpose(out._ptr(), buf.p, shape[0], shape[1],
                               DTYPES[D.dtype]))
        check(lib.da_synchronize())
    if buf is not None:
        buf.free()
    return out


def _diag_scale(D, dvec, side):
    import numpy as np
    dvec = np.ascontiguousarray(np.asarray(dvec,
                                dtype=NUMPY_DTYPES[D.dtype]))
    dim = 0 if side == 0 else 1
    if D.ndims != 2 or dvec.shape != (D.dims[dim],):
        raise DArrayError("diag scale: need 2-D DArray and matching diag")
    if D.lnumel:
        lo, hi = D.lidx[dim]
        sl = np.ascontiguousarray(dvec[lo:hi])
        buf = _Buf(max(sl.nbytes, 1))
        check(lib.da_h2d(buf.p, sl.ctypes.data_as(ctypes.c_void_p),
                         sl.nbytes))
        check(lib.da_diag_scale(D._ptr(), D.lshape[0], D.lshape[1],
                                buf.p, side, DTYPES[D.dtype]))
        check(lib.da_synchronize())
        buf.free()
    return D


def ddiag_lmul(dvec, D):
    """lmul!(Diagonal(d), DA) — linalg.jl:169-177 (row scaling; each
    rank receives only its slice of d, the DestinationSerializer
    pattern)."""
    return _diag_scale(D, dvec, 0)


def ddiag_rmul(D, dvec):
    """rmul!(DA, Diagonal(d)) — linalg.jl:179-187 (column scaling)."""
    return _diag_scale(D, dvec, 1)
