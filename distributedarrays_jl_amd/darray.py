"""DArray — the distributed-array type, SPMD over one rank per MI355X.

Mirrors the reference's type and constructor surface
(/root/reference/src/darray.jl):
  DArray{T,N,A}              darray.jl:25-55   -> DArray (localpart in HBM)
  DArray(init, dims, ...)    darray.jl:76-174  -> DArray(init=..., ...)
  dzeros/dones/dfill         darray.jl:468-494
  drand/drandn               darray.jl:502-532 (philox per BASELINE.md)
  distribute                 darray.jl:544-555
  localpart/localindices     darray.jl:330-337, :394-400
  close / d_closeall         darray.jl:46-49, core.jl:67-103
  fill!/rand!                darray.jl:822-834

Execution model: every rank runs the same program (SPMD); metadata
(dims, cuts, indices) is computed identically everywhere; only the local
chunk lives on this rank's GPU.  Collective ops must be called by all
ranks.  There is no CPU fallback: all compute goes through
libdarray_hip.so (fails loudly without a GPU).
"""
import numpy as np

from . import _ffi, comm, geometry
from ._ffi import check, lib
from ._opcodes import DTYPES, NUMPY_DTYPES, DTYPE_SIZE, RAND_KINDS
import ctypes

_registry = {}
_next_id = [0]
_allow_scalar = [True]


def allowscalar(flag):
    """allowscalar(flag) — darray.jl:638-640: gate element-wise scalar
    indexing (the reference's tests disable it to force distributed
    fast paths)."""
    _allow_scalar[0] = bool(flag)


def _auto_init():
    if not comm.initialized():
        comm.init()


class DArray:
    """Block-distributed dense array; localpart is an HBM chunk."""

    def __init__(self, dims, dtype="f64", dist=None, init=None,
                 ranks=None, _alloc=True):
        _auto_init()
        rank, nr = comm.rank_info()
        dims = tuple(int(d) for d in dims)
        if dist is None:
            dist = geometry.defaultdist(dims, nr)
        dist = tuple(int(c) for c in dist)
        np_chunks = 1
        for c in dist:
            np_chunks *= c
        if np_chunks > nr:
            raise ValueError("dist %r needs %d ranks, have %d"
                             % (dist, np_chunks, nr))
        self.dims = dims
        self.dist = dist
        self.dtype = dtype
        self.rank = rank
        self.nranks = nr
        self.nchunks = np_chunks
        self.idxs, self.cuts = geometry.chunk_indices(dims, dist)
        # chunk c is owned by ranks[c] (default identity) — mirrors the
        # reference's pids array (e.g. reducedim_initarray stores the
        # result on A.pids[region -> 1:1], mapreduce.jl:42-50)
        if ranks is None:
            ranks = list(range(np_chunks))
        if len(ranks) != np_chunks:
            raise ValueError("ranks must list one owner per chunk")
        self.ranks = ranks
        self.id = _next_id[0]
        _next_id[0] += 1
        self._chunk = None
        self.lchunk = ranks.index(rank) if rank in ranks else None
        if self.lchunk is not None:
            self.lidx = self.idxs[self.lchunk]
            self.lshape = geometry.shape_of(self.lidx)
            self.lnumel = geometry.nelems(self.lidx)
        else:
            self.lidx = tuple((0, 0) for _ in dims)
            self.lshape = tuple(0 for _ in dims)
            self.lnumel = 0
        if _alloc:
            p = ctypes.c_void_p()
            check(lib.da_alloc(max(self.lnumel, 1) * DTYPE_SIZE[dtype],
                               DTYPES[dtype], ctypes.byref(p)))
            self._chunk = p
            _registry[self.id] = self
        if init is not None:
            arr = init(self.lidx)
            arr = np.asfortranarray(arr,
                                    dtype=np.dtype(NUMPY_DTYPES[dtype]))
            if tuple(arr.shape) != self.lshape:
                raise ValueError("init returned shape %r, want %r"
                                 % (arr.shape, self.lshape))
            self.set_localpart(arr)

    @classmethod
    def from_chunk_sizes(cls, sizes, dtype="f64"):
        """1-D DArray with explicit (possibly ragged) per-rank chunk
        sizes — the samplesort result shape (sort.jl returns a DArray
        whose distribution follows the splitter boundaries)."""
        _auto_init()
        rank, nr = comm.rank_info()
        if len(sizes) != nr:
            raise ValueError("need one size per rank")
        total = sum(sizes)
        d = cls((total,), dtype, (nr,), _alloc=False)
        # override the even cuts with the ragged ones (1-based, ref style)
        cuts = [1]
        for s in sizes:
            cuts.append(cuts[-1] + int(s))
        d.cuts = [cuts]
        d.idxs = [((cuts[i] - 1, cuts[i + 1] - 1),) for i in range(nr)]
        d.lchunk = rank
        d.lidx = d.idxs[rank]
        d.lshape = (int(sizes[rank]),)
        d.lnumel = int(sizes[rank])
        p = ctypes.c_void_p()
        check(lib.da_alloc(max(d.lnumel, 1) * DTYPE_SIZE[dtype],
                           DTYPES[dtype], ctypes.byref(p)))
        d._chunk = p
        _registry[d.id] = d
        return d

    # ---- lifetime (darray.jl:46-49 -> core.jl:67-103) ----
    def close(self):
        if self._chunk is not None:
            check(lib.da_free(self._chunk))
            self._chunk = None
            _registry.pop(self.id, None)

    def __del__(self):
        try:
            if self._chunk is not None and lib is not None:
                lib.da_free(self._chunk)
                self._chunk = None
                _registry.pop(self.id, None)
        except Exception:
            pass

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    # ---- metadata ----
    @property
    def shape(self):
        return self.dims

    @property
    def size(self):
        n = 1
        for d in self.dims:
            n *= d
        return n

    @property
    def ndims(self):
        return len(self.dims)

    def samedist(self, other):
        """Identical chunk layout: same boxes AND owners (comparing idxs,
        not dist — a ragged from_chunk_sizes DVector shares dist with an
        evenly-cut one but not the cut points)."""
        return (self.dims == other.dims and self.dtype == other.dtype
                and list(self.idxs) == list(other.idxs)
                and self.ranks == other.ranks)

    def _ptr(self):
        if self._chunk is None:
            raise _ffi.DArrayError("use after close (darray id %d)" % self.id)
        return self._chunk

    def at_byte(self, off):
        return ctypes.c_void_p(self._ptr().value + int(off))

    # ---- localpart access (darray.jl:330-337, :394-400) ----
    def localindices(self):
        return self.lidx

    def localpart(self):
        """D2H copy of the local chunk as a numpy array (column-major)."""
        out = np.empty(self.lshape,
                       dtype=np.dtype(NUMPY_DTYPES[self.dtype]), order="F")
        if self.lnumel:
            check(lib.da_d2h(self._ptr(), out.ctypes.data_as(ctypes.c_void_p),
                             self.lnumel * DTYPE_SIZE[self.dtype]))
        return out

    def set_localpart(self, arr):
        """H2D upload into the local chunk (d[:L] = v, darray.jl:378-382)."""
        arr = np.asfortranarray(arr, dtype=np.dtype(NUMPY_DTYPES[self.dtype]))
        if tuple(arr.shape) != self.lshape:
            raise ValueError("shape mismatch")
        if self.lnumel:
            check(lib.da_h2d(self._ptr(),
                             arr.ctypes.data_as(ctypes.c_void_p),
                             self.lnumel * DTYPE_SIZE[self.dtype]))
        return self

    # ---- device-side content ops (darray.jl:822-834) ----
    def fill_(self, v):
        if self.dtype == "i64" and abs(int(v)) > (1 << 53):
            # the ABI carries the fill value as a double
            raise _ffi.DArrayError(
                "fill_: |i64 value| > 2^53 not representable through the "
                "double fill parameter")
        if self.lnumel:
            check(lib.da_fill(self._ptr(), float(v), self.lnumel,
                              DTYPES[self.dtype]))
        return self

    def rand_(self, kind="uniform", seed_base=1234):
        """Per-rank philox seed = seed_base + rank (BASELINE.md protocol,
        mirroring test/runtests.jl:23)."""
        if self.lnumel:
            check(lib.da_rand(self._ptr(), self.lnumel, DTYPES[self.dtype],
                              seed_base + self.rank, RAND_KINDS[kind], 0))
        return self

    def similar(self, dtype=None):
        """A new uninitialized DArray with THIS array's exact layout:
        same cut boxes (incl. ragged from_chunk_sizes cuts) and the
        same chunk->rank owners — so elementwise ops on the result stay
        aligned and local (Julia's similar preserves the
        distribution)."""
        d = DArray(self.dims, dtype or self.dtype, self.dist,
                   ranks=list(self.ranks), _alloc=False)
        d.cuts = [list(c) for c in self.cuts]
        d.idxs = list(self.idxs)
        d.lchunk = self.lchunk
        d.lidx = self.lidx
        d.lshape = self.lshape
        d.lnumel = self.lnumel
        p = ctypes.c_void_p()
        check(lib.da_alloc(max(d.lnumel, 1) * DTYPE_SIZE[d.dtype],
                           DTYPES[d.dtype], ctypes.byref(p)))
        d._chunk = p
        _registry[d.id] = d
        return d

    def copy(self):
        out = self.similar()
        if self.lnumel:
            check(lib.da_d2d(out._ptr(), self._ptr(),
                             self.lnumel * DTYPE_SIZE[self.dtype]))
        return out

    # ---- conversion (Array(::DArray); collect) ----
    def collect(self):
        """Full array on every rank (Array(::DArray), darray.jl:574ff).

        Cross-rank chunk movement is a DEVICE path: one grouped RCCL
        broadcast per chunk from its owner over xGMI (the reference's
        remotecall gather re-expressed), then one D2H of each chunk
        into the assembled host array.  Collective at nranks>1."""
        npdt = np.dtype(NUMPY_DTYPES[self.dtype])
        out = np.zeros(self.dims, dtype=npdt, order="F")
        if self.nranks == 1:
            # (the fast path must be gated on the WORLD size, not the
            # chunk count: at nranks>1 every rank must join the
            # broadcasts below or the collective mismatches)
            if self.lnumel:
                sl = tuple(slice(lo, hi) for lo, hi in self.lidx)
                out[sl] = self.localpart()
            return out
        from . import ops
        esz = DTYPE_SIZE[self.dtype]
        bufs = []
        for c in range(self.nchunks):
            nel = geometry.nelems(self.idxs[c])
            buf = ops._Buf(max(nel, 1) * esz) if nel else None
            if buf is not None and self.ranks[c] == self.rank:
                check(lib.da_d2d(buf.p, self._ptr(), nel * esz))
            bufs.append(buf)
        check(lib.da_group_start())
        for c, buf in enumerate(bufs):
            if buf is not None:
                check(lib.da_bcast(buf.p,
                                   geometry.nelems(self.idxs[c]) * esz,
                                   self.ranks[c]))
        check(lib.da_group_end())
        for c, buf in enumerate(bufs):
            if buf is None:
                continue
            shape = geometry.shape_of(self.idxs[c])
            host = np.empty(shape, dtype=npdt, order="F")
            check(lib.da_d2h(buf.p, host.ctypes.data_as(ctypes.c_void_p),
                             host.size * esz))
            sl = tuple(slice(lo, hi) for lo, hi in self.idxs[c])
            out[sl] = host
            buf.free()
        return out

    def getindex(self, *point):
        """Scalar D[i, j, ...] (darray.jl:645ff): owner reads one
        element; result broadcast via the control plane at nranks>1.
        Gated by allowscalar."""
        if not _allow_scalar[0]:
            raise _ffi.DArrayError(
                "scalar indexing disallowed (allowscalar(False))")
        sub = geometry.locate(self.cuts, point)
        owner_chunk = geometry.grid_rank(sub, self.dist)
        owner = self.ranks[owner_chunk]
        val = None
        if self.rank == owner:
            off = 0
            mul = 1
            for dd in range(len(self.dims)):
                off += (point[dd] - self.lidx[dd][0]) * mul
                mul *= self.lshape[dd]
            out = np.empty(1, dtype=np.dtype(NUMPY_DTYPES[self.dtype]))
            check(lib.da_d2h(self.at_byte(off * DTYPE_SIZE[self.dtype]),
                             out.ctypes.data_as(ctypes.c_void_p),
                             DTYPE_SIZE[self.dtype]))
            val = out[0]
        if self.nranks > 1:
            import torch.distributed as td
            lst = [None] * self.nranks
            td.all_gather_object(lst, val)
            val = lst[owner]
        return val

    def setindex(self, value, *point):
        """Scalar D[i, j, ...] = v (darray.jl:700ff); collective."""
        if not _allow_scalar[0]:
            raise _ffi.DArrayError(
                "scalar indexing disallowed (allowscalar(False))")
        sub = geometry.locate(self.cuts, point)
        owner = self.ranks[geometry.grid_rank(sub, self.dist)]
        if self.rank == owner:
            off = 0
            mul = 1
            for dd in range(len(self.dims)):
                off += (point[dd] - self.lidx[dd][0]) * mul
                mul *= self.lshape[dd]
            v = np.array([value], dtype=np.dtype(NUMPY_DTYPES[self.dtype]))
            check(lib.da_h2d(self.at_byte(off * DTYPE_SIZE[self.dtype]),
                             v.ctypes.data_as(ctypes.c_void_p),
                             DTYPE_SIZE[self.dtype]))
        return value

    def __getitem__(self, key):
        """D[i, j] / D[a:b, c:d] — the reference's scalar and
        contiguous-range getindex (darray.jl:637-820, within this
        build's range scope; strided/fancy indexing is out of scope).
        Collective at nranks>1: every rank must index identically.
        Scalars gate on allowscalar; ranges gather via the makelocal
        box path and return a numpy array on every rank."""
        if not isinstance(key, tuple):
            key = (key,)
        if len(key) != self.ndims:
            raise _ffi.DArrayError(
                "indexing needs %d subscripts" % self.ndims)
        if all(isinstance(k, (int, np.integer)) for k in key):
            return self.getindex(*(int(k) for k in key))
        from . import ops
        ranges = []
        for d, k in enumerate(key):
            if isinstance(k, (int, np.integer)):
                ranges.append((int(k), int(k) + 1))
            elif isinstance(k, slice):
                if k.step not in (None, 1):
                    raise _ffi.DArrayError(
                        "strided ranges are out of scope")
                lo = 0 if k.start is None else int(k.start)
                hi = self.dims[d] if k.stop is None else int(k.stop)
                ranges.append((lo, hi))
            else:
                raise _ffi.DArrayError("bad subscript %r" % (k,))
        out = ops.dgetindex(self, *ranges)
        squeeze = tuple(d for d, k in enumerate(key)
                        if isinstance(k, (int, np.integer)))
        return out.squeeze(axis=squeeze) if squeeze else out

    def __setitem__(self, key, value):
        """Scalar D[i, j] = v (darray.jl:700ff); collective."""
        if not isinstance(key, tuple):
            key = (key,)
        if not all(isinstance(k, (int, np.integer)) for k in key):
            raise _ffi.DArrayError(
                "only scalar setindex is supported (range assignment "
                "is the out-of-scope SubDArray machinery)")
        self.setindex(value, *(int(k) for k in key))

    def __eq__(self, other):
        """== is elementwise-all equality (test/darray.jl:84-129);
        mismatched cuts localize `other` onto self's boxes via the
        makelocal gather (collective — the branch is metadata-determined
        so every rank takes the same path)."""
        if isinstance(other, np.ndarray):
            # d == a::AbstractArray (darray.jl:403-414): each rank
            # compares its localpart with the matching slice
            if self.dims != tuple(other.shape):
                return False
            sl = tuple(slice(lo, hi) for lo, hi in self.lidx)
            same = bool(np.array_equal(self.localpart(), other[sl]))
            import torch.distributed as td
            if self.nranks > 1 and td.is_initialized():
                flags = [None] * self.nranks
                td.all_gather_object(flags, same)
                return all(flags)
            return same
        if isinstance(other, DArray):
            if self.dims != other.dims:
                return False
            a = self.localpart()
            if self.samedist(other):
                b = other.localpart()
            else:
                from . import ops
                buf, shape = ops.gather_box(other, ops._dest_boxes(self))
                if buf is not None:
                    b = np.empty(shape, order="F",
                                 dtype=np.dtype(NUMPY_DTYPES[other.dtype]))
                    if b.size:
                        check(lib.da_d2h(
                            buf.p, b.ctypes.data_as(ctypes.c_void_p),
                            b.size * DTYPE_SIZE[other.dtype]))
                    buf.free()
                else:
                    b = np.empty(self.lshape, order="F",
                                 dtype=np.dtype(NUMPY_DTYPES[other.dtype]))
            import torch.distributed as td
            same = bool(np.array_equal(a, b))
            if self.nranks > 1 and td.is_initialized():
                flags = [None] * self.nranks
                td.all_gather_object(flags, same)
                return all(flags)
            return same
        return NotImplemented

    def __hash__(self):
        return hash(self.id)

    def __repr__(self):
        return ("DArray(dims=%r, dist=%r, dtype=%s, rank=%d/%d, lshape=%r)"
                % (self.dims, self.dist, self.dtype, self.rank, self.nranks,
                   self.lshape))


# ---- convenience constructors (darray.jl:460-532) ----
def dzeros(dims, dtype="f64", dist=None):
    return DArray(dims, dtype, dist).fill_(0.0)


def dones(dims, dtype="f64", dist=None):
    return DArray(dims, dtype, dist).fill_(1.0)


def dfill(v, dims, dtype="f64", dist=None):
    return DArray(dims, dtype, dist).fill_(v)


def drand(dims, dtype="f64", dist=None, seed_base=1234):
    return DArray(dims, dtype, dist).rand_("uniform", seed_base)


def drandn(dims, dtype="f64", dist=None, seed_base=1234):
    return DArray(dims, dtype, dist).rand_("normal", seed_base)


def distribute(a, dist=None):
    """Local (replicated) array -> DArray; each rank uploads its own
    slice — the SPMD analog of the DestinationSerializer scatter
    (darray.jl:544-555): only the local slice touches the wire/PCIe."""
    a = np.asarray(a)
    dtmap = {np.dtype("float64"): "f64", np.dtype("float32"): "f32",
             np.dtype("int64"): "i64"}
    dt = dtmap.get(a.dtype)
    if dt is None:
        raise ValueError("unsupported dtype %s" % a.dtype)
    return DArray(a.shape, dt, dist,
                  init=lambda idx: a[tuple(slice(lo, hi) for lo, hi in idx)])


def dfromfunction(f, dims, dtype="f64", dist=None):
    """@DArray [f(i, j, ...) for i=..., j=...] — the comprehension
    sugar of the reference's docs (docs/src/index.md; `@DArray` macro,
    darray.jl:214-231): each rank builds its chunk by evaluating f over
    its GLOBAL index ranges (0-based here, numpy fromfunction style)."""
    def init(idx):
        return np.asfortranarray(np.fromfunction(
            lambda *loc: f(*[l + lo for l, (lo, _) in zip(loc, idx)]),
            geometry.shape_of(idx)))
    return DArray(dims, dtype, dist, init=init)


def localpart(d):
    return d.localpart()


def localindices(d):
    return d.localindices()


def ddata(value, dtype="f64"):
    """One value per rank (ddata, darray.jl:120-148): a DVector of
    length nranks whose rank-r element is that rank's value."""
    _auto_init()
    nr = comm.rank_info()[1]
    d = DArray((nr,), dtype, (nr,))
    arr = np.asfortranarray(np.array([value],
                                     dtype=np.dtype(NUMPY_DTYPES[dtype])))
    d.set_localpart(arr)
    return d


def dgather(d):
    """gather(d::DArray{T,1}) — darray.jl:150-157: one element per
    chunk collected to every rank (control-plane gather)."""
    return d.collect()


def locate(d, *point):
    """locate(d, I...) — darray.jl:448-456 (0-based chunk coords)."""
    return geometry.locate(d.cuts, point)


def d_closeall():
    """core.jl:98-103."""
    for d in list(_registry.values()):
        d.close()


def bytes_in_use():
    return int(lib.da_bytes_in_use())
