"""Broadcast expression trees — the analog of the reference's
`Broadcasted` materialization (/root/reference/src/broadcast.jl:65-98):
an arbitrary composition of the unary/binary op tables over DArrays and
scalars is fused into ONE kernel pass per chunk (da_expr), with
per-operand localisation (bclocal/makelocal, broadcast.jl:140-152) for
mismatched cuts and stride-0 expansion for Julia-broadcast singleton
dims (the `a .- mean(a, dims=1)` shape, pinned at
/root/reference/test/darray.jl:880-912).

Usage (mirrors `D .= A .- M .* sin.(C)`):

    from distributedarrays_jl_amd import expr as E
    e = E.ref(A) - E.ref(M) * E.sin(E.ref(C))
    E.materialize_(D, e)        # in-place (Base.materialize!)
    G = E.materialize(e)        # allocating (Base.materialize / copy)

Numerics: da_expr shares the scalar functor tables with da_map/da_map2
(csrc/mapops.hpp), so a fused chain is bit-identical to the equivalent
sequence of single-op kernels."""
import ctypes

from ._ffi import check, lib, DArrayError
from ._opcodes import (DTYPES, MAP_OP, MAP2_OP,
                       I64_MAP_OPS, I64_MAP2_OPS)

K_UNARY, K_ARG, K_CONST, K_BINARY = 0, 1, 2, 3

MAXLEN, MAXARGS, MAXCONSTS, MAXND, MAXSTACK = 40, 6, 6, 4, 8


class Expr:
    """Node of a broadcast tree (scalar functions of element values)."""

    def __add__(self, o):
        return Binary("add", self, wrap(o))

    def __radd__(self, o):
        return Binary("add", wrap(o), self)

    def __sub__(self, o):
        return Binary("sub", self, wrap(o))

    def __rsub__(self, o):
        return Binary("sub", wrap(o), self)

    def __mul__(self, o):
        return Binary("mul", self, wrap(o))

    def __rmul__(self, o):
        return Binary("mul", wrap(o), self)

    def __truediv__(self, o):
        return Binary("div", self, wrap(o))

    def __rtruediv__(self, o):
        return Binary("div", wrap(o), self)

    def __pow__(self, o):
        return Binary("pow", self, wrap(o))

    def __rpow__(self, o):
        return Binary("pow", wrap(o), self)

    def __mod__(self, o):
        return Binary("mod", self, wrap(o))

    def __neg__(self):
        return Unary("neg", self)

    def __abs__(self):
        return Unary("abs", self)


class Ref(Expr):
    """A DArray leaf."""

    def __init__(self, d):
        self.d = d


class Lit(Expr):
    """A scalar leaf (broadcast singleton, broadcast.jl:124-133)."""

    def __init__(self, v):
        self.v = float(v)


class Unary(Expr):
    def __init__(self, op, x):
        if op not in MAP_OP:
            raise DArrayError("expr: unknown unary op %r" % op)
        self.op, self.x = op, x


class Binary(Expr):
    def __init__(self, op, a, b):
        if op not in MAP2_OP:
            raise DArrayError("expr: unknown binary op %r" % op)
        self.op, self.a, self.b = op, a, b


def wrap(x):
    import numbers
    if isinstance(x, Expr):
        return x
    if isinstance(x, numbers.Real):    # int/float incl. numpy scalars
        return Lit(x)
    # a bare DArray in an expression position
    if hasattr(x, "lidx") and hasattr(x, "dtype"):
        return Ref(x)
    raise DArrayError("expr: cannot broadcast over %r" % type(x))


def ref(d):
    return Ref(d)


def lit(v):
    return Lit(v)


def _make_unary(name):
    def f(x):
        return Unary(name, wrap(x))
    f.__name__ = name
    f.__doc__ = "elementwise %s.(x) in a broadcast tree" % name
    return f


for _name in MAP_OP:
    if _name not in ("identity",):
        globals()[_name] = _make_unary(_name)


def _make_binary(name):
    def f(a, b):
        return Binary(name, wrap(a), wrap(b))
    f.__name__ = name
    return f


for _name in ("min2", "max2", "atan2", "idiv", "rem", "and_", "or_",
              "xor"):
    globals()[_name] = _make_binary(_name.rstrip("_"))


def compile_expr(e):
    """Postorder walk -> (prog, args, consts); DArray leaves dedup by
    identity, scalars by value."""
    prog, args, consts = [], [], []
    argids = {}

    def walk(node):
        if isinstance(node, Ref):
            key = id(node.d)
            if key not in argids:
                if len(args) >= MAXARGS:
                    raise DArrayError("expr: more than %d distinct "
                                      "DArray operands" % MAXARGS)
                argids[key] = len(args)
                args.append(node.d)
            prog.append((K_ARG << 8) | argids[key])
        elif isinstance(node, Lit):
            import struct
            key = struct.pack("<d", node.v)   # bit pattern: -0.0 != 0.0
            k = next((i for i, c in enumerate(consts)
                      if struct.pack("<d", c) == key), None)
            if k is None:
                if len(consts) >= MAXCONSTS:
                    raise DArrayError("expr: more than %d constants"
                                      % MAXCONSTS)
                k = len(consts)
                consts.append(node.v)
            prog.append((K_CONST << 8) | k)
        elif isinstance(node, Unary):
            walk(node.x)
            prog.append((K_UNARY << 8) | MAP_OP[node.op])
        elif isinstance(node, Binary):
            walk(node.a)
            walk(node.b)
            prog.append((K_BINARY << 8) | MAP2_OP[node.op])
        else:
            raise DArrayError("expr: bad node %r" % type(node))

    walk(e)
    if len(prog) > MAXLEN:
        raise DArrayError("expr: program longer than %d" % MAXLEN)
    # stack-depth check (the library validates again)
    depth = mx = 0
    for ins in prog:
        kind = ins >> 8
        depth += 1 if kind in (K_ARG, K_CONST) else \
            (-1 if kind == K_BINARY else 0)
        mx = max(mx, depth)
    if mx > MAXSTACK:
        raise DArrayError("expr: stack depth %d > %d" % (mx, MAXSTACK))
    return prog, args, consts


def _validate(dest, prog, args, consts):
    nd = dest.ndims
    for a in args:
        if a.dtype != dest.dtype:
            raise DArrayError("expr: operand dtype %s != dest %s"
                              % (a.dtype, dest.dtype))
        if a.ndims != nd:
            raise DArrayError("expr: operand ndims %d != dest %d"
                              % (a.ndims, nd))
        for d in range(nd):
            if a.dims[d] != dest.dims[d] and a.dims[d] != 1:
                raise DArrayError(
                    "expr: operand dim %d is %d, dest %d (only "
                    "singleton dims broadcast)" %
                    (d, a.dims[d], dest.dims[d]))
    if dest.dtype == "i64":
        for ins in prog:
            kind, idx = ins >> 8, ins & 0xFF
            if kind == K_UNARY:
                name = [k for k, v in MAP_OP.items() if v == idx][0]
                if name not in I64_MAP_OPS:
                    raise DArrayError("expr: op %r invalid for i64" % name)
            if kind == K_BINARY:
                name = [k for k, v in MAP2_OP.items() if v == idx][0]
                if name not in I64_MAP2_OPS:
                    raise DArrayError("expr: op %r invalid for i64" % name)
        for c in consts:
            # NB: plain `abs` is shadowed by the generated elementwise
            # function of the same name in this module's namespace
            ci = int(c)
            if ci != c or ci > (1 << 53) or ci < -(1 << 53):
                raise DArrayError("expr: i64 constant %r not exactly "
                                  "representable" % c)


def materialize_(dest, e):
    """Base.materialize!(dest, bc) — copyto!(::DArray, ::Broadcasted)
    (broadcast.jl:65-85).  Collective: every rank passes the same tree
    (metadata-deterministic branches)."""
    from .ops import gather_box, _same_layout

    prog, args, consts = compile_expr(e)
    _validate(dest, prog, args, consts)
    nd = dest.ndims
    if nd > MAXND:
        raise DArrayError("expr: ndims %d > %d" % (nd, MAXND))

    full = [a.dims == dest.dims for a in args]
    aligned = [full[i] and _same_layout(dest, a)
               for i, a in enumerate(args)]
    flat = all(aligned)
    if not flat and dest.dtype == "i64":
        raise DArrayError("expr: i64 supports only aligned full-shape "
                          "operands (strided variant is float-only)")

    bufs = [None] * len(args)
    ptrs = [None] * len(args)
    strides = [None] * len(args)
    # destination-box projection per operand (bclocal: singleton dims
    # clamp to (0, 1), broadcast.jl:140-152 + _bcview :103-120)
    for i, a in enumerate(args):
        if aligned[i]:
            ptrs[i] = a._ptr()
            t, ss = 1, []
            for d in range(nd):
                ss.append(t)
                t *= dest.lshape[d]
            strides[i] = ss
            continue
        boxes = [None] * a.nranks
        for c, r in enumerate(dest.ranks):
            box = dest.idxs[c]
            boxes[r] = tuple((0, 1) if a.dims[d] == 1 else box[d]
                             for d in range(nd))
        buf, bshape = gather_box(a, boxes)
        bufs[i] = buf
        if buf is not None:
            ptrs[i] = buf.p
            t, ss = 1, []
            for d in range(nd):
                ss.append(t if a.dims[d] != 1 else 0)
                t *= bshape[d]
            strides[i] = ss

    if dest.lnumel:
        n = dest.lnumel
        parr = (ctypes.c_int32 * len(prog))(*prog)
        dims_arr = (ctypes.c_uint64 * max(nd, 1))(*dest.lshape)
        src_arr = (ctypes.c_void_p * max(len(args), 1))(
            *[p.value if p is not None else 0 for p in ptrs])
        cons_arr = (ctypes.c_double * max(len(consts), 1))(*consts)
        if flat:
            sarr = None
        else:
            flat_strides = []
            for ss in strides:
                flat_strides.extend(ss if ss is not None else [0] * nd)
            sarr = (ctypes.c_uint64 * max(len(flat_strides), 1))(
                *flat_strides)
        check(lib.da_expr(parr, len(prog), dest._ptr(), dims_arr, nd,
                          src_arr, sarr, len(args), cons_arr,
                          len(consts), n, DTYPES[dest.dtype]))
    if any(b is not None for b in bufs):
        check(lib.da_synchronize())
    for b in bufs:
        if b is not None:
            b.free()
    return dest


def materialize(e):
    """Base.materialize(bc) — `copy(bc)` (broadcast.jl:91-98): allocate
    the result with the layout of the first full-shape operand (the
    `similar` overload, broadcast.jl:44-50)."""
    prog, args, consts = compile_expr(e)
    if not args:
        raise DArrayError("expr: no DArray operands")
    dims = tuple(max(a.dims[d] for a in args)
                 for d in range(args[0].ndims))
    proto = None
    for a in args:
        if a.dims == dims:
            proto = a
            break
    if proto is not None:
        dest = proto.similar()
    else:
        # outer-product broadcasting (e.g. row .+ col -> (m, n)): no
        # operand carries the result shape, so allocate it on the
        # default grid; every operand localizes via projected boxes
        from .darray import DArray
        dest = DArray(dims, args[0].dtype)
    return materialize_(dest, e)
