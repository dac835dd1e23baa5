"""Hot-path operation semantics — CPU restatement (oracle side).

Restates, with citations:
  map!/map            /root/reference/src/mapreduce.jl:3-12   (per-chunk map)
  reduce/mapreduce    /root/reference/src/mapreduce.jl:17-39  (per-chunk
      reduce, then left fold over chunk partials in procs order,
      mapreduce.jl:30-34; within-chunk float order is implementation-
      defined — docs/src/index.md:208-236)
  broadcast fused     /root/reference/src/broadcast.jl:65-98  (per-chunk
      copyto!(localpart, localized Broadcasted); aligned-cuts args need
      no communication)
  add!                /root/reference/src/linalg.jl:62-76
  axpy!/scale         /root/reference/src/linalg.jl:24-34,54-59
  matmul              /root/reference/src/linalg.jl:190-253   (block outer
      product: per (i,j,k) partial localpart(A[i,j]) * B[j,k]-panel, then
      add!-accumulation into C[i,k] over j; the reference's accumulation
      order is async-task completion order, i.e. UNSPECIFIED — this
      oracle uses ascending j, and float parity is by tolerance)

Arrays are COLUMN-MAJOR (Fortran order) throughout, mirroring Julia's
memory layout; chunks are numpy arrays in 'F' order.
"""
import numpy as np

# ---------------------------------------------------------------- map table
# Unary scalar functions, named after the reference's test list
# (/root/reference/test/darray.jl:775-800) restricted to the C-math
# subset.  Values are numpy implementations.
MAP_OPS = {
    "identity": lambda x: x.copy(),
    "neg": lambda x: -x,
    "abs": np.abs,
    "abs2": lambda x: x * x,
    "inv": lambda x: 1.0 / x,
    "sqrt": np.sqrt,
    "cbrt": np.cbrt,
    "exp": np.exp,
    "exp2": np.exp2,
    "exp10": lambda x: np.power(x.dtype.type(10), x),
    "expm1": np.expm1,
    "log": np.log,
    "log2": np.log2,
    "log10": np.log10,
    "log1p": np.log1p,
    "sin": np.sin,
    "cos": np.cos,
    "tan": np.tan,
    "asin": np.arcsin,
    "acos": np.arccos,
    "atan": np.arctan,
    "sinh": np.sinh,
    "cosh": np.cosh,
    "tanh": np.tanh,
    "asinh": np.arcsinh,
    "acosh": np.arccosh,
    "atanh": np.arctanh,
    "sinpi": lambda x: np.sin(np.pi * x),
    "cospi": lambda x: np.cos(np.pi * x),
    "floor": np.floor,
    "ceil": np.ceil,
    "round": np.round,
    "trunc": np.trunc,
    "sign": np.sign,
    "deg2rad": np.deg2rad,
    "rad2deg": np.rad2deg,
    "sec": lambda x: 1.0 / np.cos(x),
    "csc": lambda x: 1.0 / np.sin(x),
    "cot": lambda x: 1.0 / np.tan(x),
    # special functions via scipy (the oracle is test infrastructure;
    # the PRODUCT path uses OCML device builtins for these)
    "erf": lambda x: _sp().erf(x),
    "erfc": lambda x: _sp().erfc(x),
    "erfinv": lambda x: _sp().erfinv(x),
    "erfcinv": lambda x: _sp().erfcinv(x),
    "erfcx": lambda x: _sp().erfcx(x),
    "gamma": lambda x: _sp().gamma(x),
    "lgamma": lambda x: _sp().gammaln(x),
    "sinc": np.sinc,                      # numpy sinc IS Julia's sinc
    "cosc": lambda x: np.where(
        x == 0, 0.0, np.cos(np.pi * x) / np.where(x == 0, 1.0, x)
        - np.sin(np.pi * x) / (np.pi * np.where(x == 0, 1.0, x) ** 2)),
    "sind": lambda x: np.sin(np.deg2rad(x)),
    "cosd": lambda x: np.cos(np.deg2rad(x)),
    "tand": lambda x: np.tan(np.deg2rad(x)),
    "asind": lambda x: np.rad2deg(np.arcsin(x)),
    "acosd": lambda x: np.rad2deg(np.arccos(x)),
    "atand": lambda x: np.rad2deg(np.arctan(x)),
    "acot": lambda x: np.arctan(1.0 / x),
    "acotd": lambda x: np.rad2deg(np.arctan(1.0 / x)),
    "asec": lambda x: np.arccos(1.0 / x),
    "acsc": lambda x: np.arcsin(1.0 / x),
    "asech": lambda x: np.arccosh(1.0 / x),
    "acsch": lambda x: np.arcsinh(1.0 / x),
    "acoth": lambda x: np.arctanh(1.0 / x),
    # 0/1-valued predicates (the reference's Bool-array results, kept in
    # the input dtype)
    "isnan": lambda x: np.isnan(x).astype(x.dtype),
    "isinf": lambda x: np.isinf(x).astype(x.dtype),
    "isfinite": lambda x: np.isfinite(x).astype(x.dtype),
}


def _sp():
    import scipy.special
    return scipy.special

# Binary elementwise ops (mapreduce.jl:180-189 specials + broadcast forms).
# Integer div/mod/rem follow Julia: div = trunc, mod = floored, rem = trunc.
MAP2_OPS = {
    "add": np.add,
    "sub": np.subtract,
    "mul": np.multiply,
    "div": np.divide,          # float /
    "min2": np.minimum,        # NaN-propagating on both sides below
    "max2": np.maximum,
    "idiv": lambda a, b: (np.sign(a) * np.sign(b) * (np.abs(a) // np.abs(b))).astype(a.dtype),
    "mod": np.mod,             # floored, like Julia mod
    "rem": np.fmod,            # truncated, like Julia rem / C %
    "and": np.bitwise_and,
    "or": np.bitwise_or,
    "xor": np.bitwise_xor,
    "pow": np.power,
    "atan2": np.arctan2,
}

# mapreduce f table (mapop) and op table (redop)
MAPRED_FS = {
    "identity": lambda x: x,
    "abs": np.abs,
    "abs2": lambda x: x * x,
    # predicate maps feeding all/any/count (mapreduce.jl:97-131):
    # 0/1-valued; count = (pred, add), any = (pred, max)==1,
    # all = (pred, min)==1
    "isnan": lambda x: (np.isnan(x) if x.dtype.kind == "f"
                        else np.zeros_like(x)).astype(x.dtype),
    "isfinite": lambda x: (np.isfinite(x) if x.dtype.kind == "f"
                           else np.ones_like(x)).astype(x.dtype),
    "nonzero": lambda x: (x != 0).astype(x.dtype),
}
RED_OPS = {
    "add": (np.add, lambda dt: dt.type(0)),
    "mul": (np.multiply, lambda dt: dt.type(1)),
    "min": (np.minimum, lambda dt: (np.iinfo(dt).max if dt.kind in "iu"
                                    else dt.type(np.inf))),
    "max": (np.maximum, lambda dt: (np.iinfo(dt).min if dt.kind in "iu"
                                    else dt.type(-np.inf))),
}


def oracle_map(op, x):
    return np.asfortranarray(MAP_OPS[op](x))


def oracle_map2(op, a, b):
    return np.asfortranarray(MAP2_OPS[op](a, b))


def oracle_chunk_reduce(mapop, redop, chunk):
    """Per-chunk mapreduce (the worker-side hot loop, mapreduce.jl:31).

    Within-chunk order is implementation-defined (see module docstring);
    numpy's pairwise sum stands in for Julia Base's pairwise mapreduce.
    Integer results are exact in any order (mod 2^64).
    """
    f = MAPRED_FS[mapop]
    dt = chunk.dtype
    opf, init = RED_OPS[redop]
    if chunk.size == 0:
        return init(dt)
    v = f(chunk.ravel(order="F"))
    if redop == "add":
        return v.sum(dtype=dt)
    if redop == "mul":
        return v.prod(dtype=dt)
    if redop == "min":
        return v.min()
    return v.max()


def oracle_reduce(mapop, redop, chunks):
    """Cross-chunk fold: left fold over chunk partials in procs order
    (mapreduce.jl:30-34: asyncmap preserves procs(d) order; the final
    `reduce(op, results)` is a sequential left fold on the caller)."""
    partials = [oracle_chunk_reduce(mapop, redop, c) for c in chunks]
    opf, init = RED_OPS[redop]
    nonempty = [p for p in partials]
    acc = nonempty[0]
    for p in nonempty[1:]:
        acc = opf(acc, p)
    return acc


def oracle_bcast_fma(a, b, c):
    """D .= A .* B .+ c — separate multiply then add, as Julia Base
    broadcast evaluates it (no fma contraction); broadcast.jl:65-85."""
    return np.asfortranarray(a * b + np.asarray(c, a.dtype))


def oracle_axpy(alpha, x, y):
    """axpy! (linalg.jl:24-34): y += alpha*x elementwise."""
    return np.asfortranarray(y + np.asarray(alpha, y.dtype) * x)


def oracle_add(dest, src, scale=1.0):
    """add! (linalg.jl:62-76): dest += scale*src (scale==1 fast path
    adds without multiplying — numerically identical)."""
    if scale == 1.0:
        return np.asfortranarray(dest + src)
    return np.asfortranarray(dest + np.asarray(scale, dest.dtype) * src)


def oracle_scale(a, s):
    """rmul! (linalg.jl:54-59): a *= s elementwise."""
    return np.asfortranarray(a * np.asarray(s, a.dtype))


def oracle_reduce_dims(mapop, redop, chunks, idxs, dims, red_axes,
                       dtype=None):
    """mapreduce(f, op, A; dims=red_axes) — mapreduce.jl:42-94:
    per-chunk reduce with keepdims (mapreducedim_within), then combine
    chunk partials along the reduced axes in ascending chunk-coordinate
    order onto the lowest-coordinate slab (mapreducedim_between!).
    Returns the full (keepdims) result array."""
    import numpy as np
    f = MAPRED_FS[mapop]
    opf, init = RED_OPS[redop]
    nd = len(dims)
    rdims = tuple(1 if a in red_axes else dims[a] for a in range(nd))
    dt = chunks[0].dtype if dtype is None else np.dtype(dtype)
    out = np.full(rdims, init(dt), dtype=dt, order="F")
    for ch, idx in zip(chunks, idxs):
        if ch.size == 0:
            continue
        part = f(ch)
        for a in sorted(red_axes):
            if redop == "add":
                part = part.sum(axis=a, keepdims=True, dtype=dt)
            elif redop == "mul":
                part = part.prod(axis=a, keepdims=True, dtype=dt)
            elif redop == "min":
                part = part.min(axis=a, keepdims=True)
            else:
                part = part.max(axis=a, keepdims=True)
        sl = tuple(slice(0, 1) if a in red_axes else slice(lo, hi)
                   for a, (lo, hi) in enumerate(idx))
        out[sl] = opf(out[sl], part)
    return np.asfortranarray(out)


# ------------------------------------------------------------ chunk helpers
def make_chunks(arr, idxs):
    """Slice a global array into chunks per chunk_idxs output."""
    out = []
    for idx in idxs:
        sl = tuple(slice(lo, hi) for lo, hi in idx)
        out.append(np.asfortranarray(arr[sl]))
    return out


def assemble(chunks, idxs, dims, dtype):
    """Inverse of make_chunks (Array(::DArray), darray.jl:574ff)."""
    out = np.zeros(dims, dtype=dtype, order="F")
    for ch, idx in zip(chunks, idxs):
        sl = tuple(slice(lo, hi) for lo, hi in idx)
        out[sl] = ch
    return out


# ------------------------------------------------------------------ matmul
def oracle_matmul_blocked(A, B, row_cuts, inner_cuts, col_cuts,
                          alpha=1.0, beta=0.0, C0=None):
    """C = alpha*A*B + beta*C by the reference's block outer product
    (_matmatmul!, linalg.jl:190-253):

      for j (A's column blocks), k (C's column blocks):
        Bjk = B[Acuts2[j]-range, Ccuts[k]-range]            (linalg.jl:215)
        for i: R[i,j,k] = A[i,j] * Bjk                      (linalg.jl:224)
      beta-scale C                                          (linalg.jl:232-240)
      C[i,k] += alpha * R[i,j,k]  accumulated over j        (linalg.jl:243-251)

    cuts are 1-based reference-style cut vectors.  Accumulation over j is
    ascending (one valid schedule of the reference's async tasks).
    """
    from .geometry import chunk_ranges_1d
    rr = chunk_ranges_1d(row_cuts)
    ir = chunk_ranges_1d(inner_cuts)
    cr = chunk_ranges_1d(col_cuts)
    dt = A.dtype
    m, n = A.shape[0], B.shape[1]
    C = (np.zeros((m, n), dtype=dt, order="F") if C0 is None
         else np.asfortranarray(C0.copy()))
    if beta == 0.0:
        C[:] = 0.0
    elif beta != 1.0:
        C *= dt.type(beta)
    for i0, i1 in rr:
        for k0, k1 in cr:
            for j0, j1 in ir:
                part = A[i0:i1, j0:j1] @ B[j0:j1, k0:k1]
                if alpha == 1.0:
                    C[i0:i1, k0:k1] += part
                else:
                    C[i0:i1, k0:k1] += dt.type(alpha) * part
    return np.asfortranarray(C)
