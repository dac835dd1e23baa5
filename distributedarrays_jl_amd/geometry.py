"""Chunk geometry of the product DArray — same partition rule as the
reference (/root/reference/src/darray.jl:251-307).

This is the PRODUCT-side implementation; oracle/geometry.py is the
independent restatement used to check it (tests/test_geometry.py compares
the two over many shapes and pins `defaultdist(50,4) == [1,14,27,39,51]`,
/root/reference/test/darray.jl:66).
"""


def _prime_factors(n):
    fs = []
    d = 2
    while d * d <= n:
        if n % d == 0:
            fs.append(d)
            while n % d == 0:
                n //= d
        d += 1
    if n > 1:
        fs.append(n)
    return fs


def defaultdist(dims, nranks):
    """Chunks per dimension (darray.jl:251-276): allocate largest prime
    factor to largest dim, ties to the highest dim; np divides by the
    factor even when unallocatable."""
    dims = list(dims)
    chunks = [1] * len(dims)
    np_ = int(nranks)
    fs = sorted(_prime_factors(np_), reverse=True)
    k = 0
    while np_ > 1:
        if np_ % fs[k] != 0:
            k += 1
            if k >= len(fs):
                break
        fac = fs[k]
        mx = max(dims)
        dno = len(dims) - 1 - dims[::-1].index(mx)
        if dims[dno] >= fac:
            dims[dno] //= fac
            chunks[dno] *= fac
        np_ //= fac
    return chunks


def cuts1d(sz, nc):
    """1-based cut vector (darray.jl:279-296)."""
    sz, nc = int(sz), int(nc)
    if sz >= nc:
        chunk, rem = divmod(sz, nc)
        return [i * chunk + 1 + min(i, rem) for i in range(nc + 1)]
    return list(range(1, sz + 2)) + [0] * (nc - sz)


def ranges1d(cuts):
    """Half-open 0-based ranges per chunk from a 1-based cut vector."""
    out = []
    for i in range(len(cuts) - 1):
        lo, nxt = cuts[i], cuts[i + 1]
        if lo == 0 or nxt == 0 or nxt < lo:
            out.append((0, 0))
        else:
            out.append((lo - 1, nxt - 1))
    return out


def chunk_indices(dims, dist):
    """(idxs, cuts): idxs[rank] = per-dim half-open ranges; rank order is
    Julia column-major over the dist grid (darray.jl:159-162,299-307)."""
    cuts = [cuts1d(d, c) for d, c in zip(dims, dist)]
    rngs = [ranges1d(c) for c in cuts]
    n = len(dims)
    total = 1
    for c in dist:
        total *= c
    idxs = []
    for lin in range(total):
        rem = lin
        sub = []
        for d in range(n):
            sub.append(rem % dist[d])
            rem //= dist[d]
        idxs.append(tuple(rngs[d][sub[d]] for d in range(n)))
    return idxs, cuts


def grid_pos(rank, dist):
    """Column-major grid coordinates of a rank."""
    sub = []
    rem = rank
    for c in dist:
        sub.append(rem % c)
        rem //= c
    return tuple(sub)


def grid_rank(sub, dist):
    """Inverse of grid_pos."""
    r, mul = 0, 1
    for s, c in zip(sub, dist):
        r += s * mul
        mul *= c
    return r


def shape_of(idx):
    return tuple(hi - lo for lo, hi in idx)


def nelems(idx):
    n = 1
    for lo, hi in idx:
        n *= hi - lo
    return n


def locate(cuts, point):
    """Chunk coordinates holding 0-based point (darray.jl:448-456)."""
    import bisect
    out = []
    for c, p in zip(cuts, point):
        fi = bisect.bisect_right(c, p + 1) - 1
        if fi >= len(c) - 1:
            raise ValueError("element not contained in array")
        out.append(fi)
    return tuple(out)


def intersect1d(a, b):
    lo = max(a[0], b[0])
    hi = min(a[1], b[1])
    return (lo, hi) if hi > lo else None
