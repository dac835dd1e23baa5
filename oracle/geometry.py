"""Chunk geometry — line-by-line restatement of the reference partitioner.

Sources (cited per function):
  /root/reference/src/darray.jl:251-276  defaultdist(dims, pids)
  /root/reference/src/darray.jl:279-296  defaultdist(sz::Int, nc::Int)
  /root/reference/src/darray.jl:299-307  chunk_idxs(dims, chunks)
  Primes.factor call site: darray.jl:255 (replaced by trial division,
  per SURVEY.md §2 — only the set of distinct prime factors is used).

Known-answer pin: defaultdist_1d(50, 4) == [1, 14, 27, 39, 51]
(/root/reference/test/darray.jl:66).

Indexing convention: cuts are 1-based start indices exactly as the
reference stores them (cuts[d][i] = first 1-based index of chunk i);
ranges returned by chunk_idxs are half-open 0-based (lo, hi) pairs
equivalent to Julia's cuts[i]:cuts[i+1]-1.
"""


def factor(n):
    """Distinct prime factors of n, ascending (Primes.factor keys)."""
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


def defaultdist_dims(dims, np_):
    """How many chunks per dimension for np_ processes.

    Restates darray.jl:251-276: repeatedly allocate the largest prime
    factor to the largest dimension; ties resolve to the HIGHEST dim
    (findlast, darray.jl:268); np is divided by the factor even when no
    dimension can absorb it (darray.jl:273 sits outside the `if`).
    """
    dims = list(dims)
    chunks = [1] * len(dims)
    np_ = int(np_)
    f = sorted(factor(np_), reverse=True)
    k = 0
    while np_ > 1:
        if np_ % f[k] != 0:
            k += 1
            if k >= len(f):
                break
        fac = f[k]
        d = max(dims)
        # resolve ties to highest dim (findlast)
        dno = len(dims) - 1 - dims[::-1].index(d)
        if dims[dno] >= fac:
            dims[dno] //= fac
            chunks[dno] *= fac
        np_ //= fac
    return chunks


def defaultdist_1d(sz, nc):
    """1-based start indices dividing sz into nc chunks (darray.jl:279-296).

    sz >= nc: even split, remainder spread over the FIRST `rem` chunks.
    sz <  nc: [1, 2, .., sz+1] padded with zeros (empty trailing chunks).
    """
    sz, nc = int(sz), int(nc)
    if sz >= nc:
        chunk, rem = divmod(sz, nc)
        return [i * chunk + 1 + min(i, rem) for i in range(nc + 1)]
    return list(range(1, sz + 2)) + [0] * (nc - sz)


def chunk_ranges_1d(cuts):
    """Half-open 0-based (lo, hi) per chunk from 1-based cuts.

    Julia chunk i covers cuts[i]:cuts[i+1]-1 (darray.jl:304); for the
    sz<nc padded-zero cuts this yields empty ranges for trailing chunks.
    """
    out = []
    for i in range(len(cuts) - 1):
        lo, nxt = cuts[i], cuts[i + 1]
        if lo == 0 or nxt == 0 or nxt < lo:
            out.append((0, 0))
        else:
            out.append((lo - 1, nxt - 1))
    return out


def chunk_idxs(dims, chunks):
    """(idxs, cuts) for dividing dims into chunks (darray.jl:299-307).

    idxs is a flat list in Julia column-major chunk order (the order in
    which pids/ranks are assigned, darray.jl:159-162 reshape); each entry
    is a tuple of per-dim half-open 0-based (lo, hi) ranges.
    """
    cuts = [defaultdist_1d(d, c) for d, c in zip(dims, chunks)]
    ranges = [chunk_ranges_1d(c) for c in cuts]
    n = len(dims)
    idxs = []
    total = 1
    for c in chunks:
        total *= c
    for lin in range(total):
        # column-major (first dim fastest), as CartesianIndices iterates
        rem = lin
        sub = []
        for d in range(n):
            sub.append(rem % chunks[d])
            rem //= chunks[d]
        idxs.append(tuple(ranges[d][sub[d]] for d in range(n)))
    return idxs, cuts


def chunk_shape(idx):
    """Shape of a chunk given its tuple of (lo, hi) ranges."""
    return tuple(hi - lo for lo, hi in idx)
