"""Communication plans for the distributed matmul — pure data, no GPU.

Re-expresses the reference's _matmatmul! dataflow
(/root/reference/src/linalg.jl:190-253) over xGMI point-to-point:

  reference                              MI355X-native
  ---------                              -------------
  caller slices B[Acuts2[j], Ccuts[k]]   b-slab all-to-all: each rank
  (DArray getindex => remote gather,     (i,j) receives the B row-slab
   linalg.jl:215)                        Acuts2[j] x (all cols) from B's
                                         owners via grouped ncclSend/Recv
  R[i,j,k] = remotecall owner(A[i,j])    local MFMA GEMM on the slab panel
  (linalg.jl:218-226)
  C-owner fetches R[i,j,k], add!         partial-exchange: rank (i,j)
  (linalg.jl:243-251)                    sends partial k to C-owner (i,k);
                                         owner adds in ascending j

Plans are deterministic sorted lists so every rank derives the identical
send/recv schedule (RCCL grouped calls must pair up).  Tested on CPU
(tests/test_plan.py, tests/test_gloo.py) against the oracle.
"""
from . import geometry


def c_grid(A_dist, B_dist):
    """C's process grid: procs(A)[:, 1:q], q = min(J, B cols-chunks)
    (linalg.jl:266-272)."""
    I, J = A_dist
    K = min(J, B_dist[1] if len(B_dist) > 1 else 1)
    return (I, K)


def a_rank_pos(r, A_dist):
    """Rank r's (i, j) in A's column-major grid; None if r holds no A."""
    I, J = A_dist
    if r >= I * J:
        return None
    return (r % I, r // I)


def slab_rows(A_cuts2, j):
    """Rows of B needed by A-column j: Acuts[j]:Acuts[j+1]-1
    (linalg.jl:213-215)."""
    return geometry.ranges1d(A_cuts2)[j]


def bslab_plan(A_dist, A_cuts2, B_dims, B_dist, B_idxs):
    """All-to-all pieces (src, dst, rows, cols) in GLOBAL coordinates:
    rank dst=(i,j) needs B rows slab_rows(j) x all cols; owner src holds
    block B_idxs[src].  Sorted deterministically."""
    I, J = A_dist
    nB = 1
    for c in B_dist:
        nB *= c
    pieces = []
    for dst in range(I * J):
        i, j = dst % I, dst // I
        rneed = slab_rows(A_cuts2, j)
        for src in range(nB):
            rows_s, cols_s = B_idxs[src][0], B_idxs[src][1]
            inter = geometry.intersect1d(rneed, rows_s)
            if inter is not None and cols_s[1] > cols_s[0]:
                pieces.append((src, dst, inter, cols_s))
    pieces.sort()
    return pieces


def partial_plan(A_dist, K):
    """(src, dst, k) triples: rank (i,j) ships its k-th partial product
    to C-owner (i,k) = rank i + I*k (linalg.jl:243-251); j==k stays
    local.  Sorted deterministically."""
    I, J = A_dist
    moves = []
    for r in range(I * J):
        i, j = r % I, r // I
        for k in range(K):
            owner = i + I * k
            if owner != r:
                moves.append((r, owner, k))
    moves.sort()
    return moves


def halo_plan(src_idxs, src_ranks, boxes):
    """Pieces (src_rank, dst_rank, box) for the makelocal halo gather
    (darray.jl:351-368: requested indices not fully local are fetched
    from the owning chunks; here via grouped ncclSend/Recv).

    src_idxs: chunk boxes of the source DArray; src_ranks[c] = owner.
    boxes[r]: the box rank r requests (None = no request).  Returns a
    deterministic sorted list in GLOBAL coordinates."""
    pieces = []
    for dst, box in enumerate(boxes):
        if box is None:
            continue
        for c, idx in enumerate(src_idxs):
            inter = []
            ok = True
            for (blo, bhi), (clo, chi) in zip(box, idx):
                lo, hi = max(blo, clo), min(bhi, chi)
                if hi <= lo:
                    ok = False
                    break
                inter.append((lo, hi))
            if ok:
                pieces.append((src_ranks[c], dst, tuple(inter)))
    pieces.sort()
    return pieces


def accumulate_order(J):
    """Owner-side add! order over j (ascending — one valid schedule of
    the reference's async accumulation, linalg.jl:243-251)."""
    return list(range(J))
