"""The matmul communication plan, executed with numpy (test-only CPU
executor), reproduces the oracle's block-outer-product result — this is
the N>1 dataflow of ops.dmatmul tested without a GPU."""
import numpy as np
import pytest

from distributedarrays_jl_amd import geometry as pg, plan
from oracle import philox, ops as oops


def simulate_matmul(m, kk, n, nranks, alpha=1.0):
    """Execute plan.bslab_plan / plan.partial_plan with numpy chunks,
    mimicking ops.dmatmul step by step (same buffers, same order)."""
    A = np.asfortranarray(philox.fill_uniform_f64(m * kk, 1)
                          .reshape(m, kk, order="F"))
    B = np.asfortranarray(philox.fill_uniform_f64(kk * n, 2)
                          .reshape(kk, n, order="F"))
    A_dist = tuple(pg.defaultdist((m, kk), nranks))
    B_dist = tuple(pg.defaultdist((kk, n), nranks))
    A_idxs, A_cuts = pg.chunk_indices((m, kk), A_dist)
    B_idxs, B_cuts = pg.chunk_indices((kk, n), B_dist)
    I, J = A_dist
    K = plan.c_grid(A_dist, B_dist)[1]
    C_dist = (I, K)
    C_idxs, C_cuts = pg.chunk_indices((m, n), C_dist)
    ccols = pg.ranges1d(C_cuts[1])

    # per-rank local chunks
    def blk(arr, idx):
        return np.asfortranarray(
            arr[tuple(slice(lo, hi) for lo, hi in idx)])

    A_loc = [blk(A, A_idxs[r]) for r in range(I * J)]
    B_loc = [blk(B, B_idxs[r]) for r in range(len(B_idxs))]

    # b-slab exchange ("wire" = plain dict keyed by (src,dst,piece))
    pieces = plan.bslab_plan(A_dist, A_cuts[1], (kk, n), B_dist, B_idxs)
    slabs = {}
    for r in range(I * J):
        i, j = r % I, r // I
        rlo, rhi = plan.slab_rows(A_cuts[1], j)
        slabs[r] = np.zeros((rhi - rlo, n), order="F")
    for (src, dst, rows, cols) in pieces:
        srows, scols = B_idxs[src]
        piece = B_loc[src][rows[0] - srows[0]:rows[1] - srows[0],
                           cols[0] - scols[0]:cols[1] - scols[0]]
        j = dst // I
        rlo, _ = plan.slab_rows(A_cuts[1], j)
        slabs[dst][rows[0] - rlo:rows[1] - rlo, cols[0]:cols[1]] = piece

    # local partial GEMMs
    partials = {}
    for r in range(I * J):
        for k in range(K):
            clo, chi = ccols[k]
            partials[(r, k)] = A_loc[r] @ slabs[r][:, clo:chi]

    # partial exchange + ascending-j accumulation
    C_loc = [np.zeros(pg.shape_of(C_idxs[r]), order="F")
             for r in range(I * K)]
    for r in range(I * K):
        i, myk = r % I, r // I
        for j in plan.accumulate_order(J):
            src = i + I * j
            C_loc[r] += alpha * partials[(src, myk)]

    C = np.zeros((m, n), order="F")
    for r in range(I * K):
        sl = tuple(slice(lo, hi) for lo, hi in C_idxs[r])
        C[sl] = C_loc[r]
    return A, B, C, (A_cuts, B_cuts, C_cuts)


@pytest.mark.parametrize("nranks", [1, 2, 4, 8])
@pytest.mark.parametrize("shape", [(64, 48, 32), (60, 50, 40), (128, 128, 128)])
def test_plan_matmul_vs_oracle(nranks, shape):
    m, kk, n = shape
    A, B, C, (A_cuts, B_cuts, C_cuts) = simulate_matmul(m, kk, n, nranks)
    ref = oops.oracle_matmul_blocked(A, B, A_cuts[0], A_cuts[1], C_cuts[1])
    assert np.allclose(C, ref, rtol=1e-12)
    assert np.allclose(C, A @ B, rtol=1e-12)


def test_plan_alpha():
    A, B, C, _ = simulate_matmul(64, 64, 64, 4, alpha=2.5)
    assert np.allclose(C, 2.5 * (A @ B), rtol=1e-12)


def test_plans_are_deterministic_and_paired():
    A_dist = (2, 4)
    cuts2 = pg.cuts1d(16384, 4)
    B_idxs, _ = pg.chunk_indices((16384, 16384), (2, 4))
    p1 = plan.bslab_plan(A_dist, cuts2, (16384, 16384), (2, 4), B_idxs)
    p2 = plan.bslab_plan(A_dist, cuts2, (16384, 16384), (2, 4), B_idxs)
    assert p1 == p2
    # every send has a matching recv (same tuple seen from both ends)
    for (src, dst, rows, cols) in p1:
        assert 0 <= src < 8 and 0 <= dst < 8
    moves = plan.partial_plan((2, 4), 4)
    sends = [(s, d) for s, d, _ in moves]
    assert len(sends) == len(set((s, d, k) for s, d, k in moves))
    # each C owner receives exactly J-1 partials
    from collections import Counter
    cnt = Counter(d for _, d, _ in moves)
    for owner in range(8):
        assert cnt[owner] == 3


def test_partial_plan_per_k_partition():
    """overlap mode issues one group per k (ascending); the per-k
    partition must cover partial_plan exactly once."""
    for dist, K in [((2, 4), 4), ((2, 2), 2), ((2, 1), 1), ((1, 2), 2)]:
        moves = plan.partial_plan(dist, K)
        parts = []
        for k in range(K):
            parts.extend(mv for mv in moves if mv[2] == k)
        assert sorted(parts) == sorted(moves)
        # ascending-k issue order is identical on every rank by
        # construction (k loop), so pairing matches the single group


def test_partial_exchange_skip_pairing():
    """The zero-size skip conditions in ops.dmatmul's partial exchange
    pair consistently: the sender's byte count (A row-block height x
    C column-block width) IS the receiver's C.lnumel, so both sides
    drop exactly the same moves — over random geometries including
    degenerate cuts (sz < chunks)."""
    import numpy as np
    from distributedarrays_jl_amd import geometry as pg, plan
    rng = np.random.default_rng(7)
    for _ in range(200):
        m = int(rng.integers(1, 20))
        n = int(rng.integers(1, 20))
        I = int(rng.integers(1, 4))
        J = int(rng.integers(1, 4))
        K = int(rng.integers(1, J + 1))
        rrows = pg.ranges1d(pg.cuts1d(m, I))
        ccols = pg.ranges1d(pg.cuts1d(n, K))
        moves = plan.partial_plan((I, J), K)
        for (src, dst, k) in moves:
            i_src = src % I
            i_dst = dst % I
            assert i_src == i_dst          # partials move within a row
            send_nb = (rrows[i_src][1] - rrows[i_src][0]) * \
                (ccols[k][1] - ccols[k][0])
            kk_dst = dst // I
            assert kk_dst == k             # to the k-th column owner
            recv_nb = (rrows[i_dst][1] - rrows[i_dst][0]) * \
                (ccols[kk_dst][1] - ccols[kk_dst][0])
            assert send_nb == recv_nb      # skip iff both skip
