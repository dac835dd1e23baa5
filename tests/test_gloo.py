"""Multi-process CPU tests (gloo, world_size 2) of the distributed
logic: the cross-rank reduction fold and the matmul exchange schedule
execute with torch.distributed isend/irecv exactly as ops.dmatmul issues
them with grouped RCCL — same plans, same pairing, same order.  The
numpy executor here is test infrastructure (the product path is
GPU-only)."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as td
import torch.multiprocessing as mp

from distributedarrays_jl_amd import geometry as pg, plan
from oracle import philox, ops as oops

WORLD = 2


def _init(rank, tmpfile, world=WORLD):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    td.init_process_group("gloo", init_method="file://%s" % tmpfile,
                          rank=rank, world_size=world)


def _reduce_worker(rank, tmpfile, q):
    try:
        _init(rank, tmpfile)
        n = 100001
        idxs, cuts = pg.chunk_indices((n,), (WORLD,))
        lo, hi = idxs[rank][0]
        # per-rank philox chunk (seed = 1234 + rank, BASELINE.md protocol)
        chunk = philox.fill_uniform_f64(hi - lo, seed=1234 + rank)
        part = torch.tensor([chunk.sum()], dtype=torch.float64)
        td.all_reduce(part, op=td.ReduceOp.SUM)
        # oracle: same chunks, left fold in rank order
        chunks = [philox.fill_uniform_f64(
            pg.ranges1d(cuts[0])[r][1] - pg.ranges1d(cuts[0])[r][0],
            seed=1234 + r) for r in range(WORLD)]
        ref = oops.oracle_reduce("identity", "add", chunks)
        ok = abs(part.item() - ref) / abs(ref) < 1e-12
        # integer exactness across ranks
        ichunks = [philox.fill_int64(1000, seed=7 + r) for r in range(WORLD)]
        ipart = torch.tensor([int(ichunks[rank].sum())], dtype=torch.int64)
        td.all_reduce(ipart, op=td.ReduceOp.SUM)
        iref = oops.oracle_reduce("identity", "add", ichunks)
        ok = ok and (ipart.item() == int(iref))
        q.put((rank, ok, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        if td.is_initialized():
            td.destroy_process_group()


def _matmul_worker(rank, tmpfile, q, world=WORLD):
    try:
        _init(rank, tmpfile, world)
        WORLD_ = world
        m, kk, n = 64, 48, 32
        A = np.asfortranarray(philox.fill_uniform_f64(m * kk, 1)
                              .reshape(m, kk, order="F"))
        B = np.asfortranarray(philox.fill_uniform_f64(kk * n, 2)
                              .reshape(kk, n, order="F"))
        A_dist = tuple(pg.defaultdist((m, kk), WORLD_))
        B_dist = tuple(pg.defaultdist((kk, n), WORLD_))
        A_idxs, A_cuts = pg.chunk_indices((m, kk), A_dist)
        B_idxs, B_cuts = pg.chunk_indices((kk, n), B_dist)
        I, J = A_dist
        K = plan.c_grid(A_dist, B_dist)[1]
        C_idxs, C_cuts = pg.chunk_indices((m, n), (I, K))
        ccols = pg.ranges1d(C_cuts[1])

        def blk(arr, idx):
            return np.asfortranarray(
                arr[tuple(slice(lo, hi) for lo, hi in idx)])

        A_loc = blk(A, A_idxs[rank]) if rank < I * J else None
        B_loc = blk(B, B_idxs[rank]) if rank < len(B_idxs) else None

        # --- b-slab exchange with isend/irecv (mirrors grouped RCCL) ---
        pieces = plan.bslab_plan(A_dist, A_cuts[1], (kk, n), B_dist, B_idxs)
        i, j = rank % I, rank // I
        rlo, rhi = plan.slab_rows(A_cuts[1], j)
        slab = np.zeros((rhi - rlo, n), order="F")
        reqs, stage = [], []
        for tag, (src, dst, rows, cols) in enumerate(pieces):
            srows, scols = B_idxs[src]
            if src == rank and dst == rank:
                piece = B_loc[rows[0] - srows[0]:rows[1] - srows[0],
                              cols[0] - scols[0]:cols[1] - scols[0]]
                slab[rows[0] - rlo:rows[1] - rlo, cols[0]:cols[1]] = piece
            elif src == rank:
                piece = np.ascontiguousarray(
                    B_loc[rows[0] - srows[0]:rows[1] - srows[0],
                          cols[0] - scols[0]:cols[1] - scols[0]])
                t = torch.from_numpy(piece)
                stage.append(t)
                reqs.append(td.isend(t, dst, tag=tag))
            elif dst == rank:
                t = torch.zeros((rows[1] - rows[0], cols[1] - cols[0]),
                                dtype=torch.float64)
                stage.append(t)
                reqs.append((td.irecv(t, src, tag=tag),
                             (t, rows, cols)))
        for rq in reqs:
            if isinstance(rq, tuple):
                rq[0].wait()
                t, rows, cols = rq[1]
                slab[rows[0] - rlo:rows[1] - rlo,
                     cols[0]:cols[1]] = t.numpy()
            else:
                rq.wait()

        # --- local partials + PER-K exchange rounds (the DA_MM_OVERLAP
        # schedule of ops.dmatmul: every rank issues round k in
        # ascending order; this validates that pairing with a real
        # multiprocess transport) ---
        partials = [A_loc @ slab[:, ccols[k][0]:ccols[k][1]]
                    for k in range(K)]
        moves = plan.partial_plan(A_dist, K)
        got = {}
        for k in range(K):
            reqs = []
            for tag, (src, dst, kk) in enumerate(moves):
                if kk != k:
                    continue
                if src == rank:
                    t = torch.from_numpy(
                        np.ascontiguousarray(partials[kk]))
                    reqs.append((td.isend(t, dst, tag=1000 + tag),
                                 None, t))
                elif dst == rank:
                    shp = pg.shape_of(C_idxs[rank])
                    t = torch.zeros(shp, dtype=torch.float64)
                    reqs.append((td.irecv(t, src, tag=1000 + tag),
                                 (src, kk), t))
            for rq, key, t in reqs:
                rq.wait()
                if key is not None:
                    got[key] = t.numpy()
        C_loc = np.zeros(pg.shape_of(C_idxs[rank]), order="F")
        myk = rank // I
        for jj in plan.accumulate_order(J):
            src = i + I * jj
            C_loc += partials[myk] if src == rank else got[(src, myk)]

        gathered = [None] * WORLD_
        td.all_gather_object(gathered, C_loc)
        C = np.zeros((m, n), order="F")
        for r in range(I * K):
            sl = tuple(slice(lo, hi) for lo, hi in C_idxs[r])
            C[sl] = gathered[r]
        ref = oops.oracle_matmul_blocked(A, B, A_cuts[0], A_cuts[1],
                                         C_cuts[1])
        ok = np.allclose(C, ref, rtol=1e-12) and np.allclose(C, A @ B,
                                                             rtol=1e-12)
        q.put((rank, bool(ok), None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        if td.is_initialized():
            td.destroy_process_group()


def _spawn(fn, tmp_path, world=WORLD, extra=()):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    tmpfile = str(tmp_path / "rdv")
    procs = [ctx.Process(target=fn, args=(r, tmpfile, q) + tuple(extra))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok, err in results:
        assert ok, "rank %d failed: %s" % (rank, err)


@pytest.mark.timeout(300)
def test_gloo_reduce_world2(tmp_path):
    _spawn(_reduce_worker, tmp_path)


@pytest.mark.timeout(300)
def test_gloo_matmul_world2(tmp_path):
    _spawn(_matmul_worker, tmp_path)


@pytest.mark.timeout(300)
def test_gloo_matmul_world4(tmp_path):
    # 2x2 grid: real B-panel all-to-all AND partial exchange (K=2)
    _spawn(_matmul_worker, tmp_path, world=4, extra=(4,))


@pytest.mark.timeout(300)
def test_gloo_matmul_world8(tmp_path):
    # 2x4 grid — the exact cfg-4 bench geometry (16384^2 over 8 GPUs)
    _spawn(_matmul_worker, tmp_path, world=8, extra=(8,))


def _dims_reduce_worker(rank, tmpfile, q, world=4):
    """world-4 numpy execution of ops.dreduce_dims' exchange schedule
    (partials to the reduced-coord-0 owner, ascending-source combine)."""
    try:
        _init(rank, tmpfile, world)
        dims, dist = (24, 20), (2, 2)
        x = np.asfortranarray(philox.fill_uniform_f64(480, 9)
                              .reshape(dims, order="F"))
        idxs, cuts = pg.chunk_indices(dims, dist)
        for red in [(0,), (1,), (0, 1)]:
            loc = x[tuple(slice(lo, hi) for lo, hi in idxs[rank])]
            part = loc.sum(axis=red, keepdims=True)
            # owner groups exactly as ops.dreduce_dims builds them
            groups = {}
            for src in range(4):
                sub = list(pg.grid_pos(src, dist))
                for a in red:
                    sub[a] = 0
                groups.setdefault(pg.grid_rank(sub, dist), []).append(src)
            sub = list(pg.grid_pos(rank, dist))
            for a in red:
                sub[a] = 0
            my_owner = pg.grid_rank(sub, dist)
            reqs, got = [], {}
            if my_owner != rank:
                t = torch.from_numpy(np.ascontiguousarray(part))
                reqs.append((td.isend(t, my_owner, tag=hash(red) % 97),
                             None, t))
            if rank in groups:
                for src in groups[rank]:
                    if src != rank:
                        t = torch.zeros(part.shape, dtype=torch.float64)
                        reqs.append((td.irecv(t, src, tag=hash(red) % 97),
                                     src, t))
            for rq, key, t in reqs:
                rq.wait()
                if key is not None:
                    got[key] = t.numpy()
            if rank in groups:
                acc = None
                for src in groups[rank]:      # ascending source order
                    p = part if src == rank else got[src]
                    acc = p.copy() if acc is None else acc + p
                # owner's slab must equal the reference reduction
                my_nonred = tuple(
                    slice(0, 1) if a in red else
                    slice(idxs[rank][a][0], idxs[rank][a][1])
                    for a in range(2))
                ref = x.sum(axis=red, keepdims=True)[my_nonred]
                assert np.allclose(acc, ref, rtol=1e-12), red
        q.put((rank, True, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        if td.is_initialized():
            td.destroy_process_group()


def _sort_worker(rank, tmpfile, q, world=4):
    """world-4 numpy execution of ops.dsort's segment all-to-all."""
    try:
        _init(rank, tmpfile, world)
        n = 4003
        idxs, _ = pg.chunk_indices((n,), (world,))
        lo, hi = idxs[rank][0]
        full = philox.fill_uniform_f64(n, 33)
        mine = np.sort(full[lo:hi])
        # sampled splitters (same protocol as ops.dsort)
        s = min(64, mine.size)
        sel = ((np.arange(s) + 0.5) * mine.size / s).astype(np.int64)
        samples = mine[sel]
        gathered = [None] * world
        td.all_gather_object(gathered, samples)
        allsamp = np.sort(np.concatenate(gathered))
        splitters = allsamp[[(i + 1) * allsamp.size // world
                             for i in range(world - 1)]]
        edges = [0] + [int(np.searchsorted(mine, sp)) for sp in
                       splitters] + [mine.size]
        segs = [edges[j + 1] - edges[j] for j in range(world)]
        allsegs = [None] * world
        td.all_gather_object(allsegs, segs)
        recv_sizes = [allsegs[src][rank] for src in range(world)]
        reqs, got = [], {}
        for dst in range(world):
            if dst != rank and segs[dst]:
                t = torch.from_numpy(
                    np.ascontiguousarray(mine[edges[dst]:edges[dst + 1]]))
                reqs.append((td.isend(t, dst, tag=dst), None, t))
        for src in range(world):
            if src != rank and recv_sizes[src]:
                t = torch.zeros(recv_sizes[src], dtype=torch.float64)
                reqs.append((td.irecv(t, src, tag=rank), src, t))
        for rq, key, t in reqs:
            rq.wait()
            if key is not None:
                got[key] = t.numpy()
        parts = []
        for src in range(world):
            if src == rank:
                parts.append(mine[edges[rank]:edges[rank + 1]])
            elif src in got:
                parts.append(got[src])
        result = np.sort(np.concatenate(parts)) if parts else \
            np.empty(0)
        gathered = [None] * world
        td.all_gather_object(gathered, result)
        assert np.array_equal(np.concatenate(gathered), np.sort(full))
        q.put((rank, True, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        if td.is_initialized():
            td.destroy_process_group()


@pytest.mark.timeout(300)
def test_gloo_dims_reduce_world4(tmp_path):
    _spawn(_dims_reduce_worker, tmp_path, world=4, extra=(4,))


@pytest.mark.timeout(300)
def test_gloo_sort_world4(tmp_path):
    _spawn(_sort_worker, tmp_path, world=4, extra=(4,))
