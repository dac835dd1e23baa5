"""makelocal halo gather (darray.jl:351-368):
plan coverage on CPU, gloo world-2 execution of a mismatched-cuts map!,
and GPU pack/unpack of the local-copy path."""
import numpy as np
import pytest

from distributedarrays_jl_amd import geometry as pg, plan
from oracle import philox


def test_halo_plan_covers_box():
    idxs, _ = pg.chunk_indices((40, 30), (2, 2))
    ranks = list(range(4))
    box = ((5, 35), (3, 28))
    pieces = plan.halo_plan(idxs, ranks, [box, None, None, None])
    grid = np.zeros((40, 30), dtype=int)
    for (src, dst, b) in pieces:
        assert dst == 0
        sl = tuple(slice(lo, hi) for lo, hi in b)
        grid[sl] += 1
    inside = grid[5:35, 3:28]
    assert (inside == 1).all()
    assert grid.sum() == inside.size


def test_halo_plan_local_only():
    idxs, _ = pg.chunk_indices((100,), (2,))
    pieces = plan.halo_plan(idxs, [0, 1], [((0, 50),), ((50, 100),)])
    assert all(src == dst for src, dst, _ in pieces)


def _mismatched_map_worker(rank, tmpfile, q):
    """world-2 numpy execution of map_general's gather schedule."""
    import os
    import torch
    import torch.distributed as td
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        td.init_process_group("gloo", init_method="file://%s" % tmpfile,
                              rank=rank, world_size=2)
        dims = (8, 12)
        x = np.asfortranarray(philox.fill_uniform_f64(96, 3)
                              .reshape(dims, order="F"))
        src_dist, dst_dist = (2, 1), (1, 2)
        s_idxs, _ = pg.chunk_indices(dims, src_dist)
        d_idxs, _ = pg.chunk_indices(dims, dst_dist)
        src_loc = x[tuple(slice(lo, hi) for lo, hi in s_idxs[rank])].copy(order="F")
        boxes = [d_idxs[0], d_idxs[1]]
        pieces = plan.halo_plan(s_idxs, [0, 1], boxes)
        mybox = boxes[rank]
        oshape = tuple(hi - lo for lo, hi in mybox)
        out = np.zeros(oshape, order="F")
        reqs = []
        for tag, (src, dst, b) in enumerate(pieces):
            sl_in_src = tuple(slice(lo - s_idxs[src][d][0],
                                    hi - s_idxs[src][d][0])
                              for d, (lo, hi) in enumerate(b))
            if src == rank and dst == rank:
                sl_out = tuple(slice(lo - mybox[d][0], hi - mybox[d][0])
                               for d, (lo, hi) in enumerate(b))
                out[sl_out] = src_loc[sl_in_src]
            elif src == rank:
                t = torch.from_numpy(
                    np.ascontiguousarray(src_loc[sl_in_src]))
                reqs.append((td.isend(t, dst, tag=tag), None, t, b))
            elif dst == rank:
                t = torch.zeros([hi - lo for lo, hi in b],
                                dtype=torch.float64)
                reqs.append((td.irecv(t, src, tag=tag), "recv", t, b))
        for rq, kind, t, b in reqs:
            rq.wait()
            if kind == "recv":
                sl_out = tuple(slice(lo - mybox[d][0], hi - mybox[d][0])
                               for d, (lo, hi) in enumerate(b))
                out[sl_out] = t.numpy()
        got = np.sin(out)
        ref = np.sin(x[tuple(slice(lo, hi) for lo, hi in mybox)])
        q.put((rank, bool(np.allclose(got, ref, rtol=1e-15)), None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, repr(e)))
    finally:
        import torch.distributed as td2
        if td2.is_initialized():
            td2.destroy_process_group()


@pytest.mark.timeout(300)
def test_gloo_mismatched_map_world2(tmp_path):
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    tmpfile = str(tmp_path / "rdv")
    procs = [ctx.Process(target=_mismatched_map_worker,
                         args=(r, tmpfile, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok, err in results:
        assert ok, "rank %d: %s" % (rank, err)


@pytest.mark.gpu
def test_gpu_gather_box_local():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    x = np.asfortranarray(philox.fill_uniform_f64(50 * 30, 5)
                          .reshape(50, 30, order="F"))
    d = dja.distribute(x)
    box = ((7, 41), (3, 27))
    buf, shape = dja.gather_box(d, [box])
    assert shape == (34, 24)
    import ctypes
    out = np.empty(shape, dtype=np.float64, order="F")
    from distributedarrays_jl_amd._ffi import lib, check
    check(lib.da_d2h(buf.p, out.ctypes.data_as(ctypes.c_void_p),
                     out.size * 8))
    assert np.array_equal(out, x[7:41, 3:27])
    buf.free()
    # 1-D case
    v = philox.fill_uniform_f64(1000, 6)
    dv = dja.distribute(v)
    buf, shape = dja.gather_box(dv, [((100, 900),)])
    out = np.empty(shape, dtype=np.float64, order="F")
    check(lib.da_d2h(buf.p, out.ctypes.data_as(ctypes.c_void_p),
                     out.size * 8))
    assert np.array_equal(out, v[100:900])
    buf.free()
    d.close(); dv.close()


@pytest.mark.gpu
def test_gpu_map_general_aligned_fastpath():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    x = philox.fill_uniform_f64(4096, 8)
    d = dja.distribute(x)
    o = d.similar()
    dja.map_general("abs2", o, d)
    assert np.array_equal(o.localpart(), x * x)
    o.close(); d.close()


@pytest.mark.gpu
@pytest.mark.parametrize("shape", [(64, 64), (50, 30), (33, 97), (1, 7)])
def test_gpu_dtranspose(shape):
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    m, n = shape
    x = np.asfortranarray(philox.fill_uniform_f64(m * n, 21)
                          .reshape(shape, order="F"))
    d = dja.distribute(x)
    t = dja.dtranspose(d)
    assert t.dims == (n, m)
    assert np.array_equal(t.collect(), np.asfortranarray(x.T))
    t.close(); d.close()


@pytest.mark.gpu
def test_gpu_diag_scale():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    m, n = 40, 30
    x = np.asfortranarray(philox.fill_uniform_f64(m * n, 22)
                          .reshape(m, n, order="F"))
    dl = philox.fill_uniform_f64(m, 23)
    dr = philox.fill_uniform_f64(n, 24)
    d = dja.distribute(x)
    dja.ddiag_lmul(dl, d)
    ref = dl[:, None] * x
    assert np.array_equal(d.localpart(), ref)
    dja.ddiag_rmul(d, dr)
    ref = ref * dr[None, :]
    assert np.array_equal(d.localpart(), ref)
    d.close()


@pytest.mark.gpu
def test_gpu_gather_box_3d():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    shape = (20, 16, 12)
    x = np.asfortranarray(philox.fill_uniform_f64(int(np.prod(shape)), 31)
                          .reshape(shape, order="F"))
    d = dja.distribute(x)
    box = ((3, 17), (2, 14), (1, 11))
    buf, bshape = dja.gather_box(d, [box])
    assert bshape == (14, 12, 10)
    import ctypes
    from distributedarrays_jl_amd._ffi import lib, check
    out = np.empty(bshape, dtype=np.float64, order="F")
    check(lib.da_d2h(buf.p, out.ctypes.data_as(ctypes.c_void_p),
                     out.size * 8))
    assert np.array_equal(out, x[3:17, 2:14, 1:11])
    buf.free()
    d.close()
