"""End-to-end rehearsal of bench.py's multi-rank branches on CPU
(fakelib transport): the driver runs `bench.py --gpus 4/8` for the
first time at round end, so the world-4 cfg3 2-D branch, the world-8
cfg5 leg, the gemm leg's exchange schedules and the rank-0 JSON
contract are exercised here at tiny sizes first."""
import json
import os
import sys

import pytest
import torch.multiprocessing as mp


def _bench_worker(rank, tmpfile, q, world):
    try:
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(29000 + world)
        import fakelib
        fakelib.install()
        sys.argv = ["bench.py", "--gpus", str(world), "--steps", "2",
                    "--warmup", "1", "--elems", str(1 << 12),
                    "--gemm-n", "96"]
        import io
        import contextlib
        import bench
        buf = io.StringIO()
        with contextlib.redirect_stdout(buf):
            bench.main()
        out = buf.getvalue().strip()
        if rank == 0:
            rec = json.loads(out.splitlines()[-1])
            assert rec["n_gpus"] == world
            assert rec["value"] > 0
            assert rec["scaling"] == "weak"
            extra = rec["extra"]
            assert extra["map_sin_gbs"] > 0
            assert extra["bcast_fma_gbs"] > 0
            if world == 4:
                # the 2-D column-split cfg3 branch (32768^2 at the
                # default --elems; side scales with the rehearsal size)
                assert "dist (1, 4)" in extra["bcast_fma_config"], \
                    extra["bcast_fma_config"]
            if world == 8:
                assert ("mapreduce_abs2_f32_gbs" in extra
                        or "cfg5_error" in extra), extra
                assert "cfg5_error" not in extra, extra["cfg5_error"]
            assert "gemm_error" not in extra, extra.get("gemm_error")
            assert extra["gemm_tflops"] > 0
        q.put((rank, True, None))
    except SystemExit as e:
        q.put((rank, e.code in (0, None), "SystemExit %r" % e.code))
    except Exception:
        import traceback
        q.put((rank, False, traceback.format_exc()[-2000:]))


def _spawn(tmp_path, world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_bench_worker,
                         args=(r, str(tmp_path / "rdv"), q, world))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=420) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok, err in sorted(results):
        assert ok, "rank %d failed:\n%s" % (rank, err)


@pytest.mark.timeout(600)
def test_bench_world2(tmp_path):
    _spawn(tmp_path, 2)


@pytest.mark.timeout(600)
def test_bench_world4_cfg3_branch(tmp_path):
    _spawn(tmp_path, 4)


@pytest.mark.timeout(600)
def test_bench_world8_cfg5_branch(tmp_path):
    _spawn(tmp_path, 8)
