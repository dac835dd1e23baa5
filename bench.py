#!/usr/bin/env python3
"""bench.py — measures BASELINE.json's metric on MI355X.

Primary line (N=1 default): cfg2 — `sum(D)` on a 2^28-element Float64
DVector per GPU (weak scaling: global vector is N * 2^28).  `value` is
whole-job GB/s of the sum leg (8 B/elem algorithmic traffic, inputs
resident in HBM).  The same JSON line carries `extra` legs:
map!(sin,D,D) GB/s (cfg2), D .= A.*B .+ c GB/s (cfg3 single-node form),
DArray*DArray 16384^2 TFLOP/s (cfg4), and at N=8 mapreduce(abs2,+,f32)
(cfg5).  `roofline` covers the dominant kernel of the primary leg (the
reduce stage-1 kernel, HBM-bound); `cpu_baseline` is the C/OpenMP oracle
restatement (kind="port") timed on the host cores of the same box.

Launch: python bench.py --gpus N --steps K --warmup W
(N>1 via torch.distributed.run, one rank per GPU; RANK/WORLD_SIZE from
env; gloo is used for control barriers, RCCL via the C ABI for data).
"""
import argparse
import ctypes
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def log(msg):
    print(msg, file=sys.stderr, flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--no-gemm", action="store_true")
    ap.add_argument("--elems", type=int, default=1 << 28,
                    help="per-GPU f64 elements for the sum/map legs")
    ap.add_argument("--gemm-n", type=int, default=16384)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    if world != args.gpus and "WORLD_SIZE" in os.environ:
        args.gpus = world

    # import order matters: our package pins /opt/rocm's HIP runtime
    # before torch loads its bundled one (see _ffi.py)
    import distributedarrays_jl_amd as dja
    from distributedarrays_jl_amd._ffi import lib, check
    import torch
    import torch.distributed as td
    if world > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        td.init_process_group("gloo", rank=rank, world_size=world)
    dja.comm.init()

    def barrier():
        check(lib.da_synchronize())
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        if world > 1:
            td.barrier()

    def max_over_ranks(t):
        if world > 1:
            tt = torch.tensor([t], dtype=torch.float64)
            td.all_reduce(tt, op=td.ReduceOp.MAX)
            return float(tt.item())
        return t

    def ev_pair():
        e0, e1 = ctypes.c_void_p(), ctypes.c_void_p()
        check(lib.da_event_create(ctypes.byref(e0)))
        check(lib.da_event_create(ctypes.byref(e1)))
        return e0, e1

    def ev_ms(e0, e1):
        ms = ctypes.c_float()
        check(lib.da_event_elapsed(e0, e1, ctypes.byref(ms)))
        return float(ms.value)

    K, W = args.steps, args.warmup
    n = args.elems
    extra = {}

    # ---------------- cfg2 primary: sum(D), 2^28 f64 per GPU ----------
    log("[bench] alloc + drand %d f64 elems/GPU" % n)
    D = dja.DArray((n * world,), "f64", (world,))
    D.rand_()
    for _ in range(W):
        dja.dsum(D)
    # dominant-kernel duration via HIP events on the library stream,
    # probed OUTSIDE the timed region (events cost a few us per record)
    e0, e1 = ev_pair()
    kern_ms = []
    for _ in range(max(5, W)):
        check(lib.da_event_record(e0))
        dja.dsum(D)
        check(lib.da_event_record(e1))
        kern_ms.append(ev_ms(e0, e1))
    barrier()
    t0 = time.perf_counter()
    for _ in range(K):
        s = dja.dsum(D)
    barrier()
    t_sum = max_over_ranks(time.perf_counter() - t0)
    sum_gbs = world * n * 8.0 * K / t_sum / 1e9
    kms = sorted(kern_ms)[len(kern_ms) // 2]
    log("[bench] sum: %.1f GB/s whole-job, %.3f ms/step (launch %.3f ms), "
        "value=%r" % (sum_gbs, t_sum / K * 1e3, kms, s))

    # ---------------- cfg2: map!(sin, D, D) ----------------------------
    M = dja.DArray((n * world,), "f64", (world,))
    M.rand_()
    for _ in range(W):
        dja.map_("sin", M, M)
    barrier()
    t0 = time.perf_counter()
    for _ in range(K):
        dja.map_("sin", M, M)
    barrier()
    t_map = max_over_ranks(time.perf_counter() - t0)
    extra["map_sin_gbs"] = world * n * 16.0 * K / t_map / 1e9  # R+W
    M.close()
    log("[bench] map!(sin): %.1f GB/s" % extra["map_sin_gbs"])

    # ---------------- cfg3: D .= A .* B .+ c ---------------------------
    # At world==4 this is EXACTLY BASELINE cfg3: 32768^2 Float64 on a
    # 1-D column distribution (2 GiB/GPU); other worlds weak-scale the
    # same bytes as a DVector.
    if world == 4:
        # side^2 == 4 * elems: 32768^2 at the default 2^28/GPU
        side = 2 * int(round(n ** 0.5))
        shp, dist3 = (side, side), (1, 4)
    else:
        shp, dist3 = (n * world,), (world,)
    A3 = dja.DArray(shp, "f64", dist3); A3.rand_()
    B3 = dja.DArray(shp, "f64", dist3); B3.rand_()
    D3 = dja.dzeros(shp, "f64", dist3)
    n3 = A3.size // world
    for _ in range(W):
        dja.broadcast_fma(D3, A3, B3, 0.5)
    barrier()
    t0 = time.perf_counter()
    for _ in range(K):
        dja.broadcast_fma(D3, A3, B3, 0.5)
    barrier()
    t_bc = max_over_ranks(time.perf_counter() - t0)
    extra["bcast_fma_gbs"] = world * n3 * 24.0 * K / t_bc / 1e9  # 2R+1W
    extra["bcast_fma_config"] = "%r f64 dist %r" % (shp, dist3)
    for d in (A3, B3, D3):
        d.close()
    log("[bench] bcast fma: %.1f GB/s" % extra["bcast_fma_gbs"])

    # ---------------- fused broadcast composition (da_expr JIT) --------
    # D .= sin.(A) .+ B .* c in ONE kernel (nested broadcast,
    # test/darray.jl:880-912 analog) — algorithmic 24 B/elem.
    try:
        from distributedarrays_jl_amd import expr as E
        Ax = dja.DArray((n * world,), "f64", (world,)); Ax.rand_()
        Bx = dja.DArray((n * world,), "f64", (world,)); Bx.rand_()
        Dx = dja.dzeros((n * world,), "f64", (world,))
        ex = E.sin(E.ref(Ax)) + E.ref(Bx) * 0.5
        for _ in range(W):
            E.materialize_(Dx, ex)
        barrier()
        t0 = time.perf_counter()
        for _ in range(K):
            E.materialize_(Dx, ex)
        barrier()
        t_ex = max_over_ranks(time.perf_counter() - t0)
        extra["expr_fused_gbs"] = world * n * 24.0 * K / t_ex / 1e9
        extra["expr_jit_state"] = int(lib.da_expr_jit_state())
        for d in (Ax, Bx, Dx):
            d.close()
        log("[bench] expr sin-fused: %.1f GB/s (jit state %d)"
            % (extra["expr_fused_gbs"], extra["expr_jit_state"]))
    except Exception as e:
        extra["expr_error"] = repr(e)[:200]
        log("[bench] expr leg failed: %r" % (e,))

    # ---------------- cfg5 (N==8): mapreduce(abs2,+,f32 2^31) ----------
    if world == 8:
        try:
            n5 = 8 * n     # 2^31 global at the default 2^28/GPU
            F = dja.DArray((n5,), "f32", (world,))
            F.rand_()
            for _ in range(W):
                dja.mapreduce("abs2", "add", F)
            barrier()
            t0 = time.perf_counter()
            for _ in range(K):
                dja.mapreduce("abs2", "add", F)
            barrier()
            t5 = max_over_ranks(time.perf_counter() - t0)
            extra["mapreduce_abs2_f32_gbs"] = n5 * 4.0 * K / t5 / 1e9
            F.close()
            log("[bench] cfg5: %.1f GB/s"
                % extra["mapreduce_abs2_f32_gbs"])
        except Exception as e:
            extra["cfg5_error"] = repr(e)[:200]
            log("[bench] cfg5 leg failed: %r" % (e,))

    # ---------------- cfg4: DArray*DArray 16384^2 f64 -------------------
    # (the primary `value` is already measured; an extra-leg failure must
    # never kill the run.  If the comm/compute-overlapped exchange path
    # fails at N>1, retry once with the proven single-group schedule.)
    if not args.no_gemm:
        def gemm_leg():
            gn = args.gemm_n
            GA = dja.DArray((gn, gn), "f64"); GA.rand_()
            GB = dja.DArray((gn, gn), "f64"); GB.rand_()
            gsteps = max(1, min(K, 3))
            C = dja.dmatmul(GA, GB)  # warmup
            C.close()
            barrier()
            t0 = time.perf_counter()
            for _ in range(gsteps):
                C = dja.dmatmul(GA, GB)
                C.close()
            barrier()
            t_mm = max_over_ranks(time.perf_counter() - t0)
            extra["gemm_tflops"] = 2.0 * gn ** 3 * gsteps / t_mm / 1e12
            extra["gemm_config"] = "%d^2 x %d^2 f64, %d ranks" \
                % (gn, gn, world)
            GA.close(); GB.close()
            log("[bench] gemm: %.2f TFLOP/s" % extra["gemm_tflops"])
        try:
            gemm_leg()
        except Exception as e:
            log("[bench] gemm leg failed (%r); retrying without overlap"
                % (e,))
            os.environ["DA_MM_OVERLAP"] = "0"
            try:
                gemm_leg()
                extra["gemm_note"] = "overlap disabled after failure"
            except Exception as e2:
                extra["gemm_error"] = repr(e2)[:200]
                log("[bench] gemm leg failed again: %r" % (e2,))

    D.close()

    # ---------------- informational legs (N==1 only) --------------------
    if world == 1 and not args.no_gemm:
        gn = args.gemm_n
        GA = dja.DArray((gn, gn), "f32"); GA.rand_()
        GB = dja.DArray((gn, gn), "f32"); GB.rand_()
        C = dja.dmatmul(GA, GB); C.close()
        barrier()
        t0 = time.perf_counter()
        for _ in range(3):
            C = dja.dmatmul(GA, GB); C.close()
        barrier()
        extra["gemm_f32_tflops"] = 2.0 * gn ** 3 * 3 / \
            (time.perf_counter() - t0) / 1e12
        GA.close(); GB.close()
        log("[bench] gemm f32: %.1f TFLOP/s" % extra["gemm_f32_tflops"])

        S = dja.drand((1 << 27,), "f64")
        r = dja.dsort(S); r.close()
        barrier()
        t0 = time.perf_counter()
        for _ in range(3):
            r = dja.dsort(S); r.close()
        barrier()
        extra["sort_gkeys_per_s"] = (1 << 27) * 3 / \
            (time.perf_counter() - t0) / 1e9
        S.close()
        log("[bench] sort: %.2f Gkeys/s" % extra["sort_gkeys_per_s"])

    # ---------------- CPU baseline (rank 0, N==1 only) ------------------
    cpu_baseline = None
    if rank == 0 and world == 1:
        from oracle import cpu_baseline as cbl
        clib = cbl.load()
        import numpy as np
        h = np.empty(n, dtype=np.float64)
        clib.cb_fill_uniform_f64(cbl.ptr(h), n, 1234)
        clib.cb_sum_f64(cbl.ptr(h), n)  # warm
        reps = []
        for _ in range(5):
            t0 = time.perf_counter()
            clib.cb_sum_f64(cbl.ptr(h), n)
            reps.append(time.perf_counter() - t0)
        tc = sorted(reps)[2]
        cpu_baseline = {
            "value": n * 8.0 / tc / 1e9, "unit": "GB/s",
            "cores": int(clib.cb_num_threads()), "kind": "port",
            "sample": "full 2^28-elem f64 sum, median of 5 (%.0f ms each)"
                      % (tc * 1e3),
        }
        del h
        log("[bench] cpu baseline: %.1f GB/s on %d cores"
            % (cpu_baseline["value"], cpu_baseline["cores"]))

    # ---------------- roofline of the dominant kernel -------------------
    traffic = None
    tf = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "profiles", "traffic.json")
    if os.path.exists(tf):
        try:
            traffic = json.load(open(tf)).get("reduce_bytes_per_launch")
        except Exception:
            traffic = None
    achieved = n * 8.0 / (kms / 1e3) / 1e9  # algorithmic GB/s per launch
    roofline = {
        "bound": "hbm", "achieved": achieved, "peak": 8000.0,
        "unit": "GB/s", "frac": achieved / 8000.0, "traffic": traffic,
    }

    result = {
        "metric": "sum(DArray) GB/s",
        "value": sum_gbs,
        "unit": "GB/s",
        "n_gpus": world,
        "steps": K,
        "warmup": W,
        "ms_per_step": t_sum / K * 1e3,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "f64",
        "data": "synthetic",
        "config": {
            "workload": "cfg2: sum(drand(Float64, 2^28)) per GPU, "
                        "1-D block distribution",
            "per_gpu_elems": n,
            "global_elems": n * world,
        },
        "roofline": roofline,
        "cpu_baseline": cpu_baseline,
        "extra": extra,
    }
    if rank == 0:
        print(json.dumps(result), flush=True)
    if world > 1:
        td.destroy_process_group()


if __name__ == "__main__":
    main()
