"""Generate tests/golden/golden.npz — frozen known-answer fixtures for
the exactly-reproducible parts of the hot path (philox streams, chunk
geometry, integer reductions, fma broadcast).  The oracle is pinned
against these (tests/test_golden.py) and the GPU path is pinned
bit-exact against the oracle (tests/test_gpu_parity.py), closing the
chain.  Regenerate only with a documented protocol change:

    python tests/golden/generate.py
"""
import os
import sys

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
sys.path.insert(0, ROOT)

from oracle import philox, geometry, ops  # noqa: E402


def main():
    out = {}
    # philox streams (protocol of BASELINE.md / oracle/philox.py)
    for seed in (1234, 1235, 77):
        out["philox_f64_%d" % seed] = philox.fill_uniform_f64(64, seed)
        out["philox_f32_%d" % seed] = philox.fill_uniform_f32(64, seed)
        out["philox_i64_%d" % seed] = philox.fill_int64(64, seed)
    out["philox_f64_1234_off1000"] = philox.fill_uniform_f64(
        16, 1234, offset=1000)

    # geometry (darray.jl:251-307; includes the reference pin 50/4)
    cuts = []
    cases = [(50, 4), (100, 8), (7, 3), (2, 4), (2 ** 28, 8), (16384, 2)]
    for sz, nc in cases:
        cuts.append(geometry.defaultdist_1d(sz, nc))
    out["cuts_cases"] = np.array([c for c in cases], dtype=np.int64)
    maxlen = max(len(c) for c in cuts)
    out["cuts_values"] = np.array(
        [c + [-1] * (maxlen - len(c)) for c in cuts], dtype=np.int64)
    out["dist_16384sq_8"] = np.array(
        geometry.defaultdist_dims([16384, 16384], 8), dtype=np.int64)

    # integer reductions (wrap-exact, any order)
    with np.errstate(over="ignore"):
        x = philox.fill_int64(4096, 42)
        idxs, _ = geometry.chunk_idxs([4096], [4])
        chunks = ops.make_chunks(x, idxs)
        out["i64_input_seed42_head"] = x[:32]
        out["i64_sum"] = np.int64(ops.oracle_reduce("identity", "add",
                                                    chunks))
        out["i64_max"] = np.int64(ops.oracle_reduce("identity", "max",
                                                    chunks))
        out["i64_min"] = np.int64(ops.oracle_reduce("identity", "min",
                                                    chunks))

    # exact float elementwise (mul-then-add, no fma)
    a = philox.fill_uniform_f64(128, 7)
    b = philox.fill_uniform_f64(128, 8)
    out["fma_a_seed7"] = a
    out["fma_out"] = ops.oracle_bcast_fma(a, b, 0.25)
    out["abs2_out"] = ops.oracle_map("abs2", a)

    np.savez_compressed(os.path.join(os.path.dirname(
        os.path.abspath(__file__)), "golden.npz"), **out)
    print("wrote golden.npz with %d arrays" % len(out))


if __name__ == "__main__":
    main()
