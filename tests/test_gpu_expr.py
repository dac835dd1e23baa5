"""GPU parity of the fused broadcast-composition kernel (da_expr) vs the
oracle evaluator on identical seeded inputs — mirrors the reference's
broadcast testset incl. nested broadcast
(/root/reference/test/darray.jl:880-912) and the exact/tolerance split
of the scalar-math suite (:775-800)."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dja():
    import distributedarrays_jl_amd as dja
    dja.comm.init()
    yield dja
    dja.d_closeall()


from oracle import philox
import oracle.expr as oexpr
from distributedarrays_jl_amd import expr as E


def _mk(dja, arr):
    import distributedarrays_jl_amd as _d
    d = _d.DArray(arr.shape, {np.dtype("float64"): "f64",
                              np.dtype("float32"): "f32",
                              np.dtype("int64"): "i64"}[arr.dtype])
    d.set_localpart(np.asfortranarray(arr))
    return d


def test_flat_bitexact_chain(dja):
    """+,-,*,/,abs chains are bit-exact (shared functors,
    -ffp-contract=off)."""
    n = 100003
    A = philox.fill_uniform_f64(n, 1)
    B = philox.fill_uniform_f64(n, 2)
    da, db = _mk(dja, A), _mk(dja, B)
    e = (E.ref(da) + E.ref(db)) / (abs(E.ref(da)) + 1.0) - \
        E.ref(db) * 0.25
    prog, args, consts = E.compile_expr(e)
    ref = oexpr.evaluate(prog, [A, B], consts)
    out = E.materialize(e)
    assert np.array_equal(out.localpart(), ref)
    da.close(); db.close(); out.close()


def test_single_op_matches_da_map(dja):
    """A one-op program is bit-identical to the da_map kernel."""
    n = 65537
    x = philox.fill_uniform_f64(n, 3)
    d = _mk(dja, x)
    for op in ("sin", "exp", "sqrt", "abs2"):
        m = dja.dmap(op, d)
        e = E.materialize(getattr(E, op)(E.ref(d)))
        assert np.array_equal(m.localpart(), e.localpart()), op
        m.close(); e.close()
    d.close()


def test_nested_broadcast_sin(dja):
    """D .= sin.(A) .+ B .* c — one kernel, algorithmic traffic only
    (the VERDICT-r1 headline gap)."""
    n = 1 << 20
    A = philox.fill_uniform_f64(n, 4)
    B = philox.fill_uniform_f64(n, 5)
    da, db = _mk(dja, A), _mk(dja, B)
    D = dja.DArray((n,), "f64")
    e = E.sin(E.ref(da)) + E.ref(db) * 0.5
    prog, args, consts = E.compile_expr(e)
    E.materialize_(D, e)
    ref = oexpr.evaluate(prog, [A, B], consts)
    got = D.localpart()
    # sin is OCML (few-ulp); the + and * around it are exact given the
    # same sin values, so compare against oracle-with-numpy-sin at ulp
    # tolerance
    assert np.allclose(got, ref, rtol=5e-16, atol=5e-16)
    da.close(); db.close(); D.close()


def test_dims_expanded_operand(dja):
    """a .- mean(a, dims=1): the (1, ncols) operand expands via stride-0
    (the pinned reference form, test/darray.jl:885-897)."""
    nr, nc = 64, 40
    A = philox.fill_uniform_f64(nr * nc, 6).reshape((nr, nc), order="F")
    da = _mk(dja, A)
    M = dja.dmean_dims(da, (0,))        # (1, nc) DArray
    D = dja.DArray((nr, nc), "f64")
    E.materialize_(D, E.ref(da) - E.ref(M))
    ref = A - M.collect()               # same mean values the GPU made
    assert np.array_equal(D.localpart(), ref)
    # nested with dims-expansion: g = a .- m .* sin.(c)
    C = dja.DArray((nr, nc), "f64")
    E.materialize_(C, E.ref(da) - E.ref(M))     # c = a .- m
    G = dja.DArray((nr, nc), "f64")
    E.materialize_(G, E.ref(da) - E.ref(M) * E.sin(E.ref(C)))
    refc = A - M.collect()
    refg = A - M.collect() * np.sin(refc)
    assert np.allclose(G.localpart(), refg, rtol=5e-16, atol=5e-16)
    for d in (da, M, D, C, G):
        d.close()


def test_scalar_only_broadcast(dja):
    """a .= 3 .+ abs2.(zeros-shaped view) — in-place constant fill form
    (test/darray.jl:899-905)."""
    nr, nc = 32, 16
    Z = _mk(dja, np.zeros((nr, nc)))
    A = dja.DArray((nr, nc), "f64")
    E.materialize_(A, 3.0 + E.abs2(E.ref(Z)))
    assert (A.localpart() == 3.0).all()
    Z.close(); A.close()


def test_f32_and_i64_flat(dja):
    n = 50001
    xf = philox.fill_uniform_f32(n, 7)
    d = _mk(dja, xf)
    e = E.ref(d) * np.float64(0.5) + 1.0
    prog, args, consts = E.compile_expr(e)
    ref = oexpr.evaluate(prog, [xf], consts, np.dtype("float32"))
    out = E.materialize(e)
    assert np.array_equal(out.localpart(), ref)
    d.close(); out.close()
    xi = philox.fill_int64(n, 8) % 1000
    di = _mk(dja, xi)
    ei = abs(E.ref(di)) * 3.0 - 5.0
    progi, argsi, constsi = E.compile_expr(ei)
    refi = oexpr.evaluate(progi, [xi], constsi, np.dtype("int64"))
    outi = E.materialize(ei)
    assert np.array_equal(outi.localpart(), refi)
    di.close(); outi.close()


def test_aliased_inplace(dja):
    """a .= f.(a, …): destination aliases a source (elementwise-safe)."""
    n = 30011
    A = philox.fill_uniform_f64(n, 9)
    B = philox.fill_uniform_f64(n, 10)
    da, db = _mk(dja, A), _mk(dja, B)
    E.materialize_(da, E.ref(da) * E.ref(db) + 0.125)
    assert np.array_equal(da.localpart(), A * B + 0.125)
    da.close(); db.close()


def test_expr_vs_bcast_fma_bitexact(dja):
    """The generic interpreter reproduces the specialized cfg-3 kernel
    bit-for-bit."""
    n = 1 << 18
    A = philox.fill_uniform_f64(n, 11)
    B = philox.fill_uniform_f64(n, 12)
    da, db = _mk(dja, A), _mk(dja, B)
    D1 = dja.DArray((n,), "f64")
    dja.broadcast_fma(D1, da, db, 0.5)
    D2 = dja.DArray((n,), "f64")
    E.materialize_(D2, E.ref(da) * E.ref(db) + 0.5)
    assert np.array_equal(D1.localpart(), D2.localpart())
    for d in (da, db, D1, D2):
        d.close()


def test_jit_vs_interpreter_bitexact(dja):
    """The hipRTC-compiled kernel and the interpreter share the functor
    tables — results must be bit-identical.  DA_EXPR_JIT is read per
    call, so both paths run in this process."""
    import os
    from distributedarrays_jl_amd._ffi import lib
    n = 200003
    A = philox.fill_uniform_f64(n, 21)
    B = philox.fill_uniform_f64(n, 22)
    da, db = _mk(dja, A), _mk(dja, B)
    e = E.sin(E.ref(da)) + E.ref(db) * 0.5 - E.exp(E.ref(da) * -1.0)
    out_jit = E.materialize(e)
    jit_state = int(lib.da_expr_jit_state())
    os.environ["DA_EXPR_JIT"] = "0"
    try:
        out_interp = E.materialize(e)
    finally:
        del os.environ["DA_EXPR_JIT"]
    assert np.array_equal(out_jit.localpart(), out_interp.localpart())
    # the JIT must actually be ACTIVE on the GPU box (2), not silently
    # failed (-1) — a fallback here means lost performance
    assert jit_state == 2, \
        "expr JIT not active: state %d, err %r" % (
            jit_state, lib.da_expr_jit_errstr())
    # strided variant too (dims-expanded operand)
    nr, nc = 48, 32
    X = _mk(dja, philox.fill_uniform_f64(nr * nc, 23)
            .reshape((nr, nc), order="F"))
    M = dja.dmean_dims(X, (0,))
    D1 = dja.DArray((nr, nc), "f64")
    E.materialize_(D1, E.ref(X) - E.ref(M))
    os.environ["DA_EXPR_JIT"] = "0"
    try:
        D2 = dja.DArray((nr, nc), "f64")
        E.materialize_(D2, E.ref(X) - E.ref(M))
    finally:
        del os.environ["DA_EXPR_JIT"]
    assert np.array_equal(D1.localpart(), D2.localpart())
    for d in (da, db, out_jit, out_interp, X, M, D1, D2):
        d.close()


def test_validation_errors(dja):
    from distributedarrays_jl_amd._ffi import DArrayError
    d = dja.dzeros((16, 8))
    bad = dja.dzeros((16, 7))
    with pytest.raises(DArrayError):
        E.materialize_(d, E.ref(bad) + 1.0)
    d.close(); bad.close()


def test_outer_product_broadcast(dja):
    """row .* col -> (m, n): no full-shape operand (both stride-0
    expand from projected boxes)."""
    m, n = 48, 32
    grow = philox.fill_uniform_f64(n, 30).reshape((1, n), order="F")
    gcol = philox.fill_uniform_f64(m, 31).reshape((m, 1), order="F")
    R = _mk(dja, grow)
    C = _mk(dja, gcol)
    O = E.materialize(E.ref(C) * E.ref(R) + 1.0)
    assert O.dims == (m, n)
    assert np.array_equal(O.localpart(), gcol * grow + 1.0)
    O.close(); R.close(); C.close()
