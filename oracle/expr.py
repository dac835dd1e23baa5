"""Broadcast-composition semantics — CPU restatement (oracle side).

Restates the reference's Broadcasted-tree materialization
(/root/reference/src/broadcast.jl:65-98: the per-worker
`copyto!(localpart(dest), bclocal(bc))` evaluates the whole tree
elementwise in one pass, with Julia's singleton-dim expansion;
nested broadcast pinned at /root/reference/test/darray.jl:880-912).

Evaluates the SAME postfix encoding the product ships to da_expr
(include/darray_hip.h: kind = ins>>8 — 0 unary, 1 push-arg,
2 push-const, 3 binary), using the oracle's numpy functor tables, so a
test can compare the GPU kernel against this on identical inputs.
Constants are cast to the computation dtype before use (Julia converts
broadcast scalars to the promoted element type)."""
import numpy as np

from . import ops as oops

# opcode index -> name tables mirror include/darray_hip.h enum order
# (cross-checked against the header by tests/test_abi.py via
# distributedarrays_jl_amd/_opcodes.py)
MAP_NAMES = [
    "identity", "neg", "abs", "abs2", "inv",
    "sqrt", "cbrt", "exp", "exp2", "exp10",
    "expm1", "log", "log2", "log10", "log1p",
    "sin", "cos", "tan", "asin", "acos", "atan",
    "sinh", "cosh", "tanh", "asinh", "acosh",
    "atanh", "sinpi", "cospi", "floor", "ceil",
    "round", "trunc", "sign", "deg2rad", "rad2deg",
    "sec", "csc", "cot",
    "erf", "erfc", "erfinv", "erfcinv", "erfcx",
    "gamma", "lgamma", "sinc", "cosc",
    "sind", "cosd", "tand", "asind", "acosd",
    "atand", "acot", "acotd", "asec", "acsc",
    "asech", "acsch", "acoth",
    "isnan", "isinf", "isfinite",
]
MAP2_NAMES = [
    "add", "sub", "mul", "div", "min2",
    "max2", "idiv", "mod", "rem", "and",
    "or", "xor", "pow", "atan2",
]


def evaluate(prog, args, consts, dtype=None):
    """Evaluate a postfix broadcast program over numpy arrays.

    args may have singleton dims (numpy broadcasting == Julia's
    expansion rule for the shapes the product accepts)."""
    if dtype is None:
        dtype = args[0].dtype if args else np.dtype("float64")
    dtype = np.dtype(dtype)
    stack = []
    for ins in prog:
        kind, idx = ins >> 8, ins & 0xFF
        if kind == 1:
            stack.append(args[idx])
        elif kind == 2:
            stack.append(dtype.type(consts[idx]))
        elif kind == 0:
            stack.append(oops.MAP_OPS[MAP_NAMES[idx]](
                np.asarray(stack.pop(), dtype=dtype)))
        elif kind == 3:
            b = stack.pop()
            a = stack.pop()
            stack.append(oops.MAP2_OPS[MAP2_NAMES[idx]](a, b))
        else:
            raise ValueError("bad instruction kind %d" % kind)
    assert len(stack) == 1
    return np.asarray(stack[0], dtype=dtype)
