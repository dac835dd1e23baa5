"""Opcode tables — must match include/darray_hip.h enums exactly
(tests/test_abi.py cross-checks names against the header)."""

DTYPES = {"f64": 0, "f32": 1, "i64": 2}

MAP_OPS = [
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
MAP_OP = {name: i for i, name in enumerate(MAP_OPS)}

MAP2_OPS = [
    "add", "sub", "mul", "div", "min2",
    "max2", "idiv", "mod", "rem", "and",
    "or", "xor", "pow", "atan2",
]
MAP2_OP = {name: i for i, name in enumerate(MAP2_OPS)}

RED_OPS = {"add": 0, "mul": 1, "min": 2, "max": 3}
RED_FS = {"identity": 0, "abs": 1, "abs2": 2, "isnan": 3, "isfinite": 4,
          "nonzero": 5}
RAND_KINDS = {"uniform": 0, "normal": 1}

I64_MAP_OPS = {"identity", "neg", "abs", "abs2", "sign"}
I64_MAP2_OPS = {"add", "sub", "mul", "idiv", "mod", "rem",
                "and", "or", "xor", "min2", "max2"}

NUMPY_DTYPES = {"f64": "float64", "f32": "float32", "i64": "int64"}
DTYPE_SIZE = {"f64": 8, "f32": 4, "i64": 8}
