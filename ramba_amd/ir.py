"""Fused-group op IR.

The reference records each deferred op as a Python source fragment
(ramba/ramba.py:8383 add_op codelines, executed after Numba JIT).  Since our
executor is a HIP kernel generator, the ops are recorded as a small structured
expression IR instead; `codegen.py` lowers it to gfx950 HIP source and
`oracle/numpy_backend.py` interprets the same IR with NumPy (the parity
oracle).  The op vocabulary mirrors the reference's op tables
(ramba/ramba.py:7893-7918 binops, 7949-7973 unaries, 7981-7993 reductions).
"""

from dataclasses import dataclass
from typing import Any

import numpy as np


# ---------------------------------------------------------------------------
# dtype helpers (NumPy 2 promotion semantics; the reference's unify_args
# bottoms out in numpy promotion too)
# ---------------------------------------------------------------------------

def np_dtype(dt):
    return np.dtype(dt)


def promote(a_dtype, b):
    """Promote an array dtype with either another dtype or a python scalar
    (NEP-50 weak-scalar rules, which np.result_type implements for python
    types)."""
    if isinstance(b, np.dtype):
        return np.result_type(a_dtype, b)
    return np.result_type(a_dtype, b)


BOOL = np.dtype(np.bool_)
I64 = np.dtype(np.int64)
F64 = np.dtype(np.float64)


# ---------------------------------------------------------------------------
# expression nodes; every node carries its result dtype
# ---------------------------------------------------------------------------

@dataclass(frozen=True)
class Expr:
    pass


@dataclass(frozen=True)
class Ref(Expr):
    """Reference to a group variable: an array operand var or a temp."""
    name: str
    dtype: Any


@dataclass(frozen=True)
class ScalarArg(Expr):
    """A runtime scalar (named kernel argument; analog of the reference's
    pickled `use_other` vars, ramba.py:8103 — keeps the kernel cache key
    value-independent)."""
    name: str
    dtype: Any


@dataclass(frozen=True)
class Const(Expr):
    """A compile-time constant baked into the kernel source (used for small
    integer pow exponents so `x**2` lowers to `x*x`, matching NumPy's
    small-int fast path)."""
    value: Any
    dtype: Any


@dataclass(frozen=True)
class Iota(Expr):
    """index[axis] + global_start[axis] (int64) — the arange codeline
    (ramba/ramba.py:8952-8960), the bit-exactness anchor."""
    axis: int
    dtype: Any = I64


@dataclass(frozen=True)
class Bin(Expr):
    op: str
    a: Expr
    b: Expr
    dtype: Any = None


@dataclass(frozen=True)
class Un(Expr):
    op: str
    a: Expr
    dtype: Any = None


@dataclass(frozen=True)
class Cast(Expr):
    a: Expr
    dtype: Any = None


@dataclass(frozen=True)
class Where(Expr):
    c: Expr
    a: Expr
    b: Expr
    dtype: Any = None


# binop name -> (numpy ufunc name, is_comparison/bool-result)
BINOPS = {
    "add": np.add, "sub": np.subtract, "mul": np.multiply,
    "div": np.divide, "floordiv": np.floor_divide, "mod": np.mod,
    "pow": np.power, "minimum": np.minimum, "maximum": np.maximum,
    "gt": np.greater, "lt": np.less, "ge": np.greater_equal,
    "le": np.less_equal, "eq": np.equal, "ne": np.not_equal,
    "logical_and": np.logical_and, "logical_or": np.logical_or,
    "logical_xor": np.logical_xor,
    "bitand": np.bitwise_and, "bitor": np.bitwise_or,
    "bitxor": np.bitwise_xor, "lshift": np.left_shift,
    "rshift": np.right_shift,
}

BOOL_RESULT_BINOPS = {"gt", "lt", "ge", "le", "eq", "ne",
                      "logical_and", "logical_or", "logical_xor"}

UNOPS = {
    "abs": np.abs, "square": np.square, "sqrt": np.sqrt,
    "sin": np.sin, "cos": np.cos, "tan": np.tan,
    "sinh": np.sinh, "cosh": np.cosh, "tanh": np.tanh,
    "arcsin": np.arcsin, "arccos": np.arccos, "arctan": np.arctan,
    "neg": np.negative, "exp": np.exp, "log": np.log,
    "floor": np.floor, "ceil": np.ceil, "trunc": np.trunc,
    "rint": np.rint, "sign": np.sign,
    "isnan": np.isnan, "isinf": np.isinf, "isfinite": np.isfinite,
    "logical_not": np.logical_not, "invert": np.invert,
}

FLOAT_UNOPS = {"sqrt", "sin", "cos", "tan", "sinh", "cosh", "tanh",
               "arcsin", "arccos", "arctan", "exp", "log"}
BOOL_RESULT_UNOPS = {"isnan", "isinf", "isfinite", "logical_not"}

# reductions: name -> (binop used to combine, init value sentinel)
# reference: array_simple_reductions (ramba/ramba.py:7981-7993); init 2/-2
# mean "+inf"/"-inf" resolved per dtype (getminmax).
REDUCTIONS = {
    "sum": ("add", 0),
    "prod": ("mul", 1),
    "min": ("minimum", "maxval"),
    "max": ("maximum", "minval"),
    "all": ("logical_and", True),
    "any": ("logical_or", False),
}


def reduction_init(kind, dtype):
    comb, init = REDUCTIONS[kind]
    if init == "maxval":
        if np.issubdtype(dtype, np.floating):
            return np.inf
        return np.iinfo(dtype).max
    if init == "minval":
        if np.issubdtype(dtype, np.floating):
            return -np.inf
        return np.iinfo(dtype).min
    return init


# ---------------------------------------------------------------------------
# statements
# ---------------------------------------------------------------------------

@dataclass
class Assign:
    target: str            # var name (array operand var or temp)
    expr: Expr


@dataclass
class ReductionSpec:
    """Axis-less reduction fused into the group (the precode/body/postcode
    structure of internal_reduction1_executor, ramba/ramba.py:5789-5807).

    body statement `acc = combine(acc, expr)` lives in Group.statements;
    this spec declares acc's init and the per-rank partial output slot."""
    acc: str                # temp name of the accumulator
    kind: str               # sum/prod/min/max/all/any
    dtype: Any
    slot: int               # index into the group's partial-output slots
