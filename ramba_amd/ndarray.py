"""NumPy-compatible API surface over the fused-deferred engine.

The in-scope slice of the reference's `ramba.ndarray`
(ramba/ramba.py:5409-5591 object model; op tables 7842-7993; creation
routines 8552-8991; slice getitem/setitem 6429-6620/6143-6296).  Every
elementwise / reduction op records IR into the current fused group via
`deferred.add_op`; nothing executes until `sync()` / a reduction / `asarray`.
"""

import numbers

import numpy as np

from . import deferred, ir
from .common import default_border, default_divisions
from .shardview import View


# ---------------------------------------------------------------------------
# dtype helpers
# ---------------------------------------------------------------------------

def _dtype_of(x):
    if isinstance(x, ndarray):
        return x.dtype
    if isinstance(x, ir.Expr):
        return x.dtype
    return x  # python scalar: weak promotion via np.result_type


_rdt_cache = {}


def _result_dtype(op, a, b):
    if op in ir.BOOL_RESULT_BINOPS:
        return ir.BOOL
    da, db = _dtype_of(a), _dtype_of(b)
    # NEP-50: python scalar TYPES are weak (value-independent), so the
    # result is cacheable on (op, kinds)
    ka = da if isinstance(da, np.dtype) else type(da)
    kb = db if isinstance(db, np.dtype) else type(db)
    try:
        key = (op == "div", ka, kb)
        rt = _rdt_cache.get(key)
    except TypeError:
        rt = None
        key = None
    if rt is None:
        rt = np.result_type(da, db)
        if op == "div" and rt.kind in "bui":
            rt = np.dtype(np.float64)
        if key is not None:
            _rdt_cache[key] = rt
    return rt


def _shape_of(x):
    if isinstance(x, ndarray):
        return x.shape
    if isinstance(x, ir.Expr):
        return ()
    return ()


def _bcast(x, shape):
    if isinstance(x, ndarray) and x.shape != shape:
        return x.broadcast_to(shape)
    return x


# ---------------------------------------------------------------------------
# ndarray
# ---------------------------------------------------------------------------

# hot-path caches: iterating workloads re-execute the same source lines,
# so slice-view composition and dtype resolution repeat exactly
_getitem_cache = {}

class ndarray:
    __slots__ = ("bdarray", "view", "readonly", "__weakref__")

    def __init__(self, bd, view, readonly=False):
        self.bdarray = bd
        self.view = view
        self.readonly = readonly
        bd.nviews += 1

    def __del__(self):
        try:
            self.bdarray.decref(deferred._state.get("runtime"))
        except Exception:
            pass

    # -- basic properties ---------------------------------------------------

    @property
    def shape(self):
        return self.view.shape

    @property
    def dtype(self):
        return self.bdarray.dtype

    @property
    def ndim(self):
        return len(self.view.shape)

    @property
    def size(self):
        n = 1
        for s in self.view.shape:
            n *= s
        return n

    @property
    def T(self):
        return self.transpose()

    def __len__(self):
        if self.ndim == 0:
            raise TypeError("len() of unsized object")
        return self.shape[0]

    def __repr__(self):
        return (f"ramba_amd.ndarray(shape={self.shape}, dtype={self.dtype}, "
                f"gid={self.bdarray.gid})")

    # -- views ---------------------------------------------------------------

    def _with_view(self, view, readonly=None):
        return ndarray(self.bdarray, view,
                       self.readonly if readonly is None else readonly)

    def broadcast_to(self, shape):
        return self._with_view(self.view.broadcast_to(shape), readonly=True)

    def broadcastable_to(self, shape):
        try:
            self.view.broadcast_to(shape)
            return True
        except ValueError:
            return False

    def transpose(self, *axes):
        if len(axes) == 1 and isinstance(axes[0], (tuple, list)):
            axes = tuple(axes[0])
        return self._with_view(self.view.transpose(axes if axes else None))

    def astype(self, dtype):
        dtype = np.dtype(dtype)
        out = _new_result(self.shape, dtype)
        deferred.add_op(out, "=", ir.Cast(self, dtype), empty_like=empty_like)
        return out

    def copy(self):
        return self.astype(self.dtype)

    # -- indexing ------------------------------------------------------------

    def __getitem__(self, index):
        # compressing boolean getitem: a[mask] -> 1-D array of the selected
        # elements in C order (reference maskarray getitem; dynamic-shape
        # sync point).  Value and mask are first copied onto one fresh
        # common partition through the fused engine, then each rank
        # compacts its core box (runtime.mask_compact_op).
        if isinstance(index, np.ndarray) and index.dtype == np.bool_:
            index = fromarray(index)
        if isinstance(index, ndarray) and index.dtype == np.bool_:
            if index.shape != self.shape:
                raise IndexError(
                    f"boolean index shape {index.shape} does not match "
                    f"array shape {self.shape}")
            # copy value+mask onto ONE fresh C-contiguous (axis-0 split)
            # partition through the fused engine: cross-rank C-order
            # concatenation of per-rank compactions is then always valid,
            # whatever the source partitions look like
            from .common import contiguous_divisions
            rt = deferred.get_runtime()
            cdivs = contiguous_divisions(rt.world, self.shape)
            ta_bd = deferred.bdarray(self.shape, self.dtype, cdivs,
                                     default_border, flex=False)
            ta = ndarray(ta_bd, View.identity(self.shape))
            deferred.add_op(ta, "=", self, empty_like=empty_like)
            tm_bd = deferred.bdarray(self.shape, np.dtype(np.uint8), cdivs,
                                     default_border, flex=False)
            tm = ndarray(tm_bd, View.identity(self.shape))
            deferred.add_op(tm, "=", ir.Cast(index, np.dtype(np.uint8)),
                            empty_like=empty_like)
            deferred.flush()
            out_bd = rt.mask_compact_op(ta.bdarray, tm.bdarray)
            return ndarray(out_bd, View.identity(out_bd.shape))
        if not isinstance(index, tuple):
            index = (index,)
        if all(isinstance(i, (int, np.integer)) for i in index) \
                and len(index) == self.ndim:
            # full scalar read -- a sync point
            deferred.flush()
            v = self.view
            base = [int(o) for o in v.offset]
            for d, i in enumerate(index):
                i = int(i)
                if i < 0:
                    i += self.shape[d]
                b, st = v.axis_map[d], v.steps[d]
                if b >= 0:
                    base[b] += i * st
            rt = deferred.get_runtime()
            return rt.read_element(self.bdarray, base)
        # slices are unhashable before Python 3.12, so normalise the index
        # to a hashable key (the r1 cache keyed on raw slices and silently
        # never hit on 3.10 — every getitem recomputed apply_index)
        key = None
        parts = []
        for ix in index:
            t = type(ix)
            if t is slice:
                parts.append((ix.start, ix.stop, ix.step))
            elif t is int or ix is None or ix is Ellipsis:
                parts.append(ix)
            elif isinstance(ix, (int, np.integer)):
                parts.append(int(ix))
            else:
                parts = None
                break
        if parts is not None:
            key = (id(self.view), tuple(parts))
            ent = _getitem_cache.get(key)
            if ent is not None and ent[0] is self.view:
                return self._with_view(ent[1])
        view = self.view.apply_index(index)
        if key is not None:
            if len(_getitem_cache) > 8192:
                _getitem_cache.clear()
            # the stored base-view ref pins the id against reuse
            _getitem_cache[key] = (self.view, view)
        return self._with_view(view)

    def __setitem__(self, index, value):
        # slice branch of setitem_array_executor (ramba.py:6273-6296)
        if self.readonly:
            raise ValueError("assignment destination is read-only")
        # boolean-mask write (reference maskarray guard,
        # ramba.py:8476-8478): a[mask] = v  ->  a = where(mask, v, a)
        if isinstance(index, ndarray) and index.dtype == np.bool_:
            if not index.broadcastable_to(self.shape):
                raise IndexError("mask shape does not match array shape")
            mask = index.broadcast_to(self.shape) \
                if index.shape != self.shape else index
            if isinstance(value, np.ndarray):
                value = fromarray(value)
            if isinstance(value, ndarray) and value.shape != self.shape:
                value = value.broadcast_to(self.shape)
            dt = self.dtype
            deferred.add_op(self, "=",
                            ir.Where(mask, ir.Cast(value, dt)
                                     if isinstance(value, ndarray) else value,
                                     self, dt),
                            empty_like=empty_like)
            return
        view = self[index] if not (isinstance(index, tuple) and index == ()) \
            else self
        if isinstance(view, (np.generic, numbers.Number)):
            raise IndexError("scalar element assignment is out of scope")
        if isinstance(value, ndarray):
            if value.shape != view.shape:
                value = value.broadcast_to(view.shape)
            if (value.bdarray.gid == view.bdarray.gid
                    and value.view == view.view):
                return  # self-assignment no-op (ramba.py:6287-6291)
            deferred.add_op(view, "=", value, empty_like=empty_like)
        elif isinstance(value, (numbers.Number, np.generic)):
            deferred.add_op(view, "=", value, empty_like=empty_like)
        elif isinstance(value, np.ndarray):
            deferred.add_op(view, "=", fromarray(value), empty_like=empty_like)
        else:
            raise TypeError(f"cannot assign {type(value)}")

    # -- numpy interop (reference __array_ufunc__/__array_function__,
    #    ramba/ramba.py:6860/6825: real-numpy calls on ramba arrays stay
    #    distributed instead of silently gathering) ------------------------

    _UFUNC_BIN = {"add": "__add__", "subtract": "__sub__",
                  "multiply": "__mul__", "true_divide": "__truediv__",
                  "divide": "__truediv__", "floor_divide": "__floordiv__",
                  "remainder": "__mod__", "power": "__pow__",
                  "minimum": "minimum", "maximum": "maximum",
                  "greater": "__gt__", "less": "__lt__",
                  "greater_equal": "__ge__", "less_equal": "__le__",
                  "equal": "__eq__", "not_equal": "__ne__",
                  "logical_and": "logical_and", "logical_or": "logical_or",
                  "logical_xor": "logical_xor", "bitwise_and": "__and__",
                  "bitwise_or": "__or__", "bitwise_xor": "__xor__",
                  "left_shift": "__lshift__", "right_shift": "__rshift__",
                  "matmul": "__matmul__"}
    _UFUNC_UN = {"sin": "sin", "cos": "cos", "tan": "tan", "sinh": "sinh",
                 "cosh": "cosh", "tanh": "tanh", "arcsin": "arcsin",
                 "arccos": "arccos", "arctan": "arctan", "exp": "exp",
                 "log": "log", "sqrt": "sqrt", "square": "square",
                 "absolute": "__abs__", "negative": "__neg__",
                 "floor": "floor", "ceil": "ceil", "trunc": "trunc",
                 "rint": "rint", "sign": "sign", "isnan": "isnan",
                 "isinf": "isinf", "isfinite": "isfinite",
                 "logical_not": "logical_not", "invert": "__invert__"}

    def __array_ufunc__(self, ufunc, method, *inputs, **kwargs):
        if kwargs.get("out") is not None:
            return NotImplemented
        name = ufunc.__name__
        if method == "__call__":
            if len(inputs) == 2 and name in self._UFUNC_BIN:
                a, b = inputs
                if a is self:
                    return getattr(self, self._UFUNC_BIN[name])(b)
                if isinstance(a, (numbers.Number, np.generic)) \
                        or isinstance(a, np.ndarray):
                    # reversed: scalar/ndarray OP self
                    m = self._UFUNC_BIN[name]
                    rm = "__r" + m[2:] if m.startswith("__") else None
                    if rm and hasattr(self, rm):
                        return getattr(self, rm)(a)
                    if isinstance(b, ndarray):
                        return getattr(fromarray(np.asarray(a))
                                       if isinstance(a, np.ndarray) else a,
                                       m, None) or NotImplemented
                return NotImplemented
            if len(inputs) == 1 and name in self._UFUNC_UN:
                return getattr(self, self._UFUNC_UN[name])()
            return NotImplemented
        if method == "reduce" and not kwargs.get("axis", None) \
                and len(inputs) == 1 and inputs[0] is self:
            red = {"add": "sum", "multiply": "prod", "minimum": "min",
                   "maximum": "max", "logical_and": "all",
                   "logical_or": "any"}.get(name)
            if red:
                return getattr(self, red)()
        return NotImplemented

    # np.where / np.sum / … on ramba arrays: dispatch through the deferred
    # path instead of silently gathering (reference __array_function__,
    # ramba/ramba.py:6825).
    _ARRAY_FUNC = {}   # filled after the class body (needs module fns)

    def __array_function__(self, func, types, args, kwargs):
        handler = self._ARRAY_FUNC.get(func.__name__)
        if handler is None:
            return NotImplemented
        return handler(*args, **kwargs)

    # -- conversion ----------------------------------------------------------

    def __float__(self):
        assert self.size == 1, "only 1-element arrays convert to scalars"
        return float(self.asarray().reshape(())[()])

    def __int__(self):
        assert self.size == 1
        return int(self.asarray().reshape(())[()])

    def __bool__(self):
        assert self.size == 1, ("truth value of a multi-element array is "
                                "ambiguous")
        return bool(self.asarray().reshape(())[()])

    def asarray(self):
        deferred.flush()
        rt = deferred.get_runtime()
        return rt.gather_view(self.bdarray, self.view)

    def __array__(self, dtype=None, copy=None):
        a = self.asarray()
        if dtype is not None:
            a = a.astype(dtype)
        return a

    # -- elementwise ops ------------------------------------------------------

    def _binop(self, rhs, op, reverse=False, out_dtype=None):
        if isinstance(rhs, (list, tuple)):
            rhs = np.asarray(rhs)
        if isinstance(rhs, np.ndarray) and rhs.shape == ():
            rhs = rhs[()]
        if isinstance(rhs, np.ndarray):
            rhs = fromarray(rhs)
        if not isinstance(rhs, (ndarray, numbers.Number, np.generic)):
            return NotImplemented
        a, b = (rhs, self) if reverse else (self, rhs)
        sa, sb = _shape_of(a), _shape_of(b)
        if sa == sb or sb == ():
            shape = sa          # fast path: np.broadcast_shapes is ~10 us
        elif sa == ():
            shape = sb
        else:
            shape = np.broadcast_shapes(sa, sb)
        dt = _result_dtype(op, a, b) if out_dtype is None else out_dtype
        a, b = _bcast(a, shape), _bcast(b, shape)
        if op == "pow" and isinstance(b, (int, np.integer)) \
                and not isinstance(b, (bool, np.bool_)) and b < 0 \
                and np.dtype(dt).kind in "iu":
            # match NumPy/the oracle: the HIP rt_ipow loop would silently
            # return 1 here
            raise ValueError(
                "Integers to negative integer powers are not allowed.")
        if op == "pow" and isinstance(b, (int, np.integer)) \
                and not isinstance(b, (bool, np.bool_)) and 0 <= b <= 8:
            # small-int exponent baked as a compile-time constant so the
            # kernel emits a multiply chain (x**2 == x*x, NumPy's fast path)
            b = ir.Const(int(b), ir.I64)
        out = _new_result(shape, dt)
        deferred.add_op(out, "=", ir.Bin(op, a, b, dt), empty_like=empty_like)
        return out

    def _unop(self, op):
        if op in ir.FLOAT_UNOPS:
            dt = np.dtype(np.float32) if self.dtype == np.float32 \
                else np.dtype(np.float64)
        elif op in ir.BOOL_RESULT_UNOPS:
            dt = ir.BOOL
        else:
            dt = self.dtype
        out = _new_result(self.shape, dt)
        deferred.add_op(out, "=", ir.Un(op, self, dt), empty_like=empty_like)
        return out

    def _iop(self, rhs, aop):
        if self.readonly:
            raise ValueError("output array is read-only")
        if isinstance(rhs, np.ndarray):
            rhs = fromarray(rhs)
        if isinstance(rhs, ndarray) and rhs.shape != self.shape:
            rhs = rhs.broadcast_to(self.shape)
        deferred.add_op(self, aop, rhs, empty_like=empty_like)
        return self

    # division -> multiplication-by-reciprocal rewrite (ramba.py:6121-6126);
    # it changes numerics, so the oracle and kernels share it by construction.
    def __truediv__(self, rhs):
        if isinstance(rhs, (numbers.Number, np.generic)) \
                and not isinstance(rhs, (bool, np.bool_)):
            return self._binop(1.0 / rhs, "mul",
                               out_dtype=_result_dtype("div", self, rhs))
        if isinstance(rhs, np.ndarray):
            rhs = fromarray(rhs)
        if isinstance(rhs, ndarray):
            recip = rhs.__rtruediv__(1.0)
            return self._binop(recip, "mul",
                               out_dtype=_result_dtype("div", self, recip))
        return NotImplemented

    def __rtruediv__(self, lhs):
        return self._binop(lhs, "div", reverse=True)

    def __itruediv__(self, rhs):
        return self._iop(rhs, "/=")

    # -- reductions (axis=None; reference array_unaryop reduction path,
    #    ramba.py:5890-5920 + internal_reduction2b 5852-5863) ---------------

    def _reduce(self, kind, dtype=None, axis=None, keepdims=False):
        dt = self.dtype if dtype is None else np.dtype(dtype)
        if dtype is None and kind in ("sum", "prod") \
                and dt.kind in "bi" and dt.itemsize < 8:
            # NumPy promotes small-int/bool sums to the platform int
            dt = np.dtype(np.int64)
        if kind in ("all", "any"):
            dt = ir.BOOL
        # axis normalisation (reference array_unaryop, ramba.py:5896-5903:
        # an axis tuple covering every dim collapses to the axis-None path)
        if axis is not None:
            axes = (axis,) if isinstance(axis, (int, np.integer)) else \
                tuple(axis)
            axes = tuple(sorted(a + self.ndim if a < 0 else a for a in axes))
            if len(axes) == self.ndim:
                axis = None
        if kind in ("min", "max"):
            # NumPy parity (fuzz v6 seed 5235): no identity for min/max —
            # raises iff a REDUCED extent is zero (axis=None: any extent)
            red = range(self.ndim) if axis is None else axes
            if any(self.shape[i] == 0 for i in red):
                raise ValueError(
                    "zero-size array to reduction operation "
                    f"{'minimum' if kind == 'min' else 'maximum'} "
                    "which has no identity")
        if axis is None:
            src = self
            if kind in ("all", "any") and self.dtype != ir.BOOL:
                src = ir.Un("logical_not",
                            ir.Un("logical_not", self, ir.BOOL), ir.BOOL)
            pend = deferred.add_reduction(src, kind, dt)
            deferred.flush()
            rt = deferred.get_runtime()
            val = rt.finish_reduction(pend)
            if keepdims:
                return np.asarray(val).reshape((1,) * self.ndim)
            return val
        # axis reduction
        deferred.flush()
        rt = deferred.get_runtime()
        kd_shape = tuple(1 if i in axes else self.shape[i]
                         for i in range(self.ndim))
        out_bd = rt.reduce_axes_op(self, axes, kind, dt, kd_shape)
        res = ndarray(out_bd, View.identity(kd_shape))
        if not keepdims:
            res = res[tuple(0 if i in axes else slice(None)
                            for i in range(self.ndim))]
        return res

    def sum(self, axis=None, dtype=None, keepdims=False, **kw):
        return self._reduce("sum", dtype, axis=axis, keepdims=keepdims)

    def prod(self, axis=None, dtype=None, keepdims=False, **kw):
        return self._reduce("prod", dtype, axis=axis, keepdims=keepdims)

    def min(self, axis=None, keepdims=False, **kw):
        return self._reduce("min", axis=axis, keepdims=keepdims)

    def max(self, axis=None, keepdims=False, **kw):
        return self._reduce("max", axis=axis, keepdims=keepdims)

    def all(self, axis=None, keepdims=False, **kw):
        return self._reduce("all", axis=axis, keepdims=keepdims)

    def any(self, axis=None, keepdims=False, **kw):
        return self._reduce("any", axis=axis, keepdims=keepdims)

    def clip(self, lo=None, hi=None, **kw):
        """reference TestBasic clip (test_distributed_array.py)."""
        out = self
        if lo is not None:
            out = out.maximum(lo)
        if hi is not None:
            out = out.minimum(hi)
        return out

    def cumsum(self, axis=None, dtype=None, **kw):
        """Cumulative sum (reference scumulative/cumsum,
        ramba.py:10057-10171, 9675: 1-D, or N-D along an explicit axis —
        the reference's scumulative asserts a concrete axis for N-D)."""
        dt = self.dtype if dtype is None else np.dtype(dtype)
        if dtype is None and dt.kind in "bi" and dt.itemsize < 8:
            dt = np.dtype(np.int64)   # NumPy platform-int promotion
        deferred.flush()
        rt = deferred.get_runtime()
        if self.ndim == 1 and axis in (None, 0, -1):
            out_bd = rt.cumsum_op(self, dt)
            return ndarray(out_bd, View.identity(self.shape))
        if axis is None:
            raise ValueError(
                "cumsum on an N-D array needs an explicit axis "
                "(reference scumulative, ramba.py:10061)")
        ax = int(axis)
        if ax < 0:
            ax += self.ndim
        if not (0 <= ax < self.ndim):
            raise ValueError(f"cumsum: axis {axis} out of range")
        src = self if np.dtype(self.dtype) == dt else self.astype(dt)
        deferred.flush()
        out_bd = rt.cumsum_axis_op(src, ax, dt)
        return ndarray(out_bd, View.identity(self.shape))

    def mean(self, axis=None, keepdims=False, **kw):
        dt = np.dtype(np.float32) if self.dtype == np.float32 \
            else np.dtype(np.float64)
        s = self._reduce("sum", dt, axis=axis, keepdims=keepdims)
        if axis is None:
            return s / self.size
        axes = (axis,) if isinstance(axis, (int, np.integer)) else \
            tuple(axis)
        cnt = 1
        for a in axes:
            cnt *= self.shape[a + self.ndim if a < 0 else a]
        return s * (1.0 / cnt)

    # -- nan-aware / tolerance / axis-permutation vocabulary (the remaining
    #    names of the reference's mod_to_array export list, ramba.py:9697;
    #    all expressed through the fused elementwise/reduction machinery) --

    def nansum(self, axis=None, dtype=None, keepdims=False, **kw):
        if np.dtype(self.dtype).kind != "f":
            return self.sum(axis=axis, dtype=dtype, keepdims=keepdims)
        cleaned = where(self.isnan(), 0.0, self).astype(
            self.dtype if dtype is None else dtype)
        return cleaned.sum(axis=axis, keepdims=keepdims)

    def nanmean(self, axis=None, keepdims=False, **kw):
        if np.dtype(self.dtype).kind != "f":
            return self.mean(axis=axis, keepdims=keepdims)
        valid = self.isnan().logical_not()
        cnt = valid.astype(np.int64).sum(axis=axis, keepdims=keepdims)
        tot = self.nansum(axis=axis, keepdims=keepdims)
        return tot / cnt

    def isposinf(self):
        return self == np.inf

    def isneginf(self):
        return self == -np.inf

    def isclose(self, other, rtol=1e-5, atol=1e-8, **kw):
        # numpy semantics: |a-b| <= atol + rtol*|b| where both finite,
        # exact equality otherwise (inf==inf True, NaN always False)
        if isinstance(other, np.ndarray):
            other = fromarray(other)
        diff_ok = abs(self - other) <= (atol + rtol * abs(other))
        if isinstance(other, ndarray):
            finite = self.isfinite().logical_and(other.isfinite())
        else:
            finite = self.isfinite() if np.isfinite(other)                 else self.isfinite().logical_and(False)
        return where(finite, diff_ok, self == other)

    def allclose(self, other, rtol=1e-5, atol=1e-8, **kw):
        return bool(self.isclose(other, rtol=rtol, atol=atol).all())

    def var(self, axis=None, ddof=0, keepdims=False, **kw):
        m = self.mean(axis=axis, keepdims=True) if axis is not None             else self.mean()
        dev2 = (self - m) * (self - m)
        if axis is None:
            n = self.size
            return dev2.sum() / (n - ddof)
        axes = (axis,) if isinstance(axis, (int, np.integer)) else             tuple(axis)
        n = 1
        for a in axes:
            n *= self.shape[a + self.ndim if a < 0 else a]
        return dev2.sum(axis=axis, keepdims=keepdims) * (1.0 / (n - ddof))

    def std(self, axis=None, ddof=0, keepdims=False, **kw):
        v = self.var(axis=axis, ddof=ddof, keepdims=keepdims)
        return v ** 0.5 if isinstance(v, ndarray) else float(v) ** 0.5

    def swapaxes(self, i, j):
        n = self.ndim
        i = i + n if i < 0 else i
        j = j + n if j < 0 else j
        perm = list(range(n))
        perm[i], perm[j] = perm[j], perm[i]
        return self.transpose(perm)

    def moveaxis(self, source, destination):
        n = self.ndim
        src = [source] if isinstance(source, (int, np.integer))             else list(source)
        dst = [destination] if isinstance(destination, (int, np.integer))             else list(destination)
        src = [a + n if a < 0 else a for a in src]
        dst = [a + n if a < 0 else a for a in dst]
        order = [a for a in range(n) if a not in src]
        for d, s_ in sorted(zip(dst, src)):
            order.insert(d, s_)
        return self.transpose(order)

    def rollaxis(self, axis, start=0):
        n = self.ndim
        axis = axis + n if axis < 0 else axis
        if start < 0:
            start += n
        if axis < start:
            start -= 1
        axes = list(range(n))
        axes.remove(axis)
        axes.insert(start, axis)
        return self.transpose(axes)


deferred.NDARRAY_CLS = ndarray   # fast leaf test in deferred._subst


# -- attach the op tables (reference make_method loops, ramba.py:7893-7973) --

_BINOP_METHODS = {
    "__add__": "add", "__sub__": "sub", "__mul__": "mul",
    "__floordiv__": "floordiv", "__mod__": "mod", "__pow__": "pow",
    "__gt__": "gt", "__lt__": "lt", "__ge__": "ge", "__le__": "le",
    "__eq__": "eq", "__ne__": "ne",
    "__and__": "bitand", "__or__": "bitor", "__xor__": "bitxor",
    "__lshift__": "lshift", "__rshift__": "rshift",
}
_RBINOP_METHODS = {
    "__radd__": "add", "__rsub__": "sub", "__rmul__": "mul",
    "__rfloordiv__": "floordiv", "__rmod__": "mod", "__rpow__": "pow",
}
_IBINOP_METHODS = {
    "__iadd__": "+=", "__isub__": "-=", "__imul__": "*=",
    "__ifloordiv__": "//=", "__imod__": "%=", "__ipow__": "**=",
}
_UNOP_METHODS = {
    "__abs__": "abs", "abs": "abs", "square": "square", "sqrt": "sqrt",
    "sin": "sin", "cos": "cos", "tan": "tan", "sinh": "sinh",
    "cosh": "cosh", "tanh": "tanh", "arcsin": "arcsin",
    "arccos": "arccos", "arctan": "arctan", "__neg__": "neg",
    "exp": "exp", "log": "log", "isnan": "isnan", "isinf": "isinf",
    "isfinite": "isfinite", "logical_not": "logical_not",
    "__invert__": "invert", "floor": "floor", "ceil": "ceil",
    "trunc": "trunc", "rint": "rint", "sign": "sign",
}

for _m, _op in _BINOP_METHODS.items():
    setattr(ndarray, _m,
            (lambda op: lambda self, rhs: self._binop(rhs, op))(_op))
for _m, _op in _RBINOP_METHODS.items():
    setattr(ndarray, _m,
            (lambda op: lambda self, lhs: self._binop(lhs, op, reverse=True))(_op))
for _m, _op in _IBINOP_METHODS.items():
    setattr(ndarray, _m,
            (lambda aop: lambda self, rhs: self._iop(rhs, aop))(_op))
for _m, _op in _UNOP_METHODS.items():
    setattr(ndarray, _m,
            (lambda op: lambda self: self._unop(op))(_op))

for _m in ("minimum", "maximum"):
    setattr(ndarray, _m,
            (lambda op: lambda self, rhs: self._binop(rhs, op))(_m))
for _m in ("logical_and", "logical_or", "logical_xor"):
    setattr(ndarray, _m,
            (lambda op: lambda self, rhs: self._binop(rhs, op))(_m))


# ---------------------------------------------------------------------------
# creation (reference ramba.py:8552-8991)
# ---------------------------------------------------------------------------

def _world():
    return deferred.get_runtime().world


def _new_bd(shape, dtype, divisions=None, border=None, flex=True):
    shape = (shape,) if isinstance(shape, (int, np.integer)) else tuple(shape)
    dtype = np.dtype(np.float64 if dtype is None else dtype)
    if divisions is None:
        divisions = default_divisions(_world(), shape)
    if border is None:
        border = default_border
    return deferred.bdarray(shape, dtype, divisions, border, flex=flex)


def _new_result(shape, dtype, divisions=None):
    bd = _new_bd(shape, dtype, divisions=divisions)
    return ndarray(bd, View.identity(bd.shape))


def empty(shape, dtype=None, local_border=None, distribution=None, **kw):
    bd = _new_bd(shape, dtype, divisions=distribution, border=local_border)
    arr = ndarray(bd, View.identity(bd.shape))
    deferred.register_empty(arr)
    return arr


def empty_like(a, dtype=None, **kw):
    return empty(a.shape, a.dtype if dtype is None else dtype)


def full(shape, value, dtype=None, **kw):
    if dtype is None:
        if isinstance(value, (bool, np.bool_)):
            dtype = np.bool_
        elif isinstance(value, (int, np.integer)):
            dtype = np.int64
        else:
            dtype = np.float64
    arr = _new_result((shape,) if isinstance(shape, (int, np.integer))
                      else tuple(shape), np.dtype(dtype))
    deferred.add_op(arr, "=", value, empty_like=empty_like)
    return arr


def zeros(shape, dtype=None, **kw):
    return full(shape, 0, dtype=np.float64 if dtype is None else dtype)


def ones(shape, dtype=None, **kw):
    return full(shape, 1, dtype=np.float64 if dtype is None else dtype)


def zeros_like(a, dtype=None, **kw):
    return zeros(a.shape, a.dtype if dtype is None else dtype)


def ones_like(a, dtype=None, **kw):
    return ones(a.shape, a.dtype if dtype is None else dtype)


def full_like(a, value, dtype=None, **kw):
    return full(a.shape, value, dtype=a.dtype if dtype is None else dtype)


def arange(start, stop=None, step=None, dtype=None, **kw):
    """reference arange_executor (ramba.py:8952-8960):
    `res = start + step * (index[0] + global_start[0])` -- exact int64."""
    if stop is None:
        size = int(start)
        expr = ir.Iota(0)
    elif step is None:
        size = int(stop - start)
        expr = ir.Bin("add", start, ir.Iota(0), None)
    else:
        size = int((stop - start + step - 1) // step) if step > 0 else \
            int((stop - start + step + 1) // step)
        expr = ir.Bin("add", start, ir.Bin("mul", step, ir.Iota(0), None),
                      None)
    size = max(0, size)
    if dtype is None:
        if any(isinstance(x, (float, np.floating))
               for x in (start, stop, step) if x is not None):
            dtype = np.float64
        else:
            dtype = np.int64
    arr = _new_result((size,), np.dtype(dtype))
    deferred.add_op(arr, "=", expr, empty_like=empty_like)
    return arr


def linspace(start, stop, num=50, endpoint=True, retstep=False, dtype=None):
    # reference ramba.py:8977-8991
    assert num > 0
    length = stop - start
    step = length / (num - 1) if endpoint else length / num
    res = arange(num) * step + start
    if dtype is not None:
        res = res.astype(dtype)
    if retstep:
        return res, step
    return res


def eye(n, m=None, k=0, dtype=None, **kw):
    """Identity / shifted-diagonal matrix (reference creation routines,
    ramba.py:8688-8991)."""
    m = n if m is None else m
    dt = np.dtype(np.float64 if dtype is None else dtype)
    out = _new_result((int(n), int(m)), dt)
    cond = ir.Bin("eq", ir.Bin("add", ir.Iota(0), int(k), ir.I64),
                  ir.Iota(1), ir.BOOL)
    deferred.add_op(out, "=", ir.Where(cond, 1, 0, dt),
                    empty_like=empty_like)
    return out


def fromfunction(function, shape, dtype=None, **kw):
    """Iota-expression fill (analog of the reference's string fillers,
    create_array_executor ramba.py:8563).  `function` receives symbolic
    index expressions supporting the ndarray op vocabulary."""
    shape = tuple(int(s) for s in shape)
    dt = np.dtype(np.float64 if dtype is None else dtype)
    syms = [_IndexSym(ir.Iota(d)) for d in range(len(shape))]
    expr = function(*syms)
    if isinstance(expr, _IndexSym):
        expr = expr.e
    arr = _new_result(shape, dt)
    deferred.add_op(arr, "=", expr, empty_like=empty_like)
    return arr


class _IndexSym:
    """Symbolic index used by fromfunction."""

    def __init__(self, e):
        self.e = e

    def _mk(self, op, other, reverse=False):
        o = other.e if isinstance(other, _IndexSym) else other
        a, b = (o, self.e) if reverse else (self.e, o)
        dt = ir.BOOL if op in ir.BOOL_RESULT_BINOPS else \
            np.result_type(_dtype_of(a), _dtype_of(b))
        return _IndexSym(ir.Bin(op, a, b, dt))

    def __add__(self, o):
        return self._mk("add", o)

    def __radd__(self, o):
        return self._mk("add", o, True)

    def __sub__(self, o):
        return self._mk("sub", o)

    def __rsub__(self, o):
        return self._mk("sub", o, True)

    def __mul__(self, o):
        return self._mk("mul", o)

    def __rmul__(self, o):
        return self._mk("mul", o, True)

    def __mod__(self, o):
        return self._mk("mod", o)


def fromarray(a):
    """Distribute a host numpy array (tests / small inputs only)."""
    a = np.asarray(a)
    if a.shape == ():
        return a[()]
    arr = empty(a.shape, a.dtype)
    deferred.flush()
    rt = deferred.get_runtime()
    rt.scatter_numpy(arr.bdarray, a)
    return arr


def array(a, dtype=None):
    a = np.asarray(a, dtype=dtype)
    return fromarray(a)


# ---------------------------------------------------------------------------
# module-level functions
# ---------------------------------------------------------------------------

def sync():
    deferred.flush()
    rt = deferred.get_runtime()
    rt.backend.sync()


def asarray(a):
    if isinstance(a, ndarray):
        return a.asarray()
    return np.asarray(a)


def where(cond, a, b):
    """Elementwise select (ramba.py:9755-9831 `where`, the in-scope part)."""
    shape = np.broadcast_shapes(_shape_of(cond), _shape_of(a), _shape_of(b))
    da, db = _dtype_of(a), _dtype_of(b)
    dt = np.result_type(da, db)
    cond, a, b = _bcast(cond, shape), _bcast(a, shape), _bcast(b, shape)
    out = _new_result(shape, dt)
    deferred.add_op(out, "=", ir.Where(cond, a, b, dt), empty_like=empty_like)
    return out


def _module_unop(name):
    def f(x, **kw):
        if isinstance(x, ndarray):
            return getattr(x, name)()
        return getattr(np, name)(x, **kw)
    f.__name__ = name
    return f


sin = _module_unop("sin")
cos = _module_unop("cos")
tan = _module_unop("tan")
sinh = _module_unop("sinh")
cosh = _module_unop("cosh")
tanh = _module_unop("tanh")
arcsin = _module_unop("arcsin")
arccos = _module_unop("arccos")
arctan = _module_unop("arctan")
exp = _module_unop("exp")
log = _module_unop("log")
sqrt = _module_unop("sqrt")
square = _module_unop("square")
absolute = _module_unop("abs")
isnan = _module_unop("isnan")
isinf = _module_unop("isinf")
isfinite = _module_unop("isfinite")
logical_not = _module_unop("logical_not")
floor = _module_unop("floor")
ceil = _module_unop("ceil")
trunc = _module_unop("trunc")
rint = _module_unop("rint")
sign = _module_unop("sign")


def clip(a, lo, hi, **kw):
    if isinstance(a, ndarray):
        return a.clip(lo, hi)
    return np.clip(a, lo, hi)


def _module_reduction(name):
    def f(x, axis=None, **kw):
        if isinstance(x, ndarray):
            return getattr(x, name)(axis=axis, **kw)
        return getattr(np, name)(x, axis=axis, **kw)
    f.__name__ = name
    return f


def _module_binop(name):
    def f(a, b, **kw):
        if isinstance(a, ndarray):
            return getattr(a, name)(b)
        if isinstance(b, ndarray):
            return getattr(b, name)(a)
        return getattr(np, name)(a, b, **kw)
    f.__name__ = name
    return f


sum = _module_reduction("sum")


def cumsum(x, axis=None, dtype=None):
    if isinstance(x, ndarray):
        return x.cumsum(axis=axis, dtype=dtype)
    return np.cumsum(x, axis=axis, dtype=dtype)

prod = _module_reduction("prod")
amin = _module_reduction("min")
amax = _module_reduction("max")
minimum = _module_binop("minimum")
maximum = _module_binop("maximum")
logical_and = _module_binop("logical_and")
logical_or = _module_binop("logical_or")
logical_xor = _module_binop("logical_xor")
power = _module_binop("__pow__")


def broadcast_to(a, shape):
    if isinstance(a, ndarray):
        return a.broadcast_to(shape)
    return np.broadcast_to(a, shape)


def transpose(a, axes=None):
    if isinstance(a, ndarray):
        return a.transpose(axes) if axes else a.transpose()
    return np.transpose(a, axes)


# numpy dtype aliases on the module (reference exports these)
float64 = np.float64
float32 = np.float32
int64 = np.int64
int32 = np.int32
int16 = np.int16
int8 = np.int8
uint8 = np.uint8
bool_ = np.bool_
pi = np.pi
e = np.e
inf = np.inf
nan = np.nan


# __array_function__ dispatch table: real-numpy module calls on ramba
# arrays route to the deferred implementations above.
def _af_reduce(meth):
    def h(a, axis=None, dtype=None, out=None, keepdims=False, **kw):
        if out is not None:
            raise NotImplementedError("out= is not supported")
        if meth in ("min", "max", "any", "all"):
            return getattr(a, meth)(axis=axis, keepdims=keepdims)
        return getattr(a, meth)(axis=axis, dtype=dtype, keepdims=keepdims)
    return h


ndarray._ARRAY_FUNC.update({
    "where": lambda c, x=None, y=None: where(c, x, y),
    "sum": _af_reduce("sum"),
    "prod": _af_reduce("prod"),
    "amin": _af_reduce("min"), "min": _af_reduce("min"),
    "amax": _af_reduce("max"), "max": _af_reduce("max"),
    "any": _af_reduce("any"), "all": _af_reduce("all"),
    "mean": lambda a, axis=None, **kw: a.mean(
        axis=axis, keepdims=kw.get("keepdims", False)),
    "cumsum": lambda a, axis=None, dtype=None, **kw: a.cumsum(
        axis=axis, dtype=dtype),
    "clip": lambda a, lo=None, hi=None, **kw: a.clip(lo, hi),
    "transpose": lambda a, axes=None: transpose(a, axes),
    "broadcast_to": lambda a, shape, **kw: broadcast_to(a, shape),
    "shape": lambda a: a.shape,
    "ndim": lambda a: a.ndim,
    "size": lambda a: a.size,
})


# remaining mod_to_array names (reference ramba.py:9697): delegate to the
# method when given a ramba array, to numpy otherwise.  These are exported
# through the PACKAGE namespace (ramba_amd/__init__.py), NOT injected into
# this module's globals — `all`/`any`/`abs`/`sum` must keep their builtin
# meaning inside this file.
def _module_delegate(name):
    def f(a, *args, **kwargs):
        if isinstance(a, ndarray):
            return getattr(a, name)(*args, **kwargs)
        if name == "abs":
            import builtins
            return builtins.abs(a)
        return getattr(np, name)(a, *args, **kwargs)
    f.__name__ = name
    return f


_MOD_DELEGATES = ("mean", "nanmean", "nansum", "isneginf", "isposinf",
                  "all", "any", "isclose", "allclose", "swapaxes",
                  "moveaxis", "rollaxis", "var", "std", "prod", "abs")


ndarray._ARRAY_FUNC.update({
    "nansum": lambda a, axis=None, **kw: a.nansum(axis=axis),
    "nanmean": lambda a, axis=None, **kw: a.nanmean(axis=axis),
    "isclose": lambda a, b, rtol=1e-5, atol=1e-8, **kw:
        a.isclose(b, rtol=rtol, atol=atol),
    "allclose": lambda a, b, rtol=1e-5, atol=1e-8, **kw:
        a.allclose(b, rtol=rtol, atol=atol),
    "swapaxes": lambda a, i, j: a.swapaxes(i, j),
    "moveaxis": lambda a, s, d: a.moveaxis(s, d),
    "rollaxis": lambda a, ax, start=0: a.rollaxis(ax, start),
    "var": lambda a, axis=None, ddof=0, **kw: a.var(axis=axis, ddof=ddof),
    "std": lambda a, axis=None, ddof=0, **kw: a.std(axis=axis, ddof=ddof),
})


# ---------------------------------------------------------------------------
# matmul / expand_dims / triu / meshgrid / item — the remaining user-facing
# surface of the reference around the hot path
# ---------------------------------------------------------------------------

def expand_dims(a, axis):
    """reference expand_dims (ramba.py:9438)."""
    if not isinstance(a, ndarray):
        return np.expand_dims(a, axis)
    nd = a.ndim + 1
    axis = axis + nd if axis < 0 else axis
    idx = (slice(None),) * axis + (None,) + (slice(None),) * (a.ndim - axis)
    return a[idx]


def matmul(a, b):
    """reference matmul (ramba.py:6953): every case lowers to the
    broadcast-views × fused-multiply × axis-sum composition the reference
    itself uses for its N-D path (`(a*b).sum(axis=-2)`).  The K-axis
    product is materialised before the axis reduction (no dense-GEMM MFMA
    path — `north_star` has no dense contraction in scope), so this is
    meant for the reference's moderate-size matmul uses, not as a BLAS."""
    if isinstance(a, np.ndarray):
        a = fromarray(a)
    if isinstance(b, np.ndarray):
        b = fromarray(b)
    if not isinstance(a, ndarray) or not isinstance(b, ndarray):
        raise ValueError("matmul cannot be used with scalar arguments")
    ashape, bshape = a.shape, b.shape
    if a.ndim == 1 and b.ndim == 1:
        if ashape[0] != bshape[0]:
            raise ValueError(f"matmul: mismatched shapes {ashape}/{bshape}")
        return (a * b).sum()
    if b.ndim == 1:
        if ashape[-1] != bshape[0]:
            raise ValueError(f"matmul: mismatched shapes {ashape}/{bshape}")
        return (a * b.broadcast_to(ashape)).sum(axis=-1)
    if a.ndim == 1:
        if ashape[0] != bshape[-2]:
            raise ValueError(f"matmul: mismatched shapes {ashape}/{bshape}")
        aa = expand_dims(a, -1).broadcast_to(bshape)
        return (aa * b).sum(axis=-2)
    if ashape[-1] != bshape[-2]:
        raise ValueError(f"matmul: mismatched shapes {ashape}/{bshape}")
    aa = expand_dims(a, -1)
    bb = expand_dims(b, -3)
    shp = np.broadcast_shapes(ashape[:-2], bshape[:-2]) \
        + ashape[-2:-1] + bshape[-2:]
    return (aa.broadcast_to(shp) * bb.broadcast_to(shp)).sum(axis=-2)


def _nd_matmul(self, rhs):
    return matmul(self, rhs)


def _nd_rmatmul(self, lhs):
    return matmul(lhs, self)


def _nd_item(self, *args):
    if not args:
        assert self.size == 1, "item(): array has more than one element"
        return self.asarray().reshape(-1)[0].item()
    if len(args) == 1 and isinstance(args[0], tuple):
        args = args[0]
    if len(args) == 1 and self.ndim != 1:
        return self.asarray().reshape(-1)[int(args[0])].item()
    v = self[tuple(int(x) for x in args)]
    return v.item() if isinstance(v, np.generic) else v


ndarray.__matmul__ = _nd_matmul
ndarray.__rmatmul__ = _nd_rmatmul
ndarray.item = _nd_item


def triu(m, k=0):
    """Upper triangle (reference triu, ramba.py:9053/:2091; 2-D only like
    the reference worker)."""
    if not isinstance(m, ndarray):
        return np.triu(m, k)
    assert m.ndim == 2, "triu: 2-D only (reference triu_executor)"
    ij = fromfunction(lambda i, j: j - i, m.shape, dtype=np.int64)
    return where(ij >= k, m, zeros(m.shape, dtype=m.dtype))


def tril(m, k=0):
    if not isinstance(m, ndarray):
        return np.tril(m, k)
    assert m.ndim == 2, "tril: 2-D only"
    ij = fromfunction(lambda i, j: j - i, m.shape, dtype=np.int64)
    return where(ij <= k, m, zeros(m.shape, dtype=m.dtype))


def meshgrid(*xi, copy=True, sparse=False, indexing="ij"):
    """reference meshgrid (ramba.py:9028): 'ij' indexing only, returns ONE
    stacked (len(xi), n1, ..., nk) distributed array (the reference stacks
    np.meshgrid output into a single backing array, :3921-3927)."""
    if indexing != "ij":
        raise ValueError(f"Unsupported meshgrid indexing option {indexing}")
    if sparse or not copy:
        raise ValueError("Unsupported meshgrid sparse/copy option")
    xs = [np.asarray(x) for x in xi]
    if any(x.ndim != 1 for x in xs):
        raise ValueError("Unsupported argument to meshgrid")
    if not all(x.dtype == xs[0].dtype for x in xs):
        raise ValueError("Mis-matching dtypes to meshgrid")
    nd = len(xs)
    sizes = tuple(len(x) for x in xs)
    out = empty((nd,) + sizes, dtype=xs[0].dtype)
    for i, x in enumerate(xs):
        idx = tuple(slice(None) if d == i else None for d in range(nd))
        out[i] = fromarray(x)[idx].broadcast_to(sizes)
    return out


def pad(arr, pad_width, mode="constant", **kwargs):
    """reference pad (ramba.py:9400; worker 9280): constant mode — a
    fresh constant-filled array with the source written into its core
    through the fused engine (the reference instead grows the edge
    shards' distributions and np.pads per worker)."""
    if isinstance(arr, np.ndarray):
        arr = fromarray(arr)
    nd = arr.ndim
    if isinstance(pad_width, (int, np.integer)):
        pad_width = ((int(pad_width),) * 2,) * nd
    else:
        pad_width = tuple(pad_width)
        if pad_width and not isinstance(pad_width[0], (tuple, list)):
            pad_width = (tuple(int(x) for x in pad_width),) * nd
        else:
            pad_width = tuple(tuple(int(x) for x in p) for p in pad_width)
    assert len(pad_width) == nd
    if mode != "constant":
        raise NotImplementedError(
            f"pad mode {mode!r} (constant only; the reference defers to "
            "per-worker np.pad for other modes)")
    cval = kwargs.get("constant_values", 0)
    newshape = tuple(s + lo + hi
                     for s, (lo, hi) in zip(arr.shape, pad_width))
    out = full(newshape, cval, dtype=arr.dtype)
    out[tuple(slice(lo, lo + s)
              for s, (lo, hi) in zip(arr.shape, pad_width))] = arr
    return out


class _MGrid:
    """reference mgrid (MgridGen, ramba.py:9001-9017): dense index grids;
    one slice -> a 1-D array, k slices -> a stacked (k, n1..nk) array
    (the reference materialises np.mgrid per shard, :3909-3927)."""

    def __getitem__(self, index):
        if not isinstance(index, tuple):
            index = (index,)
        axes = []
        for sl in index:
            assert isinstance(sl, slice), "mgrid takes slices"
            step = sl.step if sl.step is not None else 1
            start = sl.start if sl.start is not None else 0
            if isinstance(step, complex):
                axes.append(np.linspace(start, sl.stop, int(abs(step))))
            else:
                axes.append(np.arange(start, sl.stop, step))
        dt = np.result_type(*axes)
        axes = [a.astype(dt) for a in axes]
        if len(axes) == 1:
            return fromarray(axes[0])
        return meshgrid(*axes, indexing="ij")


mgrid = _MGrid()


ndarray._ARRAY_FUNC.update({
    "matmul": lambda a, b, **kw: matmul(a, b),
    "dot": lambda a, b, **kw: matmul(a, b),
    "expand_dims": lambda a, axis: expand_dims(a, axis),
    "triu": lambda m, k=0: triu(m, k),
    "tril": lambda m, k=0: tril(m, k),
    "pad": lambda a, w, mode="constant", **kw: pad(a, w, mode=mode, **kw),
})


# ---------------------------------------------------------------------------
# reshape / ravel / flatten (reference ramba.py:6716-6721 + the reshape
# worker 2409-2492; always a copy here, like the reference's general path)
# ---------------------------------------------------------------------------

def _nd_reshape(self, *newshape):
    if len(newshape) == 1 and isinstance(newshape[0], (tuple, list)):
        newshape = tuple(newshape[0])
    newshape = [int(x) for x in newshape]
    neg = [i for i, x in enumerate(newshape) if x < 0]
    if neg:
        assert len(neg) == 1, "can only specify one unknown dimension"
        known = 1
        for i, x in enumerate(newshape):
            if i != neg[0]:
                known *= x
        assert known > 0 and self.size % known == 0, \
            f"cannot reshape array of size {self.size} into {newshape}"
        newshape[neg[0]] = self.size // known
    newshape = tuple(newshape)
    if newshape == self.shape:
        return self.copy()
    deferred.flush()
    rt = deferred.get_runtime()
    src = self
    # interval exchange needs C-contiguous source shards; repartition
    # through the fused engine (a plain assignment onto an array with
    # prescribed axis-0 divisions) when the schedule split trailing axes
    from .runtime import _flat_interval
    from .shardview import exec_boxes as _eb
    lbs = _eb(src.view, src.bdarray.divisions)
    if any(b is not None and _flat_interval(b, src.view.shape) is None
           for b in lbs):
        from .common import contiguous_divisions
        tmp_bd = deferred.bdarray(
            self.shape, self.dtype,
            contiguous_divisions(rt.world, self.shape),
            default_border, flex=False)
        tmp = ndarray(tmp_bd, View.identity(self.shape))
        deferred.add_op(tmp, "=", self, empty_like=empty_like)
        deferred.flush()
        src = tmp
    out_bd = rt.reshape_op(src, newshape)
    return ndarray(out_bd, View.identity(newshape))


def _nd_ravel(self):
    return self.reshape(self.size)


ndarray.reshape = _nd_reshape
ndarray.reshape_copy = _nd_reshape
ndarray.ravel = _nd_ravel
ndarray.flatten = _nd_ravel


def reshape(a, newshape):
    if isinstance(a, ndarray):
        return a.reshape(newshape)
    return np.reshape(a, newshape)


def ravel(a):
    if isinstance(a, ndarray):
        return a.ravel()
    return np.ravel(a)


ndarray._ARRAY_FUNC.update({
    "reshape": lambda a, shape, **kw: a.reshape(shape),
    "ravel": lambda a, **kw: a.ravel(),
})


# ---------------------------------------------------------------------------
# joining / splitting / squeezing (reference ramba.py:9455-9620:
# concatenate, stack, split, squeeze — vstack/block etc. are unimplemented
# comments in the reference and stay out of scope here too)
# ---------------------------------------------------------------------------

def concatenate(arrayseq, axis=0, out=None, **kwargs):
    """reference concatenate (ramba.py:9541): same-dtype parts written
    into region views of a fresh array through the fused engine."""
    assert out is None, "concatenate: out= unsupported"
    parts = [fromarray(np.asarray(a)) if not isinstance(a, ndarray) else a
             for a in arrayseq]
    assert parts, "need at least one array"
    nd = parts[0].ndim
    axis = axis + nd if axis < 0 else axis
    out_shape = list(parts[0].shape)
    for p in parts[1:]:
        assert p.ndim == nd, "concatenate: rank mismatch"
        assert p.dtype == parts[0].dtype, \
            "concatenate: dtype mismatch (reference asserts equal dtypes)"
        for i in range(nd):
            if i == axis:
                out_shape[i] += p.shape[i]
            else:
                assert out_shape[i] == p.shape[i], \
                    f"concatenate: shape mismatch on axis {i}"
    res = empty(tuple(out_shape), dtype=parts[0].dtype)
    pos = 0
    for p in parts:
        idx = tuple(slice(pos, pos + p.shape[i]) if i == axis
                    else slice(None) for i in range(nd))
        res[idx] = p
        pos += p.shape[axis]
    return res


def stack(arrays, axis=0, out=None):
    """reference stack (ramba.py:9581)."""
    assert out is None
    parts = [fromarray(np.asarray(a)) if not isinstance(a, ndarray) else a
             for a in arrays]
    assert all(p.shape == parts[0].shape for p in parts)
    nd_out = parts[0].ndim + 1
    axis = axis + nd_out if axis < 0 else axis
    return concatenate([expand_dims(p, axis) for p in parts], axis=axis)


def split(arr, indices_or_sections, axis=0):
    """reference split (ramba.py:9611): equal integer sections only,
    returns slice VIEWS (zero copy)."""
    ashape = arr.shape
    nd = len(ashape)
    if axis > nd:
        raise ValueError("Wrong axis")
    if not isinstance(indices_or_sections, numbers.Integral):
        raise ValueError("split with indices not implemented.")
    axis_len = ashape[axis]
    if axis_len % indices_or_sections != 0:
        raise ValueError(
            f"Cannot evenly divide array dimension of length {axis_len} "
            f"into {indices_or_sections} equal sections.")
    k = axis_len // indices_or_sections
    return [arr[tuple(slice(None) if y != axis else slice(x * k, (x + 1) * k)
                      for y in range(nd))]
            for x in range(indices_or_sections)]


def squeeze(a, axis=None):
    """reference squeeze (ramba.py:9455) — but as a zero-copy VIEW (integer
    index drops the axis in the affine view algebra)."""
    if not isinstance(a, ndarray):
        return np.squeeze(a, axis)
    ashape = a.shape
    if axis is None:
        axis = tuple(i for i in range(len(ashape)) if ashape[i] == 1)
    if not isinstance(axis, (list, tuple)):
        axis = (axis,)
    axis = tuple(x + len(ashape) if x < 0 else x for x in axis)
    if not all(ashape[x] == 1 for x in axis):
        raise ValueError("cannot squeeze out an axis with size not equal "
                         "to 1")
    idx = tuple(0 if i in axis else slice(None) for i in range(len(ashape)))
    return a[idx]


def dot(a, b, out=None):
    """reference dot (ramba.py:6933)."""
    assert out is None
    return matmul(a, b)


ndarray.dot = lambda self, b: matmul(self, b)
ndarray.squeeze = lambda self, axis=None: squeeze(self, axis)

ndarray._ARRAY_FUNC.update({
    "concatenate": lambda seq, axis=0, **kw: concatenate(seq, axis=axis),
    "stack": lambda seq, axis=0, **kw: stack(seq, axis=axis),
    "split": lambda a, n, axis=0: split(a, n, axis),
    "squeeze": lambda a, axis=None: squeeze(a, axis),
})


# ---------------------------------------------------------------------------
# smap / smap_index / sreduce — the reference's custom-function mapping API
# (ramba.py:9926/9930/9979).  The reference pickles the function to workers
# and Numba-JITs it per element; here the function is TRACED through the
# NumPy-compatible deferred surface instead (its body must stay inside the
# supported op vocabulary — per-element Python control flow would need the
# reference's source JIT and is out of scope; the fused kernel you get is
# the same one the reference's codegen would emit for such bodies).
# ---------------------------------------------------------------------------

def _as_traceable(func):
    if isinstance(func, str):
        # the reference accepts lambda SOURCE strings (ramba.py:9877)
        return eval(func)   # noqa: S307 — user-supplied code, like func_loads
    return func


def _index_arrays(shape):
    nd = len(shape)
    return tuple(fromfunction(lambda *c, _i=i: c[_i], shape,
                              dtype=np.int64) for i in range(nd))


def smap(func, *args, dtype=None, parallel=True, axis=None, imports=()):
    func = _as_traceable(func)
    res = func(*args)
    if isinstance(res, np.ndarray):
        res = fromarray(res)
    if dtype is not None and isinstance(res, ndarray) \
            and np.dtype(res.dtype) != np.dtype(dtype):
        res = res.astype(dtype)
    return res


def smap_index(func, *args, dtype=None, parallel=True, imports=()):
    func = _as_traceable(func)
    first = next(a for a in args if isinstance(a, ndarray))
    idx = _index_arrays(first.shape)
    res = func(idx, *args)
    if dtype is not None and isinstance(res, ndarray) \
            and np.dtype(res.dtype) != np.dtype(dtype):
        res = res.astype(dtype)
    return res


def _probe_reducer(reducer):
    """Identify a scalar reducer lambda by probing (3, 5) — add/mul/min/max
    give distinct results.  Arbitrary reducers would need the reference's
    per-element JIT."""
    try:
        r = float(reducer(np.float64(3.0), np.float64(5.0)))
    except Exception as e:
        raise NotImplementedError(f"unsupported sreduce reducer: {e}")
    kind = {8.0: "sum", 15.0: "prod", 3.0: "min", 5.0: "max"}.get(r)
    if kind is None:
        raise NotImplementedError(
            "sreduce reducer must behave like add/mul/min/max "
            f"(probe (3,5) -> {r})")
    return kind


def sreduce(func, reducer, identity, *args, parallel=True):
    func = _as_traceable(func)
    kind = _probe_reducer(_as_traceable(reducer))
    mapped = func(*args)
    if not isinstance(mapped, ndarray):
        mapped = fromarray(np.asarray(mapped))
    return getattr(mapped, {"sum": "sum", "prod": "prod", "min": "min",
                            "max": "max"}[kind])()


def sreduce_index(func, reducer, identity, *args, parallel=True):
    func = _as_traceable(func)
    kind = _probe_reducer(_as_traceable(reducer))
    first = next(a for a in args if isinstance(a, ndarray))
    idx = _index_arrays(first.shape)
    mapped = func(idx, *args)
    return getattr(mapped, {"sum": "sum", "prod": "prod", "min": "min",
                            "max": "max"}[kind])()


# small numpy-compatible accessors
def _nd_tolist(self):
    return self.asarray().tolist()


ndarray.tolist = _nd_tolist
ndarray.itemsize = property(lambda self: np.dtype(self.dtype).itemsize)
ndarray.nbytes = property(
    lambda self: self.size * np.dtype(self.dtype).itemsize)
