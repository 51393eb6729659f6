"""Op x operand-type matrix (reference TestOps,
/root/reference/ramba/tests/test_distributed_array.py:262): every binop and
unop over dtype combinations, array-array / array-scalar / reversed."""

import numpy as np
import pytest

from conftest import run_both

BIN_METHODS = ["__add__", "__sub__", "__mul__", "__floordiv__", "__mod__",
               "__pow__", "__gt__", "__lt__", "__ge__", "__le__", "__eq__",
               "__ne__"]
INT_ONLY = ["__and__", "__or__", "__xor__", "__lshift__", "__rshift__"]
UNOPS_F = ["sqrt", "sin", "cos", "tan", "tanh", "exp", "arctan"]
UNOPS_ANY = ["__neg__", "__abs__", "square"]


def _mk(np_, dtype, positive=False):
    a = np_.arange(1, 201) if positive else np_.arange(-100, 100)
    if np.dtype(dtype).kind == "f":
        return (a * 0.25).astype(dtype) if dtype == np.float32 \
            else a * 0.25
    return a.astype(dtype) if dtype != np.int64 else a


class TestBinopMatrix:
    @pytest.mark.parametrize("op", BIN_METHODS)
    @pytest.mark.parametrize("dt", [np.int64, np.float64, np.float32,
                                    np.int32])
    def test_array_array(self, ra, op, dt):
        if op in ("__floordiv__", "__mod__", "__pow__"):
            def impl(np_):
                a = _mk(np_, dt, positive=True)
                b = _mk(np_, dt, positive=True) + 1
                if op == "__pow__" and np.dtype(dt).kind == "i":
                    b = b % 5  # keep int powers small
                return getattr(a, op)(b)
        else:
            def impl(np_):
                a = _mk(np_, dt)
                b = _mk(np_, dt) * 3 + 1
                return getattr(a, op)(b)
        tol = 1e-5 if dt == np.float32 else (1e-12 if np.dtype(dt).kind == "f"
                                             else None)
        run_both(impl, ra, tol=tol)

    @pytest.mark.parametrize("op", BIN_METHODS)
    def test_array_scalar_and_reversed(self, ra, op):
        def impl(np_):
            a = _mk(np_, np.float64, positive=True)
            r1 = getattr(a, op)(3.0)
            rop = "__r" + op[2:] if op in ("__add__", "__sub__", "__mul__",
                                           "__floordiv__", "__mod__",
                                           "__pow__") else None
            if rop:
                r2 = getattr(a, rop)(7.0)
                return r1 * 1.0 + r2
            return r1
        run_both(impl, ra, tol=1e-12)

    @pytest.mark.parametrize("op", INT_ONLY)
    def test_int_bitops(self, ra, op):
        def impl(np_):
            a = np_.arange(512)
            b = np_.arange(512) % 7 + 1
            if op in ("__lshift__", "__rshift__"):
                b = b % 5
            return getattr(a, op)(b)
        run_both(impl, ra)

    def test_mixed_dtype_promotion(self, ra):
        def impl(np_):
            a = np_.arange(100)                       # int64
            b = np_.arange(100).astype(np.int32)
            c = (np_.arange(100) * 0.5)               # float64
            d = (np_.arange(100) * 0.25).astype(np.float32)
            return (a + b) * 1.0 + (c + d) + (b + d)
        r, n = run_both(impl, ra, tol=1e-6)
        assert r.dtype == n.dtype


class TestUnopMatrix:
    @pytest.mark.parametrize("op", UNOPS_F)
    @pytest.mark.parametrize("dt", [np.float64, np.float32])
    def test_float_unop(self, ra, op, dt):
        def impl(np_):
            a = _mk(np_, dt, positive=True)
            return getattr(np_, op)(a)
        run_both(impl, ra, tol=3e-5 if dt == np.float32 else 1e-12)

    @pytest.mark.parametrize("op", UNOPS_ANY)
    @pytest.mark.parametrize("dt", [np.int64, np.float64])
    def test_any_unop(self, ra, op, dt):
        def impl(np_):
            a = _mk(np_, dt)
            if op == "square":
                return np_.square(a)
            return getattr(a, op)()
        run_both(impl, ra)

    def test_log_arcsin_arccos_domain(self, ra):
        def impl(np_):
            a = np_.arange(1, 100) * 0.01
            return np_.log(a) + np_.arcsin(a) + np_.arccos(a)
        run_both(impl, ra, tol=1e-12)

    def test_is_predicates(self, ra):
        def impl(np_):
            a = np_.arange(100) * 1.0 - 50
            b = np_.log(a)   # nans for negatives, -inf at 0
            return (np_.isnan(b).sum(), np_.isinf(b).sum(),
                    np_.isfinite(b).sum())
        with np.errstate(all="ignore"):
            res_r = impl(ra)
            res_n = impl(np)
        assert tuple(int(x) for x in res_r) == tuple(int(x) for x in res_n)

    def test_invert_logicalnot(self, ra):
        def impl(np_):
            a = np_.arange(64)
            return (~a) + np_.logical_not(a % 3).astype(np.int64)
        run_both(impl, ra)


GPU_SAMPLE = [("__add__", np.float32), ("__mod__", np.int64),
              ("__floordiv__", np.int32), ("__pow__", np.float64),
              ("__lshift__", np.int64)]


@pytest.mark.gpu
class TestOpMatrixGpu:
    @pytest.mark.parametrize("op,dt", GPU_SAMPLE)
    def test_binop_gpu(self, ra_gpu, op, dt):
        def impl(np_):
            a = _mk(np_, dt, positive=True)
            b = _mk(np_, dt, positive=True) % 13 + 1
            if op == "__lshift__":
                b = b % 5
            return getattr(a, op)(b)
        tol = 1e-5 if dt == np.float32 else (1e-12 if np.dtype(dt).kind == "f"
                                             else None)
        run_both(impl, ra_gpu, tol=tol)

    def test_full_matrix_gpu(self, ra_gpu):
        """One big combined expression exercising most ops in one kernel."""
        def impl(np_):
            a = np_.arange(1, 10_001)
            f = a * 0.125
            return ((a % 97) + (a // 7) + a ** 2 % 1000
                    + np_.where(a % 2 == 0, a, -a)
                    + (np_.sqrt(f) + np_.tanh(f)
                       + np_.exp(-f * 0.001)).astype(np.int64))
        run_both(impl, ra_gpu)


class TestRoundingOps:
    @pytest.mark.parametrize("op", ["floor", "ceil", "trunc", "rint",
                                    "sign"])
    def test_float(self, ra, op):
        def impl(np_):
            a = np_.arange(-100, 100) * 0.37
            return getattr(np_, op)(a)
        run_both(impl, ra)

    @pytest.mark.parametrize("op", ["floor", "ceil", "sign"])
    def test_int(self, ra, op):
        def impl(np_):
            a = np_.arange(-50, 50)
            return getattr(np_, op)(a)
        run_both(impl, ra)

    def test_clip(self, ra):
        def impl(np_):
            a = np_.arange(200) - 100
            if np_ is np:
                return np.clip(a, -20, 55)
            return a.clip(-20, 55)
        run_both(impl, ra)
