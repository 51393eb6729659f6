"""Seeded random in-scope program generator for parity fuzzing (v6).

Generates programs over the supported vocabulary (creation, elementwise
chains, slice views, setitem, masked writes, reductions, axis reductions,
cumsum, astype, transpose/broadcast) as DATA, then interprets them against
either NumPy or ramba_amd — the same run_both pattern as the reference's
tests, driven by a generator instead of hand-written cases.  Deterministic
per seed; failures reproduce with the seed number.

v6 restores the DISCONTINUOUS float ops (`%`, `//`, floor, sign,
float→int cast) via per-value EXACTNESS tracking:
a value is "exact" when both interpretations compute it bit-identically,
which holds for integer arithmetic and for float chains built purely from
IEEE correctly-rounded ops (+ - * / sqrt abs neg min max; the HIP kernels
compile with -ffp-contract=off, csrc/ramba_rt.cpp:99).  Discontinuous ops
are drawn only on exact values, where comparison is sound at any
tolerance.  What taints exactness depends on the comparison PAIR:

- mode="numpy" (engine vs direct NumPy): true division taints (the
  engine's div→mul-by-reciprocal rewrite, shared with the reference,
  ramba.py:6121, costs 1 ulp vs NumPy's division) and the transcendentals
  taint (rt_sincos etc. vs libm on the HIP pair);
- mode="oracle" (HIP engine vs the CPU-oracle backend through the SAME
  frontend, tests/test_gpu_parity.py): both sides apply the same
  rewrites, so division stays exact; only the transcendentals taint
  (different implementations) plus association-order ops (float cumsum).
"""

import numpy as np


FLOAT_UN = ["sin", "cos", "sqrt", "tanh", "exp_neg", "arctan"]
# IEEE correctly-rounded unaries: identical on every backend
EXACT_UN = {"sqrt", "neg", "abs", "square"}
ANY_UN = ["neg", "abs", "square"]
# discontinuous unaries: drawn only on exact values (v6)
DISC_UN = ["floor", "sign"]
BIN = ["add", "sub", "mul", "maximum", "minimum", "div1"]
# discontinuous binaries: exact operands only (v6); for integers that is
# always true, for floats only untainted chains
DISC_BIN = ["mod1", "floordiv1"]
INT_BIN = ["bitand", "bitxor"]


def _apply_un(np_, x, op):
    if op == "sin":
        return np_.sin(x)
    if op == "cos":
        return np_.cos(x)
    if op == "sqrt":
        return np_.sqrt(np_.absolute(x) if hasattr(x, "asarray")
                        else np.abs(x))
    if op == "tanh":
        return np_.tanh(x)
    if op == "exp_neg":
        a = np_.absolute(x) if hasattr(x, "asarray") else np.abs(x)
        return np_.exp(-(a * 0.001))
    if op == "arctan":
        return np_.arctan(x)
    if op == "neg":
        return -x
    if op == "abs":
        return abs(x)
    if op == "square":
        return np_.square(x)
    if op == "floor":
        return np_.floor(x)
    if op == "sign":
        return np_.sign(x)
    raise AssertionError(op)


def _apply_bin(np_, a, b, op):
    if op == "add":
        return a + b
    if op == "sub":
        return a - b
    if op == "mul":
        return a * b
    if op == "maximum":
        return np_.maximum(a, b)
    if op == "minimum":
        return np_.minimum(a, b)
    if op == "mod1":
        return a % (abs(b) + 1)
    if op == "div1":
        return a / (abs(b) + 1.25)
    if op == "floordiv1":
        return a // (abs(b) + 1)
    if op == "bitand":
        return a & b
    if op == "bitxor":
        return a ^ b
    raise AssertionError(op)


def _rand_slice(rng, n):
    if n < 4:
        return slice(None)
    a = int(rng.integers(0, n // 3))
    b = int(rng.integers(2 * n // 3, n))
    step = int(rng.choice([1, 1, 1, 2, 3, -1]))
    if step < 0:
        return slice(b - 1, None if a == 0 else a - 1, step)
    return slice(a, b, step)


def build_program(seed, mode="numpy"):
    """Returns (impl, tol): impl(np_) runs the program; tol for comparison.

    mode: "numpy" = compared against direct NumPy (div taints exactness);
          "oracle" = compared against the rewrite-consistent CPU-oracle
          backend (div exact; transcendentals/assoc-order taint).
    """
    assert mode in ("numpy", "oracle")
    master = np.random.default_rng(seed)
    ndim = int(master.integers(1, 4))   # 1-, 2- or 3-D base arrays
    two_d = ndim == 2
    n0 = int(master.integers(37, 400))
    n1 = int(master.integers(17, 120))
    n2 = int(master.integers(5, 23))
    nsteps = int(master.integers(3, 9))
    # pre-draw all decisions so both interpretations agree
    plan_seed = int(master.integers(0, 2 ** 31))

    div_taints = (mode == "numpy")

    def impl(np_):
        rng = np.random.default_rng(plan_seed)
        is_np = np_ is np
        pool = []     # (value, kind, exact) kind: 'i' int64, 'f' f64, 'g' f32
        scalars = []

        def mk_base():
            if ndim == 3:
                a = np_.fromfunction(
                    lambda x, y, z: x * 23 + y * 5 + z, (n1, n2, 11),
                    dtype=np.int64)
            elif two_d:
                a = np_.fromfunction(
                    lambda x, y: x * 7 + y * 3, (n0, n1), dtype=np.int64)
            else:
                a = np_.arange(n0 * 2)
            return a

        pool.append((mk_base(), "i", True))
        pool.append(((mk_base() * 3 + 1) % 101, "i", True))
        pool.append((mk_base() * 0.001953125, "f", True))  # exact binary

        for _ in range(nsteps):
            action = rng.choice(
                ["un", "bin", "view", "setitem", "mask", "reduce",
                 "axred", "cumsum", "astype", "where", "clip", "mean",
                 "outer", "axcumsum", "maskget", "fdisc"])
            i = int(rng.integers(0, len(pool)))
            val, kind, exact = pool[i]
            if action == "un":
                if kind == "i":
                    op = str(rng.choice(ANY_UN + DISC_UN))
                else:
                    op = str(rng.choice(FLOAT_UN + ANY_UN))
                r = _apply_un(np_, val, op)
                ex = exact and (op in EXACT_UN or op in DISC_UN
                                or kind == "i")
                pool.append((r, "f" if op in FLOAT_UN else kind, ex))
            elif action == "fdisc":
                # v6: discontinuous float unaries on EXACT floats only
                if kind in ("f", "g") and exact:
                    op = str(rng.choice(DISC_UN))
                    pool.append((_apply_un(np_, val, op), kind, True))
            elif action == "bin":
                j = int(rng.integers(0, len(pool)))
                v2, k2, ex2 = pool[j]
                shape1 = val.shape
                shape2 = v2.shape
                if shape1 != shape2:
                    v2 = float(rng.uniform(0.5, 3.0)) \
                        if (kind != "i" or k2 != "i") \
                        else int(rng.integers(1, 7))
                    k2 = "f" if isinstance(v2, float) else "i"
                    ex2 = True
                ops = list(BIN)
                if kind == "i" and k2 == "i":
                    ops += INT_BIN + DISC_BIN
                elif exact and ex2:
                    # v6: float %, // sound on exact operands
                    ops += DISC_BIN
                op = str(rng.choice(ops))
                r = _apply_bin(np_, val, v2, op)
                ex = exact and ex2 and not (op == "div1" and div_taints)
                # result-kind: int only for int op int (seed 598751: the
                # old label kept 'i' for int x float32('g'), letting a
                # float64 value reach the integer-only op pool — both
                # engines then raise identically, a generator-validity
                # bug, not a parity gap)
                if kind == "i" and k2 == "i":
                    rk = "f" if op == "div1" else "i"
                elif "f" in (kind, k2) or op == "div1":
                    rk = "f"
                else:
                    rk = "g"
                pool.append((r, rk, ex))
            elif action == "view":
                if val.ndim == 1:
                    sl = _rand_slice(rng, val.shape[0])
                    pool.append((val[sl], kind, exact))
                elif val.ndim == 2:
                    sl = (_rand_slice(rng, val.shape[0]),
                          _rand_slice(rng, val.shape[1]))
                    v = val[sl]
                    if rng.integers(0, 3) == 0:
                        v = v.T
                    pool.append((v, kind, exact))
                else:
                    sl = tuple(_rand_slice(rng, val.shape[d])
                               for d in range(3))
                    v = val[sl]
                    if rng.integers(0, 3) == 0:
                        perm = list(rng.permutation(3))
                        v = v.transpose(perm)
                    pool.append((v, kind, exact))
            elif action == "setitem":
                # write a computed value into a slice of a FRESH array
                if is_np:
                    tgt = np.zeros(val.shape,
                                   dtype=np.float64 if kind != "g"
                                   else np.float32)
                else:
                    tgt = np_.zeros(val.shape,
                                    dtype=np.float64 if kind != "g"
                                    else np.float32)
                if val.ndim == 1 and val.shape[0] >= 8:
                    a = int(rng.integers(1, 3))
                    tgt[a:-a] = val[a:-a] * 0.5
                else:
                    tgt[...] = val * 0.5 if val.ndim == 1 else val * 0.5
                pool.append((tgt, "f" if kind != "g" else "g", exact))
            elif action == "mask":
                if kind == "i":
                    m = (val % 5) == 0
                    w = val.copy()
                    w[m] = -7
                    pool.append((w, kind, exact))
            elif action == "reduce":
                op = str(rng.choice(["sum", "max", "min"]))
                if op != "sum" and 0 in val.shape:
                    continue   # min/max of empty raises (both sides)
                r = getattr(val, op)()
                scalars.append(float(r))
            elif action == "axred":
                if val.ndim >= 2 and min(val.shape) >= 2:
                    # v6 also draws axis TUPLES
                    if val.ndim == 3 and rng.integers(0, 3) == 0:
                        axs = rng.permutation(3)[:2]
                        ax = tuple(sorted(int(a) for a in axs))
                    else:
                        ax = int(rng.integers(0, val.ndim))
                    r = val.sum(axis=ax)
                    # float axis-sum association differs per backend
                    pool.append((r, "i" if kind == "i" else "f",
                                 exact and kind == "i"))
            elif action == "cumsum":
                if val.ndim == 1 and val.shape[0] > 0 and kind == "i":
                    pool.append((val.cumsum(), "i", exact))
            elif action == "astype":
                if kind == "i" and exact:
                    # v6: int -> f64 is exact (values < 2^52)
                    pool.append((val.astype(np.float64), "f", True))
                elif exact and kind in ("f", "g") \
                        and rng.integers(0, 2) == 0:
                    # v6: float->int truncation restored on exact floats
                    # (seed 28097's class is sound here)
                    clipped = np_.minimum(np_.maximum(val, -1e9), 1e9)
                    pool.append((clipped.astype(np.int64), "i", True))
                else:
                    # float narrows to f32 (continuous, correctly rounded)
                    pool.append((val.astype(np.float32), "g", exact))
            elif action == "where":
                c = float(rng.uniform(-5.0, 50.0))
                r = np_.where(val > c, val, -val)
                pool.append((r, kind, exact))
            elif action == "clip":
                lo = float(rng.uniform(-10.0, 0.0))
                hi = float(rng.uniform(1.0, 100.0))
                if kind == "i":
                    r = val.clip(int(lo), int(hi))
                else:
                    r = val.clip(lo, hi)
                pool.append((r, kind, exact))
            elif action == "mean":
                if 0 in val.shape:
                    continue
                if kind != "i":
                    scalars.append(float(val.mean()))
                else:
                    scalars.append(float(val.sum()))
            elif action == "outer":
                if val.ndim == 1 and val.shape[0] <= 200:
                    r = val[:, None] * (val[None, :] + 1)
                    pool.append((r, kind, exact))
            elif action == "axcumsum":
                if val.ndim >= 2 and kind in ("f", "i"):
                    ax = int(rng.integers(0, val.ndim))
                    # float scan association differs per backend
                    pool.append((val.cumsum(axis=ax), kind,
                                 exact and kind == "i"))
            elif action == "maskget":
                # integer-valued membership only: a threshold on computed
                # floats would make selection itself ulp-sensitive
                if kind == "i":
                    k = int(rng.integers(2, 9))
                    pool.append((val[(val % k) == 0], kind, exact))

        # result: flattened concat of the last few pool values + scalars
        outs = []
        for (v, k, _) in pool[-4:]:
            a = v.asarray() if hasattr(v, "asarray") else np.asarray(v)
            outs.append(np.asarray(a, dtype=np.float64).reshape(-1))
        outs.append(np.asarray(scalars, dtype=np.float64))
        return np.concatenate(outs) if outs else np.zeros(0)

    return impl, 1e-4


def check_seed(ra_module, seed, mode="numpy"):
    impl, tol = build_program(seed, mode=mode)
    with np.errstate(all="ignore"):
        got = impl(ra_module)
        ref = impl(np)
    np.testing.assert_allclose(got, ref, rtol=tol, atol=tol,
                               err_msg=f"fuzz seed {seed} (mode={mode})")


def check_seed_vs_oracle(ra_module, seed):
    """HIP engine vs the CPU-oracle backend through the SAME frontend:
    rewrite-consistent on both sides, so the v6 discontinuous float ops
    compare soundly.  Swaps the global runtime to a
    fresh NumpyBackend for the reference leg, then swaps back."""
    import ramba_amd as ra
    from ramba_amd.runtime import Runtime
    import sys, os
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    from oracle.numpy_backend import NumpyBackend

    impl, tol = build_program(seed, mode="oracle")
    rt_prod = ra._deferred.get_runtime()
    with np.errstate(all="ignore"):
        got = impl(ra_module)
        nb = NumpyBackend()
        rt_cpu = Runtime(nb, rank=0, world=1)
        nb.attach(rt_cpu)
        ra._deferred.set_runtime(rt_cpu)
        try:
            ref = impl(ra_module)
        finally:
            ra._deferred.set_runtime(rt_prod)
    np.testing.assert_allclose(got, ref, rtol=tol, atol=tol,
                               err_msg=f"fuzz seed {seed} (vs oracle)")
