"""Fused-group IR -> gfx950 HIP source.

The analog of the reference's function-string emission
(`deferred_op.execute`, ramba/ramba.py:8115-8265: codelines -> a
`numba.pndindex` loop, sha-hashed for the JIT cache) — but emitting HIP for
hand-tuned CDNA4 execution instead of Python for Numba:

- 256-thread blocks (4 waves of 64), grid-stride over the innermost axis,
  outer axes on blockIdx.y/z with grid-stride loops;
- vectorised loads/stores (`ext_vector_type`, 16 B per lane) for unit-stride
  aligned operands — the coalescing rule of the CDNA4 guide (G2/G13);
- dead arrays live in registers (the live_gids rule, ramba.py:8123);
- `sin`+`cos` of one operand fuse into a single `sincos` range reduction;
- axis-less reductions: per-thread accumulator -> 64-wide `__shfl_down`
  wave tree -> LDS across the 4 waves -> one partial per workgroup -> a
  one-block finish kernel (replaces internal_reduction1/2b,
  ramba.py:5789-5863; the cross-rank step is RCCL allreduce in the runtime).

The kernel cache key is the structural hash of (statements, operand stride
classes, dtypes, vec width) — runtime values (pointers, extents, strides,
scalars) are kernel arguments, so iterating workloads reuse one kernel.
"""

import hashlib
import os
import struct

import numpy as np

from . import ir

CTYPE = {
    np.dtype(np.bool_): "unsigned char",
    np.dtype(np.uint8): "unsigned char",
    np.dtype(np.int8): "signed char",
    np.dtype(np.int16): "short",
    np.dtype(np.int32): "int",
    np.dtype(np.int64): "long long",
    np.dtype(np.float32): "float",
    np.dtype(np.float64): "double",
}

def ctype(dt):
    return CTYPE[np.dtype(dt)]


# ---------------------------------------------------------------------------
# operand structural classes
# ---------------------------------------------------------------------------

class OpndClass:
    __slots__ = ("name", "dtype", "inner", "outer_g", "rel", "base",
                 "delta", "krow")
    # inner: 'v' aligned vector (unit stride, rel 0), 'u' unit stride with a
    #        structural misalignment `rel` relative to the anchor (reads use
    #        two aligned vector loads + constant lane extract), 'z' zero
    #        stride, 'g' general stride
    # base/delta: same-array fold — this operand addresses base's buffer at
    #        a small structural offset delta, so overlapping vector loads
    #        CSE (a 5-pt stencil's x±1 shifted loads share the center's
    #        aligned vectors — probe_stencil measured the win)

    def __init__(self, name, dtype, inner, outer_g, rel=0, base=None,
                 delta=0, krow=0):
        self.name = name
        self.dtype = np.dtype(dtype)
        self.inner = inner
        self.outer_g = outer_g   # tuple of bools: outer axis stride passed?
        self.rel = rel
        self.base = base
        self.delta = delta
        self.krow = krow   # symbolic row delta: addr uses (i0 + krow) * s0


def classify_plan(plan, vec):
    """Structural classification of every live operand.

    The vectorised grid is ALIGNED to the anchor operand (the first
    unit-stride written stream): its stores become aligned vector stores;
    unit-stride reads at a different alignment carry a structural `rel`
    shift and are served by two aligned vector loads with a constant lane
    extract (measured 2.4-3x over per-lane scalar loads, which produce a
    stride-V lane pattern -- tools/probe_stencil).

    Returns (classes, novec, anchor_name): novec means a written stream
    cannot be aligned (or a temp operand would read out of bounds) and the
    kernel must be built with vec=1."""
    nd = len(plan.itershape)
    written = {st.target for st in plan.statements}
    anchor = None
    w_off = None
    for op in plan.operands:
        inner = op.strides[nd - 1] if nd else 0
        if inner == 1 and op.name in written:
            anchor, w_off = op.name, op.offset0
            break
    if anchor is None:
        for op in plan.operands:
            inner = op.strides[nd - 1] if nd else 0
            if inner == 1:
                anchor, w_off = op.name, op.offset0
                break
    if w_off is None:
        w_off = 0
    out = []
    novec = False
    for op in plan.operands:
        st = op.strides
        inner = st[nd - 1] if nd else 0
        rel = 0
        if inner == 1 and vec > 1:
            rel = (op.offset0 - w_off) % vec
            icl = "v" if rel == 0 else "u"
            if rel != 0 and (op.name in written or op.kind == "temp"):
                # cannot mis-store / temp buffers carry no safety pads
                novec = True
        elif inner == 1:
            icl = "u"
        elif inner == 0:
            icl = "z"
        else:
            icl = "g"
        outer_g = tuple(st[d] != 0 for d in range(nd - 1))
        out.append(OpndClass(op.name, op.dtype, icl, outer_g, rel))
    # same-array offset folding: container operands of the same backing
    # array with identical strides and a small offset difference share the
    # base operand's pointer/offset/stride args; their own offset becomes
    # a structural literal delta
    by_group = {}
    for c, op in zip(out, plan.operands):
        if op.kind != "container" or c.inner not in ("v", "u"):
            continue
        gkey = (id(op.bd), op.strides)
        bases = by_group.setdefault(gkey, [])
        folded = False
        s0 = op.strides[0] if nd == 2 and len(op.strides) == 2 else 0
        for (bc, bop) in bases:
            delta = op.offset0 - bop.offset0
            krow = 0
            if s0 > 0:
                # nearest small row multiple (enables the unrolled y-block
                # cross-row load CSE -- tools/probe_stencil yunroll4)
                kr = int(round(delta / s0))
                if 0 < abs(kr) <= 4 and abs(delta - kr * s0) <= 4 * vec:
                    krow = kr
                    delta = delta - kr * s0
            if abs(delta) <= 4 * vec:
                c.base = bc.name
                c.delta = int(delta)
                c.krow = int(krow)
                folded = True
                break
        if not folded:
            bases.append((c, op))
    return out, novec, anchor


def plan_structure(plan):
    """(vec, classes, anchor, key) — shared by generate() and the backend's
    kernel-cache fast path."""
    vec = decide_vec(plan)
    classes, novec, anchor = classify_plan(plan, vec)
    if novec and vec > 1:
        vec = 1
        classes, _, anchor = classify_plan(plan, 1)
    key = structural_key(plan, classes, vec)
    return vec, classes, anchor, key


# ---------------------------------------------------------------------------
# SSA renaming + use analysis
# ---------------------------------------------------------------------------

def _rename(e, version):
    if isinstance(e, ir.Ref):
        v = version.get(e.name, 0)
        return ir.Ref(f"{e.name}@{v}", e.dtype)
    if isinstance(e, ir.Bin):
        return ir.Bin(e.op, _rename(e.a, version), _rename(e.b, version),
                      e.dtype)
    if isinstance(e, ir.Un):
        return ir.Un(e.op, _rename(e.a, version), e.dtype)
    if isinstance(e, ir.Cast):
        return ir.Cast(_rename(e.a, version), e.dtype)
    if isinstance(e, ir.Where):
        return ir.Where(_rename(e.c, version), _rename(e.a, version),
                        _rename(e.b, version), e.dtype)
    return e


def ssa_statements(statements, acc_names):
    """Returns list of (versioned_target|None-for-acc, acc_name|None, expr)"""
    version = {}
    out = []
    for st in statements:
        if st.target in acc_names:
            # acc = comb(acc, src): rename only the src side
            assert isinstance(st.expr, ir.Bin)
            src = _rename(st.expr.b, version)
            out.append((None, (st.target, st.expr.op), src))
        else:
            e = _rename(st.expr, version)
            v = version.get(st.target, 0) + 1
            version[st.target] = v
            out.append((f"{st.target}@{v}", None, e))
    return out, version


def collect_nodes(e, bag):
    bag[e] = bag.get(e, 0) + 1
    if isinstance(e, ir.Bin):
        collect_nodes(e.a, bag)
        collect_nodes(e.b, bag)
    elif isinstance(e, (ir.Un, ir.Cast)):
        collect_nodes(e.a, bag)
    elif isinstance(e, ir.Where):
        collect_nodes(e.c, bag)
        collect_nodes(e.a, bag)
        collect_nodes(e.b, bag)


# ---------------------------------------------------------------------------
# expression emission (per lane)
# ---------------------------------------------------------------------------

class LaneEmitter:
    def __init__(self, gen, lane_tag, idx_exprs):
        self.gen = gen
        self.tag = lane_tag          # suffix for locals
        self.idx = idx_exprs         # C index expr per axis (strings)
        self.memo = {}               # node -> C var / expr string
        self.lines = []
        self.n = 0

    def fresh(self, ct, init):
        self.n += 1
        v = f"t{self.n}{self.tag}"
        self.lines.append(f"      {ct} {v} = {init};")
        return v

    def emit(self, e):
        if e in self.memo:
            return self.memo[e]
        r = self._emit(e)
        self.memo[e] = r
        return r

    def _emit(self, e):
        g = self.gen
        if isinstance(e, ir.Ref):
            return g.lane_value(self, e.name, e.dtype)
        if isinstance(e, ir.ScalarArg):
            ct = ctype(e.dtype)
            if np.dtype(e.dtype).kind == "f":
                return f"(({ct})a.{e.name})"
            return f"(({ct})a.{e.name})"
        if isinstance(e, ir.Const):
            ct = ctype(e.dtype)
            v = e.value
            if isinstance(v, float):
                return f"(({ct}){v!r})"
            return f"(({ct}){int(v)})"
        if isinstance(e, ir.Iota):
            return f"((long long)({self.idx[e.axis]}))"
        if isinstance(e, ir.Cast):
            a = self.emit(e.a)
            src_dt = _dt(e.a)
            ct = ctype(e.dtype)
            if np.dtype(e.dtype) == np.dtype(np.bool_) and \
                    np.dtype(src_dt) != np.dtype(np.bool_):
                return self.fresh("unsigned char", f"({a}) != 0")
            return self.fresh(ct, f"({ct})({a})")
        if isinstance(e, ir.Where):
            c = self.emit(e.c)
            a = self.emit(e.a)
            b = self.emit(e.b)
            ct = ctype(e.dtype)
            return self.fresh(ct, f"({c}) ? ({ct})({a}) : ({ct})({b})")
        if isinstance(e, ir.Un):
            return self._emit_un(e)
        if isinstance(e, ir.Bin):
            return self._emit_bin(e)
        raise TypeError(f"bad node {e!r}")

    def _emit_un(self, e):
        # sincos pairing: if both sin(X) and cos(X) are wanted, emit one call
        g = self.gen
        if e.op in ("sin", "cos"):
            other = ir.Un("cos" if e.op == "sin" else "sin", e.a, e.dtype)
            if ctype(e.dtype) == "double" and other not in g.wanted:
                # SOLO fp64 sin/cos: rt_sincos (Cody-Waite fast path) and
                # discard the other result — measurably cheaper than the
                # generic ocml sin/cos for |x|<1e6 (the mixed pipeline's
                # producer recompute is transcendental-bound)
                x = self.emit(e.a)
                self.n += 1
                s, c = f"sc_s{self.n}{self.tag}", f"sc_c{self.n}{self.tag}"
                self.lines.append(
                    f"      double {s}, {c}; "
                    f"rt_sincos((double)({x}), &{s}, &{c});")
                return s if e.op == "sin" else c
            if other in g.wanted and other not in self.memo:
                x = self.emit(e.a)
                ct = ctype(e.dtype)
                fn = "sincosf" if ct == "float" else "rt_sincos"
                self.n += 1
                s, c = f"sc_s{self.n}{self.tag}", f"sc_c{self.n}{self.tag}"
                self.lines.append(
                    f"      {ct} {s}, {c}; {fn}(({ct})({x}), &{s}, &{c});")
                sin_node = e if e.op == "sin" else other
                cos_node = e if e.op == "cos" else other
                self.memo[sin_node] = s
                self.memo[cos_node] = c
                return self.memo[e]
        a = self.emit(e.a)
        src_dt = np.dtype(_dt(e.a))
        ct = ctype(e.dtype)
        f32 = ct == "float"
        sfx = "f" if f32 else ""
        op = e.op
        simple = {"sqrt": "sqrt", "sin": "sin", "cos": "cos", "tan": "tan",
                  "sinh": "sinh", "cosh": "cosh", "tanh": "tanh",
                  "arcsin": "asin", "arccos": "acos", "arctan": "atan",
                  "exp": "exp", "log": "log"}
        if op in ("floor", "ceil", "trunc", "rint"):
            if src_dt.kind in "iub":
                return self.fresh(ct, f"({a})")
            fn = {"floor": "floor", "ceil": "ceil", "trunc": "trunc",
                  "rint": "rint"}[op]
            return self.fresh(ct, f"{fn}{sfx}(({ct})({a}))")
        if op == "sign":
            if src_dt.kind == "f":
                return self.fresh(
                    ct, f"__builtin_isnan((double)({a})) ? ({ct})({a}) : "
                        f"(({a}) > 0 ? ({ct})1 : (({a}) < 0 ? ({ct})-1 "
                        f": ({ct})0))")
            return self.fresh(ct, f"(({a}) > 0) - (({a}) < 0)")
        if op in simple:
            return self.fresh(ct, f"{simple[op]}{sfx}(({ct})({a}))")
        if op == "neg":
            return self.fresh(ct, f"-({a})")
        if op == "abs":
            if src_dt.kind == "f":
                return self.fresh(ct, f"fabs{sfx}({a})")
            if src_dt.kind == "u" or src_dt == np.dtype(np.bool_):
                return self.fresh(ct, f"({a})")
            return self.fresh(ct, f"(({a}) < 0 ? -({a}) : ({a}))")
        if op == "square":
            return self.fresh(ct, f"({a}) * ({a})")
        if op == "isnan":
            return self.fresh("unsigned char",
                              f"__builtin_isnan((double)({a})) ? 1 : 0"
                              if src_dt.kind == "f" else "0")
        if op == "isinf":
            return self.fresh("unsigned char",
                              f"__builtin_isinf((double)({a})) ? 1 : 0"
                              if src_dt.kind == "f" else "0")
        if op == "isfinite":
            return self.fresh("unsigned char",
                              f"__builtin_isfinite((double)({a})) ? 1 : 0"
                              if src_dt.kind == "f" else "1")
        if op == "logical_not":
            return self.fresh("unsigned char", f"(({a}) != 0) ? 0 : 1")
        if op == "invert":
            if src_dt == np.dtype(np.bool_):
                return self.fresh("unsigned char", f"(({a}) != 0) ? 0 : 1")
            return self.fresh(ct, f"~({a})")
        raise NotImplementedError(f"unop {op}")

    def _emit_bin(self, e):
        ct = ctype(e.dtype)
        op = e.op
        # pow with small constant integer exponent -> multiply chain
        # (matches NumPy's small-int fast path)
        if op == "pow" and isinstance(e.b, ir.Const) \
                and float(e.b.value) == int(e.b.value) \
                and 0 <= int(e.b.value) <= 8:
            n = int(e.b.value)
            a = self.emit(e.a)
            av = self.fresh(ctype(_dt(e.a)), f"{a}")
            if n == 0:
                return self.fresh(ct, "1")
            cur = av
            for _ in range(n - 1):
                cur2 = self.fresh(ctype(_dt(e.a)), f"({cur}) * ({av})")
                cur = cur2
            return self.fresh(ct, f"({ct})({cur})")
        a = self.emit(e.a)
        b = self.emit(e.b)
        adt, bdt = np.dtype(_dt(e.a)), np.dtype(_dt(e.b))
        f32 = ct == "float"
        sfx = "f" if f32 else ""
        infix = {"add": "+", "sub": "-", "mul": "*",
                 "bitand": "&", "bitor": "|", "bitxor": "^",
                 "lshift": "<<", "rshift": ">>"}
        if op in infix:
            return self.fresh(ct, f"(({ct})({a})) {infix[op]} (({ct})({b}))")
        cmp = {"gt": ">", "lt": "<", "ge": ">=", "le": "<=", "eq": "==",
               "ne": "!="}
        if op in cmp:
            wt = ctype(np.result_type(adt, bdt))
            return self.fresh("unsigned char",
                              f"((({wt})({a})) {cmp[op]} (({wt})({b}))) ? 1 : 0")
        if op == "div":
            return self.fresh(ct, f"(({ct})({a})) / (({ct})({b}))")
        if op == "floordiv":
            if np.dtype(e.dtype).kind == "f":
                return self.fresh(
                    ct, f"floor{sfx}((({ct})({a})) / (({ct})({b})))")
            return self.fresh(ct, f"rt_fdiv((long long)({a}), "
                                  f"(long long)({b}))")
        if op == "mod":
            if np.dtype(e.dtype).kind == "f":
                return self.fresh(ct, f"rt_fmod{sfx}(({ct})({a}), "
                                      f"({ct})({b}))")
            return self.fresh(ct, f"rt_imod((long long)({a}), "
                                  f"(long long)({b}))")
        if op == "pow":
            if np.dtype(e.dtype).kind in "iu":
                return self.fresh(ct, f"rt_ipow((long long)({a}), "
                                      f"(long long)({b}))")
            return self.fresh(ct, f"pow{sfx}(({ct})({a}), ({ct})({b}))")
        if op in ("minimum", "maximum"):
            lt = "<" if op == "minimum" else ">"
            if np.dtype(e.dtype).kind == "f":
                # NaN-propagating (NumPy minimum/maximum semantics)
                return self.fresh(
                    ct, f"(__builtin_isnan((double)({a})) || (({ct})({a}))"
                        f" {lt} (({ct})({b}))) ? ({ct})({a}) : ({ct})({b})")
            return self.fresh(
                ct, f"((({ct})({a})) {lt} (({ct})({b}))) ? "
                    f"({ct})({a}) : ({ct})({b})")
        if op in ("logical_and", "logical_or", "logical_xor"):
            cop = {"logical_and": "&&", "logical_or": "||"}.get(op)
            if cop:
                return self.fresh("unsigned char",
                                  f"((({a}) != 0) {cop} (({b}) != 0)) ? 1 : 0")
            return self.fresh("unsigned char",
                              f"((({a}) != 0) != (({b}) != 0)) ? 1 : 0")
        raise NotImplementedError(f"binop {op}")


def _dt(e):
    return e.dtype


# ---------------------------------------------------------------------------
# the kernel generator
# ---------------------------------------------------------------------------

PREAMBLE = r"""
typedef long long i64;
typedef unsigned long long u64;
typedef __attribute__((ext_vector_type(2))) double d2_t;
typedef __attribute__((ext_vector_type(4))) float f4_t;
typedef __attribute__((ext_vector_type(2))) long long l2_t;
typedef __attribute__((ext_vector_type(2))) float f2_t;
typedef __attribute__((ext_vector_type(2))) int i2_t;
typedef __attribute__((ext_vector_type(4))) int i4_t;
typedef __attribute__((ext_vector_type(2))) short s2_t;
typedef __attribute__((ext_vector_type(4))) short s4_t;
typedef __attribute__((ext_vector_type(2))) unsigned char b2_t;
typedef __attribute__((ext_vector_type(4))) unsigned char b4_t;
typedef __attribute__((ext_vector_type(2))) signed char c2_t;
typedef __attribute__((ext_vector_type(4))) signed char c4_t;

__device__ __forceinline__ i64 rt_fdiv(i64 a, i64 b) {
    i64 q = a / b, r = a % b;
    return (r != 0 && ((r < 0) != (b < 0))) ? q - 1 : q;
}
__device__ __forceinline__ i64 rt_imod(i64 a, i64 b) {
    i64 r = a % b;
    return (r != 0 && ((r < 0) != (b < 0))) ? r + b : r;
}
__device__ __forceinline__ double rt_fmod(double a, double b) {
    double r = fmod(a, b);
    return (r != 0.0 && ((r < 0.0) != (b < 0.0))) ? r + b : r;
}
__device__ __forceinline__ float rt_fmodf(float a, float b) {
    float r = fmodf(a, b);
    return (r != 0.0f && ((r < 0.0f) != (b < 0.0f))) ? r + b : r;
}
__device__ __forceinline__ i64 rt_ipow(i64 b, i64 e) {
    i64 r = 1;
    while (e > 0) { if (e & 1) r *= b; b *= b; e >>= 1; }
    return r;
}

// Fast fp64 sincos: fdlibm-style Cody-Waite reduction by pi/2 (medium
// path, valid while |fn| < 2^20 i.e. |x| < ~1.6e6) + the classic
// __kernel_sin/__kernel_cos minimax polynomials; falls back to ocml
// sincos (full Payne-Hanek) outside the fast range.  ~45 fp64 ops for
// BOTH results vs ~2x that for two ocml calls — the hot chain is at the
// HBM/VALU crossover (SURVEY §7), so this is a throughput lever, and it
// stays well inside the 1e-12 parity tolerance (abs err ~1 ulp).
__device__ __forceinline__ void rt_sincos(double x, double *sr, double *cr) {
    double ax = __builtin_fabs(x);
    if (!(ax < 1.0e6)) { sincos(x, sr, cr); return; }
    const double invpio2 = 6.36619772367581382433e-01;
    const double pio2_1 = 1.57079632673412561417e+00;
    const double pio2_1t = 6.07710050650619224932e-11;
    const double pio2_2 = 6.07710050630396597660e-11;
    const double pio2_2t = 2.02226624879595063154e-21;
    double fn = __builtin_rint(x * invpio2);
    int n = (int)fn;
    // fn*pio2_1 is exact while |fn| < 2^20 (pio2_1 carries 33 bits), so
    // y0+y1 ~ x - fn*pi/2 with ~1e-20 absolute error -- far inside the
    // 1e-12 parity tolerance (DESIGN.md §6)
    double r = __builtin_fma(-fn, pio2_1, x);
    double w = fn * pio2_1t;
    double y0 = r - w;
    // second/third Cody-Waite stages when cancellation near a multiple
    // of pi/2 ate the leading bits (fdlibm __ieee754_rem_pio2 medium
    // path): keeps RELATIVE accuracy of sin/cos near their zeros, not
    // just the 1e-12 absolute parity bar
    long long hx = (long long)__builtin_bit_cast(unsigned long long, x);
    long long hy = (long long)__builtin_bit_cast(unsigned long long, y0);
    int bits_lost = (int)((hx >> 52) & 0x7ff) - (int)((hy >> 52) & 0x7ff);
    double y1;
    if (bits_lost > 16) {
        double t = r;
        w = fn * pio2_2;
        r = t - w;
        w = __builtin_fma(fn, pio2_2t, -((t - r) - w));
        y0 = r - w;
        hy = (long long)__builtin_bit_cast(unsigned long long, y0);
        bits_lost = (int)((hx >> 52) & 0x7ff) - (int)((hy >> 52) & 0x7ff);
        if (bits_lost > 49) {
            const double pio2_3  = 2.02226624871116645580e-21;
            const double pio2_3t = 8.47842766036889956997e-32;
            t = r;
            w = fn * pio2_3;
            r = t - w;
            w = __builtin_fma(fn, pio2_3t, -((t - r) - w));
            y0 = r - w;
        }
        y1 = (r - y0) - w;
    } else {
        y1 = (r - y0) - w;
    }
    // kernel_sin(y0, y1)
    const double S1 = -1.66666666666666324348e-01;
    const double S2 = 8.33333333332248946124e-03;
    const double S3 = -1.98412698298579493134e-04;
    const double S4 = 2.75573137070700676789e-06;
    const double S5 = -2.50507602534068634195e-08;
    const double S6 = 1.58969099521155010221e-10;
    double z = y0 * y0;
    double v = z * y0;
    double rs = S2 + z * (S3 + z * (S4 + z * (S5 + z * S6)));
    double ks = y0 - ((z * (0.5 * y1 - v * rs) - y1) - v * S1);
    // kernel_cos(y0, y1)
    const double C1 = 4.16666666666666019037e-02;
    const double C2 = -1.38888888888741095749e-03;
    const double C3 = 2.48015872894767294178e-05;
    const double C4 = -2.75573143513906633035e-07;
    const double C5 = 2.08757232129817482790e-09;
    const double C6 = -1.13596475577881948265e-11;
    double rc = z * (C1 + z * (C2 + z * (C3 + z * (C4 + z * (C5 + z * C6)))));
    double hz = 0.5 * z;
    double wc = 1.0 - hz;
    double kc = wc + (((1.0 - wc) - hz) + (z * rc - y0 * y1));
    switch (n & 3) {
        case 0: *sr = ks;  *cr = kc;  break;
        case 1: *sr = kc;  *cr = -ks; break;
        case 2: *sr = -ks; *cr = -kc; break;
        default: *sr = -kc; *cr = ks; break;
    }
}
"""


class GeneratedKernel:
    """Source + arg-packing recipe for one fused-group structure."""

    def __init__(self, key, source, kname_main, kname_finish, finish_source,
                 fields, vec, nd, nred):
        self.key = key
        self.source = source
        self.kname_main = kname_main
        self.kname_finish = kname_finish
        self.finish_source = finish_source
        self.fields = fields     # list of (kind, payload) for packing
        self.vec = vec
        self.nd = nd
        self.nred = nred
        self.handle = None
        self.finish_handle = None
        self.anchor = None
        self.yblock = 1


def structural_key(plan, classes, vec):
    h = hashlib.sha256()
    h.update(f"nd={len(plan.itershape)};vec={vec};".encode())
    for c in classes:
        h.update(f"op:{c.name}:{c.dtype}:{c.inner}:{c.rel}:{c.base}:"
                 f"{c.delta}:{c.krow}:{c.outer_g};".encode())
    for n, (v, dt) in sorted(plan.scalars.items()):
        h.update(f"sc:{n}:{dt};".encode())
    for n, dt in sorted(plan.dead_vars.items()):
        h.update(f"dv:{n}:{dt};".encode())
    for st in plan.statements:
        h.update(repr((st.target, st.expr)).encode())
    for r in plan.reductions:
        h.update(f"red:{r.acc}:{r.kind}:{r.dtype};".encode())
    return h.hexdigest()[:24]


class KernelGen:
    def __init__(self, plan, classes, vec):
        self.plan = plan
        self.classes = {c.name: c for c in classes}
        self.class_list = classes
        self.vec = vec
        self.nd = len(plan.itershape)
        self.acc_specs = {r.acc: r for r in plan.reductions}
        # SSA
        self.ssa, self.final_version = ssa_statements(
            plan.statements, set(self.acc_specs))
        self.wanted = {}
        for (_, _, e) in self.ssa:
            collect_nodes(e, self.wanted)
        # which operands are read (version-0 refs used) / written
        self.read_ops = set()
        self.written_ops = {}
        for node in self.wanted:
            if isinstance(node, ir.Ref):
                base, v = node.name.split("@")
                if base in self.classes and int(v) == 0:
                    self.read_ops.add(base)
        for (tgt, accn, _) in self.ssa:
            if tgt is not None:
                base, v = tgt.split("@")
                if base in self.classes:
                    self.written_ops[base] = max(
                        self.written_ops.get(base, 0), int(v))

    # -- per-lane value resolution -----------------------------------------

    def lane_value(self, em, versioned, dtype):
        base, v = versioned.split("@")
        if int(v) == 0:
            if base in self.classes:
                return em.gen._load_name(em, base)
            if base in self.plan.dead_vars:
                raise AssertionError(f"dead var {base} read before write")
            raise AssertionError(f"unknown var {versioned}")
        return f"r_{base}_{v}{em.tag}"

    def _load_name(self, em, name):
        return f"ld_{name}{em.tag}"

    # -- addressing ---------------------------------------------------------

    def eff(self, name):
        """arg-owning operand: the fold base, or the operand itself."""
        c = self.classes[name]
        return c.base if c.base is not None else name

    def addr_parts(self, name, inner_expr, row=None):
        """(symbolic parts, constant) of the element index into `name`'s
        buffer — the constant is kept separate so equal addresses CSE.

        `row` = (base_expr, ky): in the unrolled y-block the row index is
        base_expr + ky, so ky and the fold's krow collapse into one
        literal and equal absolute rows produce IDENTICAL part strings
        (true cross-row vector sharing)."""
        c = self.classes[name]
        eff = self.eff(name)
        parts = [f"a.{eff}_off"]
        const = int(c.delta)
        for d in range(self.nd - 1):
            if c.outer_g[d]:
                if d == 0 and row is not None:
                    base_expr, ky = row
                    k = ky + c.krow
                    parts.append(
                        f"({base_expr} + {k}) * a.{eff}_s{d}" if k else
                        f"({base_expr}) * a.{eff}_s{d}")
                elif d == 0 and c.krow:
                    parts.append(f"(i0 + ({c.krow})) * a.{eff}_s{d}")
                else:
                    parts.append(f"i{d} * a.{eff}_s{d}")
        if c.inner in ("v", "u"):
            parts.append(f"({inner_expr})")
        elif c.inner == "g":
            parts.append(f"({inner_expr}) * a.{eff}_sx")
        return parts, const

    def addr_expr(self, name, inner_expr, extra_const=0, row=None):
        parts, const = self.addr_parts(name, inner_expr, row=row)
        const += extra_const
        if const:
            parts = parts + [f"({const})"]
        return " + ".join(parts)

    # -- body generation ------------------------------------------------------

    def gen_lane_body(self, tag, inner_expr, indent, vec_lane=None,
                      svp="", row=None):
        """Emit loads + statements + stores for ONE element.

        vec_lane: (vecvar-suffix, lane index) when inside the vectorised
        body -- loads come from preloaded vectors."""
        idx = []
        for d in range(self.nd - 1):
            if d == 0 and row is not None:
                idx.append(f"a.gs0 + {row[0]} + {row[1]}"
                           if row[1] else f"a.gs0 + {row[0]}")
            else:
                idx.append(f"a.gs{d} + i{d}")
        idx.append(f"a.gs{self.nd-1} + ({inner_expr})" if self.nd else "0")
        em = LaneEmitter(self, tag, idx)
        # loads
        for name in sorted(self.read_ops):
            c = self.classes[name]
            ct = ctype(c.dtype)
            ldv = self._load_name(em, name)
            if vec_lane is not None and c.inner == "v":
                em.lines.append(
                    f"      {ct} {ldv} = "
                    f"{self.preload[name][1]}[{vec_lane}];")
            elif vec_lane is not None and c.inner == "u" and c.rel:
                k = vec_lane + c.rel
                _, ul, uh = self.preload[name]
                src = f"{ul}[{k}]" if k < self.vec \
                    else f"{uh}[{k - self.vec}]"
                em.lines.append(f"      {ct} {ldv} = {src};")
            else:
                em.lines.append(
                    f"      {ct} {ldv} = a.{self.eff(name)}_p["
                    f"{self.addr_expr(name, inner_expr, row=row)}];")
        # statements
        last_val = {}
        for (tgt, accinfo, expr) in self.ssa:
            if accinfo is not None:
                accn, comb = accinfo
                spec = self.acc_specs[accn]
                src = em.emit(expr)
                ct = ctype(spec.dtype)
                em.lines.append(
                    f"      acc_{accn} = "
                    f"{self.comb_expr(comb, spec.dtype, f'acc_{accn}', src)};")
            else:
                base, v = tgt.split("@")
                val = em.emit(expr)
                dt = self.classes[base].dtype if base in self.classes \
                    else self.plan.dead_vars[base]
                ct = ctype(dt)
                em.lines.append(f"      {ct} r_{base}_{v}{tag} = "
                                f"({ct})({val});")
                last_val[base] = f"r_{base}_{v}{tag}"
        # stores
        for name, lastv in sorted(self.written_ops.items()):
            c = self.classes[name]
            final = f"r_{name}_{lastv}{tag}"
            if vec_lane is not None and c.inner == "v":
                em.lines.append(f"      sv{svp}_{name}[{vec_lane}] = "
                                f"{final};")
            else:
                em.lines.append(
                    f"      a.{self.eff(name)}_p["
                    f"{self.addr_expr(name, inner_expr, row=row)}] = "
                    f"{final};")
        pad = " " * (indent - 6)
        return "\n".join(pad + ln.lstrip() if False else
                         (" " * indent) + ln.strip() for ln in em.lines)

    def comb_expr(self, comb, dtype, a, b):
        ct = ctype(dtype)
        if comb == "add":
            return f"({a}) + ({b})"
        if comb == "mul":
            return f"({a}) * ({b})"
        if comb in ("minimum", "maximum"):
            lt = "<" if comb == "minimum" else ">"
            if np.dtype(dtype).kind == "f":
                return (f"(__builtin_isnan((double)({a})) || (({a}) {lt} "
                        f"({b}))) ? ({ct})({a}) : ({ct})({b})")
            return f"(({a}) {lt} ({b})) ? ({ct})({a}) : ({ct})({b})"
        if comb == "logical_and":
            return f"((({a}) != 0) && (({b}) != 0)) ? 1 : 0"
        if comb == "logical_or":
            return f"((({a}) != 0) || (({b}) != 0)) ? 1 : 0"
        raise NotImplementedError(comb)


    def emit_quad(self, L, body_ind, tagp, row=None, shared=None):
        """Preloads + per-lane bodies + vector stores for one (x, row)
        position.  `tagp` keeps locals unique across unrolled rows.
        `row`/`shared`: unrolled y-blocks share one load-CSE dict across
        the rows, with addresses keyed by the folded absolute row — the
        3 column vectors of a 5-pt stencil load 6 times per 4 rows
        instead of 12 (the probe's ycarry structure, generically)."""
        V = self.vec
        self.preload = {}
        cse = {} if shared is None else shared
        nload = self._nload = getattr(self, "_nload", [0])             if shared is not None else [0]

        def vec_load(name, extra_const):
            parts, const = self.addr_parts(name, "vb", row=row)
            const += extra_const
            kk = (self.eff(name), tuple(parts), const)
            v = cse.get(kk)
            if v is None:
                vt = self.vec_type(self.classes[name].dtype)
                nload[0] += 1
                v = f"vq{nload[0]}" if shared is not None \
                    else f"vv{tagp}_{nload[0]}"
                addr = " + ".join(parts + ([f"({const})"] if const else []))
                L.append(f"{' '*body_ind}const {vt} {v} = "
                         f"*(const {vt}*)&a.{self.eff(name)}_p[{addr}];")
                cse[kk] = v
            return v

        for name in sorted(self.read_ops):
            c = self.classes[name]
            if c.inner == "v":
                self.preload[name] = ("v", vec_load(name, 0))
            elif c.inner == "u" and V > 1 and c.rel:
                self.preload[name] = (
                    "u", vec_load(name, -c.rel),
                    vec_load(name, -c.rel + V))
        for name in sorted(self.written_ops):
            c = self.classes[name]
            if c.inner == "v":
                vt = self.vec_type(c.dtype)
                L.append(f"{' '*body_ind}{vt} sv{tagp}_{name};")
        for lane in range(V):
            L.append(f"{' '*body_ind}{{ // lane {lane}")
            L.append(self.gen_lane_body(f"{tagp}_L{lane}", f"vb + {lane}",
                                        body_ind + 2, vec_lane=lane,
                                        svp=tagp, row=row))
            L.append(f"{' '*body_ind}}}")
        for name, _ in sorted(self.written_ops.items()):
            c = self.classes[name]
            if c.inner == "v":
                vt = self.vec_type(c.dtype)
                L.append(f"{' '*body_ind}*({vt}*)&a.{self.eff(name)}_p["
                         f"{self.addr_expr(name, 'vb', row=row)}] = "
                         f"sv{tagp}_{name};")

    # -- full source -----------------------------------------------------------

    def generate(self, key):
        nd, V = self.nd, self.vec
        L = []
        L.append(PREAMBLE)
        # args struct
        L.append("struct Args {")
        fields = []
        for d in range(nd):
            L.append(f"  i64 n{d};")
            fields.append(("iter_n", d))
        for d in range(nd):
            L.append(f"  i64 gs{d};")
            fields.append(("iter_gs", d))
        for c in self.class_list:
            if c.base is not None:
                continue  # folded: shares the base operand's args
            ct = ctype(c.dtype)
            L.append(f"  {ct}* __restrict__ {c.name}_p;")
            fields.append(("ptr", c.name))
            L.append(f"  i64 {c.name}_off;")
            fields.append(("off", c.name))
            for d in range(nd - 1):
                if c.outer_g[d]:
                    L.append(f"  i64 {c.name}_s{d};")
                    fields.append(("stride", (c.name, d)))
            if c.inner == "g":
                L.append(f"  i64 {c.name}_sx;")
                fields.append(("stride", (c.name, nd - 1)))
        for n in sorted(self.plan.scalars):
            _, dt = self.plan.scalars[n]
            if np.dtype(dt).kind == "f":
                L.append(f"  double {n};")
                fields.append(("scalar_f", n))
            else:
                L.append(f"  i64 {n};")
                fields.append(("scalar_i", n))
        if V > 1:
            L.append("  i64 lead;")
            fields.append(("lead", None))
        nred = len(self.plan.reductions)
        if nred:
            L.append("  char* partials;")
            fields.append(("partials", None))
            L.append("  i64 npartials;")
            fields.append(("npartials", None))
        L.append("};")

        kname = f"k_{key}"
        L.append(f'extern "C" __global__ void __launch_bounds__(256) '
                 f"{kname}(Args a) {{")
        # accumulators
        for spec in self.plan.reductions:
            ct = ctype(spec.dtype)
            init = self.init_literal(spec)
            L.append(f"  {ct} acc_{spec.acc} = {init};")
        # outer loops; nd==2 kernels with row-delta folds use an unrolled
        # y-block so LLVM CSEs the overlapping row vectors across rows
        # (tools/probe_stencil yunroll4: +23% over the flat structure)
        yblock = 4 if (nd == 2 and V > 1
                       and any(c.krow for c in self.class_list)) else 1
        self.yblock = yblock
        ind = 2
        if nd == 3:
            L.append(f"{' '*ind}for (i64 i0 = blockIdx.z; i0 < a.n0; "
                     f"i0 += gridDim.z) {{")
            ind += 2
        elif nd == 4:
            # leading axes collapsed onto blockIdx.z
            L.append(f"{' '*ind}for (i64 zz = blockIdx.z; zz < a.n0 * a.n1;"
                     f" zz += gridDim.z) {{")
            ind += 2
            L.append(f"{' '*ind}const i64 i0 = zz / a.n1;")
            L.append(f"{' '*ind}const i64 i1 = zz % a.n1;")
        if nd >= 2 and yblock == 1:
            oy = nd - 2
            L.append(f"{' '*ind}for (i64 i{oy} = blockIdx.y; i{oy} < a.n{oy};"
                     f" i{oy} += gridDim.y) {{")
            ind += 2
        x = nd - 1
        lead = "a.lead" if V > 1 else "0"
        L.append(f"{' '*ind}i64 vb = {lead} + ((i64)blockIdx.x * 256 "
                 f"+ threadIdx.x) * {V};")
        L.append(f"{' '*ind}const i64 xs = (i64)gridDim.x * 256 * {V};")
        L.append(f"{' '*ind}for (; vb + {V} <= a.n{x}; vb += xs) {{")
        body_ind = ind + 2
        if yblock == 1:
            self.emit_quad(L, body_ind, "")
        else:
            L.append(f"{' '*body_ind}i64 yb0 = (i64)blockIdx.y * {yblock};")
            L.append(f"{' '*body_ind}const i64 ys = (i64)gridDim.y * "
                     f"{yblock};")
            L.append(f"{' '*body_ind}for (; yb0 + {yblock} <= a.n0; "
                     f"yb0 += ys) {{")
            # NOTE: no braces between rows — the shared vq* load vars
            # must stay in scope across the unrolled rows
            shared = {}
            self._nload = [0]
            for ky in range(yblock):
                self.emit_quad(L, body_ind + 2, f"_Y{ky}",
                               row=("yb0", ky), shared=shared)
            L.append(f"{' '*body_ind}}}")
            L.append(f"{' '*body_ind}for (i64 i0r = yb0; i0r < a.n0; "
                     f"++i0r) {{ const i64 i0 = i0r;")
            self.emit_quad(L, body_ind + 2, "_YT")
            L.append(f"{' '*body_ind}}}")
        L.append(f"{' '*ind}}}")
        # scalar edges: [0, lead) prologue and [tstart, n) tail
        if V > 1:
            yloop = yblock > 1
            def edge(cond, lo, hi, tag):
                L.append(f"{' '*ind}{cond} {{")
                e_ind = ind + 2
                if yloop:
                    L.append(f"{' '*e_ind}for (i64 yb0 = (i64)blockIdx.y * "
                             f"{yblock}; yb0 < a.n0; yb0 += (i64)gridDim.y "
                             f"* {yblock})")
                    L.append(f"{' '*e_ind}for (i64 i0 = yb0; i0 < a.n0 && "
                             f"i0 < yb0 + {yblock}; ++i0) {{")
                    e_ind += 2
                L.append(f"{' '*e_ind}for (i64 te = {lo}; te < {hi}; "
                         f"++te) {{")
                L.append(self.gen_lane_body(tag, "te", e_ind + 2))
                L.append(f"{' '*e_ind}}}")
                if yloop:
                    e_ind -= 2
                    L.append(f"{' '*e_ind}}}")
                L.append(f"{' '*ind}}}")
            L.append(f"{' '*ind}const i64 pe = a.lead < a.n{x} ? a.lead : "
                     f"a.n{x};")
            edge("if (a.lead > 0 && blockIdx.x == 0 && threadIdx.x == 0)",
                 "0", "pe", "_P")
            L.append(f"{' '*ind}const i64 tstart = a.n{x} > a.lead ? "
                     f"a.lead + (a.n{x} - a.lead) / {V} * {V} : a.n{x};")
            edge(f"if (vb == tstart && tstart < a.n{x})",
                 "tstart", f"a.n{x}", "_T")
        if nd >= 2 and yblock == 1:
            ind -= 2
            L.append(f"{' '*ind}}}")
        if nd >= 3:
            ind -= 2
            L.append(f"{' '*ind}}}")
        # reduction epilogue: wave shfl tree + LDS across 4 waves
        if nred:
            L.append("  {")
            L.append("    const int lane = threadIdx.x & 63;")
            L.append("    const int wid = threadIdx.x >> 6;")
            for spec in self.plan.reductions:
                ct = ctype(spec.dtype)
                an = f"acc_{spec.acc}"
                comb, _ = ir.REDUCTIONS[spec.kind]
                L.append(f"    for (int o = 32; o > 0; o >>= 1) {{")
                L.append(f"      {ct} other = {self.shfl_down(spec.dtype, an, 'o')};")
                L.append(f"      {an} = "
                         f"{self.comb_expr(comb, spec.dtype, an, 'other')};")
                L.append("    }")
                L.append(f"    __shared__ {ct} lds_{spec.acc}[4];")
                L.append(f"    if (lane == 0) lds_{spec.acc}[wid] = {an};")
            L.append("    __syncthreads();")
            L.append("    if (threadIdx.x == 0) {")
            L.append("      i64 pidx = (i64)blockIdx.x + (i64)gridDim.x * "
                     "((i64)blockIdx.y + (i64)gridDim.y * blockIdx.z);")
            for ri, spec in enumerate(self.plan.reductions):
                ct = ctype(spec.dtype)
                comb, _ = ir.REDUCTIONS[spec.kind]
                L.append(f"      {ct} v{ri} = lds_{spec.acc}[0];")
                for w in range(1, 4):
                    L.append(f"      v{ri} = "
                             f"{self.comb_expr(comb, spec.dtype, f'v{ri}', f'lds_{spec.acc}[{w}]')};")
                L.append(f"      *({ct}*)(a.partials + (i64){ri} * "
                         f"a.npartials * 8 + pidx * 8) = v{ri};")
            L.append("    }")
            L.append("  }")
        L.append("}")

        kname_finish = None
        finish_source = None
        if nred:
            kname_finish = f"kf_{key}"
            finish_source = PREAMBLE + "\n" + self.gen_finish(kname_finish)
        return "\n".join(L), kname, kname_finish, finish_source, fields

    def vec_type(self, dt):
        ct = ctype(dt)
        V = self.vec
        m = {("double", 2): "d2_t", ("float", 4): "f4_t",
             ("long long", 2): "l2_t", ("float", 2): "f2_t",
             ("int", 2): "i2_t", ("int", 4): "i4_t",
             ("short", 2): "s2_t", ("short", 4): "s4_t",
             ("unsigned char", 2): "b2_t", ("unsigned char", 4): "b4_t",
             ("signed char", 2): "c2_t", ("signed char", 4): "c4_t"}
        vt = m.get((ct, V))
        assert vt is not None, f"no vector type for {ct} x{V}"
        return vt

    def shfl_down(self, dtype, var, off):
        if np.dtype(dtype).itemsize < 4:
            ct = ctype(dtype)
            return f"({ct})__shfl_down((int)({var}), {off}, 64)"
        return f"__shfl_down({var}, {off}, 64)"

    def init_literal(self, spec):
        v = ir.reduction_init(spec.kind, spec.dtype)
        ct = ctype(spec.dtype)
        if isinstance(v, float) and np.isinf(v):
            return ("-__builtin_inf()" if v < 0 else "__builtin_inf()") \
                if ct == "double" else \
                ("-__builtin_inff()" if v < 0 else "__builtin_inff()")
        if isinstance(v, bool):
            return "1" if v else "0"
        return f"({ct})({v!r})"

    def gen_finish(self, kname):
        """One-block finish kernel folding the per-workgroup partials into
        per-rank outputs (device scalars)."""
        L = []
        L.append("struct FArgs { char* partials; i64 npartials; "
                 + " ".join(f"char* out{i};" for i in
                            range(len(self.plan.reductions)))
                 + " };")
        L.append(f'extern "C" __global__ void __launch_bounds__(256) '
                 f"{kname}(FArgs f) {{")
        L.append("  const int lane = threadIdx.x & 63;")
        L.append("  const int wid = threadIdx.x >> 6;")
        for ri, spec in enumerate(self.plan.reductions):
            ct = ctype(spec.dtype)
            comb, _ = ir.REDUCTIONS[spec.kind]
            init = self.init_literal(spec)
            L.append(f"  {{")
            L.append(f"    {ct} acc = {init};")
            L.append(f"    for (i64 i = threadIdx.x; i < f.npartials; "
                     f"i += 256) {{")
            L.append(f"      {ct} pv = *({ct}*)(f.partials + (i64){ri} * "
                     f"f.npartials * 8 + i * 8);")
            L.append(f"      acc = {self.comb_expr(comb, spec.dtype, 'acc', 'pv')};")
            L.append("    }")
            L.append("    for (int o = 32; o > 0; o >>= 1) {")
            L.append(f"      {ct} other = {self.shfl_down(spec.dtype, 'acc', 'o')};")
            L.append(f"      acc = {self.comb_expr(comb, spec.dtype, 'acc', 'other')};")
            L.append("    }")
            L.append(f"    __shared__ {ct} lds[4];")
            L.append("    if (lane == 0) lds[wid] = acc;")
            L.append("    __syncthreads();")
            L.append("    if (threadIdx.x == 0) {")
            L.append(f"      {ct} v = lds[0];")
            for w in range(1, 4):
                L.append(f"      v = {self.comb_expr(comb, spec.dtype, 'v', f'lds[{w}]')};")
            L.append(f"      *({ct}*)f.out{ri} = v;")
            L.append("    }")
            L.append("    __syncthreads();")
            L.append("  }")
        L.append("}")
        return "\n".join(L)


def decide_vec(plan):
    """Vector width from the widest operand dtype (16 B per lane target)."""
    sizes = [np.dtype(o.dtype).itemsize for o in plan.operands] or [8]
    max_es = max(sizes)
    return 2 if max_es == 8 else (4 if max_es == 4 else 1)


def generate(plan):
    """plan -> GeneratedKernel (source + packing recipe)."""
    nd = len(plan.itershape)
    if nd < 1 or nd > 4:
        raise NotImplementedError(f"{nd}-d iteration spaces (shardview is "
                                  "<=4-D, SURVEY §8 a8)")
    vec, classes, anchor, key = plan_structure(plan)
    gen = KernelGen(plan, classes, vec)
    source, kmain, kfinish, finish_source, fields = gen.generate(key)
    gk = GeneratedKernel(key, source, kmain, kfinish, finish_source,
                         fields, vec, nd, len(plan.reductions))
    gk.anchor = anchor
    gk.yblock = getattr(gen, "yblock", 1)
    return gk


def pack_args(gk, plan, ptr_of):
    """Pack the Args struct per gk.fields (one struct.pack call; the
    format string is cached on the kernel).  ptr_of(name) -> device
    address of an operand buffer; special names '__partials__'."""
    opmap = {o.name: o for o in plan.operands}
    lead = 0
    if gk.anchor is not None and gk.vec > 1:
        lead = (-opmap[gk.anchor].offset0) % gk.vec
    fmt = getattr(gk, "_pack_fmt", None)
    if fmt is None:
        codes = {"iter_n": "q", "iter_gs": "q", "ptr": "Q", "off": "q",
                 "stride": "q", "scalar_f": "d", "scalar_i": "q",
                 "lead": "q", "partials": "Q", "npartials": "q"}
        fmt = "<" + "".join(codes[k] for (k, _) in gk.fields)
        gk._pack_fmt = fmt
    vals = []
    ap = vals.append
    scalars = plan.scalars
    for kind, payload in gk.fields:
        if kind == "iter_n":
            ap(plan.itershape[payload])
        elif kind == "iter_gs":
            ap(plan.global_start[payload])
        elif kind == "ptr":
            ap(ptr_of(payload))
        elif kind == "off":
            ap(opmap[payload].offset0)
        elif kind == "stride":
            name, d = payload
            ap(opmap[name].strides[d])
        elif kind == "scalar_f":
            ap(float(scalars[payload][0]))
        elif kind == "scalar_i":
            ap(int(scalars[payload][0]))
        elif kind == "lead":
            ap(lead)
        elif kind == "partials":
            ap(ptr_of("__partials__"))
        elif kind == "npartials":
            ap(ptr_of("__npartials__"))
        else:
            raise AssertionError(kind)
    return struct.pack(fmt, *vals)


def pack_finish_args(partials_ptr, npartials, out_ptrs):
    out = bytearray()
    out += struct.pack("<Q", partials_ptr)
    out += struct.pack("<q", npartials)
    for p in out_ptrs:
        out += struct.pack("<Q", p)
    return bytes(out)


# ---------------------------------------------------------------------------
# axis-reduction kernels (sum/prod/min/max/any/all along
# axes; replaces the reference's axis_reduce loops, ramba/ramba.py:8231-8244)
# ---------------------------------------------------------------------------

def generate_axis_reduce(nd, axes, in_dtype, out_dtype, kind,
                         chunked=False):
    """Local phase: reduce a strided view over `axes` into a contiguous
    partial buffer covering the local out box.

    Two shapes:
    - the innermost view axis is kept: lanes map to contiguous out elements
      (coalesced reads), reduced axes become serial inner loops;
    - the innermost view axis is reduced: one 64-lane wave per out element,
      lanes split the innermost reduction (coalesced), `__shfl_down` tree
      combines (the wave idiom of the CDNA4 guide, Appendix B).

    Args struct (packed by pack_axis_reduce_args):
      i64 oe{j} for each out axis (out extents, in axis order)
      i64 ke{j} for each reduced axis (extents)
      T* in; i64 in_off; i64 in_s{d} for every view axis d
      O* out  (contiguous over out extents, row-major)
    """
    axes = tuple(sorted(axes))
    out_axes = [d for d in range(nd) if d not in axes]
    it = ctype(in_dtype)
    ot = ctype(out_dtype)
    comb, _ = ir.REDUCTIONS[kind]
    lane_split = (nd - 1) in axes
    assert not (chunked and (lane_split or len(axes) != 1))

    def combc(a, b):
        if comb == "add":
            return f"({a}) + ({b})"
        if comb == "mul":
            return f"({a}) * ({b})"
        if comb in ("minimum", "maximum"):
            lt = "<" if comb == "minimum" else ">"
            if np.dtype(out_dtype).kind == "f":
                return (f"(__builtin_isnan((double)({a})) || (({a}) {lt} "
                        f"({b}))) ? ({ot})({a}) : ({ot})({b})")
            return f"(({a}) {lt} ({b})) ? ({ot})({a}) : ({ot})({b})"
        if comb == "logical_and":
            return f"((({a}) != 0) && (({b}) != 0)) ? 1 : 0"
        return f"((({a}) != 0) || (({b}) != 0)) ? 1 : 0"

    init = _axinit(kind, out_dtype)
    L = [PREAMBLE]
    L.append("struct AxArgs {")
    fields = []
    for j, d in enumerate(out_axes):
        L.append(f"  i64 oe{j};")
        fields.append(("oe", d))
    for j, d in enumerate(axes):
        L.append(f"  i64 ke{j};")
        fields.append(("ke", d))
    L.append(f"  {it}* __restrict__ in;")
    fields.append(("in_ptr", None))
    L.append("  i64 in_off;")
    fields.append(("in_off", None))
    for d in range(nd):
        L.append(f"  i64 in_s{d};")
        fields.append(("in_s", d))
    L.append(f"  {ot}* __restrict__ out;")
    fields.append(("out_ptr", None))
    if chunked:
        # split the (single) reduced axis into nchunk ragged chunks of
        # clen; chunk is an extra slowest OUT axis (parallelism for
        # small-nout reductions like sum(axis=0))
        L.append("  i64 nchunk; i64 clen; i64 ktot;")
        fields.append(("chunk", None))
    L.append("};")

    key = hashlib.sha256(
        f"axred:{nd}:{axes}:{in_dtype}:{out_dtype}:{kind}:{chunked}"
        .encode()).hexdigest()[:20]
    kname = f"ax_{key}"
    L.append(f'extern "C" __global__ void __launch_bounds__(256) '
             f"{kname}(AxArgs a) {{")
    # total out elements and per-thread mapping
    tot = " * ".join([f"a.oe{j}" for j in range(len(out_axes))]) or "1"
    if chunked:
        tot = f"({tot}) * a.nchunk"
    L.append(f"  const i64 nout = {tot};")
    if lane_split:
        L.append("  const int lane = threadIdx.x & 63;")
        L.append("  i64 w = ((i64)blockIdx.x * 256 + threadIdx.x) >> 6;")
        L.append("  const i64 ws = ((i64)gridDim.x * 256) >> 6;")
        L.append("  for (; w < nout; w += ws) {")
        L.append("    i64 rem = w;")
    else:
        L.append("  i64 o = (i64)blockIdx.x * 256 + threadIdx.x;")
        L.append("  const i64 os = (i64)gridDim.x * 256;")
        L.append("  for (; o < nout; o += os) {")
        L.append("    i64 rem = o;")
    # decompose out index (row-major over out extents; chunk slowest)
    L.append("    i64 base = a.in_off;")
    for j in range(len(out_axes) - 1, -1, -1):
        d = out_axes[j]
        L.append(f"    {{ i64 ix = rem % a.oe{j}; rem /= a.oe{j}; "
                 f"base += ix * a.in_s{d}; }}")
    if chunked:
        L.append("    const i64 chunk = rem;")
    L.append(f"    {ot} acc = {init};")
    # reduction loops
    if lane_split:
        inner = axes[-1]
        outer_red = axes[:-1]
        ind = "    "
        for j, d in enumerate(outer_red):
            L.append(f"{ind}for (i64 k{j} = 0; k{j} < a.ke{j}; ++k{j}) {{")
            ind += "  "
        ki = len(axes) - 1
        L.append(f"{ind}for (i64 kk = lane; kk < a.ke{ki}; kk += 64) {{")
        addr = "base" + "".join(
            f" + k{j} * a.in_s{d}" for j, d in enumerate(outer_red))
        if kind in ("all", "any"):
            L.append(f"{ind}  {ot} v = (a.in[{addr} + kk * a.in_s{inner}]"
                     f" != 0) ? 1 : 0;")
        else:
            L.append(f"{ind}  {ot} v = ({ot})a.in[{addr} + kk * "
                     f"a.in_s{inner}];")
        L.append(f"{ind}  acc = {combc('acc', 'v')};")
        L.append(f"{ind}}}")
        for j in range(len(outer_red)):
            ind = ind[:-2]
            L.append(f"{ind}}}")
        # wave tree
        L.append("    for (int off = 32; off > 0; off >>= 1) {")
        if np.dtype(out_dtype).itemsize < 4:
            L.append(f"      {ot} other = ({ot})__shfl_down((int)acc, off, 64);")
        else:
            L.append(f"      {ot} other = __shfl_down(acc, off, 64);")
        L.append(f"      acc = {combc('acc', 'other')};")
        L.append("    }")
        L.append("    if (lane == 0) a.out[w] = acc;")
    elif chunked:
        d = axes[0]
        L.append("    const i64 k0 = chunk * a.clen;")
        L.append("    const i64 k1 = k0 + a.clen < a.ktot ? k0 + a.clen "
                 ": a.ktot;")
        L.append("    for (i64 kk = k0; kk < k1; ++kk) {")
        if kind in ("all", "any"):
            L.append(f"      {ot} v = (a.in[base + kk * a.in_s{d}] != 0) "
                     "? 1 : 0;")
        else:
            L.append(f"      {ot} v = ({ot})a.in[base + kk * a.in_s{d}];")
        L.append(f"      acc = {combc('acc', 'v')};")
        L.append("    }")
        L.append("    a.out[o] = acc;")
    else:
        ind = "    "
        for j, d in enumerate(axes):
            L.append(f"{ind}for (i64 k{j} = 0; k{j} < a.ke{j}; ++k{j}) {{")
            ind += "  "
        addr = "base" + "".join(
            f" + k{j} * a.in_s{d}" for j, d in enumerate(axes))
        if kind in ("all", "any"):
            L.append(f"{ind}{ot} v = (a.in[{addr}] != 0) ? 1 : 0;")
        else:
            L.append(f"{ind}{ot} v = ({ot})a.in[{addr}];")
        L.append(f"{ind}acc = {combc('acc', 'v')};")
        for j in range(len(axes)):
            ind = ind[:-2]
            L.append(f"{ind}}}")
        L.append("    a.out[o] = acc;")
    L.append("  }")
    L.append("}")
    return key, "\n".join(L), kname, fields, lane_split


def _axinit(kind, dtype):
    v = ir.reduction_init(kind, dtype)
    ct = ctype(dtype)
    if isinstance(v, float) and np.isinf(v):
        return ("-__builtin_inf()" if v < 0 else "__builtin_inf()") \
            if ct == "double" else \
            ("-__builtin_inff()" if v < 0 else "__builtin_inff()")
    if isinstance(v, bool):
        return "1" if v else "0"
    return f"({ct})({v!r})"


def pack_axis_reduce_args(fields, out_extents_by_axis, red_extents_by_axis,
                          in_ptr, in_off, in_strides, out_ptr,
                          chunk_spec=None):
    out = bytearray()
    for kind, d in fields:
        if kind == "oe":
            out += struct.pack("<q", out_extents_by_axis[d])
        elif kind == "ke":
            out += struct.pack("<q", red_extents_by_axis[d])
        elif kind == "in_ptr":
            out += struct.pack("<Q", in_ptr)
        elif kind == "in_off":
            out += struct.pack("<q", in_off)
        elif kind == "in_s":
            out += struct.pack("<q", in_strides[d])
        elif kind == "out_ptr":
            out += struct.pack("<Q", out_ptr)
        elif kind == "chunk":
            nchunk, clen, ktot = chunk_spec
            out += struct.pack("<qqq", nchunk, clen, ktot)
    return bytes(out)


# ---------------------------------------------------------------------------
# staged/tiled kernel (cross-stage fusion, BASELINE configs[4]): one
# workgroup computes an index-pure producer stage over a tile+halo
# FOOTPRINT into LDS (recomputing halo values instead of exchanging
# them), stores the producer's live outputs for the footprint cells its
# containers back, then computes the consumer stage over the tile from
# LDS.  See ramba_amd/staged.py for the orchestration and the
# sequential-fallback contract.
# ---------------------------------------------------------------------------

TILE_H = 16
TILE_W = 64


class _StageGen:
    """Minimal `gen` shim for LaneEmitter: resolves Refs for one stage."""

    def __init__(self, stmts, resolve0, dead):
        self.resolve0 = resolve0      # fn(base_name) -> C expr (version 0)
        self.dead = dead
        self.ssa, self.final_version = ssa_statements(stmts, set())
        self.wanted = {}
        for (_, _, e) in self.ssa:
            collect_nodes(e, self.wanted)

    def lane_value(self, em, versioned, dtype):
        base, v = versioned.split("@")
        if int(v) == 0:
            r = self.resolve0(base, em)
            if r is None:
                raise AssertionError(f"unresolvable var {base}")
            return r
        return f"r_{base}_{v}{em.tag}"

    def emit_into(self, em):
        """Emit all statements; returns {base: final C value}."""
        for (tgt, _, e) in self.ssa:
            val = em.emit(e)
            base, v = tgt.split("@")
            ct = ctype(e.dtype)
            em.lines.append(f"      {ct} r_{base}_{v}{em.tag} = {val};")
        out = {}
        for base, v in self.final_version.items():
            out[base] = f"r_{base}_{v}{em.tag}"
        return out


def staged_tiled_key(desc):
    """Structural cache key WITHOUT source generation (the per-step hot
    path looks the kernel up by this before building anything)."""
    return hashlib.sha256(repr((
        [(st.target, st.expr) for st in desc["s1_stmts"]],
        [(st.target, st.expr) for st in desc["s2_stmts"]],
        desc["staged"], desc["s1_stores"], sorted(desc["readers"].items()),
        desc["s2_ops"], sorted((n, str(d)) for n, d in
                               desc["scalars"].items()),
        desc.get("tk_reds", []),
        desc["E0"], desc["E1"],
        os.environ.get("RAMBA_TK_TH", str(TILE_H)),
        os.environ.get("RAMBA_TK_CW", "128"),
        os.environ.get("RAMBA_TK_SEG", "4"))).encode()).hexdigest()[:24]


def generate_staged_tiled(desc):
    """desc (structural only; runtime values go through the fields):
      s1_stmts, s2_stmts : lists of ir.Assign (stage1 names p_-prefixed)
      staged : [(lds, dtype, live)]  # one per staged gid, emit order;
               `lds` is also the stage-1 var base name (p_<writer var>)
      s1_stores : [(var, dtype)]     # non-staged live stage-1 outputs
      readers : {stage2_var: (lds, dr0, dr1)}
      s2_ops : [(var, dtype, written)]
      scalars : {name: np.dtype}     # merged, stage-1 names p_-prefixed
      E0, E1 : global footprint extents (>= all reader deltas)
    returns (key, source, kname, fields): fields = ordered packing names.
    """
    E0, E1 = desc["E0"], desc["E1"]
    TH = int(os.environ.get("RAMBA_TK_TH", str(TILE_H)))
    TW = int(os.environ.get("RAMBA_TK_CW", "128"))
    SEG = int(os.environ.get("RAMBA_TK_SEG", "4"))
    RING = TH + E0
    FH, FW = RING, TW + E1

    key = staged_tiled_key(desc)
    kname = f"tk_{key}"

    fields = [("n0", "q"), ("n1", "q"), ("gs0", "q"), ("gs1", "q"),
              ("gb0", "q"), ("gb1", "q"), ("N0", "q"), ("N1", "q")]
    L = [PREAMBLE]
    L.append("struct TkArgs {")
    L.append("  i64 n0, n1;      // consumer exec-box extents")
    L.append("  i64 gs0, gs1;    // consumer global start (Iota)")
    L.append("  i64 gb0, gb1;    // base coord of footprint origin at k=0")
    L.append("  i64 N0, N1;      // producer-array global shape")
    for (lds, dt, live) in desc["staged"]:
        if live:
            for f in (f"{lds}_ptr", f"{lds}_off", f"{lds}_s0", f"{lds}_s1",
                      f"{lds}_lo0", f"{lds}_hi0", f"{lds}_lo1",
                      f"{lds}_hi1"):
                fields.append((f, "Q" if f.endswith("ptr") else "q"))
            L.append(f"  {ctype(dt)}* __restrict__ {lds}_ptr; "
                     f"i64 {lds}_off, {lds}_s0, {lds}_s1, "
                     f"{lds}_lo0, {lds}_hi0, {lds}_lo1, {lds}_hi1;")
    for (var, dt) in desc["s1_stores"]:
        for f in (f"{var}_ptr", f"{var}_off", f"{var}_s0", f"{var}_s1",
                  f"{var}_lo0", f"{var}_hi0", f"{var}_lo1", f"{var}_hi1"):
            fields.append((f, "Q" if f.endswith("ptr") else "q"))
        L.append(f"  {ctype(dt)}* __restrict__ {var}_ptr; "
                 f"i64 {var}_off, {var}_s0, {var}_s1, "
                 f"{var}_lo0, {var}_hi0, {var}_lo1, {var}_hi1;")
    for (var, dt, written) in desc["s2_ops"]:
        for f in (f"{var}_ptr", f"{var}_off", f"{var}_s0", f"{var}_s1"):
            fields.append((f, "Q" if f.endswith("ptr") else "q"))
        L.append(f"  {ctype(dt)}* __restrict__ {var}_ptr; "
                 f"i64 {var}_off, {var}_s0, {var}_s1;")
    for n in sorted(desc["scalars"]):
        dt = desc["scalars"][n]
        if np.dtype(dt).kind == "f":
            fields.append((n, "d"))
            L.append(f"  double {n};")
        else:
            fields.append((n, "q"))
            L.append(f"  i64 {n};")
    tk_reds = desc.get("tk_reds", [])
    for ri, (wvar, dt) in enumerate(tk_reds):
        L.append(f"  {ctype(dt)}* __restrict__ red{ri}_ptr;")
        fields.append((f"red{ri}_ptr", "Q"))
        # base-coordinate addressing of the reduction SOURCE array + up
        # to 4 rim boxes (core minus written image): the kernel folds
        # the unwritten rim cells into the same partials, so the fused
        # sum needs no extra launches at all
        L.append(f"  {ctype(dt)}* __restrict__ rsrc{ri}_ptr; "
                 f"i64 rsrc{ri}_off, rsrc{ri}_s0, rsrc{ri}_s1, "
                 f"rim{ri}_n;")
        fields += [(f"rsrc{ri}_ptr", "Q"), (f"rsrc{ri}_off", "q"),
                   (f"rsrc{ri}_s0", "q"), (f"rsrc{ri}_s1", "q"),
                   (f"rim{ri}_n", "q")]
        for k in range(4):
            L.append(f"  i64 rim{ri}_{k}_lo0, rim{ri}_{k}_hi0, "
                     f"rim{ri}_{k}_lo1, rim{ri}_{k}_hi1;")
            fields += [(f"rim{ri}_{k}_lo0", "q"), (f"rim{ri}_{k}_hi0", "q"),
                       (f"rim{ri}_{k}_lo1", "q"), (f"rim{ri}_{k}_hi1", "q")]
    L.append("};")

    L.append(f'extern "C" __global__ void __launch_bounds__(256) '
             f"{kname}(TkArgs a) {{")
    for ri, (wvar, dt) in enumerate(tk_reds):
        L.append(f"  {ctype(dt)} red{ri}_acc = ({ctype(dt)})0;")
    # rolling LDS ring down column strips (same idea as the load-tiled
    # stencil): the producer recomputes only TH NEW rows per tile, the
    # E0 overlap rows stay in the ring — recompute amortises from
    # (TH+E0)/TH to ~1 + E0/(SEG*TH)
    for (lds, dt, live) in desc["staged"]:
        L.append(f"  __shared__ {ctype(dt)} lds_{lds}[{RING}]"
                 f"[{FW} + 1];")
    L.append(f"  const i64 strips = (a.n1 + {TW} - 1) / {TW};")
    L.append(f"  const i64 tiles0 = (a.n0 + {TH} - 1) / {TH};")
    L.append(f"  const i64 nseg = (tiles0 + {SEG} - 1) / {SEG};")
    L.append("  for (i64 work = blockIdx.x; work < strips * nseg; "
             "work += gridDim.x) {")
    L.append("    const i64 strip = work % strips;")
    L.append("    const i64 seg = work / strips;")
    L.append(f"    const i64 k1o = strip * {TW};")
    L.append(f"    const i64 t0 = seg * {SEG};")
    L.append(f"    i64 tmax_ = tiles0 - t0; "
             f"const int tmax = (int)(tmax_ < {SEG} ? tmax_ : {SEG});")
    L.append("    int base = 0;")
    L.append("    for (int t = 0; t < tmax; ++t) {")
    L.append(f"      const i64 k0o = (t0 + t) * {TH};")

    # ---- stage 1: compute NEW footprint rows into the ring ---------------
    s1gen = _StageGen(desc["s1_stmts"], lambda b_, em: None, desc["dead1"])
    L.append(f"      const int rlo = t == 0 ? 0 : {E0};")
    L.append(f"      for (int fi = threadIdx.x; fi < ({RING} - rlo) * {FW};"
             " fi += 256) {")
    L.append(f"        const int r = rlo + fi / {FW};")
    L.append(f"        const int f1 = fi % {FW};")
    L.append(f"        int slot = base + r; "
             f"if (slot >= {RING}) slot -= {RING};")
    L.append("        const i64 B0 = a.gb0 + k0o + r;")
    L.append("        const i64 B1 = a.gb1 + k1o + f1;")
    L.append("        if (B0 < 0 || B0 >= a.N0 || B1 < 0 || B1 >= a.N1) "
             "continue;")
    em1 = LaneEmitter(s1gen, "_s1", ["B0", "B1"])
    finals = s1gen.emit_into(em1)
    body = list(em1.lines)
    for (lds, dt, live) in desc["staged"]:
        body.append(f"        lds_{lds}[slot][f1] = {finals[lds]};")
        if live:
            body.append(
                f"        if (B0 >= a.{lds}_lo0 && B0 <= a.{lds}_hi0 && "
                f"B1 >= a.{lds}_lo1 && B1 <= a.{lds}_hi1)")
            body.append(
                f"          a.{lds}_ptr[a.{lds}_off + B0 * a.{lds}_s0 + "
                f"B1 * a.{lds}_s1] = {finals[lds]};")
    for (var, dt) in desc["s1_stores"]:
        body.append(
            f"        if (B0 >= a.{var}_lo0 && B0 <= a.{var}_hi0 && "
            f"B1 >= a.{var}_lo1 && B1 <= a.{var}_hi1)")
        body.append(
            f"          a.{var}_ptr[a.{var}_off + B0 * a.{var}_s0 + "
            f"B1 * a.{var}_s1] = {finals[var]};")
    L.extend(body)
    L.append("      }")
    L.append("      __syncthreads();")

    # ---- stage 2 ---------------------------------------------------------
    readers = desc["readers"]
    s2_op_names = {v for (v, _, _) in desc["s2_ops"]}
    dr0s = sorted({dr0 for (_, dr0, _) in readers.values()})

    def resolve2(base_name, em):
        if base_name in readers:
            lds, dr0, dr1 = readers[base_name]
            return (f"lds_{lds}[sl{dr0}{em.tag}]"
                    f"[q1{em.tag} + {dr1}]")
        if base_name in s2_op_names:
            return (f"a.{base_name}_ptr[a.{base_name}_off + k0{em.tag} * "
                    f"a.{base_name}_s0 + k1{em.tag} * a.{base_name}_s1]")
        return None

    s2gen = _StageGen(desc["s2_stmts"], resolve2, desc["dead2"])
    L.append(f"      for (int ti = threadIdx.x; ti < {TH} * {TW}; "
             "ti += 256) {")
    L.append(f"        const int q0_t2 = ti / {TW}, q1_t2 = ti % {TW};")
    L.append("        const i64 k0_t2 = k0o + q0_t2;")
    L.append("        const i64 k1_t2 = k1o + q1_t2;")
    L.append("        if (k0_t2 < a.n0 && k1_t2 < a.n1) {")
    for d in dr0s:
        L.append(f"        int sl{d}_t2 = base + q0_t2 + {d}; "
                 f"if (sl{d}_t2 >= {RING}) sl{d}_t2 -= {RING};")
    em2 = LaneEmitter(s2gen, "_t2",
                      ["(a.gs0 + k0_t2)", "(a.gs1 + k1_t2)"])
    finals2 = s2gen.emit_into(em2)
    L.extend(em2.lines)
    for (var, dt, written) in desc["s2_ops"]:
        if written:
            L.append(
                f"      a.{var}_ptr[a.{var}_off + k0_t2 * a.{var}_s0 + "
                f"k1_t2 * a.{var}_s1] = {finals2[var]};")
    for ri, (wvar, dt) in enumerate(tk_reds):
        L.append(f"      red{ri}_acc += ({ctype(dt)})({finals2[wvar]});")
    L.append("        }")
    L.append("      }")
    L.append("      __syncthreads();")
    L.append(f"      base += {TH}; if (base >= {RING}) base -= {RING};")
    L.append("    }")
    L.append("  }")
    if tk_reds:
        # rim cells (core minus written image) folded into the same
        # accumulators: grid-strided over the <=4 rim boxes
        for ri, (wvar, dt) in enumerate(tk_reds):
            ct = ctype(dt)
            for k in range(4):
                L.append(f"  if ({k} < a.rim{ri}_n) {{")
                L.append(f"    const i64 h0 = a.rim{ri}_{k}_hi0 - "
                         f"a.rim{ri}_{k}_lo0 + 1;")
                L.append(f"    const i64 h1 = a.rim{ri}_{k}_hi1 - "
                         f"a.rim{ri}_{k}_lo1 + 1;")
                L.append("    for (i64 i = (i64)blockIdx.x * 256 + "
                         "threadIdx.x; i < h0 * h1; "
                         "i += (i64)gridDim.x * 256) {")
                L.append(f"      const i64 r = a.rim{ri}_{k}_lo0 + i / h1;")
                L.append(f"      const i64 c = a.rim{ri}_{k}_lo1 + i % h1;")
                L.append(f"      red{ri}_acc += a.rsrc{ri}_ptr["
                         f"a.rsrc{ri}_off + r * a.rsrc{ri}_s0 + "
                         f"c * a.rsrc{ri}_s1];")
                L.append("    }")
                L.append("  }")
        L.append("  {")
        L.append("    const int lane = threadIdx.x & 63;")
        L.append("    const int wid = threadIdx.x >> 6;")
        for ri, (wvar, dt) in enumerate(tk_reds):
            ct = ctype(dt)
            L.append("    for (int o = 32; o > 0; o >>= 1)")
            L.append(f"      red{ri}_acc += "
                     f"__shfl_down(red{ri}_acc, o, 64);")
            L.append(f"    __shared__ {ct} lds_red{ri}[4];")
            L.append(f"    if (lane == 0) lds_red{ri}[wid] = red{ri}_acc;")
        L.append("    __syncthreads();")
        L.append("    if (threadIdx.x == 0) {")
        for ri, (wvar, dt) in enumerate(tk_reds):
            ct = ctype(dt)
            L.append(f"      {ct} t_red{ri} = lds_red{ri}[0] + "
                     f"lds_red{ri}[1] + lds_red{ri}[2] + lds_red{ri}[3];")
            L.append(f"      a.red{ri}_ptr[blockIdx.x] = t_red{ri};")
        L.append("    }")
        L.append("  }")
    L.append("}")
    return key, "\n".join(L), kname, fields


_TK_FMT_CACHE = {}


def pack_tk_args(fields, values):
    fid = id(fields)
    ent = _TK_FMT_CACHE.get(fid)
    if ent is None or ent[0] is not fields:
        fmt = "<" + "".join(k for (_, k) in fields)
        if len(_TK_FMT_CACHE) > 512:
            _TK_FMT_CACHE.clear()
        _TK_FMT_CACHE[fid] = ent = (fields, fmt)
    fmt = ent[1]
    return struct.pack(fmt, *[
        float(values[n]) if k == "d" else int(values[n])
        for (n, k) in fields])


# ---------------------------------------------------------------------------
# load-tiled stencil kernel: operands that read the
# SAME array at several small shifts ("stencil family") are staged
# through LDS — one cooperative tile+halo load replaces the per-element
# shifted reads, killing the 1.3-1.6x L2/HBM over-fetch the PMC showed
# at cache-resident sizes (profiles/r02_pmc_sten_*.csv).
# ---------------------------------------------------------------------------

LT_MAXD = 8


def find_stencil_families(plan):
    """Group container read operands by (buffer, strides); return
    (families, members) where families = [(anchor_off, s0, dtype,
    [(name, dr0, dr1)])] for groups of >=3 pure-shift readers with inner
    stride 1, or None if the plan does not qualify."""
    nd = len(plan.itershape)
    if nd != 2 or plan.reductions:
        return None
    written = {st.target for st in plan.statements}
    groups = {}
    for op in plan.operands:
        if op.kind != "container" or op.name in written:
            continue
        if len(op.strides) != 2 or op.strides[1] != 1 \
                or op.strides[0] <= 0:
            continue
        groups.setdefault((id(op.bd), op.strides), []).append(op)
    fams = []
    for (key, ops) in groups.items():
        if len(ops) < 3:
            continue
        s0 = ops[0].strides[0]
        base = min(o.offset0 for o in ops)
        mem = []
        ok = True
        for o in ops:
            d = o.offset0 - base
            dr0 = int(round(d / s0))
            dr1 = d - dr0 * s0
            if not (0 <= dr0 <= LT_MAXD and abs(dr1) <= LT_MAXD):
                ok = False
                break
            mem.append((o.name, dr0, dr1))
        if not ok:
            continue
        dr1_min = min(m[2] for m in mem)
        anchor = base + dr1_min
        mem = [(n, a, b - dr1_min) for (n, a, b) in mem]
        fams.append((anchor, s0, str(ops[0].dtype), mem, ops[0].name))
    return fams or None


LT_TH = 16            # tile rows (TH=32's 36 KB LDS halves occupancy: 2.24 vs 1.58 ms at 30000^2 — r02 sweep)
LT_TXCH = 64          # column chunks per tile (one per lane of a wave?)


def generate_load_tiled(plan, fams):
    """v3: 16 B/lane vectorised, LDS-ring rolling stencil.  Workgroups
    own (column-strip, row-segment) pairs of SEG consecutive tiles and
    march down the strip carrying the E0 overlap rows in an LDS ring
    (slot = input_row % (TH+E0)) — the row-halo re-read amortises from
    (TH+E0)/TH per tile to ~1+E0/(SEG*TH) per segment.  16x256 tiles,
    18 KB LDS (f32), 8 blocks/CU."""
    nd = 2
    items = [np.dtype(dt).itemsize for (a, s, dt, mem, rep) in fams]
    for op in plan.operands:
        if any(op.name == n for (a, s, dt, mem, rep) in fams
               for (n, _, _) in mem):
            continue
        items.append(np.dtype(op.dtype).itemsize)
    V = max(1, 16 // max(items))
    # adaptive tile geometry (same-box sweeps, profiles/README r02):
    # wider strips cut the per-strip boundary-line over-fetch
    # (~strips*128B per array row), shorter tiles keep LDS/occupancy;
    # SEG shrinks when the (strips x segments) grid would starve 256 CUs
    n0g, n1g = plan.itershape
    nch_d = 128 if n1g >= 8192 else 64
    th_d = 8 if nch_d == 128 else LT_TH
    TH = int(os.environ.get("RAMBA_LT_TH", str(th_d)))
    NCH = int(os.environ.get("RAMBA_LT_NCH", str(nch_d)))
    CW = NCH * V                     # tile output columns
    strips_d = -(-n1g // CW)
    tiles0_d = -(-n0g // TH)
    seg_d = 1
    for cand in (2, 4, 8):
        if -(-tiles0_d // cand) * strips_d >= 6144:
            seg_d = cand
    SEG = int(os.environ.get("RAMBA_LT_SEG", str(seg_d)))

    fam_members = {}
    fam_ext = []
    for fi, (anchor, s0, dt, mem, rep) in enumerate(fams):
        for (n, dr0, dr1) in mem:
            fam_members[n] = (fi, dr0, dr1)
        fam_ext.append((max(m[1] for m in mem), max(m[2] for m in mem)))
    E0g = max(e[0] for e in fam_ext)     # shared ring depth
    RING = TH + E0g

    other_ops = [op for op in plan.operands if op.name not in fam_members]
    written = {st.target for st in plan.statements}

    key = hashlib.sha256(repr((
        "loadtiled3", [(st.target, st.expr) for st in plan.statements],
        [(fi, dt, sorted(mem)) for fi, (a, s, dt, mem, rep)
         in enumerate(fams)], fam_ext,
        [(o.name, str(o.dtype), o.name in written, o.strides[1] == 0)
         for o in other_ops],
        sorted((n, str(dt)) for n, (v, dt) in plan.scalars.items()),
        sorted((n, str(dt)) for n, dt in plan.dead_vars.items()),
        TH, NCH, V, SEG)).encode()).hexdigest()[:24]
    kname = f"lt_{key}"

    fields = [("n0", "q"), ("n1", "q"), ("gs0", "q"), ("gs1", "q")]
    L = [PREAMBLE]
    L.append("struct LtArgs {")
    L.append("  i64 n0, n1, gs0, gs1;")
    for fi, (anchor, s0, dt, mem, rep) in enumerate(fams):
        L.append(f"  {ctype(dt)}* __restrict__ fam{fi}_ptr; "
                 f"i64 fam{fi}_off, fam{fi}_s0;")
        fields += [(f"fam{fi}_ptr", "Q"), (f"fam{fi}_off", "q"),
                   (f"fam{fi}_s0", "q")]
    for o in other_ops:
        L.append(f"  {ctype(o.dtype)}* __restrict__ {o.name}_ptr; "
                 f"i64 {o.name}_off, {o.name}_s0, {o.name}_s1;")
        fields += [(f"{o.name}_ptr", "Q"), (f"{o.name}_off", "q"),
                   (f"{o.name}_s0", "q"), (f"{o.name}_s1", "q")]
    for n in sorted(plan.scalars):
        dt = plan.scalars[n][1]
        if np.dtype(dt).kind == "f":
            fields.append((n, "d"))
            L.append(f"  double {n};")
        else:
            fields.append((n, "q"))
            L.append(f"  i64 {n};")
    L.append("};")

    L.append(f'extern "C" __global__ void __launch_bounds__(256) '
             f"{kname}(LtArgs a) {{")
    pitches = []
    for fi, (E0, E1) in enumerate(fam_ext):
        dt = fams[fi][2]
        FW = CW + E1
        pitch = -(-FW // V) * V + V
        pitches.append(pitch)
        L.append(f"  __shared__ {ctype(dt)} lds_f{fi}"
                 f"[{RING}][{pitch}];")
    L.append(f"  const i64 strips = (a.n1 + {CW} - 1) / {CW};")
    L.append(f"  const i64 tiles0 = (a.n0 + {TH} - 1) / {TH};")
    L.append(f"  const i64 nseg = (tiles0 + {SEG} - 1) / {SEG};")
    L.append("  const int tx = threadIdx.x & 63;")
    L.append("  const int ty = threadIdx.x >> 6;")
    L.append("  for (i64 work = blockIdx.x; work < strips * nseg; "
             "work += gridDim.x) {")
    L.append("    const i64 strip = work % strips;")
    L.append("    const i64 seg = work / strips;")
    L.append(f"    const i64 k1o = strip * {CW};")
    L.append(f"    const i64 t0 = seg * {SEG};")
    L.append(f"    i64 tmax_ = tiles0 - t0; "
             f"const int tmax = (int)(tmax_ < {SEG} ? tmax_ : {SEG});")
    L.append("    int base = 0;   // ring slot of input row k0o")
    L.append("    for (int t = 0; t < tmax; ++t) {")
    L.append(f"      const i64 k0o = (t0 + t) * {TH};")

    # ---- fill: first tile loads RING rows, later tiles TH new rows ------
    # new input rows for tile t: t==0 -> [k0o, k0o+RING), else
    # [k0o+E0g, k0o+TH+E0g); clamped to < n0 + E0g
    for fi, (E0, E1) in enumerate(fam_ext):
        dt = fams[fi][2]
        ct = ctype(dt)
        FW = CW + E1
        vct = {("double", 2): "d2_t", ("float", 4): "f4_t",
               ("long long", 2): "l2_t", ("int", 4): "i4_t",
               ("short", 4): "s4_t"}.get((ct, V))
        L.append(f"      {{ const int rlo = t == 0 ? 0 : {E0g};")
        L.append(f"        i64 rhi_ = a.n0 + {E0} - k0o; "
                 f"const int rhi = (int)(rhi_ < {RING} ? rhi_ : {RING});")
        L.append(f"        i64 fw_ = a.n1 - k1o + {E1}; "
                 f"const int fw = (int)(fw_ < {FW} ? fw_ : {FW});")
        L.append(f"        const int nch = (fw + {V} - 1) / {V};")
        L.append("        for (int r = rlo + ty; r < rhi; r += 4) {")
        L.append(f"          int slot = base + r; "
                 f"if (slot >= {RING}) slot -= {RING};")
        L.append(f"          const i64 rb = a.fam{fi}_off + (k0o + r) * "
                 f"a.fam{fi}_s0 + k1o;")
        L.append("          for (int c = tx; c < nch; c += 64) {")
        L.append(f"            const int col = c * {V};")
        if vct:
            L.append(f"            if (col + {V} <= fw) {{")
            L.append(f"              {vct} v; __builtin_memcpy(&v, "
                     f"&a.fam{fi}_ptr[rb + col], sizeof(v));")
            L.append(f"              __builtin_memcpy(&lds_f{fi}[slot]"
                     f"[col], &v, sizeof(v));")
            L.append("            } else {")
            L.append(f"              for (int j = 0; j < {V}; ++j) "
                     f"if (col + j < fw) lds_f{fi}[slot][col + j] = "
                     f"a.fam{fi}_ptr[rb + col + j];")
            L.append("            }")
        else:
            L.append(f"            for (int j = 0; j < {V}; ++j) "
                     f"if (col + j < fw) lds_f{fi}[slot][col + j] = "
                     f"a.fam{fi}_ptr[rb + col + j];")
        L.append("          }")
        L.append("        } }")
    L.append("      __syncthreads();")

    # ---- stage 2 ---------------------------------------------------------
    def mk_resolver(jexpr):
        def resolve0(base_name, em):
            if base_name in fam_members:
                fi, dr0, dr1 = fam_members[base_name]
                return (f"lds_f{fi}[sl{dr0}{em.tag}]"
                        f"[tx * {V} + {jexpr} + {dr1}]")
            for o in other_ops:
                if o.name == base_name:
                    return (f"a.{base_name}_ptr[a.{base_name}_off + k0_t * "
                            f"a.{base_name}_s0 + (k1_t + {jexpr}) * "
                            f"a.{base_name}_s1]")
            return None
        return resolve0

    dr0s = sorted({dr0 for (fi, dr0, dr1) in fam_members.values()})
    L.append(f"      for (int r2 = ty; r2 < {TH}; r2 += 4) {{")
    L.append("        const i64 k0_t = k0o + r2;")
    L.append("        if (k0_t >= a.n0) break;")
    L.append(f"        const i64 k1_t = k1o + tx * {V};")
    for dr0 in dr0s:
        L.append(f"        int sl{dr0}_jf = base + r2 + {dr0}; "
                 f"if (sl{dr0}_jf >= {RING}) sl{dr0}_jf -= {RING};")
    L.append(f"        const bool full = (k1_t + {V} <= a.n1);")
    L.append("        if (full) {")
    # per-j emissions share the slot vars: alias per tag
    store_vecs = {}
    for j in range(V):
        for dr0 in dr0s:
            L.append(f"        const int sl{dr0}_j{j} = sl{dr0}_jf;")
        genj = _StageGen(plan.statements, mk_resolver(str(j)),
                         plan.dead_vars)
        em = LaneEmitter(genj, f"_j{j}",
                         ["(a.gs0 + k0_t)", f"(a.gs1 + k1_t + {j})"])
        finals = genj.emit_into(em)
        L.extend(em.lines)
        for o in other_ops:
            if o.name in written:
                store_vecs.setdefault(o.name, []).append(finals[o.name])
    for o in other_ops:
        if o.name not in written:
            continue
        ct = ctype(o.dtype)
        vct = {("double", 2): "d2_t", ("float", 4): "f4_t",
               ("long long", 2): "l2_t", ("int", 4): "i4_t",
               ("short", 4): "s4_t", ("signed char", 4): "c4_t",
               ("unsigned char", 4): "b4_t"}.get((ct, V))
        vals = store_vecs[o.name]
        if vct and o.strides[1] == 1:
            L.append(f"        {{ {vct} sv; " + " ".join(
                f"sv[{j}] = {vals[j]};" for j in range(V)))
            L.append(f"          __builtin_memcpy(&a.{o.name}_ptr["
                     f"a.{o.name}_off + k0_t * a.{o.name}_s0 + k1_t], "
                     f"&sv, sizeof(sv)); }}")
        else:
            for j in range(V):
                L.append(f"        a.{o.name}_ptr[a.{o.name}_off + k0_t *"
                         f" a.{o.name}_s0 + (k1_t + {j}) * "
                         f"a.{o.name}_s1] = {vals[j]};")
    L.append("        } else {")
    L.append(f"          for (int j = 0; j < {V}; ++j) {{")
    L.append("            if (k1_t + j >= a.n1) break;")
    for dr0 in dr0s:
        L.append(f"            const int sl{dr0}_je = sl{dr0}_jf;")
    gene = _StageGen(plan.statements, mk_resolver("j"), plan.dead_vars)
    eme = LaneEmitter(gene, "_je",
                      ["(a.gs0 + k0_t)", "(a.gs1 + k1_t + j)"])
    finals_e = gene.emit_into(eme)
    L.extend(eme.lines)
    for o in other_ops:
        if o.name in written:
            L.append(f"            a.{o.name}_ptr[a.{o.name}_off + k0_t *"
                     f" a.{o.name}_s0 + (k1_t + j) * a.{o.name}_s1] = "
                     f"{finals_e[o.name]};")
    L.append("          }")
    L.append("        }")
    L.append("      }")
    L.append("      __syncthreads();")
    L.append(f"      base += {TH}; if (base >= {RING}) base -= {RING};")
    L.append("    }")
    L.append("  }")
    L.append("}")
    return key, "\n".join(L), kname, fields, (TH, CW, SEG)
