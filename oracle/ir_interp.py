"""NumPy interpreter for the ramba_amd fused-group IR — TEST INFRASTRUCTURE.

Restates the semantics of the reference's generated fused loop
(ramba/ramba.py:8246-8265, executed at 3768/3779) with whole-array NumPy
ufuncs.  Statement-at-a-time evaluation is equivalent to the reference's
per-element fused loop because the alias guards of add_op
(ramba/ramba.py:8434-8465), which ramba_amd mirrors, forbid any statement
from reading a shifted version of an array written earlier in the group.

Every node is evaluated in its annotated dtype — the same dtypes the HIP
code generator emits — so oracle and kernel agree by construction.
"""

import numpy as np

from ramba_amd import ir


def eval_expr(e, env, itershape, global_start):
    if isinstance(e, ir.Ref):
        return env[e.name]
    if isinstance(e, ir.ScalarArg):
        val, dt = env["__scalars__"][e.name]
        return np.asarray(val, dtype=dt)[()]
    if isinstance(e, ir.Const):
        return np.asarray(e.value, dtype=e.dtype)[()]
    if isinstance(e, ir.Iota):
        n = itershape[e.axis]
        idx = np.arange(global_start[e.axis], global_start[e.axis] + n,
                        dtype=np.int64)
        shape = [1] * len(itershape)
        shape[e.axis] = n
        return np.broadcast_to(idx.reshape(shape), itershape)
    if isinstance(e, ir.Bin):
        a = eval_expr(e.a, env, itershape, global_start)
        b = eval_expr(e.b, env, itershape, global_start)
        r = ir.BINOPS[e.op](a, b)
        return _as_dtype(r, e.dtype)
    if isinstance(e, ir.Un):
        a = eval_expr(e.a, env, itershape, global_start)
        r = ir.UNOPS[e.op](a)
        return _as_dtype(r, e.dtype)
    if isinstance(e, ir.Cast):
        a = eval_expr(e.a, env, itershape, global_start)
        return _as_dtype(a, e.dtype)
    if isinstance(e, ir.Where):
        c = eval_expr(e.c, env, itershape, global_start)
        a = eval_expr(e.a, env, itershape, global_start)
        b = eval_expr(e.b, env, itershape, global_start)
        return _as_dtype(np.where(c, a, b), e.dtype)
    raise TypeError(f"bad expr {e!r}")


def _as_dtype(x, dt):
    x = np.asarray(x)
    if x.dtype != dt:
        x = x.astype(dt)
    return x


def run_statements(plan, env):
    """Execute a KernelPlan's statements over numpy views in `env`;
    returns the reduction partials in spec order."""
    itershape = plan.itershape
    gs = plan.global_start
    acc_specs = {spec.acc: spec for spec in plan.reductions}
    acc_vals = {}
    for spec in plan.reductions:
        acc_vals[spec.acc] = np.asarray(
            ir.reduction_init(spec.kind, spec.dtype), dtype=spec.dtype)[()]

    for st in plan.statements:
        if st.target in acc_specs:
            spec = acc_specs[st.target]
            # pattern: acc = comb(acc, src)  (internal_reduction1 body,
            # ramba/ramba.py:5798-5807)
            assert isinstance(st.expr, ir.Bin) and \
                isinstance(st.expr.a, ir.Ref) and st.expr.a.name == st.target
            src = eval_expr(st.expr.b, env, itershape, gs)
            src = np.broadcast_to(src, itershape)
            comb = ir.BINOPS[st.expr.op]
            part = comb.reduce(src.reshape(-1)) if src.size else acc_vals[st.target]
            if src.size:
                part = comb(acc_vals[st.target], part)
            acc_vals[st.target] = np.asarray(part, dtype=spec.dtype)[()]
        else:
            val = eval_expr(st.expr, env, itershape, gs)
            tgt = env[st.target]
            tgt[...] = np.broadcast_to(val, tgt.shape)

    return [acc_vals[spec.acc] for spec in plan.reductions]
