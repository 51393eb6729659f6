"""Execution half of the staged/tiled path (see staged.py for the
validation half and the fallback contract).  A StagedRecipe caches the
compiled kernel handle, name lists and residual launch plans so
iterating workloads skip descriptor construction and source-key hashing
after the first step (the staged analog of Runtime's recipe cache)."""

import numpy as np

from .shardview import box_shape, box_subtract


class StagedRecipe:
    __slots__ = ("handle", "writer_names", "store_names", "s2_names",
                 "F_lo", "E", "residual_units", "lds_of_writer", "nred",
                 "vals", "vals_key", "partials")

    def __init__(self):
        self.nred = 0
        self.vals = None        # cached arg values (ptr+scalar validated)
        self.vals_key = None
        self.partials = None    # persistent reduction partials tensors
        self.handle = None
        self.writer_names = []      # g1 var names of staged writers
        self.store_names = []       # g1 var names of non-staged stores
        self.s2_names = []          # g2 var names of HBM operands
        self.F_lo = (0, 0)
        self.E = (0, 0)
        self.residual_units = None  # built lazily on first run
        self.lds_of_writer = {}


def adopt_and_alloc(rt, g, live):
    """Flex adoption + allocation (mirror of _build_and_run's prologue)."""
    eboxes = g.exec_boxes()
    nd = len(g.shape)
    adopted = None
    for oi in live.values():
        if oi.bd.is_flex and oi.bd.shape == g.shape:
            if adopted is None:
                adopted = np.zeros((rt.world, 2, nd), dtype=np.int64)
                for r, b in enumerate(eboxes):
                    if b is None:
                        adopted[r, 1, :] = -1
                    else:
                        adopted[r] = b
            oi.bd.divisions = adopted
            oi.bd.flex = False
    for oi in live.values():
        if not oi.bd.constructed:
            rt.backend.alloc_container(oi.bd, rt)
            oi.bd.constructed = True


def _keep_hbm_reads(g1, g2, hbm_read_gids):
    """Consumer-read producer arrays must outlive g1's deletes."""
    keep = [bd for bd in g1.delete_bds if bd.gid in hbm_read_gids]
    if keep:
        g1.delete_bds = [bd for bd in g1.delete_bds
                         if bd.gid not in hbm_read_gids]
        g2.delete_bds.extend(keep)


def _store_vals(vals, prefix, rt, bd):
    """ptr/off/strides + store-bounds fields for a stage-1 output
    addressed by BASE coordinates; `bounds` = (lo0,hi0,lo1,hi1)."""
    d, _, cs, pads = rt.shard_geometry(bd)
    if d is None:
        # no local shard: disable stores (empty bounds); the owning
        # ranks store their own cells.  Rank-uniform control flow.
        for f in ("ptr", "off", "s0", "s1"):
            vals[f"{prefix}_{f}"] = 0
        vals[f"{prefix}_lo0"] = vals[f"{prefix}_lo1"] = 1
        vals[f"{prefix}_hi0"] = vals[f"{prefix}_hi1"] = 0
        return None
    vals[f"{prefix}_ptr"] = rt.backend.container_addr(bd)
    vals[f"{prefix}_off"] = sum((pads[i] - int(d[0, i])) * cs[i]
                                for i in range(2))
    vals[f"{prefix}_s0"], vals[f"{prefix}_s1"] = cs[0], cs[1]
    return d


def _vals_probe(rt, rec, g1, g2, live1, live2):
    """Cheap validity key for the cached vals: every container address
    that appears in them + every scalar value (iterating workloads keep
    both stable; the staged fuzzer's per-iteration scalars invalidate)."""
    backend = rt.backend
    ptrs = []
    for wname in rec.writer_names:
        if wname in live1:
            ptrs.append(backend.container_addr(g1.arr_vars[wname].bd))
    for name in rec.store_names:
        ptrs.append(backend.container_addr(live1[name].bd))
    for name in rec.s2_names:
        ptrs.append(backend.container_addr(live2[name].bd))
    for (src, wv, pend) in g2.staged_reductions:
        ptrs.append(backend.container_addr(src.bdarray))
    scal = (tuple(v for (v, dt) in g1.scalars.values())
            + tuple(v for (v, dt) in g2.scalars.values()))
    return (tuple(ptrs), scal)


def build_vals_and_launch(rt, rec, g1, g2, live1, live2, ib2):
    backend = rt.backend
    from . import codegen
    probe = None
    try:
        probe = _vals_probe(rt, rec, g1, g2, live1, live2)
    except Exception:  # noqa: BLE001  (missing shard etc: rebuild)
        probe = None
    if (probe is not None and rec.vals is not None
            and rec.vals_key == probe):
        return _tk_launch(rt, rec, g2, rec.vals, ib2)
    vals = {}
    n0, n1 = box_shape(ib2)
    vals["n0"], vals["n1"] = n0, n1
    vals["gs0"], vals["gs1"] = int(ib2[0, 0]), int(ib2[0, 1])
    vals["gb0"] = int(ib2[0, 0]) + rec.F_lo[0]
    vals["gb1"] = int(ib2[0, 1]) + rec.F_lo[1]
    vals["N0"], vals["N1"] = g1.shape
    for wname in rec.writer_names:
        oi = g1.arr_vars[wname]
        lds = rec.lds_of_writer[wname]
        if wname not in live1:
            continue
        bd = oi.bd
        d = _store_vals(vals, lds, rt, bd)
        if d is not None:
            # staged halo stores may land in the border ring (values are
            # index-pure, so overlapping ranks store identical bytes)
            vals[f"{lds}_lo0"] = max(0, int(d[0, 0]) - bd.border)
            vals[f"{lds}_hi0"] = min(bd.shape[0] - 1,
                                     int(d[1, 0]) + bd.border)
            vals[f"{lds}_lo1"] = max(0, int(d[0, 1]) - bd.border)
            vals[f"{lds}_hi1"] = min(bd.shape[1] - 1,
                                     int(d[1, 1]) + bd.border)
    for name in rec.store_names:
        oi = live1[name]
        d = _store_vals(vals, "p_" + name, rt, oi.bd)
        if d is not None:
            var = "p_" + name
            vals[f"{var}_lo0"], vals[f"{var}_hi0"] = int(d[0, 0]), \
                int(d[1, 0])
            vals[f"{var}_lo1"], vals[f"{var}_hi1"] = int(d[0, 1]), \
                int(d[1, 1])
    for name in rec.s2_names:
        oi = live2[name]
        d, _, cs, pads = rt.shard_geometry(oi.bd)
        assert d is not None   # guaranteed by the uniform locality check
        off, strides = oi.view.operand_addressing(ib2[0], cs, d[0], pads)
        vals[f"{name}_ptr"] = backend.container_addr(oi.bd)
        vals[f"{name}_off"] = off
        vals[f"{name}_s0"], vals[f"{name}_s1"] = strides[0], strides[1]
    for n, (v, dt) in g1.scalars.items():
        vals["p_" + n] = v
    for n, (v, dt) in g2.scalars.items():
        vals[n] = v
    # fused reductions: base-coordinate source addressing + rim boxes
    # (core minus written image) folded in-kernel
    for ri, (src, wv, pend) in enumerate(g2.staged_reductions):
        bd = src.bdarray
        d, _, cs, pads = rt.shard_geometry(bd)
        assert d is not None    # written on this rank => shard exists
        vals[f"rsrc{ri}_ptr"] = backend.container_addr(bd)
        vals[f"rsrc{ri}_off"] = sum((pads[i] - int(d[0, i])) * cs[i]
                                    for i in range(2))
        vals[f"rsrc{ri}_s0"], vals[f"rsrc{ri}_s1"] = cs[0], cs[1]
        core = rt.core_box(bd, rt.rank)
        wimg = wv.image_box(ib2)
        rims = box_subtract(core, wimg) if wimg is not None else [core]
        assert len(rims) <= 4
        vals[f"rim{ri}_n"] = len(rims)
        for k in range(4):
            if k < len(rims):
                b = rims[k]
                vals[f"rim{ri}_{k}_lo0"] = int(b[0, 0])
                vals[f"rim{ri}_{k}_hi0"] = int(b[1, 0])
                vals[f"rim{ri}_{k}_lo1"] = int(b[0, 1])
                vals[f"rim{ri}_{k}_hi1"] = int(b[1, 1])
            else:
                vals[f"rim{ri}_{k}_lo0"] = 1
                vals[f"rim{ri}_{k}_hi0"] = 0
                vals[f"rim{ri}_{k}_lo1"] = 1
                vals[f"rim{ri}_{k}_hi1"] = 0
    rec.vals, rec.vals_key = vals, probe
    return _tk_launch(rt, rec, g2, vals, ib2)


def _tk_launch(rt, rec, g2, vals, ib2):
    import os
    from . import codegen
    n0, n1 = box_shape(ib2)
    th = int(os.environ.get("RAMBA_TK_TH", str(codegen.TILE_H)))
    cw = int(os.environ.get("RAMBA_TK_CW", "128"))
    seg = int(os.environ.get("RAMBA_TK_SEG", "4"))
    tiles0 = (n0 + th - 1) // th
    ntiles = ((tiles0 + seg - 1) // seg) * ((n1 + cw - 1) // cw)
    return rt.backend.tiled_launch(
        rec.handle, vals, ntiles,
        [p.dtype for (_, _, p) in g2.staged_reductions], rec=rec)


def run_residual(rt, rec, g1, live1, dead1, ib1, ib2):
    """Stage-1 coverage of the cells the tiled footprints miss."""
    from .runtime import KernelPlan, LaunchUnit
    backend = rt.backend
    if ib1 is None or not live1:
        return
    if rec.residual_units is None:
        if ib2 is not None:
            covered = np.array(
                [[int(ib2[0, 0]) + rec.F_lo[0],
                  int(ib2[0, 1]) + rec.F_lo[1]],
                 [int(ib2[1, 0]) + rec.F_lo[0] + rec.E[0],
                  int(ib2[1, 1]) + rec.F_lo[1] + rec.E[1]]],
                dtype=np.int64)
            residual = box_subtract(ib1, covered)
        else:
            residual = [ib1]
        units = []
        p = KernelPlan()
        p.scalars = dict(g1.scalars)
        p.statements = g1.statements
        p.reductions = []
        p.dead_vars = {n: oi.dtype for n, oi in dead1.items()}
        for box in residual:
            units.append(LaunchUnit(rt._address_plan(p, live1, box, {})))
        rec.residual_units = units
    for u in rec.residual_units:
        u.plan.scalars = dict(g1.scalars)
        u.plan.statements = g1.statements
        for op in u.plan.operands:
            op.bd = live1[op.name].bd
        backend.launch(u.plan, u)


def run_recipe(rt, rec, g1, g2, live1, dead1, live2, hbm_read_gids):
    """Fast path: re-run a cached StagedRecipe against fresh groups."""
    backend = rt.backend
    _keep_hbm_reads(g1, g2, hbm_read_gids)
    adopt_and_alloc(rt, g1, live1)
    adopt_and_alloc(rt, g2, live2)
    ib1 = g1.exec_boxes()[rt.rank]
    ib2 = g2.exec_boxes()[rt.rank]
    interior = None
    if ib2 is not None:
        interior = build_vals_and_launch(rt, rec, g1, g2, live1, live2,
                                         ib2)
    run_residual(rt, rec, g1, live1, dead1, ib1, ib2)
    if g2.staged_reductions:
        if ib2 is not None:
            # the tiled kernel already folded the rim: partial = interior
            for i, (src, wv, pend) in enumerate(g2.staged_reductions):
                pend.partial = np.asarray(interior[i],
                                          dtype=pend.dtype)[()]
        else:
            # no local tiles: reduce this rank's whole core (if any)
            finish_staged_reductions(rt, g2)
    backend.free_temps()
    for bd in g1.delete_bds + g2.delete_bds:
        if bd.constructed:
            backend.free_container(bd)
            bd.constructed = False
    return True


# -- staged reductions (sum(A) fused into the pair) --------------------------


def reduce_boxes_partial(rt, bd, boxes, kind, dtype):
    """Partial reduction of identity-view `bd` over base-space `boxes`
    on this rank, via the standard fused-reduce kernel (one launch per
    box; plan cached by the backend's structural key)."""
    from . import ir
    from .runtime import KernelPlan, OperandPlan
    backend = rt.backend
    acc = np.asarray(ir.reduction_init(kind, dtype), dtype=dtype)[()]
    d, _, cs, pads = rt.shard_geometry(bd)
    if d is None:
        return acc
    comb, _ = ir.REDUCTIONS[kind]
    for box in boxes:
        shape = box_shape(box)
        if 0 in shape:
            continue
        p = KernelPlan()
        p.itershape = shape
        p.global_start = tuple(int(x) for x in box[0])
        p.scalars = {}
        src = ir.Ref("vR", bd.dtype)
        if np.dtype(bd.dtype) != np.dtype(dtype):
            src = ir.Cast(src, np.dtype(dtype))
        spec = ir.ReductionSpec("accR", kind, np.dtype(dtype), 0)
        p.statements = [ir.Assign(
            "accR", ir.Bin(comb, ir.Ref("accR", np.dtype(dtype)), src,
                           np.dtype(dtype)))]
        p.reductions = [spec]
        p.dead_vars = {}
        nd = len(bd.shape)
        off0 = sum((int(box[0, i]) - int(d[0, i]) + pads[i]) * cs[i]
                   for i in range(nd))
        p.operands = [OperandPlan("vR", "container", bd, None, off0,
                                  tuple(cs), bd.dtype)]
        vals = backend.launch(p, None)
        v = vals[0]
        acc = np.asarray(ir.BINOPS[comb](acc, v) if comb in ir.BINOPS
                         else acc + v, dtype=dtype)[()]
    return acc


def finish_staged_reductions(rt, g2):
    """Fallback/no-local-tiles path: each staged-reduction partial =
    the plain reduce over this rank's whole core (the tiled path instead
    folds interior+rim in-kernel and sets the partial directly)."""
    from . import ir
    for (src, wv, pend) in g2.staged_reductions:
        bd = src.bdarray
        core = rt.core_box(bd, rt.rank)
        if core is None:
            pend.partial = np.asarray(
                ir.reduction_init(pend.kind, pend.dtype),
                dtype=pend.dtype)[()]
            continue
        pend.partial = reduce_boxes_partial(rt, bd, [core], pend.kind,
                                            pend.dtype)
