"""Execution half of the staged/tiled path (see staged.py for the
validation half and the fallback contract).  A StagedRecipe caches the
compiled kernel handle, name lists and residual launch plans so
iterating workloads skip descriptor construction and source-key hashing
after the first step (the staged analog of Runtime's recipe cache)."""

import numpy as np

from .shardview import box_shape, box_subtract


class StagedRecipe:
    __slots__ = ("handle", "writer_names", "store_names", "s2_names",
                 "F_lo", "E", "hbm_read_names", "residual_units",
                 "lds_of_writer")

    def __init__(self):
        self.handle = None
        self.writer_names = []      # g1 var names of staged writers
        self.store_names = []       # g1 var names of non-staged stores
        self.s2_names = []          # g2 var names of HBM operands
        self.F_lo = (0, 0)
        self.E = (0, 0)
        self.hbm_read_names = []    # g2 var names pinning g1 stores
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


def build_vals_and_launch(rt, rec, g1, g2, live1, live2, ib2):
    backend = rt.backend
    from . import codegen
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
    ntiles = ((n0 + codegen.TILE_H - 1) // codegen.TILE_H) \
        * ((n1 + codegen.TILE_W - 1) // codegen.TILE_W)
    backend.tiled_launch(rec.handle, vals, ntiles)


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
    if ib2 is not None:
        build_vals_and_launch(rt, rec, g1, g2, live1, live2, ib2)
    run_residual(rt, rec, g1, live1, dead1, ib1, ib2)
    backend.free_temps()
    for bd in g1.delete_bds + g2.delete_bds:
        if bd.constructed:
            backend.free_container(bd)
            bd.constructed = False
    return True
