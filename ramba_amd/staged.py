"""Fused LDS-tiled execution of a sealed producer/consumer group pair
(BASELINE configs[4] "fused-kernel": the sin producer fused into the
stencil reader through LDS staging, with halo values RECOMPUTED from
index arithmetic instead of exchanged).

The pair arrives from `deferred` (seal at alias check 1); every
validation failure returns False and the runtime executes the two groups
sequentially — the reference's own flush-at-alias order
(ramba.py:8434-8443) — so this module is a pure performance transform.
"""

import numpy as np

from . import deferred
from .shardview import box_shape, box_subtract

MAX_DELTA = 8        # per-axis LDS halo bound (elements)


def classify_readers(g2, staged_gids):
    """For each staged gid: the consumer vars reading it and their
    absolute base offsets.  Returns None if any read is not a pure
    bounded shift (steps 1, identity axis_map)."""
    readers = {}      # gid -> list[(var_name, (off0, off1))]
    nd = 2
    for name, oi in g2.arr_vars.items():
        if oi.bd.gid not in staged_gids:
            continue
        if oi.written:
            return None
        v = oi.view
        if v.ndim != nd or v.axis_map != (0, 1) or v.steps != (1, 1):
            return None
        readers.setdefault(oi.bd.gid, []).append(
            (name, (int(v.offset[0]), int(v.offset[1]))))
    return readers


def try_execute_tiled(rt, g1, g2):
    backend = rt.backend
    if len(g1.shape) != 2 or len(g2.shape) != 2:
        return False
    if g2.reductions:
        return False
    if not deferred.group_is_index_pure(g1):
        return False
    readers = classify_readers(g2, g2.staged_gids)
    if not readers:
        return False

    # staged writer vars (identity views in g1)
    writers = {}      # gid -> OperandInfo (g1's writer var)
    for name, oi in g1.arr_vars.items():
        if oi.bd.gid in readers:
            if not oi.view.is_identity_for(oi.bd.shape):
                return False
            writers[oi.bd.gid] = oi
    if set(writers) != set(readers):
        return False

    # per-gid anchor + deltas
    deltas = {}       # gid -> (a0, a1, d0, d1) anchor + extents
    for gid, rl in readers.items():
        o0 = [o[0] for (_, o) in rl]
        o1 = [o[1] for (_, o) in rl]
        a0, a1 = min(o0), min(o1)
        d0, d1 = max(o0) - a0, max(o1) - a1
        if d0 > MAX_DELTA or d1 > MAX_DELTA:
            return False
        deltas[gid] = (a0, a1, d0, d1)

    # liveness.  Consumer HBM reads of NON-staged producer outputs pin
    # them live; staged gids are recomputed in-kernel, so they stay live
    # only by their own references.
    g1_gids = {oi.bd.gid for oi in g1.arr_vars.values()}
    hbm_read_gids = set()
    for name, oi in g2.arr_vars.items():
        if oi.bd.gid in g1_gids and oi.bd.gid not in readers:
            # consumer reads a producer output from HBM at an identity
            # view only (anything else would race across tiles)
            if not oi.view.is_identity_for(oi.bd.shape):
                return False
            if not oi.written:
                hbm_read_gids.add(oi.bd.gid)
    live1, dead1 = deferred.compute_live_vars(
        g1, extra_live_gids=hbm_read_gids)
    live2, dead2 = deferred.compute_live_vars(g2)
    # keep consumer-read producer arrays alive past g1's deletes
    keep = [bd for bd in g1.delete_bds if bd.gid in hbm_read_gids]
    if keep:
        g1.delete_bds = [bd for bd in g1.delete_bds
                         if bd.gid not in hbm_read_gids]
        g2.delete_bds.extend(keep)

    # v1 restrictions on the consumer: all non-staged operands fully
    # local ON EVERY RANK (no halo exchange, no temps) — the staged path
    # never communicates, so this decision MUST be rank-uniform (a mixed
    # tiled/fallback split would deadlock: the fallback posts sends that
    # the tiled ranks never match).  Everything checked here is
    # replicated state (views, divisions), so all ranks agree.
    eb2 = g2.exec_boxes()
    ib2 = eb2[rt.rank]
    from .shardview import box_contains
    for r in range(rt.world):
        ibr = eb2[r]
        if ibr is None:
            continue
        for name, oi in live2.items():
            if oi.bd.gid in readers:
                continue
            need = oi.view.image_box(ibr)
            if need is None:
                continue
            core = rt.core_box(oi.bd, r)
            if core is None or not box_contains(core, need):
                return False
    # staged arrays: halo stores stay within the container (border ring)
    eb1 = g1.exec_boxes()
    ib1 = eb1[rt.rank]
    for gid, (a0, a1, d0, d1) in deltas.items():
        bd = writers[gid].bd
        if max(d0, d1) > bd.border:
            return False

    # global anchor/footprint over ALL readers (all gids share one
    # footprint; per-reader LDS delta = offset - F_lo)
    all_off0 = [o[0] for rl in readers.values() for (_, o) in rl]
    all_off1 = [o[1] for rl in readers.values() for (_, o) in rl]
    F_lo = (min(all_off0), min(all_off1))
    F_hi = (max(all_off0), max(all_off1))
    E = (F_hi[0] - F_lo[0], F_hi[1] - F_lo[1])
    if E[0] > MAX_DELTA or E[1] > MAX_DELTA:
        return False

    return _execute_tiled(rt, g1, g2, live1, dead1, live2, dead2,
                          writers, readers, F_lo, E, ib1, ib2)


def _rename1(e):
    from . import ir
    if isinstance(e, ir.Ref):
        return ir.Ref("p_" + e.name, e.dtype)
    if isinstance(e, ir.ScalarArg):
        return ir.ScalarArg("p_" + e.name, e.dtype)
    if isinstance(e, ir.Bin):
        return ir.Bin(e.op, _rename1(e.a), _rename1(e.b), e.dtype)
    if isinstance(e, ir.Un):
        return ir.Un(e.op, _rename1(e.a), e.dtype)
    if isinstance(e, ir.Cast):
        return ir.Cast(_rename1(e.a), e.dtype)
    if isinstance(e, ir.Where):
        return ir.Where(_rename1(e.c), _rename1(e.a), _rename1(e.b),
                        e.dtype)
    return e


def _adopt_and_alloc(rt, g, live):
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


def _execute_tiled(rt, g1, g2, live1, dead1, live2, dead2, writers,
                   readers, F_lo, E, ib1, ib2):
    from . import codegen, ir
    from .runtime import KernelPlan
    backend = rt.backend

    _adopt_and_alloc(rt, g1, live1)
    _adopt_and_alloc(rt, g2, live2)

    # ---- structural descriptor -------------------------------------------
    s1_stmts = [ir.Assign("p_" + st.target, _rename1(st.expr))
                for st in g1.statements]
    staged_list = []        # (lds_name, dtype, live)
    lds_of_gid = {}
    for gid, oi in sorted(writers.items()):
        lds = "p_" + oi.name
        lds_of_gid[gid] = lds
        staged_list.append((lds, str(oi.bd.dtype), oi.name in live1))
    s1_stores = [("p_" + name, str(oi.dtype))
                 for name, oi in sorted(live1.items())
                 if oi.bd.gid not in writers]
    reader_map = {}
    for gid, rl in readers.items():
        for (vname, off) in rl:
            reader_map[vname] = (lds_of_gid[gid],
                                 off[0] - F_lo[0], off[1] - F_lo[1])
    s2_ops = [(name, str(oi.dtype), oi.written)
              for name, oi in sorted(live2.items())
              if oi.bd.gid not in readers]
    scalars = {("p_" + n): dt for n, (v, dt) in g1.scalars.items()}
    scalars.update({n: dt for n, (v, dt) in g2.scalars.items()})
    desc = {
        "s1_stmts": s1_stmts, "s2_stmts": g2.statements,
        "staged": staged_list, "s1_stores": s1_stores,
        "readers": reader_map, "s2_ops": s2_ops, "scalars": scalars,
        "dead1": {("p_" + n): oi.dtype for n, oi in dead1.items()},
        "dead2": {n: oi.dtype for n, oi in dead2.items()},
        "E0": E[0], "E1": E[1],
    }
    handle = backend.tiled_kernel(desc)
    if handle is None:
        return False

    # ---- runtime argument values -----------------------------------------
    launched = False
    if ib2 is not None:
        vals = {}
        n0, n1 = box_shape(ib2)
        vals["n0"], vals["n1"] = n0, n1
        vals["gs0"], vals["gs1"] = int(ib2[0, 0]), int(ib2[0, 1])
        vals["gb0"] = int(ib2[0, 0]) + F_lo[0]
        vals["gb1"] = int(ib2[0, 1]) + F_lo[1]
        vals["N0"], vals["N1"] = g1.shape
        ok = True
        for gid, oi in sorted(writers.items()):
            lds = lds_of_gid[gid]
            if oi.name not in live1:
                continue
            bd = oi.bd
            d, _, cs, pads = rt.shard_geometry(bd)
            if d is None:
                # no local shard: disable stores (empty bounds); the
                # owning ranks store their own cells.  Keeps the
                # tiled/fallback decision rank-uniform.
                for f in ("ptr", "off", "s0", "s1"):
                    vals[f"{lds}_{f}"] = 0
                vals[f"{lds}_lo0"] = vals[f"{lds}_lo1"] = 1
                vals[f"{lds}_hi0"] = vals[f"{lds}_hi1"] = 0
                continue
            vals[f"{lds}_ptr"] = backend.container_addr(bd)
            vals[f"{lds}_off"] = sum((pads[i] - int(d[0, i])) * cs[i]
                                     for i in range(2))
            vals[f"{lds}_s0"], vals[f"{lds}_s1"] = cs[0], cs[1]
            vals[f"{lds}_lo0"] = max(0, int(d[0, 0]) - bd.border)
            vals[f"{lds}_hi0"] = min(bd.shape[0] - 1,
                                     int(d[1, 0]) + bd.border)
            vals[f"{lds}_lo1"] = max(0, int(d[0, 1]) - bd.border)
            vals[f"{lds}_hi1"] = min(bd.shape[1] - 1,
                                     int(d[1, 1]) + bd.border)
        for name, oi in sorted(live1.items()):
            if oi.bd.gid in writers:
                continue
            var = "p_" + name
            bd = oi.bd
            d, _, cs, pads = rt.shard_geometry(bd)
            if d is None:
                for f in ("ptr", "off", "s0", "s1"):
                    vals[f"{var}_{f}"] = 0
                vals[f"{var}_lo0"] = vals[f"{var}_lo1"] = 1
                vals[f"{var}_hi0"] = vals[f"{var}_hi1"] = 0
                continue
            vals[f"{var}_ptr"] = backend.container_addr(bd)
            vals[f"{var}_off"] = sum((pads[i] - int(d[0, i])) * cs[i]
                                     for i in range(2))
            vals[f"{var}_s0"], vals[f"{var}_s1"] = cs[0], cs[1]
            vals[f"{var}_lo0"], vals[f"{var}_hi0"] = int(d[0, 0]), \
                int(d[1, 0])
            vals[f"{var}_lo1"], vals[f"{var}_hi1"] = int(d[0, 1]), \
                int(d[1, 1])
        for name, oi in sorted(live2.items()):
            if oi.bd.gid in readers:
                continue
            d, _, cs, pads = rt.shard_geometry(oi.bd)
            if d is None:
                ok = False
                break
            off, strides = oi.view.operand_addressing(ib2[0], cs, d[0],
                                                      pads)
            vals[f"{name}_ptr"] = backend.container_addr(oi.bd)
            vals[f"{name}_off"] = off
            vals[f"{name}_s0"], vals[f"{name}_s1"] = strides[0], strides[1]
        if not ok:
            return False
        for n, (v, dt) in g1.scalars.items():
            vals["p_" + n] = v
        for n, (v, dt) in g2.scalars.items():
            vals[n] = v
        ntiles = ((n0 + codegen.TILE_H - 1) // codegen.TILE_H) \
            * ((n1 + codegen.TILE_W - 1) // codegen.TILE_W)
        backend.tiled_launch(handle, vals, ntiles)
        launched = True

    # ---- residual stage-1 coverage ---------------------------------------
    if ib1 is not None:
        if ib2 is not None:
            covered = np.array(
                [[int(ib2[0, 0]) + F_lo[0], int(ib2[0, 1]) + F_lo[1]],
                 [int(ib2[1, 0]) + F_hi_ax(F_lo, E, 0),
                  int(ib2[1, 1]) + F_hi_ax(F_lo, E, 1)]], dtype=np.int64)
            residual = box_subtract(ib1, covered)
        else:
            residual = [ib1]
        if residual and live1:
            p = KernelPlan()
            p.scalars = dict(g1.scalars)
            p.statements = g1.statements
            p.reductions = []
            p.dead_vars = {n: oi.dtype for n, oi in dead1.items()}
            for box in residual:
                ap = rt._address_plan(p, live1, box, {})
                backend.launch(ap, None)

    # ---- deletes (both groups) -------------------------------------------
    backend.free_temps()
    for bd in g1.delete_bds + g2.delete_bds:
        if bd.constructed:
            backend.free_container(bd)
            bd.constructed = False
    return True


def F_hi_ax(F_lo, E, ax):
    return F_lo[ax] + E[ax]
