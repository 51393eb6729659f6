"""SPMD runtime — the replacement for the reference's worker side
(`RemoteState.run_deferred_ops`, ramba/ramba.py:3493-3819).

One process per GPU (the analog of the reference's MPI-SPMD mode,
ramba.py:3986-3993).  Rank r owns shard r of every backing array, stored in a
container that covers its division box plus a border ring
(`LocalNdarray.bcontainer`, ramba.py:1193).  For each fused group the runtime:

1. partitions the iteration space (owner-computes over the output view's
   partition -- the reference's exec_dist subspace, ramba.py:3497),
2. plans, deterministically and identically on every rank, which sub-boxes
   must move between ranks (the overlap scan + comm_queues put/get sites,
   ramba.py:3547-3693, and getborder, ramba.py:1260) and moves them with
   torch.distributed P2P (RCCL over xGMI on GPUs, gloo in CPU tests),
3. asks the backend to run the fused kernel over the local box,
4. finishes reductions with one allreduce (replacing the reference's
   gather-to-driver, ramba.py:5745-5764/5852-5863).
"""

import os
import time

import numpy as np

from . import deferred, ir
from .common import add_time, default_border, default_divisions
from .shardview import (box_contains, box_intersect, box_shape,
                        box_subtract)


class OperandPlan:
    """How one live array var is addressed by the kernel on this rank."""
    __slots__ = ("name", "kind", "bd", "temp_key", "offset0", "strides",
                 "dtype")

    def __init__(self, name, kind, bd, temp_key, offset0, strides, dtype):
        self.name = name
        self.kind = kind            # "container" | "temp"
        self.bd = bd
        self.temp_key = temp_key
        self.offset0 = offset0
        self.strides = strides
        self.dtype = dtype


class KernelPlan:
    __slots__ = ("itershape", "global_start", "operands", "scalars",
                 "statements", "reductions", "dead_vars", "temps")

    def __init__(self):
        self.itershape = None
        self.global_start = None
        self.operands = []          # list[OperandPlan]
        self.scalars = {}           # name -> (value, dtype)
        self.statements = []
        self.reductions = []        # list[ir.ReductionSpec]
        self.dead_vars = {}         # name -> dtype (register temps)
        self.temps = {}             # temp_key -> (shape, dtype) buffers


class LaunchUnit:
    """One kernel launch over one sub-box of the rank's iteration space
    (interior or rim slab); carries its own cached kernel handle."""
    __slots__ = ("plan", "backend_kernel")

    def __init__(self, plan):
        self.plan = plan
        self.backend_kernel = None


class Recipe:
    """Cached execution geometry for one fused-group structure."""
    __slots__ = ("plan", "temp_specs", "comm_msgs", "temp_geom",
                 "adopted_divs", "backend_kernel", "units",
                 "pre_wait_units")

    def __init__(self):
        self.plan = None
        self.temp_specs = []
        self.comm_msgs = []
        self.temp_geom = {}
        self.adopted_divs = None
        self.backend_kernel = None
        # halo/compute overlap (BASELINE configs[4] "overlapped RCCL",
        # replacing the reference's serial exchange-then-execute,
        # ramba.py:3547-3693): `units` is the launch list; the first
        # `pre_wait_units` of them run between comms-post and comms-wait
        self.units = None
        self.pre_wait_units = 0


class Runtime:
    def __init__(self, backend, rank=0, world=1):
        self.backend = backend
        self.rank = rank
        self.world = world

    # ------------------------------------------------------------------
    def lo_pads(self, bd):
        """Leading pad per axis: the border ring, with the innermost axis
        padded up to a 128-byte boundary so every core row starts
        cacheline-aligned (misaligned store streams cost partial-line
        read-modify-write traffic on HBM — measured with
        tools/probe_streams)."""
        nd = len(bd.shape)
        align = max(1, 128 // bd.dtype.itemsize)
        pads = [bd.border] * nd
        # the last axis always gets at least one full granule of pad on
        # both sides: the rel-shifted vector reads of the code generator
        # (codegen.classify_plan) may touch up to vec-1 elements beyond a
        # view's range
        pads[nd - 1] = max(-(-bd.border // align) * align, align)
        return tuple(pads)

    def shard_geometry(self, bd, r=None):
        """(div box, container shape, container strides, lo pad per axis).

        Container per axis = lo_pad + core + hi_pad; the innermost axis's
        total padded extent is rounded up to the 128-B granule so OUTER
        axis strides stay line-aligned too."""
        r = self.rank if r is None else r
        d = bd.divisions[r]
        nd = len(bd.shape)
        pads = self.lo_pads(bd)
        if np.any(d[1] < d[0]):
            return None, None, None, pads
        align = max(1, 128 // bd.dtype.itemsize)
        cshape = []
        for i in range(nd):
            sz = int(d[1, i] - d[0, i] + 1) + pads[i] + bd.border
            if i == nd - 1:
                # round up + one extra granule of tail slack (rel-shifted
                # vector reads; keeps outer-axis strides line-aligned)
                sz = -(-sz // align) * align + align
            cshape.append(sz)
        cshape = tuple(cshape)
        cstrides = [1] * nd
        for i in range(nd - 2, -1, -1):
            cstrides[i] = cstrides[i + 1] * cshape[i + 1]
        return d, cshape, tuple(cstrides), pads

    def core_box(self, bd, r):
        d = bd.divisions[r]
        if np.any(d[1] < d[0]):
            return None
        return d.astype(np.int64)

    def border_box(self, bd, r):
        """core expanded by the border ring, clipped to the global array."""
        d = self.core_box(bd, r)
        if d is None:
            return None
        lo = np.maximum(d[0] - bd.border, 0)
        hi = np.minimum(d[1] + bd.border,
                        np.array(bd.shape, dtype=np.int64) - 1)
        return np.array([lo, hi])

    # ------------------------------------------------------------------
    # Execution-recipe cache: iterating workloads (the bench loops, stencil
    # sweeps) re-issue structurally identical fused groups every step; all
    # the box math, operand addressing and kernel lookup depend only on the
    # structure, so they are computed once and re-bound to the new backing
    # arrays on later steps.  The structural signature pins everything the
    # geometry depends on (views, divisions CONTENT, liveness, statements).
    # ------------------------------------------------------------------

    _divb_cache = {}

    @classmethod
    def _div_bytes(cls, divs):
        """divisions.tobytes() memoised by array identity (divisions are
        rebound, never mutated; the ref in the cache keeps the id valid)."""
        e = cls._divb_cache.get(id(divs))
        if e is not None and e[0] is divs:
            return e[1]
        b = divs.tobytes()
        if len(cls._divb_cache) > 8192:
            cls._divb_cache.clear()
        cls._divb_cache[id(divs)] = (divs, b)
        return b

    def _group_signature(self, group, live):
        parts = [group.shape, self.world,
                 ("part", group.part_view,
                  self._div_bytes(group.part_divs), group.flex)]
        for name, oi in group.arr_vars.items():
            parts.append((name, name in live, oi.bd.shape, oi.bd.dtype,
                          oi.bd.border, self._div_bytes(oi.bd.divisions),
                          oi.bd.is_flex, oi.bd.constructed, oi.view,
                          oi.written))
        parts.append(tuple(sorted((n, dt.name)
                                  for n, (v, dt) in group.scalars.items())))
        parts.append(tuple((s.acc, s.kind, s.dtype)
                           for s, _ in group.reductions))
        parts.append(tuple((st.target, st.expr) for st in group.statements))
        return tuple(parts)

    def execute_group(self, group):
        # timing accumulators: the analog of the reference's add_time
        # summaries (ramba/ramba.py:945-1022, RAMBA_TIMING)
        t0 = time.perf_counter()
        try:
            self._execute_group(group)
        finally:
            add_time("run_deferred_ops", time.perf_counter() - t0)

    def _execute_group(self, group):
        from .common import ntiming
        t0 = time.perf_counter() if ntiming else 0.0
        live, dead = deferred.compute_live_vars(group)
        cache = getattr(self, "_recipe_cache", None)
        if cache is None:
            cache = self._recipe_cache = {}
        try:
            sig = self._group_signature(group, live)
        except TypeError:
            sig = None
        if ntiming:
            add_time("eg_signature", time.perf_counter() - t0)
        recipe = cache.get(sig) if sig is not None else None
        if recipe is not None:
            self._run_recipe(recipe, group, live)
            return
        recipe = self._build_and_run(group, live, dead)
        if sig is not None and recipe is not None:
            if len(cache) > 128:
                cache.clear()
            cache[sig] = recipe

    def _run_recipe(self, recipe, group, live):
        """Re-run a cached recipe against the current group's backing
        arrays (pointers differ; geometry is identical by signature)."""
        # flex adoption + allocation
        for name, oi in live.items():
            if oi.bd.is_flex and oi.bd.shape == group.shape \
                    and recipe.adopted_divs is not None:
                oi.bd.divisions = recipe.adopted_divs
                oi.bd.flex = False
            if not oi.bd.constructed:
                self.backend.alloc_container(oi.bd, self)
                oi.bd.constructed = True
        # rebind operand backing arrays
        plan = recipe.plan
        plan.scalars = dict(group.scalars)
        for op in plan.operands:
            op.bd = live[op.name].bd
        # temps + local copies
        for (vname, shape, dtype, owner, need, part) in recipe.temp_specs:
            self.backend.alloc_temp(vname, shape, dtype)
            bd = live[owner].bd
            if part is not None and bd.constructed:
                self.backend.copy_container_to_temp(bd, self, part, vname,
                                                    need)
        state = None
        if recipe.comm_msgs:
            msgs = [(dst, src, live[owner].bd, bx, tgt)
                    for (dst, src, owner, bx, tgt) in recipe.comm_msgs]
            state = self._comms_begin(msgs, recipe.temp_geom)
        from .common import ntiming
        tl = time.perf_counter() if ntiming else 0.0
        if recipe.units is not None:
            # overlapped split: interior unit(s) launch while the
            # exchange is in flight, rim units after it lands
            for u in recipe.units:
                u.plan.scalars = plan.scalars
                for op in u.plan.operands:
                    op.bd = live[op.name].bd
            for i, u in enumerate(recipe.units):
                if i == recipe.pre_wait_units and state is not None:
                    self._comms_finish(state)
                    state = None
                self.backend.launch(u.plan, u)
            if state is not None:
                self._comms_finish(state)
                state = None
            partials = []
            if ntiming:
                add_time("eg_launch_host", time.perf_counter() - tl)
        else:
            if state is not None:
                self._comms_finish(state)
                state = None
            if plan.itershape is not None:
                partials = self.backend.launch(plan, recipe)
            else:
                partials = [np.asarray(
                    ir.reduction_init(spec.kind, spec.dtype),
                    dtype=spec.dtype)[()] for spec in plan.reductions]
            if ntiming:
                add_time("eg_launch_host", time.perf_counter() - tl)
        for (spec, pend), val in zip(group.reductions, partials):
            pend.partial = np.asarray(val, dtype=spec.dtype)[()]
        self.backend.free_temps()
        for bd in group.delete_bds:
            if bd.constructed:
                self.backend.free_container(bd)
                bd.constructed = False

    def _build_and_run(self, group, live, dead):
        eboxes = group.exec_boxes()
        recipe = Recipe()

        # flex arrays adopt the group's partition (ramba.py:8093-8101)
        nd = len(group.shape)
        adopted = None
        for oi in live.values():
            if oi.bd.is_flex and oi.bd.shape == group.shape:
                if adopted is None:
                    adopted = np.zeros((self.world, 2, nd), dtype=np.int64)
                    for r, b in enumerate(eboxes):
                        if b is None:
                            adopted[r, 1, :] = -1
                        else:
                            adopted[r] = b
                oi.bd.divisions = adopted
                oi.bd.flex = False
        recipe.adopted_divs = adopted

        # allocate + mark constructed (creation on first use, ramba.py:3506)
        for oi in live.values():
            if not oi.bd.constructed:
                self.backend.alloc_container(oi.bd, self)
                oi.bd.constructed = True

        plan = KernelPlan()
        plan.scalars = dict(group.scalars)
        plan.statements = group.statements
        plan.reductions = [s for (s, _) in group.reductions]
        plan.dead_vars = {n: oi.dtype for n, oi in dead.items()}

        # ---- per-gid needed boxes on every rank (deterministic everywhere)
        # needed[r] : gid -> (container_fill_box or None, {var: temp box})
        comm_msgs = []   # (dst, src, owner_var, box, target)
        my_plans = {}
        for r in range(self.world):
            ib = eboxes[r]
            per_gid = {}
            for gid, names in group.vars_by_gid.items():
                vars_here = [live[n] for n in names if n in live]
                if not vars_here:
                    continue
                bd = vars_here[0].bd
                owner = vars_here[0].name
                core = self.core_box(bd, r)
                bbox = self.border_box(bd, r)
                fit_boxes = []
                temp_vars = {}
                for oi in vars_here:
                    if ib is None:
                        need = None
                    else:
                        need = oi.view.image_box(ib)
                    if oi.written and need is not None:
                        assert core is not None and box_contains(core, need), \
                            "write outside owned shard (incompatible dists)"
                    if need is None:
                        continue
                    if core is not None and box_contains(core, need):
                        continue  # fully local
                    if bbox is not None and box_contains(bbox, need):
                        fit_boxes.append(need)
                    else:
                        temp_vars[oi.name] = need
                fill = None
                if fit_boxes:
                    lo = np.min([b[0] for b in fit_boxes], axis=0)
                    hi = np.max([b[1] for b in fit_boxes], axis=0)
                    fill = np.array([lo, hi])
                per_gid[gid] = (bd, owner, fill, temp_vars)
                # build transfer list
                if fill is not None:
                    missing = box_subtract(fill, core) if core is not None \
                        else [fill]
                    for mbox in missing:
                        for s in range(self.world):
                            if s == r:
                                continue
                            part = box_intersect(mbox, self.core_box(bd, s))
                            if part is not None:
                                comm_msgs.append(
                                    (r, s, owner, part, ("border", gid)))
                for vname, need in temp_vars.items():
                    for s in range(self.world):
                        part = box_intersect(need, self.core_box(bd, s))
                        if part is None:
                            continue
                        if s == r:
                            continue  # local part copied below
                        comm_msgs.append((r, s, owner, part, ("temp", vname)))
            if r == self.rank:
                my_plans = per_gid

        # ---- allocate temp operand buffers + copy local parts
        temp_geom = {}   # var -> (need box, strides)
        for gid, (bd, owner, fill, temp_vars) in my_plans.items():
            for vname, need in temp_vars.items():
                shape = box_shape(need)
                self.backend.alloc_temp(vname, shape, bd.dtype)
                strides = [1] * len(shape)
                for i in range(len(shape) - 2, -1, -1):
                    strides[i] = strides[i + 1] * shape[i + 1]
                temp_geom[vname] = (need, tuple(strides))
                core = self.core_box(bd, self.rank)
                part = box_intersect(need, core)
                recipe.temp_specs.append(
                    (vname, shape, bd.dtype, owner, need, part))
                if part is not None and bd.constructed:
                    self.backend.copy_container_to_temp(bd, self, part,
                                                        vname, need)
        recipe.temp_geom = temp_geom
        recipe.comm_msgs = comm_msgs

        # ---- launches, with the exchange posted first and (when the
        # geometry allows) the interior launched while it is in flight
        ib = eboxes[self.rank]
        split_boxes, pre_wait = self._plan_overlap_split(
            ib, comm_msgs, plan, temp_geom, live)
        state = None
        if comm_msgs:
            msgs = [(dst, src, live[owner].bd, bx, tgt)
                    for (dst, src, owner, bx, tgt) in comm_msgs]
            state = self._comms_begin(msgs, temp_geom)
        if split_boxes is not None:
            recipe.units = [
                LaunchUnit(self._address_plan(plan, live, b, temp_geom))
                for b in split_boxes]
            recipe.pre_wait_units = pre_wait
            plan.itershape = None
            for i, u in enumerate(recipe.units):
                if i == pre_wait and state is not None:
                    self._comms_finish(state)
                    state = None
                self.backend.launch(u.plan, u)
            if state is not None:
                self._comms_finish(state)
                state = None
            partials = []
        else:
            if state is not None:
                self._comms_finish(state)
                state = None
            if ib is not None:
                ap = self._address_plan(plan, live, ib, temp_geom)
                plan.itershape = ap.itershape
                plan.global_start = ap.global_start
                plan.operands = ap.operands
                partials = self.backend.launch(plan, recipe)
            else:
                plan.itershape = None
                partials = [np.asarray(
                    ir.reduction_init(spec.kind, spec.dtype),
                    dtype=spec.dtype)[()] for spec in plan.reductions]
        recipe.plan = plan

        for (spec, pend), val in zip(group.reductions, partials):
            pend.partial = np.asarray(val, dtype=spec.dtype)[()]

        self.backend.free_temps()

        # ---- deferred frees (ramba.py:8321-8328)
        for bd in group.delete_bds:
            if bd.constructed:
                self.backend.free_container(bd)
                bd.constructed = False
        return recipe

    # ------------------------------------------------------------------
    def _address_plan(self, base_plan, live, ibox, temp_geom):
        """Operand descriptors for a launch over iteration box `ibox`
        (shares statements/scalars/dead_vars with the group's plan)."""
        p = KernelPlan()
        p.itershape = box_shape(ibox)
        p.global_start = tuple(int(x) for x in ibox[0])
        p.scalars = base_plan.scalars
        p.statements = base_plan.statements
        p.reductions = base_plan.reductions
        p.dead_vars = base_plan.dead_vars
        for name, oi in live.items():
            if name in temp_geom:
                need, strides = temp_geom[name]
                off0, s = oi.view.operand_addressing(
                    ibox[0], strides, need[0], (0,) * len(strides))
                p.operands.append(OperandPlan(
                    name, "temp", oi.bd, name, off0, s, oi.dtype))
            else:
                d, cshape, cstrides, border = self.shard_geometry(oi.bd)
                if d is None:
                    # no local shard; var must be unused here
                    need = oi.view.image_box(ibox)
                    assert need is None, "operand needed but unowned"
                    p.operands.append(OperandPlan(
                        name, "container", oi.bd, None, 0,
                        (0,) * len(p.itershape), oi.dtype))
                    continue
                off0, s = oi.view.operand_addressing(
                    ibox[0], cstrides, d[0], border)
                p.operands.append(OperandPlan(
                    name, "container", oi.bd, None, off0, s, oi.dtype))
        return p

    def _plan_overlap_split(self, ib, comm_msgs, plan, temp_geom, live):
        """(boxes, pre_wait) for overlapped halo/compute, or (None, 0).

        Interior = the subset of the rank's iteration box whose reads all
        land inside locally-owned cores (no halo dependency): it launches
        while the exchange is in flight; rim slabs launch after.  A rank
        that only SENDS launches its whole box early.  Groups with
        reductions or temp-materialised operands stay sequential (the
        reference's serial order, ramba.py:3547-3693, is the fallback)."""
        from .common import overlap_exchange
        if (not overlap_exchange or ib is None or not comm_msgs
                or plan.reductions or temp_geom):
            return None, 0
        if not any(dst == self.rank for (dst, _, _, _, _) in comm_msgs):
            return [ib], 1            # send-only rank: nothing to wait for
        interior = ib
        for name, oi in live.items():
            if oi.written:
                continue              # writes are core-contained (asserted)
            need = oi.view.image_box(ib)
            if need is None:
                continue
            core = self.core_box(oi.bd, self.rank)
            if core is not None and box_contains(core, need):
                continue              # fully local operand
            pre = oi.view.preimage_box(core) if core is not None else None
            interior = box_intersect(interior, pre)
            if interior is None:
                return None, 0        # no core-only region: sequential
        rims = box_subtract(ib, interior)
        if not rims:
            return None, 0
        return [interior] + rims, 1

    def _comms_begin(self, msgs, temp_geom):
        """Pack outgoing sub-boxes and POST the exchange (non-blocking on
        the HIP/RCCL backend).  The message list is identical on every
        rank; pairwise ordering comes from a deterministic sort."""
        def key(m):
            dst, src, bd, bx, tgt = m
            return (dst, src, bd.gid, tuple(bx[0]), tuple(bx[1]), tgt[0],
                    str(tgt[1]))
        msgs = sorted(msgs, key=key)
        tc0 = time.perf_counter()
        nbytes = sum(box_shape(m[3]) and
                     int(np.prod(box_shape(m[3]))) * m[2].dtype.itemsize
                     for m in msgs)
        add_time("exchange_bytes", nbytes)
        sends, recvs = [], []
        for m in msgs:
            dst, src, bd, bx, tgt = m
            if src == self.rank:
                buf = self.backend.pack_box(bd, self, bx)
                sends.append((dst, buf))
            if dst == self.rank:
                buf = self.backend.new_message_buffer(box_shape(bx), bd.dtype)
                recvs.append((src, buf, bd, bx, tgt))
        token = self.backend.exchange_begin(
            sends, [(s, b) for (s, b, _, _, _) in recvs])
        return (token, recvs, temp_geom, tc0)

    def _comms_finish(self, state):
        """Wait for the posted exchange and unpack received boxes."""
        token, recvs, temp_geom, tc0 = state
        self.backend.exchange_finish(token)
        for (src, buf, bd, bx, tgt) in recvs:
            if tgt[0] == "border":
                self.backend.unpack_box_to_container(bd, self, bx, buf)
            else:
                vname = tgt[1]
                need, _ = temp_geom[vname]
                self.backend.unpack_box_to_temp(vname, need, bx, buf)
        add_time("part_exchange", time.perf_counter() - tc0)

    def _do_comms(self, msgs, temp_geom):
        self._comms_finish(self._comms_begin(msgs, temp_geom))

    # ------------------------------------------------------------------
    # cross-stage fusion (BASELINE configs[4]): a sealed producer group
    # (index-pure elementwise, identity writes) + its consumer group
    # (shifted reads of the producer's outputs).  The HIP backend fuses
    # the pair into ONE LDS-tiled kernel — halo values are RECOMPUTED
    # from index arithmetic, so the halo exchange for staged arrays
    # disappears entirely.  Every other case (and every non-HIP backend)
    # falls back to sequential pair execution, which is byte-identical
    # to the reference's flush-at-alias behaviour (ramba.py:8434-8443).
    # ------------------------------------------------------------------

    def execute_staged(self, g1, g2):
        t0 = time.perf_counter()
        try:
            if not self._try_execute_tiled(g1, g2):
                self._execute_pair_sequential(g1, g2)
        finally:
            add_time("run_deferred_ops", time.perf_counter() - t0)

    def _execute_pair_sequential(self, g1, g2):
        # arrays the consumer reads from HBM pin the producer's stores
        # even if their last python ref died while the pair was pending
        extra = {oi.bd.gid for oi in g2.arr_vars.values()}
        live, dead = deferred.compute_live_vars(g1, extra_live_gids=extra)
        # consumer-read gids must not be freed by the producer's deletes
        keep = [bd for bd in g1.delete_bds if bd.gid in extra]
        g1.delete_bds = [bd for bd in g1.delete_bds if bd.gid not in extra]
        g2.delete_bds.extend(keep)
        self._execute_group_with(g1, live, dead)
        self._execute_group(g2)
        if g2.staged_reductions:
            # the reduction that was fused onto the pair: the sequential
            # order recomputes it over the full local core
            from . import staged_exec
            staged_exec.finish_staged_reductions(self, g2)

    def _execute_group_with(self, group, live, dead):
        """_execute_group with a precomputed liveness split (recipe cache
        keyed on the extra-live-adjusted liveness)."""
        cache = getattr(self, "_recipe_cache", None)
        if cache is None:
            cache = self._recipe_cache = {}
        try:
            sig = self._group_signature(group, live)
        except TypeError:
            sig = None
        recipe = cache.get(sig) if sig is not None else None
        if recipe is not None:
            self._run_recipe(recipe, group, live)
            return
        recipe = self._build_and_run(group, live, dead)
        if sig is not None and recipe is not None:
            if len(cache) > 128:
                cache.clear()
            cache[sig] = recipe

    def _try_execute_tiled(self, g1, g2):
        """Fused LDS-tiled execution of the pair; False -> caller falls
        back to sequential."""
        backend = self.backend
        if not getattr(backend, "supports_staged", False):
            return False
        from . import staged
        return staged.try_execute_tiled(self, g1, g2)

    # ------------------------------------------------------------------
    def finish_reduction(self, pend):
        val = pend.partial
        if self.world > 1:
            val = self.backend.allreduce(val, pend.kind)
        return np.asarray(val, dtype=pend.dtype)[()]

    # ------------------------------------------------------------------
    def gather_view(self, bd, view):
        """Assemble the full (global) content of `view` as numpy on every
        rank — the exit/parity boundary (`asarray`, ramba.py:5735-5764)."""
        base = np.empty(bd.shape, dtype=bd.dtype)
        for r in range(self.world):
            core = self.core_box(bd, r)
            if core is None:
                continue
            if r == self.rank:
                part = self.backend.box_to_numpy(bd, self, core)
            else:
                part = None
            part = self.backend.bcast_numpy(part, root=r)
            sl = tuple(slice(int(core[0, i]), int(core[1, i]) + 1)
                       for i in range(len(bd.shape)))
            base[sl] = part
        return numpy_view(base, view)

    def container_slice(self, bd, box, r=None):
        """Container index slices covering a global base box."""
        d = bd.divisions[self.rank if r is None else r]
        nd = len(bd.shape)
        pads = self.lo_pads(bd)
        return tuple(slice(int(box[0, i] - d[0, i]) + pads[i],
                           int(box[1, i] - d[0, i]) + pads[i] + 1)
                     for i in range(nd))

    def free_shard(self, bd):
        if bd.constructed:
            self.backend.free_container(bd)
            bd.constructed = False

    def scatter_numpy(self, bd, nparr):
        """Distribute a host array: each rank copies its core slice."""
        assert tuple(nparr.shape) == bd.shape
        if not bd.constructed:
            self.backend.alloc_container(bd, self)
            bd.constructed = True
            bd.flex = False
        core = self.core_box(bd, self.rank)
        if core is not None:
            sl = tuple(slice(int(core[0, i]), int(core[1, i]) + 1)
                       for i in range(len(bd.shape)))
            self.backend.write_core_from_numpy(
                bd, self, np.ascontiguousarray(nparr[sl], dtype=bd.dtype))

    def read_element(self, bd, coords):
        box = np.array([coords, coords], dtype=np.int64)
        owner = None
        for r in range(self.world):
            core = self.core_box(bd, r)
            if core is not None and box_contains(core, box):
                owner = r
                break
        assert owner is not None
        if owner == self.rank:
            val = self.backend.box_to_numpy(bd, self, box)
        else:
            val = None
        val = self.backend.bcast_numpy(val, root=owner)
        return val.reshape(())[()]


def numpy_view(base, view):
    """Apply a View to a full numpy base array (host-side, exit boundary)."""
    itemsize = base.itemsize
    offset = 0
    strides = []
    bstr = base.strides
    for b in range(len(base.shape)):
        offset += view.offset[b] * bstr[b]
    for v in range(view.ndim):
        b, st = view.axis_map[v], view.steps[v]
        strides.append(st * bstr[b] if b >= 0 else 0)
    flat = np.lib.stride_tricks.as_strided(
        base, shape=view.shape, strides=tuple(strides),
        # numpy as_strided has no offset arg; slice the base buffer instead
    ) if offset == 0 else np.lib.stride_tricks.as_strided(
        base.reshape(-1)[offset // itemsize:], shape=view.shape,
        strides=tuple(strides))
    return flat.copy()


# ---------------------------------------------------------------------------
# axis reductions — bolted onto Runtime
# ---------------------------------------------------------------------------

def _proj_box(box, axes):
    """view-space box -> keepdims-space box (reduced axes pinned to 0)."""
    out = box.copy()
    for d in axes:
        out[0, d] = 0
        out[1, d] = 0
    return out


def reduce_axes_op(self, arr, axes, kind, out_dtype, kd_shape):
    """Distributed axis reduction: local strided-reduce kernel into a
    per-rank partial, then cross-rank combining box exchange into a fresh
    result array (replaces the reference's axis_reduce loops 8231-8244 +
    internal_reduction2 slice combine 5818-5849)."""
    from .shardview import exec_boxes as _eb
    bd, v = arr.bdarray, arr.view
    axes = tuple(sorted(axes))
    out_bd = deferred.bdarray(kd_shape, out_dtype,
                              default_divisions(self.world, kd_shape),
                              default_border, flex=False)
    self.backend.alloc_container(out_bd, self)
    out_bd.constructed = True
    ident = ir.reduction_init(kind, out_dtype)
    self.backend.fill_container(out_bd, self, ident)

    lbs = _eb(v, bd.divisions)
    lb = lbs[self.rank]
    if lb is not None:
        d_, cshape, cstrides, pads = self.shard_geometry(bd)
        off0, strides = v.operand_addressing(lb[0], cstrides, d_[0], pads)
        self.backend.axis_reduce_partial(
            bd, off0, strides, lb, axes, kind, out_dtype)

    # combining exchange plan (deterministic on every rank)
    msgs = []
    for r in range(self.world):
        if lbs[r] is None:
            continue
        pb = _proj_box(lbs[r], axes)
        for s in range(self.world):
            part = box_intersect(pb, self.core_box(out_bd, s))
            if part is not None:
                msgs.append((s, r, part))
    msgs.sort(key=lambda m: (m[0], m[1], tuple(m[2][0]), tuple(m[2][1])))

    my_pb = _proj_box(lb, axes) if lb is not None else None
    sends, recvs, locals_ = [], [], []
    for (dst, src, bx) in msgs:
        if src == self.rank and dst == self.rank:
            locals_.append(bx)
            continue
        if src == self.rank:
            rel = bx.copy()
            rel[0] -= my_pb[0]
            rel[1] -= my_pb[0]
            sends.append((dst, self.backend.pack_temp_box("__axred__", rel)))
        if dst == self.rank:
            buf = self.backend.new_message_buffer(box_shape(bx), out_dtype)
            recvs.append((src, buf, bx))
    if sends or recvs:
        self.backend.exchange(sends, [(s, b) for (s, b, _) in recvs])
    for bx in locals_:
        rel = bx.copy()
        rel[0] -= my_pb[0]
        rel[1] -= my_pb[0]
        self.backend.combine_temp_into_container(out_bd, self, bx,
                                                 "__axred__", rel, kind)
    for (src, buf, bx) in recvs:
        self.backend.combine_box_into_container(out_bd, self, bx, buf, kind)
    self.backend.free_temps()
    return out_bd


Runtime.reduce_axes_op = reduce_axes_op


# ---------------------------------------------------------------------------
# cumsum (replaces the reference scumulative local-prefix +
# sequential cross-worker fixup chain, ramba.py:10057-10171/3378-3460 — the
# chain becomes an allgather of rank totals)
# ---------------------------------------------------------------------------

def cumsum_op(self, arr, out_dtype):
    from .shardview import exec_boxes as _eb
    bd, v = arr.bdarray, arr.view
    assert v.ndim == 1, "cumsum is 1-D (reference scumulative)"
    lbs = _eb(v, bd.divisions)
    # result partitioned by the input's exec boxes (no data exchange)
    nd = 1
    divs = np.zeros((self.world, 2, nd), dtype=np.int64)
    starts = []
    for r, b in enumerate(lbs):
        if b is None:
            divs[r, 1, :] = -1
            starts.append(None)
        else:
            divs[r] = b
            starts.append(int(b[0, 0]))
    out_bd = deferred.bdarray(v.shape, out_dtype, divs, default_border,
                              flex=False)
    self.backend.alloc_container(out_bd, self)
    out_bd.constructed = True

    lb = lbs[self.rank]
    local_total = np.asarray(0, dtype=out_dtype)[()]
    if lb is not None:
        d_, cshape, cstrides, pads = self.shard_geometry(bd)
        off0, strides = v.operand_addressing(lb[0], cstrides, d_[0], pads)
        n_local = int(lb[1, 0] - lb[0, 0] + 1)
        local_total = self.backend.cumsum_local_phase12(
            bd, off0, strides[0], n_local, out_dtype)
    # cross-rank exclusive prefix of totals, ordered by global start
    totals = self.backend.allgather_scalars(local_total, out_dtype)
    offset = np.asarray(0, dtype=out_dtype)[()]
    my_start = starts[self.rank]
    if my_start is not None:
        for r in range(self.world):
            if starts[r] is not None and starts[r] < my_start:
                offset = np.asarray(offset + totals[r], dtype=out_dtype)[()]
    if lb is not None:
        dd, _, ocs, opads = self.shard_geometry(out_bd)
        out_off = opads[0]
        self.backend.cumsum_local_phase3(bd, off0, strides[0], n_local,
                                         out_bd, out_off, offset, out_dtype)
    self.backend.free_temps()
    return out_bd


Runtime.cumsum_op = cumsum_op

# ---------------------------------------------------------------------------
# boolean-mask compaction getitem (SURVEY §8f n3 second half; the
# reference's compressing `a[mask]`, ramba.py maskarray getitem).  The
# frontend first materialises value and mask onto ONE common partition
# (fresh default-partition copies through the fused engine), so here both
# arrays share divisions; each rank compacts its core box in C order and
# the result is partitioned unevenly by the per-rank counts — no data
# exchange, just an allgather of counts.
# ---------------------------------------------------------------------------

def mask_compact_op(self, bd_a, bd_m):
    assert np.array_equal(bd_a.divisions, bd_m.divisions), \
        "mask_compact_op requires co-partitioned value/mask"
    nd = len(bd_a.shape)
    order = []
    for r in range(self.world):
        cb = self.core_box(bd_a, r)
        if cb is None:
            continue
        # C-order concatenation across ranks needs each rank's box to
        # span the full extent of every axis but the first
        for dax in range(1, nd):
            if not (cb[0, dax] == 0
                    and cb[1, dax] == bd_a.shape[dax] - 1):
                raise NotImplementedError(
                    "a[mask] needs C-contiguous (axis-0 split) partitions "
                    f"for nd>={nd}; got division {cb.tolist()} of rank {r}")
        order.append((int(cb[0, 0]), r))
    order.sort()

    local, count = self.backend.mask_compact(bd_a, bd_m, self)
    counts = self.backend.allgather_scalars(np.int64(count), np.int64)
    prefix, run = {}, 0
    for (_, r) in order:
        prefix[r] = run
        run += int(counts[r])
    total = run
    divs = np.zeros((self.world, 2, 1), dtype=np.int64)
    for r in range(self.world):
        c = int(counts[r])
        if c == 0:
            divs[r, 1, 0] = -1
        else:
            divs[r, 0, 0] = prefix[r]
            divs[r, 1, 0] = prefix[r] + c - 1
    out_bd = deferred.bdarray((total,), bd_a.dtype, divs, default_border,
                              flex=False)
    self.backend.alloc_container(out_bd, self)
    out_bd.constructed = True
    if count:
        self.backend.write_local_dense(out_bd, self, local)
    self.backend.free_temps()
    return out_bd


Runtime.mask_compact_op = mask_compact_op

# ---------------------------------------------------------------------------
# axis-wise cumsum on N-D arrays (the axis half of the
# reference's scumulative, ramba.py:10057-10171 + scumulative_worker
# 3378-3440).  Local inclusive scan along the axis per rank; the
# reference's sequential cross-worker relay chain becomes ONE deterministic
# pairwise slab exchange: every rank that precedes another along the scan
# axis sends the overlap of its line-totals slab, receivers sum incoming
# slabs and broadcast-add them over their local box (rt_combine_box with
# stride 0 on the axis).
# ---------------------------------------------------------------------------

def cumsum_axis_op(self, arr, axis, out_dtype):
    from .shardview import exec_boxes as _eb
    bd, v = arr.bdarray, arr.view
    nd = v.ndim
    assert 0 <= axis < nd and nd >= 2
    assert np.dtype(bd.dtype) == np.dtype(out_dtype)  # frontend casts
    lbs = _eb(v, bd.divisions)
    divs = np.zeros((self.world, 2, nd), dtype=np.int64)
    for r, b in enumerate(lbs):
        if b is None:
            divs[r, 1, :] = -1
        else:
            divs[r] = b
    out_bd = deferred.bdarray(v.shape, out_dtype, divs, default_border,
                              flex=False)
    self.backend.alloc_container(out_bd, self)
    out_bd.constructed = True

    lb = lbs[self.rank]
    if lb is not None:
        d_, _, cstrides, pads = self.shard_geometry(bd)
        off0, strides = v.operand_addressing(lb[0], cstrides, d_[0], pads)
        od, _, ocs, opads = self.shard_geometry(out_bd)
        out_off = sum((int(lb[0, i]) - int(od[0, i]) + opads[i]) * ocs[i]
                      for i in range(nd))
        self.backend.axis_scan_local(bd, off0, strides, box_shape(lb),
                                     axis, out_bd, out_off, ocs)

    def _proj(b):
        keep = [i for i in range(nd) if i != axis]
        return np.array([[int(b[0, i]) for i in keep],
                         [int(b[1, i]) for i in keep]], dtype=np.int64)

    # deterministic pair plan on every rank
    msgs = []
    for s_ in range(self.world):          # receiver
        bs = lbs[s_]
        if bs is None:
            continue
        for r_ in range(self.world):      # sender (strictly precedes s_)
            br = lbs[r_]
            if br is None or r_ == s_:
                continue
            if int(br[1, axis]) < int(bs[0, axis]):
                inter = box_intersect(_proj(br), _proj(bs))
                if inter is not None:
                    msgs.append((s_, r_, inter))
    msgs.sort(key=lambda m: (m[0], m[1], tuple(m[2][0]), tuple(m[2][1])))

    my_lines = _proj(lb) if lb is not None else None
    sends, recvs = [], []
    for (dst, src, bx) in msgs:
        if src == self.rank:
            rel = bx.copy()
            rel[0] -= my_lines[0]
            rel[1] -= my_lines[0]
            sends.append((dst,
                          self.backend.pack_temp_box("__axcs_tot__", rel)))
        if dst == self.rank:
            recvs.append((src,
                          self.backend.new_message_buffer(box_shape(bx),
                                                          out_dtype), bx))
    if recvs:
        self.backend.axcs_init_offsets(box_shape(my_lines), out_dtype)
    if sends or recvs:
        self.backend.exchange(sends, [(s, b) for (s, b, _) in recvs])
    for (src, buf, bx) in recvs:
        rel = bx.copy()
        rel[0] -= my_lines[0]
        rel[1] -= my_lines[0]
        self.backend.axcs_accumulate(rel, buf)
    if recvs:
        self.backend.axcs_apply(out_bd, self, lb, axis)
    self.backend.free_temps()
    return out_bd


Runtime.cumsum_axis_op = cumsum_axis_op

# ---------------------------------------------------------------------------
# reshape (reference reshape worker, ramba.py:2409-2492 + frontend 6716/
# 9438-area).  The reference remaps every element's flat C-order index and
# ships per-element lists; here both sides' shards are required to cover
# C-contiguous flat INTERVALS (always true at world 1; multi-rank needs
# axis-0-only splits spanning full trailing axes), so the movement
# collapses to interval intersections + the rt_flat_copy gather/scatter
# kernels.
# ---------------------------------------------------------------------------

def _flat_interval(box, shape):
    """(lo, hi_exclusive) flat C-order range of `box` if it is contiguous
    (spans full extent on every axis but the first), else None."""
    nd = len(shape)
    for d in range(1, nd):
        if not (int(box[0, d]) == 0 and int(box[1, d]) == shape[d] - 1):
            return None
    inner = 1
    for d in range(1, nd):
        inner *= shape[d]
    return int(box[0, 0]) * inner, (int(box[1, 0]) + 1) * inner


def reshape_op(self, arr, newshape):
    bd, v = arr.bdarray, arr.view
    from .shardview import exec_boxes as _eb
    newshape = tuple(int(x) for x in newshape)
    total = 1
    for x in v.shape:
        total *= x
    ntotal = 1
    for x in newshape:
        ntotal *= x
    assert total == ntotal, f"cannot reshape {v.shape} into {newshape}"

    from .common import contiguous_divisions
    out_bd = deferred.bdarray(newshape, bd.dtype,
                              contiguous_divisions(self.world, newshape),
                              default_border, flex=False)
    self.backend.alloc_container(out_bd, self)
    out_bd.constructed = True
    if total == 0:
        return out_bd

    lbs = _eb(v, bd.divisions)
    src_iv, dst_iv = [], []
    for r in range(self.world):
        b = lbs[r]
        iv = None if b is None else _flat_interval(b, v.shape)
        if b is not None and iv is None:
            # the frontend repartitions first; reaching here is a bug
            raise AssertionError(
                "reshape_op: non-contiguous source shard "
                f"{b.tolist()} of {v.shape} (frontend must repartition)")
        src_iv.append(iv)
        ob = self.core_box(out_bd, r)
        oiv = None if ob is None else _flat_interval(ob, newshape)
        if ob is not None and oiv is None:
            raise NotImplementedError(
                "reshape needs C-contiguous destination shards")
        dst_iv.append(oiv)

    msgs = []   # (dst, src, lo, hi) flat intervals, deterministic
    for q in range(self.world):
        if dst_iv[q] is None:
            continue
        for r in range(self.world):
            if src_iv[r] is None:
                continue
            lo = max(dst_iv[q][0], src_iv[r][0])
            hi = min(dst_iv[q][1], src_iv[r][1])
            if lo < hi:
                msgs.append((q, r, lo, hi))
    msgs.sort()

    lb = lbs[self.rank]
    if lb is not None:
        d_, _, cstrides, pads = self.shard_geometry(bd)
        off0, strides = v.operand_addressing(lb[0], cstrides, d_[0], pads)
        src_shape = box_shape(lb)
    ob = self.core_box(out_bd, self.rank)
    if ob is not None:
        _, _, ocs, opads = self.shard_geometry(out_bd)
        out_off = sum(opads[i] * ocs[i] for i in range(len(newshape)))
        dst_shape = box_shape(ob)

    cont_in = self.backend._cont(bd) if lb is not None else None
    cont_out = self.backend._cont(out_bd) if ob is not None else None
    sends, recvs, locals_ = [], [], []
    for (q, r, lo, hi) in msgs:
        if r == self.rank and q == self.rank:
            locals_.append((lo, hi))
            continue
        if r == self.rank:
            buf = self.backend.flat_gather(
                cont_in, off0, strides, src_shape, lo - src_iv[r][0],
                hi - lo)
            sends.append((q, buf))
        if q == self.rank:
            recvs.append((r, self.backend.new_message_buffer(
                (hi - lo,), bd.dtype), lo))
    for (lo, hi) in locals_:
        buf = self.backend.flat_gather(cont_in, off0, strides, src_shape,
                                       lo - src_iv[self.rank][0], hi - lo)
        self.backend.flat_scatter(cont_out, out_off, ocs, dst_shape,
                                  lo - dst_iv[self.rank][0], buf)
    if sends or recvs:
        self.backend.exchange(sends, [(s, b) for (s, b, _) in recvs])
    for (r, buf, lo) in recvs:
        self.backend.flat_scatter(cont_out, out_off, ocs, dst_shape,
                                  lo - dst_iv[self.rank][0], buf)
    self.backend.free_temps()
    return out_bd


Runtime.reshape_op = reshape_op
