"""Deferred-op fusion engine — the host half of the hot path.

Re-implements the semantics of the reference's `deferred_op` class
(ramba/ramba.py:8039-8533): ops accumulate into one fused group while the
iteration shape and output partition stay compatible; RAW/WAR alias hazards
force a flush or a temp (ramba.py:8434-8465); at flush time, arrays nobody
references any more are demoted to per-element register temps (the live_gids
materialisation rule, ramba.py:8123) and the group is handed to the runtime
(the `remote_exec_all("run_deferred_ops", ...)` boundary, ramba.py:8286).
"""

import numbers

import numpy as np

from . import ir
from .common import dprint
from .shardview import exec_boxes, exec_boxes_eq


_gid_counter = [0]


def next_gid():
    # deterministic sequence gids; SPMD-safe like ramba/ramba_uuid.py
    _gid_counter[0] += 1
    return _gid_counter[0]


class bdarray:
    """Backing store descriptor (analog of ramba/ramba.py:1049-1158).

    The device/host shard itself is owned by the runtime backend, keyed by
    this object.  `nviews` counts live ndarray wrappers; when it reaches zero
    the store is freed (immediately, or after the current fused group -- the
    deferred-delete path of ramba.py:8321-8328)."""

    __slots__ = ("gid", "shape", "dtype", "divisions", "border", "constructed",
                 "nviews", "flex", "__weakref__")

    def __init__(self, shape, dtype, divisions, border, flex=True):
        self.gid = next_gid()
        self.shape = tuple(int(s) for s in shape)
        self.dtype = np.dtype(dtype)
        self.divisions = divisions
        self.border = border
        self.constructed = False
        self.nviews = 0
        # flexible distribution: a fresh, never-constructed array may still
        # adopt the fused group's partition at flush (reference flex_dist,
        # ramba/ramba.py:8093-8101 "Change distributions for any flexible
        # arrays")
        self.flex = flex

    @property
    def is_flex(self):
        return self.flex and not self.constructed

    def decref(self, runtime):
        self.nviews -= 1
        if self.nviews <= 0:
            g = current_group()
            if g is not None and self.gid in g.vars_by_gid:
                g.delete_bds.append(self)
                dprint(2, "deferred delete of gid", self.gid)
            elif self.constructed and runtime is not None:
                runtime.free_shard(self)


class OperandInfo:
    __slots__ = ("bd", "view", "dtype", "written", "name")

    def __init__(self, name, bd, view, dtype, written):
        self.name = name
        self.bd = bd
        self.view = view
        self.dtype = dtype
        self.written = written


class PendingReduction:
    """Handle returned by add_reduction; the runtime fills `partial` (this
    rank's partial, a 0-d numpy value) at group execution."""

    __slots__ = ("kind", "dtype", "partial")

    def __init__(self, kind, dtype):
        self.kind = kind
        self.dtype = dtype
        self.partial = None


class FusedGroup:
    def __init__(self, runtime, shape, part_view, part_divs, flex):
        self.runtime = runtime
        self.shape = tuple(shape)
        # the (view, divisions) pair that defines the iteration partition
        self.part_view = part_view
        self.part_divs = part_divs
        self.flex = flex
        self._exec_boxes = None
        self.statements = []
        self.arr_vars = {}          # name -> OperandInfo
        self._var_by_key = {}       # (gid, view) -> var name
        self.vars_by_gid = {}       # gid -> list of var names
        self.scalars = {}           # name -> (value, np.dtype)
        self.reductions = []        # list of (ReductionSpec, PendingReduction)
        self.keepalives = []        # strong refs to bdarrays used
        self.delete_bds = []        # bdarrays to free after execution
        self.read_views = []        # (gid, view) every read
        self.write_views = []       # (gid, view) every write
        self._counter = 0
        # producer/consumer staging (BASELINE configs[4] cross-stage
        # fusion): when a shifted read of an in-group-written INDEX-PURE
        # array arrives, the group is SEALED as this group's `producer`
        # instead of flushed; the runtime fuses the pair into one
        # LDS-tiled kernel (or falls back to sequential execution, which
        # is exactly the reference's flush-at-alias behaviour,
        # ramba.py:8434-8443)
        self.producer = None        # sealed stage-1 FusedGroup or None
        self.staged_gids = set()    # gids served from LDS recompute
        # sum(A) immediately after the pair, where the consumer wrote a
        # sub-box of A: fused into the tiled kernel (interior partials)
        # + a tiny complement reduce (runtime/staged_exec)
        self.staged_reductions = [] # [(src ndarray, written_view, pend)]

    # -- naming -------------------------------------------------------------

    _NAME_TABLES = {}

    def fresh(self, prefix):
        self._counter += 1
        tab = FusedGroup._NAME_TABLES.get(prefix)
        if tab is None:
            tab = FusedGroup._NAME_TABLES[prefix] = [
                f"{prefix}{i:04d}" for i in range(256)]
        c = self._counter
        return tab[c] if c < 256 else f"{prefix}{c:04d}"

    def exec_boxes(self):
        if self._exec_boxes is None:
            self._exec_boxes = exec_boxes_cached(self.part_view,
                                                 self.part_divs)
        return self._exec_boxes

    def adopt_partition(self, arr):
        """A non-flex participant fixes a flex group's partition
        (reference ramba.py:8470-8473 'fixing distribution')."""
        self.part_view = arr.view
        self.part_divs = arr.bdarray.divisions
        self.flex = False
        self._exec_boxes = None

    # -- operand registration ------------------------------------------------

    def arr_var(self, arr, written):
        """Same (gid, view) -> same var (reference add_gid, ramba.py:8078).
        The key MUST be view CONTENT, not identity: a write and a read of
        the same (gid, view) arriving as distinct-but-equal View objects
        must merge into ONE var, or the kernel's SSA forwarding loses the
        in-group RAW (caught by fuzz seed 39 on the HIP path — the CPU
        oracle's sequential memory interpretation hides it)."""
        gid = arr.bdarray.gid
        key = (gid, arr.view)
        name = self._var_by_key.get(key)
        if name is not None:
            if written:
                self.arr_vars[name].written = True
            return name
        name = self.fresh("v")
        oi = OperandInfo(name, arr.bdarray, arr.view, arr.dtype, written)
        self.arr_vars[name] = oi
        self._var_by_key[key] = name
        self.vars_by_gid.setdefault(gid, []).append(name)
        self.keepalives.append(arr.bdarray)
        return name

    def scalar_var(self, value):
        name = self.fresh("s")
        if isinstance(value, (bool, np.bool_)):
            dt = ir.BOOL
        elif isinstance(value, (int, np.integer)):
            dt = np.dtype(np.int64)
        elif isinstance(value, (float, np.floating)):
            dt = np.dtype(np.float64)
        else:
            dt = np.asarray(value).dtype
        self.scalars[name] = (value, dt)
        return name, dt


# ---------------------------------------------------------------------------
# module state
# ---------------------------------------------------------------------------

_eb_cache = {}


def exec_boxes_cached(view, divisions):
    """exec_boxes with an IDENTITY-keyed cache (Views are interned by the
    getitem/identity caches and divisions arrays are shared/rebound, so
    object identity is the fast, safe key; the stored refs pin the ids):
    _ensure_group would otherwise recompute the per-rank preimages for
    EVERY op of an iterating workload."""
    k = (id(view), id(divisions))
    e = _eb_cache.get(k)
    if e is not None and e[0] is view and e[1] is divisions:
        return e[2]
    r = exec_boxes(view, divisions)
    if len(_eb_cache) > 4096:
        _eb_cache.clear()
    _eb_cache[k] = (view, divisions, r)
    return r


_state = {"group": None, "runtime": None}


def set_runtime(rt):
    _state["runtime"] = rt


def get_runtime():
    rt = _state["runtime"]
    if rt is None:
        raise RuntimeError("ramba_amd runtime not initialised; call "
                           "ramba_amd.init() first")
    return rt


def current_group():
    return _state["group"]


def flush():
    """do_ops (reference ramba.py:8332): execute the pending fused group
    (or the sealed producer/consumer pair)."""
    g = _state["group"]
    if g is None:
        return
    _state["group"] = None
    if g.producer is not None:
        get_runtime().execute_staged(g.producer, g)
    else:
        get_runtime().execute_group(g)


# ---------------------------------------------------------------------------
# expression-tree substitution
# ---------------------------------------------------------------------------

NDARRAY_CLS = None   # set by ndarray.py at import (breaks the cycle)


def _is_ndarray(x):
    if NDARRAY_CLS is not None:
        return type(x) is NDARRAY_CLS
    return hasattr(x, "bdarray") and hasattr(x, "view")


def _subst(group, tree, reads_out):
    """Replace ndarray leaves with Ref(var) and python scalars with
    ScalarArg, mirroring the operand walk of add_op (ramba.py:8487-8512)."""
    if _is_ndarray(tree):
        if tree.shape == ():
            raise NotImplementedError("0-d distributed arrays")
        name = group.arr_var(tree, written=False)
        reads_out.append((tree.bdarray.gid, tree.view))
        return ir.Ref(name, tree.dtype)
    if isinstance(tree, (numbers.Number, np.bool_, np.number)):
        name, dt = group.scalar_var(tree)
        return ir.ScalarArg(name, dt)
    if isinstance(tree, (ir.Iota, ir.Const)):
        return tree
    if isinstance(tree, ir.Bin):
        a = _subst(group, tree.a, reads_out)
        b = _subst(group, tree.b, reads_out)
        return ir.Bin(tree.op, a, b, tree.dtype)
    if isinstance(tree, ir.Un):
        return ir.Un(tree.op, _subst(group, tree.a, reads_out), tree.dtype)
    if isinstance(tree, ir.Cast):
        return ir.Cast(_subst(group, tree.a, reads_out), tree.dtype)
    if isinstance(tree, ir.Where):
        return ir.Where(_subst(group, tree.c, reads_out),
                        _subst(group, tree.a, reads_out),
                        _subst(group, tree.b, reads_out), tree.dtype)
    if isinstance(tree, (ir.Ref, ir.ScalarArg)):
        return tree
    raise TypeError(f"bad expr leaf {tree!r}")


def _tree_ndarrays(tree, out):
    if _is_ndarray(tree):
        out.append(tree)
    elif isinstance(tree, ir.Bin):
        _tree_ndarrays(tree.a, out)
        _tree_ndarrays(tree.b, out)
    elif isinstance(tree, (ir.Un, ir.Cast)):
        _tree_ndarrays(tree.a, out)
    elif isinstance(tree, ir.Where):
        _tree_ndarrays(tree.c, out)
        _tree_ndarrays(tree.a, out)
        _tree_ndarrays(tree.b, out)
    return out


# ---------------------------------------------------------------------------
# add_op — the fusion decision (reference ramba.py:8383-8533)
# ---------------------------------------------------------------------------

ASSIGN_BINOP = {"=": None, "+=": "add", "-=": "sub", "*=": "mul",
                "/=": "div", "//=": "floordiv", "%=": "mod", "**=": "pow"}


def _ensure_group(arr):
    """Fusion-compatibility gate (reference ramba.py:8395-8473): flush on a
    shape mismatch, or a partition mismatch between two non-flex
    participants; a flex group adopts the first non-flex participant's
    partition."""
    g = _state["group"]
    arr_flex = arr.bdarray.is_flex
    if g is not None:
        if g.shape != arr.shape:
            dprint(2, "deferred ops shape mismatch; flushing")
            flush()
            g = None
        elif not arr_flex and not g.flex:
            boxes = exec_boxes_cached(arr.view, arr.bdarray.divisions)
            if not exec_boxes_eq(g.exec_boxes(), boxes):
                dprint(2, "deferred ops partition mismatch; flushing")
                flush()
                g = None
    if g is None:
        g = FusedGroup(get_runtime(), arr.shape, arr.view,
                       arr.bdarray.divisions, flex=arr_flex)
        _state["group"] = g
    elif not arr_flex and g.flex:
        g.adopt_partition(arr)
    return g


def _expr_refs(e, out):
    if isinstance(e, ir.Ref):
        out.append(e.name)
    elif isinstance(e, ir.Bin):
        _expr_refs(e.a, out)
        _expr_refs(e.b, out)
    elif isinstance(e, (ir.Un, ir.Cast)):
        _expr_refs(e.a, out)
    elif isinstance(e, ir.Where):
        _expr_refs(e.c, out)
        _expr_refs(e.a, out)
        _expr_refs(e.b, out)
    return out


def group_is_index_pure(g):
    """True if every statement's leaves are Iota/Const/scalars or refs to
    vars written EARLIER IN THE GROUP — i.e. the whole group is a pure
    function of the global index, recomputable on any rank (the halo-
    recompute precondition for cross-stage fusion)."""
    written = set()
    for st in g.statements:
        for name in _expr_refs(st.expr, []):
            if name not in written:
                return False
        written.add(st.target)
    return True


def _stageable(g, conflicts, write_arr, operands):
    """May `g` become the sealed producer of a new consumer group?"""
    from .common import stage_fusion
    if (not stage_fusion or g.producer is not None or g.reductions
            or len(g.shape) != 2):
        return False
    # a write-after-read hazard on this op goes through the temp dance
    # (alias check 2), which would lose the seal: don't seal then
    if any(o.bdarray.gid == write_arr.bdarray.gid
           and o.view != write_arr.view for o in operands):
        return False
    # the consumer's write target must not alias anything the producer
    # touches (the fused kernel interleaves the stages per tile)
    if write_arr.bdarray.gid in g.vars_by_gid:
        return False
    nd = len(g.shape)
    # every producer write must be an identity view (base coords == group
    # iteration coords, so halo recompute uses plain global indices)
    for (wgid, wview) in g.write_views:
        bshape = None
        for name in g.vars_by_gid.get(wgid, ()):
            bshape = g.arr_vars[name].bd.shape
            break
        if bshape is None or not wview.is_identity_for(bshape):
            return False
    # conflicting reads must be pure small shifts of the identity layout
    for (o, wgid, wview) in conflicts:
        v = o.view
        if o is write_arr:
            return False           # consumer writing a producer array
        if (v.ndim != nd or v.axis_map != tuple(range(nd))
                or v.steps != (1,) * nd):
            return False
        if any(not (0 <= off <= 16) for off in v.offset):
            return False
    return group_is_index_pure(g)


def add_op(write_arr, assign_op, rhs_tree, empty_like=None):
    """Record `write_arr <assign_op> rhs_tree` (or a pure read when
    write_arr is None is not allowed -- every statement has a target).

    `empty_like` is injected by the API layer to break the circular import
    for the WAR temp dance (reference ramba.py:8445-8465)."""
    operands = _tree_ndarrays(rhs_tree, [])
    arr = write_arr if write_arr is not None else (
        operands[0] if operands else None)
    assert arr is not None, "deferred op with no ndarray parameter"

    g = _state["group"]

    # Alias check 1 (ramba.py:8434-8443): an operand reads/writes a shifted
    # version of an array written earlier in this group.  If the group is
    # an index-pure elementwise producer and the reads are pure shifts,
    # SEAL it as the next group's producer (cross-stage fusion); else
    # flush first (the reference's behaviour).
    sealed = None
    if g is not None:
        check = operands + ([write_arr] if write_arr is not None else [])
        conflicts = []
        for o in check:
            for (wgid, wview) in g.write_views:
                if o.bdarray.gid == wgid and o.view != wview:
                    conflicts.append((o, wgid, wview))
        if conflicts:
            if write_arr is not None and _stageable(g, conflicts,
                                                    write_arr, operands):
                dprint(2, "RAW with mismatched views; sealing producer")
                sealed = g
                sealed_gids = {wgid for (_, wgid, _) in conflicts}
                _state["group"] = None
                g = None
            else:
                dprint(2, "RAW with mismatched views; flushing")
                flush()
                g = None

    # Alias check 2 (ramba.py:8445-8465): the write target is a shifted
    # version of something read (in this op or earlier in the group) ->
    # compute into a temp, flush, then assign temp to target.
    if write_arr is not None:
        conflict = any(o.bdarray.gid == write_arr.bdarray.gid
                       and o.view != write_arr.view for o in operands)
        g = _state["group"]
        if not conflict and g is not None:
            conflict = any(rgid == write_arr.bdarray.gid and
                           rview != write_arr.view
                           for (rgid, rview) in g.read_views)
        if conflict:
            dprint(2, "WAR with mismatched views; temp + flush")
            tmp = empty_like(write_arr)
            add_op(tmp, "=", rhs_tree, empty_like=empty_like)
            flush()
            add_op(write_arr, assign_op, tmp, empty_like=empty_like)
            return

    g = _ensure_group(arr)
    if sealed is not None:
        assert g.producer is None
        g.producer = sealed
        g.staged_gids = sealed_gids

    # lower compound assignment: W op= rhs  ->  W = W op rhs
    if assign_op != "=":
        binop = ASSIGN_BINOP[assign_op]
        rhs_tree = _build_binexpr(binop, write_arr, rhs_tree)

    reads = []
    expr = _subst(g, rhs_tree, reads)
    g.read_views.extend(reads)
    if write_arr is not None:
        tname = g.arr_var(write_arr, written=True)
        g.write_views.append((write_arr.bdarray.gid, write_arr.view))
        target_dtype = write_arr.dtype
    else:
        raise AssertionError("unreachable")
    if expr_dtype(expr) != target_dtype:
        expr = ir.Cast(expr, target_dtype)
    g.statements.append(ir.Assign(tname, expr))


def register_empty(arr):
    """A no-op use that guarantees construction at the next flush
    (reference: deferred no-op codeline for `empty`, ramba.py:8603)."""
    g = _ensure_group(arr)
    g.arr_var(arr, written=True)
    g.write_views.append((arr.bdarray.gid, arr.view))


def add_reduction(src_tree, kind, dtype):
    """Fuse an axis-less reduction into the group (reference
    internal_reduction1_executor, ramba.py:5789-5807) and return the pending
    per-rank partial handle.  Caller flushes and allreduces."""
    operands = _tree_ndarrays(src_tree, [])
    assert operands, "reduction of non-array"
    arr = operands[0]
    # sum(A) arriving while a sealed producer/consumer pair is pending,
    # with A's sub-box written by the consumer: fuse the reduction into
    # the pair (interior accumulated in the tiled kernel + complement
    # boxes reduced separately) instead of flushing on the shape
    # mismatch.  The caller flushes right after, so no further op can
    # slip between the pair and the reduction.
    g = _state["group"]
    if (g is not None and g.producer is not None and kind == "sum"
            and _is_ndarray(src_tree)
            and src_tree.view.is_identity_for(src_tree.bdarray.shape)
            and np.dtype(dtype) == src_tree.bdarray.dtype):
        gid = src_tree.bdarray.gid
        wvs = [wv for (wg, wv) in g.write_views if wg == gid]
        nd = len(src_tree.bdarray.shape)
        if (len(wvs) == 1 and wvs[0].ndim == nd
                and wvs[0].axis_map == tuple(range(nd))
                and wvs[0].steps == (1,) * nd
                and not any(rg == gid and rv != wvs[0]
                            for (rg, rv) in g.read_views)):
            pend = PendingReduction(kind, np.dtype(dtype))
            g.staged_reductions.append((src_tree, wvs[0], pend))
            return pend
    g = _ensure_group(arr)

    reads = []
    expr = _subst(g, src_tree, reads)
    g.read_views.extend(reads)
    if expr_dtype(expr) != dtype:
        expr = ir.Cast(expr, dtype)

    acc = g.fresh("acc")
    comb, _ = ir.REDUCTIONS[kind]
    body = ir.Bin(comb, ir.Ref(acc, dtype), expr, dtype)
    g.statements.append(ir.Assign(acc, body))
    pend = PendingReduction(kind, dtype)
    spec = ir.ReductionSpec(acc, kind, dtype, len(g.reductions))
    g.reductions.append((spec, pend))
    return pend


def _build_binexpr(op, a, b):
    # used only for the compound-assign lowering; dtype = target dtype
    return ir.Bin(op, a, b, a.dtype if hasattr(a, "dtype") else None)


def expr_dtype(e):
    if isinstance(e, (ir.Ref, ir.ScalarArg, ir.Iota)):
        return e.dtype
    return e.dtype


# ---------------------------------------------------------------------------
# liveness at flush (reference live_gids filter, ramba.py:8123)
# ---------------------------------------------------------------------------

def compute_live_vars(group, extra_live_gids=frozenset()):
    """Partition arr_vars into materialised operands and register temps.

    A var is LIVE (materialises) iff its backing array still has external
    ndarray references and is not queued for deletion, OR it was already
    constructed on the shards (preconstructed, ramba.py:8091-8097).
    Dead vars become per-element register temps: their stores are elided and
    within-group reads use the register value."""
    delete_gids = {bd.gid for bd in group.delete_bds}
    live, dead = {}, {}
    for name, oi in group.arr_vars.items():
        alive = (oi.bd.nviews > 0 and oi.bd.gid not in delete_gids) \
            or oi.bd.constructed or oi.bd.gid in extra_live_gids
        if alive:
            live[name] = oi
        else:
            dead[name] = oi
    return live, dead
