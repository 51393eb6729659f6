"""CPU oracle backend — TEST INFRASTRUCTURE (see oracle/__init__.py).

Implements the ramba_amd backend interface with NumPy shard containers and
(for world_size > 1) torch.distributed `gloo` transport, so the full SPMD
planner / halo-exchange / reduction logic runs on CPU ranks in CI.  It is
injected explicitly by tests via `ramba_amd.init(backend=NumpyBackend())`;
the product never selects it.
"""

import numpy as np

from ramba_amd import ir
from ramba_amd.shardview import box_shape
from . import ir_interp


class NumpyBackend:
    name = "numpy-oracle"

    def __init__(self):
        self.containers = {}   # gid -> np.ndarray
        self.temps = {}        # name -> np.ndarray
        self.rt = None
        self._dist = None

    def attach(self, rt):
        self.rt = rt

    # -- process group ------------------------------------------------------

    def init_process_group(self, rank, world):
        import torch.distributed as dist
        self._dist = dist
        if not dist.is_initialized():
            dist.init_process_group("gloo", rank=rank, world_size=world)

    def _d(self):
        if self._dist is None:
            import torch.distributed as dist
            self._dist = dist
        return self._dist

    # -- memory -------------------------------------------------------------

    def alloc_container(self, bd, rt):
        from ramba_amd.common import debug_poison
        _, cshape, _, _ = rt.shard_geometry(bd)
        if cshape is None:
            self.containers[bd.gid] = None
            return
        c = np.empty(cshape, dtype=bd.dtype)
        if debug_poison:
            c.view(np.uint8)[...] = 0xCC   # NaN-ish for floats, junk ints
        self.containers[bd.gid] = c

    def free_container(self, bd):
        self.containers.pop(bd.gid, None)

    def alloc_temp(self, name, shape, dtype):
        self.temps[name] = np.empty(shape, dtype=dtype)

    def free_temps(self):
        self.temps.clear()

    # -- box copies ----------------------------------------------------------

    def _cont(self, bd):
        c = self.containers.get(bd.gid)
        assert c is not None, f"shard for gid {bd.gid} not allocated"
        return c

    def copy_container_to_temp(self, bd, rt, part_box, vname, need_box):
        src = self._cont(bd)[rt.container_slice(bd, part_box)]
        lo = part_box[0] - need_box[0]
        sl = tuple(slice(int(lo[i]), int(lo[i]) + src.shape[i])
                   for i in range(src.ndim))
        self.temps[vname][sl] = src

    def pack_box(self, bd, rt, box):
        import torch
        arr = np.ascontiguousarray(self._cont(bd)[rt.container_slice(bd, box)])
        return torch.from_numpy(arr)

    def new_message_buffer(self, shape, dtype):
        import torch
        return torch.from_numpy(np.empty(shape, dtype=dtype))

    def exchange_begin(self, sends, recvs):
        """Post the pairwise exchange (gloo over host tensors)."""
        if not sends and not recvs:
            return None
        dist = self._d()
        ops = []
        for (dst, buf) in sends:
            ops.append(dist.P2POp(dist.isend, buf, dst))
        for (src, buf) in recvs:
            ops.append(dist.P2POp(dist.irecv, buf, src))
        return dist.batch_isend_irecv(ops) if ops else []

    def exchange_finish(self, token):
        if token is None:
            return
        for req in token:
            req.wait()

    def exchange(self, sends, recvs):
        self.exchange_finish(self.exchange_begin(sends, recvs))

    def unpack_box_to_container(self, bd, rt, box, buf):
        self._cont(bd)[rt.container_slice(bd, box)] = buf.numpy()

    def unpack_box_to_temp(self, vname, need_box, box, buf):
        lo = box[0] - need_box[0]
        a = buf.numpy()
        sl = tuple(slice(int(lo[i]), int(lo[i]) + a.shape[i])
                   for i in range(a.ndim))
        self.temps[vname][sl] = a

    def box_to_numpy(self, bd, rt, box):
        return np.ascontiguousarray(
            self._cont(bd)[rt.container_slice(bd, box)])

    def write_core_from_numpy(self, bd, rt, nparr):
        core = rt.core_box(bd, rt.rank)
        self._cont(bd)[rt.container_slice(bd, core)] = nparr

    # -- collectives ----------------------------------------------------------

    def bcast_numpy(self, obj, root):
        if self.rt.world == 1:
            return obj
        dist = self._d()
        lst = [obj]
        dist.broadcast_object_list(lst, src=root)
        return lst[0]

    _RED_MAP = {"sum": "SUM", "prod": "PRODUCT", "min": "MIN", "max": "MAX",
                "all": "MIN", "any": "MAX"}

    def allreduce(self, value, kind):
        import torch
        dist = self._d()
        v = np.asarray(value)
        dt = v.dtype
        if kind in ("min", "max") and dt.kind == "f":
            # same NaN-propagation contract as the HIP backend
            parts = self.allgather_scalars(value, dt)
            red = np.minimum.reduce if kind == "min" else np.maximum.reduce
            return np.asarray(red(np.asarray(parts, dtype=dt)), dtype=dt)[()]
        if dt == np.bool_:
            v = v.astype(np.uint8)
        t = torch.from_numpy(v.reshape(1).copy())
        dist.all_reduce(t, op=getattr(dist.ReduceOp, self._RED_MAP[kind]))
        out = t.numpy()[0]
        if dt == np.bool_:
            out = bool(out)
        return np.asarray(out, dtype=dt)[()]

    # -- kernel execution ------------------------------------------------------

    def launch(self, plan, recipe=None):
        env = {"__scalars__": plan.scalars}
        for op in plan.operands:
            if op.kind == "temp":
                base = self.temps[op.temp_key]
            else:
                base = self.containers.get(op.bd.gid)
                if base is None:
                    continue
            flat = base.reshape(-1)
            isz = base.itemsize
            strides = tuple(s * isz for s in op.strides)
            env[op.name] = np.lib.stride_tricks.as_strided(
                flat[op.offset0:] if op.offset0 else flat,
                shape=plan.itershape, strides=strides)
        for name, dt in plan.dead_vars.items():
            env[name] = np.empty(plan.itershape, dtype=dt)
        return ir_interp.run_statements(plan, env)

    def sync(self):
        pass

    # -- axis reductions -------------------------------------

    def fill_container(self, bd, rt, value):
        c = self.containers.get(bd.gid)
        if c is not None:
            c[...] = value

    def axis_reduce_partial(self, bd, off0, strides, lb, axes, kind,
                            out_dtype):
        from ramba_amd.shardview import box_shape as _bs
        base = self._cont(bd)
        flat = base.reshape(-1)
        isz = base.itemsize
        view = np.lib.stride_tricks.as_strided(
            flat[off0:] if off0 else flat, shape=_bs(lb),
            strides=tuple(s * isz for s in strides))
        comb, _ = ir.REDUCTIONS[kind]
        uf = ir.BINOPS[comb]
        part = uf.reduce(view.astype(out_dtype, copy=False), axis=axes,
                         keepdims=True)
        self.temps["__axred__"] = np.ascontiguousarray(part,
                                                       dtype=out_dtype)

    def pack_temp_box(self, vname, rel_box):
        import torch
        t = self.temps[vname]
        sl = tuple(slice(int(rel_box[0, i]), int(rel_box[1, i]) + 1)
                   for i in range(t.ndim))
        return torch.from_numpy(np.ascontiguousarray(t[sl]))

    def _combine(self, dst, src, kind):
        comb, _ = ir.REDUCTIONS[kind]
        dst[...] = ir.BINOPS[comb](dst, src)

    def combine_box_into_container(self, bd, rt, box, buf, kind):
        self._combine(self._cont(bd)[rt.container_slice(bd, box)],
                      buf.numpy(), kind)

    def combine_temp_into_container(self, bd, rt, box, vname, rel_box, kind):
        t = self.temps[vname]
        sl = tuple(slice(int(rel_box[0, i]), int(rel_box[1, i]) + 1)
                   for i in range(t.ndim))
        self._combine(self._cont(bd)[rt.container_slice(bd, box)], t[sl],
                      kind)

    # -- cumsum ---------------------------------------------

    def cumsum_local_phase12(self, bd, off0, stride, n, out_dtype):
        base = self._cont(bd).reshape(-1)
        isz = base.itemsize
        view = np.lib.stride_tricks.as_strided(
            base[off0:] if off0 else base, shape=(n,),
            strides=(stride * isz,))
        cs = np.cumsum(view, dtype=out_dtype)
        self.temps["__cumsum__"] = cs
        return cs[-1] if n else np.asarray(0, dtype=out_dtype)[()]

    def cumsum_local_phase3(self, bd, off0, stride, n, out_bd, out_off,
                            offset, out_dtype):
        cs = self.temps["__cumsum__"]
        out = self.containers[out_bd.gid].reshape(-1)
        out[out_off:out_off + n] = cs + offset

    def allgather_scalars(self, val, dtype):
        if self.rt.world == 1:
            return [val]
        import torch
        dist = self._d()
        t = torch.from_numpy(np.asarray([val], dtype=dtype))
        outs = [torch.empty_like(t) for _ in range(self.rt.world)]
        dist.all_gather(outs, t)
        return [o.numpy()[0] for o in outs]


def _nb_mask_compact(self, bd_a, bd_m, rt):
    d, _, _, pads_a = rt.shard_geometry(bd_a)
    if d is None:
        return None, 0
    _, _, _, pads_m = rt.shard_geometry(bd_m)
    nd = len(bd_a.shape)
    sl_a = tuple(slice(pads_a[i], pads_a[i] + int(d[1, i] - d[0, i] + 1))
                 for i in range(nd))
    sl_m = tuple(slice(pads_m[i], pads_m[i] + int(d[1, i] - d[0, i] + 1))
                 for i in range(nd))
    a = self.containers[bd_a.gid][sl_a]
    m = self.containers[bd_m.gid][sl_m]
    sel = a[m != 0]          # C order
    return sel.copy(), int(sel.size)


def _nb_write_local_dense(self, out_bd, rt, local):
    _, _, _, pads = rt.shard_geometry(out_bd)
    cont = self.containers[out_bd.gid]
    cont[pads[0]:pads[0] + local.shape[0]] = local


NumpyBackend.mask_compact = _nb_mask_compact
NumpyBackend.write_local_dense = _nb_write_local_dense


# -- axis-wise cumsum (N-D scumulative) -----------------------


def _nb_axis_scan_local(self, bd_in, off0, strides, lshape, axis, out_bd,
                        out_off, out_strides):
    base = self._cont(bd_in).reshape(-1)
    isz = base.itemsize
    view = np.lib.stride_tricks.as_strided(
        base[off0:] if off0 else base, shape=tuple(lshape),
        strides=tuple(s * isz for s in strides))
    cs = np.cumsum(view.astype(out_bd.dtype, copy=False), axis=axis)
    out = self.containers[out_bd.gid].reshape(-1)
    osz = out.itemsize
    oview = np.lib.stride_tricks.as_strided(
        out[out_off:] if out_off else out, shape=tuple(lshape),
        strides=tuple(s * osz for s in out_strides))
    oview[...] = cs
    sel = tuple(slice(None) if d != axis else -1
                for d in range(len(lshape)))
    self.temps["__axcs_tot__"] = np.ascontiguousarray(cs[sel])


def _nb_axcs_init_offsets(self, lines_shape, dtype):
    self.temps["__axcs_off__"] = np.zeros(
        lines_shape if lines_shape else (1,), dtype=dtype)


def _nb_axcs_accumulate(self, rel_box, buf):
    offs = self.temps["__axcs_off__"]
    sl = tuple(slice(int(rel_box[0, i]), int(rel_box[1, i]) + 1)
               for i in range(offs.ndim))
    offs[sl] += buf.numpy().reshape(offs[sl].shape)


def _nb_axcs_apply(self, out_bd, rt, box, axis):
    offs = self.temps["__axcs_off__"]
    dst = self._cont(out_bd)[rt.container_slice(out_bd, box)]
    dst += np.expand_dims(offs, axis)


NumpyBackend.axis_scan_local = _nb_axis_scan_local
NumpyBackend.axcs_init_offsets = _nb_axcs_init_offsets
NumpyBackend.axcs_accumulate = _nb_axcs_accumulate
NumpyBackend.axcs_apply = _nb_axcs_apply


def _nb_flat_view(cont, off0, strides, shape):
    base = cont.reshape(-1)
    isz = base.itemsize
    return np.lib.stride_tricks.as_strided(
        base[off0:] if off0 else base, shape=tuple(shape),
        strides=tuple(s * isz for s in strides))


def _nb_flat_gather(self, cont, off0, strides, shape, flat0, n):
    import torch
    v = _nb_flat_view(cont, off0, strides, shape).reshape(-1)
    return torch.from_numpy(np.ascontiguousarray(v[flat0:flat0 + n]))


def _nb_flat_scatter(self, cont, off0, strides, shape, flat0, buf):
    v = _nb_flat_view(cont, off0, strides, shape)
    flat = v.reshape(-1) if v.flags["C_CONTIGUOUS"] else None
    b = buf.numpy()
    if flat is not None:
        flat[flat0:flat0 + b.size] = b
        return
    # strided destination: walk coords
    idx = np.arange(flat0, flat0 + b.size)
    coords = np.unravel_index(idx, tuple(shape))
    v[coords] = b


NumpyBackend.flat_gather = _nb_flat_gather
NumpyBackend.flat_scatter = _nb_flat_scatter
