"""HIP product backend: gfx950 kernels via the C-ABI runtime.

Torch is plumbing only — device allocation (shard containers as raw device
tensors), the HIP stream, and the RCCL process group.  Every compute kernel
is our own HIP: fused groups JIT-compiled by `codegen` + `rt_kernel_get`
(hiprtc, gfx950), box copies by `rt_copy_box`.  FAILS LOUDLY if the
extension or a GPU is missing — no CPU fallback exists in the product path.
"""

import ctypes
import os

import numpy as np

from . import codegen
from .shardview import box_shape

LIBPATH = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_lib",
                       "libramba_rt.so")

TORCH_DTYPE = None  # filled at init


def _load_lib():
    # on-disk kernel cache: pre-populated by the CPU suite's compile
    # checks; ships with the snapshot so GPU boxes skip hiprtc entirely
    kc = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_kcache")
    os.environ.setdefault("RAMBA_KCACHE", kc)
    os.makedirs(os.environ["RAMBA_KCACHE"], exist_ok=True)
    if not os.path.exists(LIBPATH):
        raise RuntimeError(
            f"ramba_amd HIP runtime not built: {LIBPATH} missing. "
            "Run `python __graft_entry__.py build` (hipcc, gfx950). "
            "There is no CPU fallback in the product path.")
    lib = ctypes.CDLL(LIBPATH)
    lib.rt_init.argtypes = [ctypes.c_int]
    lib.rt_device_count.restype = ctypes.c_int
    lib.rt_last_error.restype = ctypes.c_char_p
    lib.rt_kernel_get.argtypes = [ctypes.c_char_p, ctypes.c_char_p,
                                  ctypes.c_char_p,
                                  ctypes.POINTER(ctypes.c_void_p)]
    lib.rt_launch.argtypes = [ctypes.c_void_p, ctypes.c_uint, ctypes.c_uint,
                              ctypes.c_uint, ctypes.c_uint, ctypes.c_size_t,
                              ctypes.c_char_p, ctypes.c_size_t]
    lib.rt_copy_box.argtypes = [ctypes.c_size_t, ctypes.c_void_p,
                                ctypes.c_void_p, ctypes.c_int,
                                ctypes.POINTER(ctypes.c_int64),
                                ctypes.POINTER(ctypes.c_int64),
                                ctypes.POINTER(ctypes.c_int64),
                                ctypes.c_int64, ctypes.c_int64, ctypes.c_int]
    lib.rt_combine_box.argtypes = [ctypes.c_size_t, ctypes.c_void_p,
                                   ctypes.c_void_p, ctypes.c_int,
                                   ctypes.POINTER(ctypes.c_int64),
                                   ctypes.POINTER(ctypes.c_int64),
                                   ctypes.POINTER(ctypes.c_int64),
                                   ctypes.c_int64, ctypes.c_int64,
                                   ctypes.c_int, ctypes.c_int]
    lib.rt_cumsum.argtypes = [ctypes.c_size_t, ctypes.c_void_p,
                              ctypes.c_int64, ctypes.c_int64, ctypes.c_int64,
                              ctypes.c_void_p, ctypes.c_int64,
                              ctypes.c_void_p, ctypes.c_int64,
                              ctypes.c_void_p, ctypes.c_double,
                              ctypes.c_int64, ctypes.c_int, ctypes.c_int]
    lib.rt_cumsum_scan.argtypes = [ctypes.c_size_t, ctypes.c_void_p,
                                   ctypes.c_int64, ctypes.c_int64,
                                   ctypes.c_int64, ctypes.c_void_p,
                                   ctypes.c_int64, ctypes.c_void_p,
                                   ctypes.c_void_p, ctypes.c_void_p,
                                   ctypes.c_void_p, ctypes.c_double,
                                   ctypes.c_int64, ctypes.c_int]
    lib.rt_mask_compact.argtypes = [ctypes.c_size_t, ctypes.c_void_p,
                                    ctypes.c_void_p, ctypes.c_void_p,
                                    ctypes.c_int,
                                    ctypes.POINTER(ctypes.c_int64),
                                    ctypes.POINTER(ctypes.c_int64),
                                    ctypes.POINTER(ctypes.c_int64),
                                    ctypes.c_void_p, ctypes.c_int64,
                                    ctypes.c_int, ctypes.c_int]
    lib.rt_axis_scan.argtypes = [ctypes.c_size_t, ctypes.c_void_p,
                                 ctypes.c_void_p, ctypes.c_void_p,
                                 ctypes.c_int,
                                 ctypes.POINTER(ctypes.c_int64),
                                 ctypes.POINTER(ctypes.c_int64),
                                 ctypes.POINTER(ctypes.c_int64),
                                 ctypes.c_int, ctypes.c_int,
                                 ctypes.c_void_p, ctypes.c_int64]
    lib.rt_flat_copy.argtypes = [ctypes.c_size_t, ctypes.c_void_p,
                                 ctypes.c_void_p, ctypes.c_int,
                                 ctypes.POINTER(ctypes.c_int64),
                                 ctypes.POINTER(ctypes.c_int64),
                                 ctypes.c_int64, ctypes.c_int64,
                                 ctypes.c_int, ctypes.c_int]
    lib.rt_stream_sync.argtypes = [ctypes.c_size_t]
    return lib


def _i64arr(vals):
    return (ctypes.c_int64 * len(vals))(*[int(v) for v in vals])


class HipBackend:
    name = "hip"

    def __init__(self, device=None):
        global TORCH_DTYPE
        import torch
        self.torch = torch
        if not torch.cuda.is_available():
            raise RuntimeError(
                "ramba_amd HIP backend requires an AMD GPU "
                "(torch.cuda.is_available() is False); no CPU fallback.")
        self.lib = _load_lib()
        # RAMBA_DEVICE overrides (multi-rank-on-one-GPU dry-runs);
        # production multi-GPU uses torchrun's LOCAL_RANK
        self.device = int(os.environ.get(
            "RAMBA_DEVICE", os.environ.get("LOCAL_RANK", "0"))) \
            if device is None else device
        torch.cuda.set_device(self.device)
        self._check(self.lib.rt_init(self.device), "rt_init")
        TORCH_DTYPE = {
            np.dtype(np.float64): torch.float64,
            np.dtype(np.float32): torch.float32,
            np.dtype(np.int64): torch.int64,
            np.dtype(np.int32): torch.int32,
            np.dtype(np.int16): torch.int16,
            np.dtype(np.int8): torch.int8,
            np.dtype(np.uint8): torch.uint8,
            np.dtype(np.bool_): torch.uint8,
        }
        self.containers = {}       # gid -> torch tensor
        self.temps = {}            # name -> torch tensor
        self.kernels = {}          # structural key -> GeneratedKernel
        self.rt = None
        self._dist = None
        # optional HIP-event kernel timing (bench roofline leg)
        self.time_kernels = False
        self.kernel_times_ms = []
        self.kernel_keys = []

    # ------------------------------------------------------------------
    def _check(self, rc, what):
        if rc != 0:
            raise RuntimeError(
                f"{what} failed: {self.lib.rt_last_error().decode()}")

    def attach(self, rt):
        self.rt = rt

    def _stream(self):
        return self.torch.cuda.current_stream().cuda_stream

    def init_process_group(self, rank, world):
        import torch.distributed as dist
        self._dist = dist
        # RCCL in production; RAMBA_PG_BACKEND=gloo lets tests run several
        # ranks against ONE GPU (gloo needs host tensors -> staging below)
        self._pg = os.environ.get("RAMBA_PG_BACKEND", "nccl")
        if not dist.is_initialized():
            dist.init_process_group(self._pg, rank=rank, world_size=world)

    def _d(self):
        if self._dist is None:
            import torch.distributed as dist
            self._dist = dist
        return self._dist

    # -- memory -------------------------------------------------------------

    def _tdt(self, dtype):
        return TORCH_DTYPE[np.dtype(dtype)]

    def alloc_container(self, bd, rt):
        from .common import debug_poison
        _, cshape, _, _ = rt.shard_geometry(bd)
        if cshape is None:
            self.containers[bd.gid] = None
            return
        t = self.torch.empty(cshape, dtype=self._tdt(bd.dtype),
                             device="cuda")
        if debug_poison:
            t.view(self.torch.uint8).fill_(0xCC)
        self.containers[bd.gid] = t

    def free_container(self, bd):
        self.containers.pop(bd.gid, None)

    def alloc_temp(self, name, shape, dtype):
        self.temps[name] = self.torch.empty(
            shape, dtype=self._tdt(dtype), device="cuda")

    def free_temps(self):
        self.temps.clear()

    # -- box copies -----------------------------------------------------------

    def _cont(self, bd):
        c = self.containers.get(bd.gid)
        assert c is not None, f"shard for gid {bd.gid} not allocated"
        return c

    def _copy(self, dst_t, dst_strides, dst_off, src_t, src_strides, src_off,
              shape, elemsize):
        nd = len(shape)
        self._check(self.lib.rt_copy_box(
            self._stream(), ctypes.c_void_p(dst_t.data_ptr()),
            ctypes.c_void_p(src_t.data_ptr()), nd, _i64arr(shape),
            _i64arr(dst_strides), _i64arr(src_strides),
            int(dst_off), int(src_off), elemsize), "rt_copy_box")

    def _box_off(self, bd, rt, box):
        d = bd.divisions[rt.rank]
        cont = self._cont(bd)
        cs = cont.stride()
        pads = rt.lo_pads(bd)
        off = 0
        for i in range(len(bd.shape)):
            off += (int(box[0, i] - d[0, i]) + pads[i]) * cs[i]
        return off, cs

    def copy_container_to_temp(self, bd, rt, part_box, vname, need_box):
        cont = self._cont(bd)
        tmp = self.temps[vname]
        shape = box_shape(part_box)
        src_off, cs = self._box_off(bd, rt, part_box)
        lo = part_box[0] - need_box[0]
        ts = tmp.stride()
        dst_off = sum(int(lo[i]) * ts[i] for i in range(len(shape)))
        self._copy(tmp, ts, dst_off, cont, cs, src_off, shape,
                   cont.element_size())

    def pack_box(self, bd, rt, box):
        cont = self._cont(bd)
        shape = box_shape(box)
        msg = self.torch.empty(shape, dtype=cont.dtype, device="cuda")
        src_off, cs = self._box_off(bd, rt, box)
        self._copy(msg, msg.stride(), 0, cont, cs, src_off, shape,
                   cont.element_size())
        return msg

    def new_message_buffer(self, shape, dtype):
        return self.torch.empty(shape, dtype=self._tdt(dtype), device="cuda")

    def exchange_begin(self, sends, recvs):
        """POST the pairwise exchange; returns a token for
        exchange_finish.  On the nccl(=RCCL) path this is non-blocking on
        both host and device: torch orders the RCCL stream after the
        current stream (our pack kernels), so interior compute launched
        after this call overlaps the wire time.  The gloo path stages
        through host buffers (tests only)."""
        if not sends and not recvs:
            return None
        dist = self._d()
        staged = getattr(self, "_pg", "nccl") != "nccl"
        hrecvs = None
        if staged:
            self.torch.cuda.synchronize()  # packs -> host staging copies
            sends = [(d, b.cpu()) for (d, b) in sends]
            hrecvs = [(s, b, b.cpu()) for (s, b) in recvs]
        ops = []
        for (dst, buf) in sends:
            ops.append(dist.P2POp(dist.isend, buf, dst))
        if staged:
            for (src, _, hb) in hrecvs:
                ops.append(dist.P2POp(dist.irecv, hb, src))
        else:
            for (src, buf) in recvs:
                ops.append(dist.P2POp(dist.irecv, buf, src))
        reqs = dist.batch_isend_irecv(ops) if ops else []
        return (reqs, hrecvs)

    def exchange_finish(self, token):
        if token is None:
            return
        reqs, hrecvs = token
        for req in reqs:
            req.wait()
        if hrecvs:
            for (_, dbuf, hb) in hrecvs:
                dbuf.copy_(hb.to(dbuf.device))

    def exchange(self, sends, recvs):
        self.exchange_finish(self.exchange_begin(sends, recvs))

    def unpack_box_to_container(self, bd, rt, box, buf):
        cont = self._cont(bd)
        shape = box_shape(box)
        dst_off, cs = self._box_off(bd, rt, box)
        self._copy(cont, cs, dst_off, buf, buf.stride(), 0, shape,
                   cont.element_size())

    def unpack_box_to_temp(self, vname, need_box, box, buf):
        tmp = self.temps[vname]
        shape = box_shape(box)
        lo = box[0] - need_box[0]
        ts = tmp.stride()
        dst_off = sum(int(lo[i]) * ts[i] for i in range(len(shape)))
        self._copy(tmp, ts, dst_off, buf, buf.stride(), 0, shape,
                   tmp.element_size())

    def box_to_numpy(self, bd, rt, box):
        return self.pack_box(bd, rt, box).cpu().numpy().astype(
            bd.dtype, copy=False).reshape(box_shape(box))

    def write_core_from_numpy(self, bd, rt, nparr):
        cont = self._cont(bd)
        src = self.torch.from_numpy(
            np.ascontiguousarray(nparr).view(
                np.uint8 if bd.dtype == np.bool_ else bd.dtype)).to("cuda")
        core = rt.core_box(bd, rt.rank)
        dst_off, cs = self._box_off(bd, rt, core)
        self._copy(cont, cs, dst_off, src, src.stride(), 0,
                   tuple(src.shape), cont.element_size())

    # -- collectives -----------------------------------------------------------

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
        dist = self._d()
        v = np.asarray(value)
        dt = v.dtype
        if kind in ("min", "max") and dt.kind == "f":
            # NCCL/RCCL MIN/MAX does not reliably propagate NaN across
            # ranks, while the in-kernel combine and NumPy do:
            # gather the per-rank partials and combine on the host
            parts = self.allgather_scalars(value, dt)
            red = np.minimum.reduce if kind == "min" else np.maximum.reduce
            return np.asarray(red(np.asarray(parts, dtype=dt)), dtype=dt)[()]
        if dt == np.bool_:
            v = v.astype(np.uint8)
        t = self.torch.from_numpy(v.reshape(1).copy())
        if getattr(self, "_pg", "nccl") == "nccl":
            t = t.to("cuda")
        dist.all_reduce(t, op=getattr(dist.ReduceOp, self._RED_MAP[kind]))
        out = t.cpu().numpy()[0]
        if dt == np.bool_:
            out = bool(out)
        return np.asarray(out, dtype=dt)[()]

    # -- kernel execution -------------------------------------------------------

    def _get_kernel(self, plan):
        # fast path: structural key without source generation
        vec, classes, anchor, key = codegen.plan_structure(plan)
        cached = self.kernels.get(key)
        if cached is not None:
            return cached
        gk = codegen.generate(plan)
        assert gk.key == key
        if int(os.environ.get("RAMBA_SHOW_CODE", "0")):
            print(f"=== kernel {gk.key} ===\n{gk.source}\n", flush=True)
        h = ctypes.c_void_p()
        self._check(self.lib.rt_kernel_get(
            gk.key.encode(), gk.source.encode(), gk.kname_main.encode(),
            ctypes.byref(h)), "rt_kernel_get")
        gk.handle = h.value
        if gk.kname_finish:
            hf = ctypes.c_void_p()
            self._check(self.lib.rt_kernel_get(
                (gk.key + "_f").encode(), gk.finish_source.encode(),
                gk.kname_finish.encode(), ctypes.byref(hf)),
                "rt_kernel_get(finish)")
            gk.finish_handle = hf.value
        self.kernels[gk.key] = gk
        return gk

    def launch(self, plan, recipe=None):
        from .common import ntiming, add_time
        import time as _time
        _t0 = _time.perf_counter() if ntiming else 0.0
        if recipe is not None and recipe.backend_kernel is not None:
            gk = recipe.backend_kernel
        else:
            gk = self._maybe_load_tiled(plan) or self._get_kernel(plan)
            if recipe is not None:
                recipe.backend_kernel = gk
        if isinstance(gk, tuple) and gk[0] == "lt":
            r = self._launch_load_tiled(gk, plan)
            if ntiming:
                add_time("hb_submit", _time.perf_counter() - _t0)
            return r
        nd = gk.nd
        shape = plan.itershape
        V = gk.vec
        nx = shape[nd - 1]
        cap = int(os.environ.get("RAMBA_GRID_CAP", "32768"))
        gx = max(1, min(cap, (nx + 256 * V - 1) // (256 * V)))
        yb = getattr(gk, "yblock", 1)
        gy = max(1, min(8192, (shape[nd - 2] + yb - 1) // yb)) \
            if nd >= 2 else 1
        if nd == 3:
            gz = max(1, min(64, shape[0]))
        elif nd >= 4:
            gz = max(1, min(64, shape[0] * shape[1]))
        else:
            gz = 1
        if gk.nred:
            # bound the partials array: the finish kernel is one block
            gy = max(1, min(gy, 8192 // gx))
            gz = max(1, min(gz, max(1, 8192 // (gx * gy))))

        partials = None
        outs = []
        np_partials = gx * gy * gz
        if gk.nred:
            partials = self.torch.empty(gk.nred * np_partials,
                                        dtype=self.torch.float64,
                                        device="cuda")
            for spec in plan.reductions:
                outs.append(self.torch.empty(
                    1, dtype=self._tdt(spec.dtype), device="cuda"))

        def ptr_of(name):
            if name == "__partials__":
                return partials.data_ptr()
            if name == "__npartials__":
                return np_partials
            if name in self.temps:
                return self.temps[name].data_ptr()
            op = next(o for o in plan.operands if o.name == name)
            return self._cont(op.bd).data_ptr()

        args = codegen.pack_args(gk, plan, ptr_of)
        if ntiming:
            add_time("hb_prep", _time.perf_counter() - _t0)
            _t0 = _time.perf_counter()
        stream = self._stream()
        if self.time_kernels:
            ev0 = self.torch.cuda.Event(enable_timing=True)
            ev1 = self.torch.cuda.Event(enable_timing=True)
            ev0.record()
        self._check(self.lib.rt_launch(
            ctypes.c_void_p(gk.handle), gx, gy, gz, 256, stream, args,
            len(args)), "rt_launch")
        if self.time_kernels:
            ev1.record()
            ev1.synchronize()
            self.kernel_times_ms.append(ev0.elapsed_time(ev1))
            self.kernel_keys.append(gk.key)
        results = []
        if gk.nred:
            fargs = codegen.pack_finish_args(
                partials.data_ptr(), np_partials,
                [o.data_ptr() for o in outs])
            self._check(self.lib.rt_launch(
                ctypes.c_void_p(gk.finish_handle), 1, 1, 1, 256, stream,
                fargs, len(fargs)), "rt_launch(finish)")
            for spec, o in zip(plan.reductions, outs):
                v = o.cpu().numpy()[0]
                results.append(np.asarray(v, dtype=spec.dtype)[()])
        if ntiming:
            add_time("hb_submit", _time.perf_counter() - _t0)
        return results

    def sync(self):
        self.torch.cuda.synchronize()


# ---------------------------------------------------------------------------
# axis reductions: local strided-reduce kernel + typed
# combining box merges (rt_combine_box)
# ---------------------------------------------------------------------------

_DT_ENUM = {"float64": 0, "float32": 1, "int64": 2, "int32": 3, "int16": 4,
            "int8": 5, "uint8": 6, "bool": 6}
_OP_ENUM = {"sum": 0, "prod": 1, "min": 2, "max": 3, "all": 4, "any": 5}


def _hb_fill_container(self, bd, rt, value):
    c = self.containers.get(bd.gid)
    if c is not None:
        c.fill_(float(value) if c.is_floating_point() else int(value))


def _hb_axred_kernel(self, nd, axes, in_dtype, out_dtype, kind,
                     chunked=False):
    from . import codegen as cg
    ck = ("axred", nd, tuple(sorted(axes)), str(in_dtype), str(out_dtype),
          kind, chunked)
    cached = self.kernels.get(ck)
    if cached is None:
        key, source, kname, fields, lane_split = cg.generate_axis_reduce(
            nd, axes, in_dtype, out_dtype, kind, chunked=chunked)
        h = ctypes.c_void_p()
        self._check(self.lib.rt_kernel_get(
            key.encode(), source.encode(), kname.encode(), ctypes.byref(h)),
            "rt_kernel_get(axred)")
        cached = (h.value, fields, lane_split)
        self.kernels[ck] = cached
    return cached


def _hb_axred_launch(self, handle, fields, lane_split, out_extents,
                     red_extents, in_ptr, in_off, in_strides, out_ptr,
                     nout, chunk_spec=None):
    from . import codegen as cg
    args = cg.pack_axis_reduce_args(fields, out_extents, red_extents,
                                    in_ptr, in_off, in_strides, out_ptr,
                                    chunk_spec=chunk_spec)
    if lane_split:
        gx = max(1, min(4096, (nout * 64 + 255) // 256))
    else:
        gx = max(1, min(4096, (nout + 255) // 256))
    self._check(self.lib.rt_launch(
        ctypes.c_void_p(handle), gx, 1, 1, 256, self._stream(), args,
        len(args)), "rt_launch(axred)")


def _hb_axis_reduce_partial(self, bd, off0, strides, lb, axes, kind,
                            out_dtype):
    from .shardview import box_shape as _bs
    nd = lb.shape[1]
    axes = tuple(sorted(axes))
    shape = _bs(lb)
    kd_shape = tuple(1 if d in axes else shape[d] for d in range(nd))
    out_extents = {d: shape[d] for d in range(nd) if d not in axes}
    red_extents = {d: shape[d] for d in axes}
    nout = 1
    for v in out_extents.values():
        nout *= v
    lane_split = (nd - 1) in axes
    K = red_extents.get(axes[0], 1) if axes else 1

    # small-nout, large-K, non-lane-split reductions (sum(axis=0)) starve
    # the chip at one thread per out element: chunk the reduced axis for
    # a parallel stage 1, then reduce the chunks (measured 320 GB/s ->
    # streaming rate)
    if (not lane_split and len(axes) == 1 and nout < 65536 and K >= 2048
            and nout >= 1):
        C = max(1, min((1 << 22) // max(nout, 1), (K + 255) // 256, 4096))
    else:
        C = 1
    out_t = self.torch.empty(kd_shape, dtype=self._tdt(out_dtype),
                             device="cuda")
    self.temps["__axred__"] = out_t
    if C > 1:
        clen = (K + C - 1) // C
        part = self.torch.empty((C, nout), dtype=self._tdt(out_dtype),
                                device="cuda")
        self.temps["__axred_s1__"] = part
        h1, f1, _ = self._hb_axred_kernel(nd, axes, bd.dtype, out_dtype,
                                          kind, chunked=True)
        self._hb_axred_launch(h1, f1, False, out_extents, red_extents,
                              self._cont(bd).data_ptr(), off0, strides,
                              part.data_ptr(), nout * C,
                              chunk_spec=(C, clen, K))
        # stage 2: reduce the chunk axis of the contiguous (C, nout) temp
        h2, f2, ls2 = self._hb_axred_kernel(2, (0,), out_dtype, out_dtype,
                                            kind)
        self._hb_axred_launch(h2, f2, ls2, {1: nout}, {0: C},
                              part.data_ptr(), 0, (nout, 1),
                              out_t.data_ptr(), nout)
        return
    handle, fields, lane_split = self._hb_axred_kernel(
        nd, axes, bd.dtype, out_dtype, kind)
    self._hb_axred_launch(handle, fields, lane_split, out_extents,
                          red_extents, self._cont(bd).data_ptr(), off0,
                          strides, out_t.data_ptr(), nout)


def _hb_pack_temp_box(self, vname, rel_box):
    from .shardview import box_shape as _bs
    t = self.temps[vname]
    shape = _bs(rel_box)
    msg = self.torch.empty(shape, dtype=t.dtype, device="cuda")
    ts = t.stride()
    src_off = sum(int(rel_box[0, i]) * ts[i] for i in range(len(shape)))
    self._copy(msg, msg.stride(), 0, t, ts, src_off, shape,
               t.element_size())
    return msg


def _hb_combine(self, dst_t, dst_strides, dst_off, src_t, shape, np_dtype,
                kind):
    self._check(self.lib.rt_combine_box(
        self._stream(), ctypes.c_void_p(dst_t.data_ptr()),
        ctypes.c_void_p(src_t.data_ptr()), len(shape), _i64arr(shape),
        _i64arr(dst_strides), _i64arr(src_t.stride()), int(dst_off), 0,
        _DT_ENUM[str(np_dtype)], _OP_ENUM[kind]), "rt_combine_box")


def _hb_combine_box_into_container(self, bd, rt, box, buf, kind):
    from .shardview import box_shape as _bs
    dst_off, cs = self._box_off(bd, rt, box)
    self._combine(self._cont(bd), cs, dst_off, buf, _bs(box), bd.dtype,
                  kind)


def _hb_combine_temp_into_container(self, bd, rt, box, vname, rel_box, kind):
    from .shardview import box_shape as _bs
    t = self.temps[vname]
    shape = _bs(rel_box)
    # pack the temp slice contiguous first (combine kernel takes any
    # strides, but reuse the packed path for simplicity)
    sl = self.pack_temp_box(vname, rel_box)
    dst_off, cs = self._box_off(bd, rt, box)
    self._combine(self._cont(bd), cs, dst_off, sl, shape, bd.dtype, kind)


HipBackend.fill_container = _hb_fill_container
HipBackend._hb_axred_kernel = _hb_axred_kernel
HipBackend._hb_axred_launch = _hb_axred_launch
HipBackend.axis_reduce_partial = _hb_axis_reduce_partial
HipBackend.pack_temp_box = _hb_pack_temp_box
HipBackend._combine = _hb_combine
HipBackend.combine_box_into_container = _hb_combine_box_into_container
HipBackend.combine_temp_into_container = _hb_combine_temp_into_container


# -- cumsum --------------------------------------------------

_CS_DT = {"float64": 0, "float32": 1, "int64": 2, "int32": 3}
_SCAN_CHUNK = 4096


def _hb_cumsum_src(self, bd, off0, stride, n, out_dtype):
    cont = self._cont(bd)
    src = cont
    if np.dtype(bd.dtype) != np.dtype(out_dtype):
        # pack (handles negative strides) then cast
        tmp = self.torch.empty(n, dtype=cont.dtype, device="cuda")
        self._copy(tmp, (1,), 0, cont, (stride,), off0, (n,),
                   cont.element_size())
        src = tmp.to(self._tdt(out_dtype))
        off0, stride = 0, 1
        self.temps["__cs_src__"] = src
    return src, off0, stride


def _hb_cumsum_local_phase12(self, bd, off0, stride, n, out_dtype):
    if str(np.dtype(out_dtype)) not in _CS_DT:
        raise NotImplementedError(f"cumsum dtype {out_dtype}")
    dt = _CS_DT[str(np.dtype(out_dtype))]
    src, off0, stride = self._hb_cumsum_src(bd, off0, stride, n, out_dtype)
    if self.rt.world == 1 and os.environ.get("RAMBA_CUMSUM", "3pass") \
            == "lookback":
        # REJECTED-by-measurement alternative, kept for reference: the
        # single-pass decoupled-lookback scan (rt_cumsum_scan) moves
        # 16 B/elem instead of 24 but its cross-chunk agent-atomic chain
        # costs more than the saved read pass — same-box A/B at 5e8 i64:
        # lookback 3.81 ms vs 3-phase 2.93 ms (hand-off pricing, CDNA4
        # guide §price list).  Default stays 3-phase.
        self.temps["__cs_state__"] = (src, off0, stride, n, 0, dt)
        return np.asarray(0, dtype=out_dtype)[()]
    nblocks = max(1, (n + _SCAN_CHUNK - 1) // _SCAN_CHUNK)
    bsums = self.torch.empty(nblocks, dtype=self._tdt(out_dtype),
                             device="cuda")
    total = self.torch.empty(1, dtype=self._tdt(out_dtype), device="cuda")
    self.temps["__cs_bsums__"] = bsums
    self.temps["__cs_state__"] = (src, off0, stride, n, nblocks, dt)
    rc = self.lib.rt_cumsum(
        self._stream(), ctypes.c_void_p(src.data_ptr()), off0, stride, n,
        None, 0, ctypes.c_void_p(bsums.data_ptr()),
        min(nblocks, 4096), None, 0.0, 0, dt, 1)
    self._check(rc, "rt_cumsum(1)")
    rc = self.lib.rt_cumsum(
        self._stream(), None, 0, 0, 0, None, 0,
        ctypes.c_void_p(bsums.data_ptr()), nblocks,
        ctypes.c_void_p(total.data_ptr()), 0.0, 0, dt, 2)
    self._check(rc, "rt_cumsum(2)")
    return np.asarray(total.cpu().numpy()[0], dtype=out_dtype)[()]


def _hb_cumsum_local_phase3(self, bd, off0, stride, n, out_bd, out_off,
                            offset, out_dtype):
    src, off0_, stride_, n_, nblocks, dt = self.temps["__cs_state__"]
    out = self._cont(out_bd)
    fb = float(offset) if np.dtype(out_dtype).kind == "f" else 0.0
    ib = int(offset) if np.dtype(out_dtype).kind != "f" else 0
    if self.rt.world == 1 and nblocks == 0:
        nchunks = max(1, (n_ + 8191) // 8192)
        # flags + ticket zeroed every call (G16 "Re-initialise every call")
        ws = self.torch.zeros(2 * nchunks + nchunks + 1,
                              dtype=self.torch.int64, device="cuda")
        agg = ws.data_ptr()
        inc = agg + 8 * nchunks
        flag = inc + 8 * nchunks
        ticket = flag + 4 * nchunks
        rc = self.lib.rt_cumsum_scan(
            self._stream(), ctypes.c_void_p(src.data_ptr()), off0_, stride_,
            n_, ctypes.c_void_p(out.data_ptr()), out_off,
            ctypes.c_void_p(agg), ctypes.c_void_p(inc),
            ctypes.c_void_p(flag), ctypes.c_void_p(ticket), fb, ib, dt)
        self._check(rc, "rt_cumsum_scan")
        return
    bsums = self.temps["__cs_bsums__"]
    rc = self.lib.rt_cumsum(
        self._stream(), ctypes.c_void_p(src.data_ptr()), off0_, stride_, n_,
        ctypes.c_void_p(out.data_ptr()), out_off,
        ctypes.c_void_p(bsums.data_ptr()), min(nblocks, 4096), None,
        fb, ib, dt, 3)
    self._check(rc, "rt_cumsum(3)")


def _hb_allgather_scalars(self, val, dtype):
    if self.rt.world == 1:
        return [val]
    dist = self._d()
    t = self.torch.from_numpy(np.asarray([val], dtype=dtype))
    if getattr(self, "_pg", "nccl") == "nccl":
        t = t.to("cuda")
    outs = [self.torch.empty_like(t) for _ in range(self.rt.world)]
    dist.all_gather(outs, t)
    return [np.asarray(o.cpu().numpy()[0], dtype=dtype)[()] for o in outs]


HipBackend._hb_cumsum_src = _hb_cumsum_src
HipBackend.cumsum_local_phase12 = _hb_cumsum_local_phase12
HipBackend.cumsum_local_phase3 = _hb_cumsum_local_phase3
HipBackend.allgather_scalars = _hb_allgather_scalars


# -- boolean-mask compaction (a[mask]) ------------------------

_MC_DT = {"float64": 0, "float32": 1, "int64": 2, "int32": 3,
          "int16": 4, "int8": 5, "uint8": 6, "bool": 6}


def _hb_mask_compact(self, bd_a, bd_m, rt):
    """Local ordered compaction of the rank's core box: returns
    (dense device tensor of selected elements, count)."""
    d, _, cstr_a, pads_a = rt.shard_geometry(bd_a)
    if d is None:
        return None, 0
    _, _, cstr_m, pads_m = rt.shard_geometry(bd_m)
    nd = len(bd_a.shape)
    shape = [int(d[1, i] - d[0, i] + 1) for i in range(nd)]
    n = 1
    for sz in shape:
        n *= sz
    if n == 0:
        return None, 0
    ca, cm = self._cont(bd_a), self._cont(bd_m)
    off_a = sum(pads_a[i] * cstr_a[i] for i in range(nd))
    off_m = sum(pads_m[i] * cstr_m[i] for i in range(nd))
    a_ptr = ca.data_ptr() + off_a * ca.element_size()
    m_ptr = cm.data_ptr() + off_m * cm.element_size()
    dtc = _MC_DT[str(np.dtype(bd_a.dtype))]
    _MC_CHUNK = 2048     # must match MC_CHUNK in ramba_rt.cpp
    nchunks = max(1, (n + _MC_CHUNK - 1) // _MC_CHUNK)
    bcounts = self.torch.empty(nchunks, dtype=self.torch.int64,
                               device="cuda")
    total = self.torch.empty(1, dtype=self.torch.int64, device="cuda")
    st = self._stream()
    rc = self.lib.rt_mask_compact(
        st, ctypes.c_void_p(a_ptr), ctypes.c_void_p(m_ptr), None, nd,
        _i64arr(shape), _i64arr(cstr_a), _i64arr(cstr_m),
        ctypes.c_void_p(bcounts.data_ptr()), nchunks, dtc, 1)
    self._check(rc, "rt_mask_compact(1)")
    rc = self.lib.rt_cumsum(
        st, None, 0, 0, 0, None, 0, ctypes.c_void_p(bcounts.data_ptr()),
        nchunks, ctypes.c_void_p(total.data_ptr()), 0.0, 0, 2, 2)
    self._check(rc, "rt_mask_compact(2)")
    count = int(total.cpu().numpy()[0])   # D2H syncs the stream
    if count == 0:
        return None, 0
    out = self.torch.empty(count, dtype=ca.dtype, device="cuda")
    rc = self.lib.rt_mask_compact(
        st, ctypes.c_void_p(a_ptr), ctypes.c_void_p(m_ptr),
        ctypes.c_void_p(out.data_ptr()), nd, _i64arr(shape),
        _i64arr(cstr_a), _i64arr(cstr_m),
        ctypes.c_void_p(bcounts.data_ptr()), nchunks, dtc, 3)
    self._check(rc, "rt_mask_compact(3)")
    return out, count


def _hb_write_local_dense(self, out_bd, rt, local):
    """Write a dense 1-D local tensor into the rank's container core."""
    _, _, _, pads = rt.shard_geometry(out_bd)
    cont = self._cont(out_bd)
    cont[pads[0]:pads[0] + local.shape[0]] = local


HipBackend.mask_compact = _hb_mask_compact
HipBackend.write_local_dense = _hb_write_local_dense


# -- axis-wise cumsum (N-D scumulative) -----------------------


def _hb_axis_scan_local(self, bd_in, off0, strides, lshape, axis, out_bd,
                        out_off, out_strides):
    """Local inclusive scan along `axis` over the rank's box; line totals
    land in temps["__axcs_tot__"] (dense C-order line-space tensor)."""
    dt = _CS_DT[str(np.dtype(out_bd.dtype))]
    ca, co = self._cont(bd_in), self._cont(out_bd)
    nd = len(lshape)
    lines_shape = tuple(lshape[d] for d in range(nd) if d != axis)
    tot = self.torch.empty(lines_shape, dtype=co.dtype, device="cuda")
    self.temps["__axcs_tot__"] = tot
    in_ptr = ca.data_ptr() + off0 * ca.element_size()
    out_ptr = co.data_ptr() + out_off * co.element_size()
    # axis != last with few lines: single thread-per-line starves the
    # chip -> chunked 3-pass variant (k1 local / k2 offsets / k3 apply)
    nlines = 1
    for d in range(nd):
        if d != axis:
            nlines *= lshape[d]
    length = lshape[axis]
    nchunks, tot2_ptr = 1, None
    if axis != nd - 1 and nlines < 262144 and length >= 64:
        # enough chunks to fill the chip, but keep >=32 elems per chunk
        nchunks = min(-(-524288 // max(1, nlines)), -(-length // 32))
        if nchunks > 1:
            tot2 = self.torch.empty(nchunks * nlines, dtype=co.dtype,
                                    device="cuda")
            self.temps["__axcs_tot2__"] = tot2
            tot2_ptr = ctypes.c_void_p(tot2.data_ptr())
    rc = self.lib.rt_axis_scan(
        self._stream(), ctypes.c_void_p(in_ptr), ctypes.c_void_p(out_ptr),
        ctypes.c_void_p(tot.data_ptr()), nd, _i64arr(lshape),
        _i64arr(strides), _i64arr(out_strides), axis, dt,
        tot2_ptr, nchunks if tot2_ptr else 1)
    self._check(rc, "rt_axis_scan")


def _hb_axcs_init_offsets(self, lines_shape, dtype):
    self._axcs_np_dtype = np.dtype(dtype)
    self.temps["__axcs_off__"] = self.torch.zeros(
        lines_shape if lines_shape else (1,), dtype=self._tdt(dtype),
        device="cuda")


def _hb_axcs_accumulate(self, rel_box, buf):
    offs = self.temps["__axcs_off__"]
    from .shardview import box_shape as _bs
    shape = _bs(rel_box)
    ts = offs.stride()
    dst_off = sum(int(rel_box[0, i]) * ts[i] for i in range(len(shape)))
    self._check(self.lib.rt_combine_box(
        self._stream(), ctypes.c_void_p(offs.data_ptr()),
        ctypes.c_void_p(buf.data_ptr()), len(shape), _i64arr(shape),
        _i64arr(ts), _i64arr(buf.stride()), dst_off, 0,
        _DT_ENUM[str(self._axcs_np_dtype)], _OP_ENUM["sum"]),
        "axcs_accumulate")


def _hb_axcs_apply(self, out_bd, rt, box, axis):
    """out[box] += offsets, broadcast along the scan axis (src stride 0)."""
    offs = self.temps["__axcs_off__"]
    from .shardview import box_shape as _bs
    cont = self._cont(out_bd)
    dst_off, cs = self._box_off(out_bd, rt, box)
    shape = _bs(box)
    ostr = list(offs.stride())
    src_strides = []
    j = 0
    for d in range(len(shape)):
        if d == axis:
            src_strides.append(0)
        else:
            src_strides.append(ostr[j])
            j += 1
    self._check(self.lib.rt_combine_box(
        self._stream(), ctypes.c_void_p(cont.data_ptr()),
        ctypes.c_void_p(offs.data_ptr()), len(shape), _i64arr(shape),
        _i64arr(cs), _i64arr(src_strides), dst_off, 0,
        _DT_ENUM[str(np.dtype(out_bd.dtype))], _OP_ENUM["sum"]),
        "axcs_apply")


HipBackend.axis_scan_local = _hb_axis_scan_local
HipBackend.axcs_init_offsets = _hb_axcs_init_offsets
HipBackend.axcs_accumulate = _hb_axcs_accumulate
HipBackend.axcs_apply = _hb_axcs_apply


# -- reshape flat gather/scatter ---------------------------------------------


def _hb_flat_gather(self, cont, off0, strides, shape, flat0, n):
    """Dense device buffer = C-order flat subrange [flat0, flat0+n) of the
    strided box at cont[off0...]."""
    dense = self.torch.empty(n, dtype=cont.dtype, device="cuda")
    p = cont.data_ptr() + off0 * cont.element_size()
    self._check(self.lib.rt_flat_copy(
        self._stream(), ctypes.c_void_p(p),
        ctypes.c_void_p(dense.data_ptr()), len(shape), _i64arr(shape),
        _i64arr(strides), int(flat0), int(n), cont.element_size(), 0),
        "rt_flat_copy(gather)")
    return dense


def _hb_flat_scatter(self, cont, off0, strides, shape, flat0, buf):
    p = cont.data_ptr() + off0 * cont.element_size()
    self._check(self.lib.rt_flat_copy(
        self._stream(), ctypes.c_void_p(p),
        ctypes.c_void_p(buf.data_ptr()), len(shape), _i64arr(shape),
        _i64arr(strides), int(flat0), int(buf.numel()),
        cont.element_size(), 1), "rt_flat_copy(scatter)")


HipBackend.flat_gather = _hb_flat_gather
HipBackend.flat_scatter = _hb_flat_scatter


# -- cross-stage fusion (staged/tiled kernel; ramba_amd/staged.py) -----------

HipBackend.supports_staged = True


def _hb_container_addr(self, bd):
    return self._cont(bd).data_ptr()


def _hb_tiled_kernel(self, desc):
    """Build (or fetch) the staged/tiled kernel; returns a handle token
    or None (fall back to sequential)."""
    from . import codegen as cg
    key = cg.staged_tiled_key(desc)
    ck = ("tiled", key)
    cached = self.kernels.get(ck)
    if cached is None:
        try:
            key2, source, kname, fields = cg.generate_staged_tiled(desc)
        except NotImplementedError:
            return None
        assert key2 == key
        if int(os.environ.get("RAMBA_SHOW_CODE", "0")):
            print(f"=== tiled kernel {key} ===\n{source}\n", flush=True)
        h = ctypes.c_void_p()
        self._check(self.lib.rt_kernel_get(
            key.encode(), source.encode(), kname.encode(),
            ctypes.byref(h)), "rt_kernel_get(tiled)")
        cached = (h.value, fields, key)
        self.kernels[ck] = cached
    return cached


def _hb_tiled_launch(self, handle, vals, ntiles, red_dtypes=None,
                     rec=None):
    from . import codegen as cg
    h, fields, tkkey = handle
    gx = max(1, min(int(os.environ.get("RAMBA_GRID_CAP", "32768")),
                    ntiles))
    red_dtypes = red_dtypes or []
    if rec is not None and rec.partials is not None \
            and len(rec.partials) == len(red_dtypes) \
            and all(t.shape[0] == gx for t in rec.partials):
        parts = rec.partials
        for ri, t in enumerate(parts):
            vals[f"red{ri}_ptr"] = t.data_ptr()
    else:
        parts = []
        for ri, dt in enumerate(red_dtypes):
            t = self.torch.empty(gx, dtype=self._tdt(dt), device="cuda")
            parts.append(t)
            vals[f"red{ri}_ptr"] = t.data_ptr()
        if rec is not None:
            rec.partials = parts
    args = cg.pack_tk_args(fields, vals)
    if self.time_kernels:
        ev0 = self.torch.cuda.Event(enable_timing=True)
        ev1 = self.torch.cuda.Event(enable_timing=True)
        ev0.record()
    self._check(self.lib.rt_launch(
        ctypes.c_void_p(h), gx, 1, 1, 256, self._stream(), args,
        len(args)), "rt_launch(tiled)")
    if self.time_kernels:
        ev1.record()
        ev1.synchronize()
        self.kernel_times_ms.append(ev0.elapsed_time(ev1))
        self.kernel_keys.append(tkkey)
    out = []
    for t, dt in zip(parts, red_dtypes):
        total = self.torch.empty(1, dtype=t.dtype, device="cuda")
        rc = self.lib.rt_cumsum(
            self._stream(), None, 0, 0, 0, None, 0,
            ctypes.c_void_p(t.data_ptr()), gx,
            ctypes.c_void_p(total.data_ptr()), 0.0, 0,
            _CS_DT[str(np.dtype(dt))], 2)
        self._check(rc, "rt_cumsum(tk partials)")
        out.append(np.asarray(total.cpu().numpy()[0], dtype=dt)[()])
    return out


HipBackend.container_addr = _hb_container_addr
HipBackend.tiled_kernel = _hb_tiled_kernel
HipBackend.tiled_launch = _hb_tiled_launch


# -- load-tiled stencil dispatch -------------------------


def _hb_maybe_load_tiled(self, plan):
    """If the plan is a qualifying 2-D stencil (>=3 same-array shifted
    readers), build the LDS load-tiled kernel instead of the vectorized
    elementwise one.  Returns ("lt", handle, fields, fam_meta, others)
    or None."""
    # measured on MI355X (profiles/README r02): at 16x256 tiles the LDS
    # kernel cuts HBM FETCH 1.56x -> 1.21-1.27x of algorithmic AND beats
    # the vectorized cross-row-sharing kernel at PRK scale (30000^2:
    # 1.562 vs 1.599 ms; 4096^2 par).  Default ON; =0 selects the
    # vectorized kernel for A/B.  (The first 32x256 tiling lost to
    # occupancy: 36 KB LDS -> 2.24 ms.)
    if os.environ.get("RAMBA_STENCIL_LDS", "1") == "0":
        return None
    if plan.reductions or len(plan.itershape) != 2:
        return None
    if plan.itershape[0] * plan.itershape[1] < (1 << 16):
        return None         # rim slabs / tiny boxes: vectorized path wins
    from . import codegen as cg
    fams = cg.find_stencil_families(plan)
    if not fams:
        return None
    key, source, kname, fields, tile = cg.generate_load_tiled(plan, fams)
    ck = ("lt", key)
    cached = self.kernels.get(ck)
    if cached is None:
        if int(os.environ.get("RAMBA_SHOW_CODE", "0")):
            print(f"=== load-tiled kernel {key} ===\n{source}\n",
                  flush=True)
        h = ctypes.c_void_p()
        self._check(self.lib.rt_kernel_get(
            key.encode(), source.encode(), kname.encode(),
            ctypes.byref(h)), "rt_kernel_get(load_tiled)")
        cached = (h.value, fields)
        self.kernels[ck] = cached
    fam_meta = [(rep, anchor, s0) for (anchor, s0, dt, mem, rep) in fams]
    member_names = {n for (a, s, dt, mem, rep) in fams for (n, _, _) in mem}
    others = [op.name for op in plan.operands
              if op.name not in member_names]
    return ("lt", cached[0], cached[1], fam_meta, others, tile, key)


def _hb_launch_load_tiled(self, gk, plan):
    from . import codegen as cg
    _, handle, fields, fam_meta, others, tile, ltkey = gk
    opmap = {o.name: o for o in plan.operands}
    vals = {"n0": plan.itershape[0], "n1": plan.itershape[1],
            "gs0": plan.global_start[0], "gs1": plan.global_start[1]}
    for fi, (rep, anchor, s0) in enumerate(fam_meta):
        op = opmap[rep]
        vals[f"fam{fi}_ptr"] = self._cont(op.bd).data_ptr() \
            if op.kind == "container" else self.temps[op.name].data_ptr()
        vals[f"fam{fi}_off"] = anchor
        vals[f"fam{fi}_s0"] = s0
    for name in others:
        op = opmap[name]
        vals[f"{name}_ptr"] = self._cont(op.bd).data_ptr() \
            if op.kind == "container" else self.temps[name].data_ptr()
        vals[f"{name}_off"] = op.offset0
        vals[f"{name}_s0"] = op.strides[0]
        vals[f"{name}_s1"] = op.strides[1]
    for n, (v, dt) in plan.scalars.items():
        vals[n] = v
    args = cg.pack_tk_args(fields, vals)
    th, cw, seg = tile
    tiles0 = (plan.itershape[0] + th - 1) // th
    ntiles = ((tiles0 + seg - 1) // seg) \
        * ((plan.itershape[1] + cw - 1) // cw)
    gx = max(1, min(int(os.environ.get("RAMBA_GRID_CAP", "32768")),
                    ntiles))
    if self.time_kernels:
        ev0 = self.torch.cuda.Event(enable_timing=True)
        ev1 = self.torch.cuda.Event(enable_timing=True)
        ev0.record()
    self._check(self.lib.rt_launch(
        ctypes.c_void_p(handle), gx, 1, 1, 256, self._stream(), args,
        len(args)), "rt_launch(load_tiled)")
    if self.time_kernels:
        ev1.record()
        ev1.synchronize()
        self.kernel_times_ms.append(ev0.elapsed_time(ev1))
        self.kernel_keys.append(ltkey)
    return []


HipBackend._maybe_load_tiled = _hb_maybe_load_tiled
HipBackend._launch_load_tiled = _hb_launch_load_tiled
