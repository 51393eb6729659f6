/* ramba_rt — C-ABI runtime for the MI355X-native Ramba hot-path backend.
 *
 * This is the drop-in boundary of SURVEY.md §8(b): the Python host lowers
 * each fused deferred-op group (the payload of the reference's
 * remote_exec_all("run_deferred_ops", ...) call, ramba/ramba.py:8286-8298)
 * to generated gfx950 HIP source + a packed argument buffer, and drives it
 * through these entry points.  No torch types cross this boundary — only
 * plain pointers and sizes.
 *
 * Reference interfaces each entry point replaces (file:line in the
 * reference repo Python-for-HPC/ramba):
 *   rt_init           — worker startup / device binding (ramba.py:10646-10725)
 *   rt_kernel_get     — FunctionMetadata JIT + sha-keyed code cache
 *                       (ramba.py:249-438, fname hash at 8260)
 *   rt_launch         — the fused-loop invocation func(global_start,
 *                       itershape, ...) (ramba.py:3768/3779)
 *   rt_copy_box       — shard sub-box pack/unpack for the part/halo
 *                       exchange (comm_queues puts at ramba.py:3656,
 *                       getborder ramba.py:1260) and gather (get_view 2160)
 *   rt_stream_sync /
 *   rt_device_sync    — worker-side completion (sync, ramba.py:9843)
 *   rt_event_*        — hot-loop timing (add_time, ramba.py:945-1022)
 *
 * Python-side binding a maintainer would write: see INTEGRATION.md
 * (ctypes stub).
 */

#ifndef RAMBA_RT_H
#define RAMBA_RT_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* All functions return 0 on success, nonzero on failure; rt_last_error()
 * returns a static string describing the last failure (fail-fast contract —
 * the analog of the reference's ("ERROR", worker, traceback) reply,
 * ramba.py:3878-3881). */

int         rt_init(int device);
int         rt_device_count(void);
const char *rt_last_error(void);

/* Compile (hiprtc, gfx950) and cache a kernel. `key` is the host-computed
 * structural hash of the fused group (the analog of the sha256 code hash at
 * ramba.py:8260); `source` is full HIP source; `kname` the extern-C kernel
 * symbol.  Returns a handle valid for the process lifetime. */
int rt_kernel_get(const char *key, const char *source, const char *kname,
                  void **out_kernel);

/* Compile-only validation of generated source (no GPU needed). */
int rt_compile_check(const char *source);

/* Launch a compiled kernel with a packed argument buffer (passed via
 * HIP_LAUNCH_PARAM_BUFFER_POINTER).  block is (bx,1,1). */
int rt_launch(void *kernel, unsigned gx, unsigned gy, unsigned gz,
              unsigned bx, uintptr_t stream, const void *args,
              size_t argsize);

/* Strided box copy between device buffers (pack/unpack/halo/gather).
 * shape/strides are in elements, nd <= 4, elemsize in {1,2,4,8}. */
int rt_copy_box(uintptr_t stream, void *dst, const void *src, int nd,
                const int64_t *shape, const int64_t *dst_strides,
                const int64_t *src_strides, int64_t dst_off, int64_t src_off,
                int elemsize);

/* Typed combining box copy: dst = dst OP src (axis-reduction merge).
 * dtype: 0=f64 1=f32 2=i64 3=i32 4=i16 5=i8 6=u8;
 * op: 0=add 1=mul 2=min 3=max 4=logical_and 5=logical_or. */
int rt_combine_box(uintptr_t stream, void *dst, const void *src, int nd,
                   const int64_t *shape, const int64_t *dst_strides,
                   const int64_t *src_strides, int64_t dst_off,
                   int64_t src_off, int dtype, int op);

/* cumsum phases (local scan; cross-rank offset fixed up by the host):
 * phase 1: block sums into `bsums[nblocks]`; phase 2: one-block exclusive
 * scan of bsums in place + grand total into `total`; phase 3: apply with
 * `fbase`/`ibase` (the rank's global offset).  dtype: 0=f64 1=f32 2=i64
 * 3=i32. */
int rt_cumsum(uintptr_t stream, const void *in, int64_t in_off,
              int64_t in_stride, int64_t n, void *out, int64_t out_off,
              void *bsums, int64_t nblocks, void *total, double fbase,
              int64_t ibase, int dtype, int phase);

/* single-pass decoupled-lookback cumsum (one read + one write per
 * element; the chained cross-workgroup hand-off follows the CDNA4
 * guide's agent-atomic discipline).  agg/inc: u64[nchunks]; flag:
 * u32[nchunks] and ticket: u32[1], both zeroed by the host before every
 * launch. */
int rt_cumsum_scan(uintptr_t stream, const void *in, int64_t in_off,
                   int64_t in_stride, int64_t n, void *out, int64_t out_off,
                   void *agg, void *inc, void *flag, void *ticket,
                   double fbase, int64_t ibase, int dtype);

/* Ordered boolean-mask compaction over the rank's local core box, C
 * iteration order (reference: the compressing boolean getitem of
 * ramba.ndarray, ramba/ramba.py maskarray path).  phase 1: per-4096-chunk
 * selected counts into bcounts (int64[nchunks]); phase 2 is
 * rt_cumsum(phase=2, dtype=2) on bcounts (exclusive scan + total);
 * phase 3: write selected elements of `a` densely into `out`.
 * dtype: 0=f64 1=f32 2=i64 3=i32 4=i16 5=i8 6=u8. */
int rt_mask_compact(uintptr_t stream, const void *a, const void *m,
                    void *out, int nd, const int64_t *shape,
                    const int64_t *a_strides, const int64_t *m_strides,
                    void *bcounts, int64_t nchunks, int dtype, int phase);

/* Flat C-order gather (scatter=0) / scatter (scatter=1) between a strided
 * local box and a dense buffer — the data movement of reshape (reference
 * flat-index remap worker, ramba/ramba.py:2409-2492). */
int rt_flat_copy(uintptr_t stream, void *boxed, void *dense, int nd,
                 const int64_t *shape, const int64_t *strides,
                 int64_t flat0, int64_t n, int elemsize, int scatter);

int rt_stream_sync(uintptr_t stream);
int rt_device_sync(void);

/* HIP-event timing for the bench's roofline leg. */
int   rt_event_create(void **ev);
int   rt_event_destroy(void *ev);
int   rt_event_record(void *ev, uintptr_t stream);
/* elapsed ms between two recorded events (synchronises on `end`) */
int   rt_event_elapsed(void *start, void *end, float *ms);

#ifdef __cplusplus
}
#endif

#endif /* RAMBA_RT_H */
