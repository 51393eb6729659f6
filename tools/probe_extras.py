"""Direct kernel-rate probe for the extension kernels (no per-iteration
allocations, device-event style timing).  Diagnoses where bench_extras'
wall time goes: kernel vs allocator/host."""

import sys
import time

import numpy as np

sys.path.insert(0, ".")
import ramba_amd as ra  # noqa: E402
from ramba_amd import deferred  # noqa: E402


def t(label, fn, nbytes, iters=3):
    import torch
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"{label}: {dt*1e3:.2f} ms, {nbytes/dt/1e12:.2f} TB/s", flush=True)


def main():
    small = "--small" in sys.argv   # profiled runs: keep rocprof cheap
    ra.init()
    rt = deferred.get_runtime()
    be = rt.backend

    n = 8192 if small else 16384
    A = ra.fromfunction(lambda i, j: (i * 3 + j) * 1e-9, (n, n))
    ra.sync()
    out = ra.zeros((n, n))
    ra.sync()
    bd, obd = A.bdarray, out.bdarray
    d_, _, cs, pads = rt.shard_geometry(bd)
    off0 = sum(pads[i] * cs[i] for i in range(2))
    od, _, ocs, opads = rt.shard_geometry(obd)
    ooff = sum(opads[i] * ocs[i] for i in range(2))
    # raw axis-scan kernel, axis 0 (chunked path) and axis 1 (waves)
    t("rt_axis_scan axis=0 16384^2", lambda: (be.axis_scan_local(
        bd, off0, cs, (n, n), 0, obd, ooff, ocs), be.free_temps()),
      n * n * 8 * 2)
    t("rt_axis_scan axis=1 16384^2", lambda: (be.axis_scan_local(
        bd, off0, cs, (n, n), 1, obd, ooff, ocs), be.free_temps()),
      n * n * 8 * 2)

    # raw flat gather / scatter on the same 2-D container
    nn = n * n
    t("rt_flat_copy gather 16384^2", lambda: be.flat_gather(
        be._cont(bd), off0, cs, (n, n), 0, nn), nn * 8 * 2)
    import torch
    dense = torch.empty(nn, dtype=torch.float64, device="cuda")
    t("rt_flat_copy scatter 16384^2", lambda: be.flat_scatter(
        be._cont(obd), ooff, ocs, (n, n), 0, dense), nn * 8 * 2)

    # raw mask compaction on prebuilt co-partitioned pair (50%)
    m = 100_000_000 if small else 500_000_000
    V = ra.arange(m) * 1.0
    M8 = (ra.arange(m) % 2).astype(np.uint8)
    ra.sync()
    t("rt_mask_compact 5e8 fp64 sel=50%", lambda: (be.mask_compact(
        V.bdarray, M8.bdarray, rt), be.free_temps())[1],
      m * (8 + 1) + m // 2 * 8)
    print("done", flush=True)


if __name__ == "__main__":
    main()
