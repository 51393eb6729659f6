"""Timings for the round-1 extension kernels at streaming scale
(axis cumsum / mask-getitem / reshape).  Not the judged bench line —
evidence for DESIGN.md §9's extra rows; run under rocprofv3 --stats for
the per-kernel summary committed to profiles/."""

import sys
import time

import numpy as np

sys.path.insert(0, ".")
import ramba_amd as ra  # noqa: E402


def timed(label, fn, algo_bytes, iters=3):
    import torch
    fn()                      # warmup (compile, allocate)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        r = fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"{label}: {dt*1e3:.2f} ms/iter, algorithmic "
          f"{algo_bytes/dt/1e12:.2f} TB/s", flush=True)
    return r


def main():
    ra.init()
    n = 16384
    # N-D axis cumsum, both mappings (thread-per-line / wave-per-line)
    A = ra.fromfunction(lambda i, j: (i * 3 + j) * 1e-9, (n, n))
    ra.sync()
    nb = n * n * 8 * 2     # read + write
    timed("cumsum(axis=0) 16384^2 fp64 [thread-per-line]",
          lambda: A.cumsum(axis=0), nb)
    timed("cumsum(axis=1) 16384^2 fp64 [wave-per-line]",
          lambda: A.cumsum(axis=1), nb)

    # mask-getitem at 1e9, ~50% selectivity: read a (8) + mask copy cycle
    # (a copy 16 + mask astype-write 1 + mask read 1) + write out (4)
    m = 1_000_000_000
    B = ra.arange(m) % 2
    Bf = B * 1.0
    ra.sync()
    timed("a[mask] 1e9 fp64 sel=50% [rt_mask_compact]",
          lambda: Bf[B == 1], m * (8 + 8 + 8 + 1 + 1 + 1) + m // 2 * 8)

    # reshape 1e9 fp64: flat gather + scatter = 2r + 2w = 32 B/elem
    C = ra.arange(m) * 1.0
    ra.sync()
    timed("reshape 1e9 -> (31250, 32000) fp64 [rt_flat_copy]",
          lambda: C.reshape(31250, 32000), m * 32)
    print("done", flush=True)


if __name__ == "__main__":
    main()
