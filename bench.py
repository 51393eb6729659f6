#!/usr/bin/env python3
"""bench.py — the judged benchmark (BASELINE.json configs[1] workload).

Measures the north-star fused elementwise chain on 1e9 fp64 elements,
sharded one block per GPU (strong scaling over N GPUs):

    A = arange(1e9) / 1000.0          (materialised once, untimed —
                                       matching sample/test-ramba.py:5-10)
    per step:  B = sin(A); C = cos(A); D = B*B + C**2; sync()

One step moves 32 B/elem of algorithmic HBM traffic per GPU (read A +
store B, C, D; dead temps stay in registers — the live_gids rule,
reference ramba/ramba.py:8123).

Run:  python bench.py [--gpus N] [--steps K] [--warmup W] [--elems E]
Multi-GPU (driver contract): torchrun --nproc-per-node N bench.py --gpus N
Rank 0 prints ONE JSON line.
"""

import argparse
import json
import os
import subprocess
import sys
import time

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

HBM_PEAK_BYTES = 8.0e12          # MI355X HBM3E spec peak


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=None)
    ap.add_argument("--warmup", type=int, default=None)
    ap.add_argument("--elems", type=int, default=1_000_000_000)
    ap.add_argument("--check", action="store_true",
                    help="verify a small slice against NumPy first")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--workload", default="flagship",
                    choices=["flagship", "reduce", "stencil", "mixed"],
                    help="flagship = BASELINE configs[1] (the judged line); "
                         "reduce = configs[2] 1e9 fp64 sum; "
                         "stencil = configs[3] 4096^2 fp32 5-pt Laplacian; "
                         "mixed = configs[4] 8192^2 fp64 "
                         "iota→sin→stencil→sum pipeline")
    ap.add_argument("--stencil-n", type=int, default=4096)
    args = ap.parse_args()
    if args.steps is None:
        # short pipelines need more steps so the one-time allocator-growth
        # and first-compile spikes stay out of the steady-state window
        # windows sized so the box's occasional random 20-40 ms stall
        # (observed ~once per run at arbitrary steps, not attributable to
        # GC or our code — gpurun_out/st2.log attribution test) cannot
        # dominate a short window or always land inside a long one
        args.steps = {"flagship": 5, "reduce": 8, "stencil": 12,
                      "mixed": 20}[args.workload]
    if args.warmup is None:
        args.warmup = {"flagship": 2, "reduce": 3, "stencil": 8,
                       "mixed": 6}[args.workload]

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world != args.gpus and world == 1 and args.gpus > 1:
        print(f"requested --gpus {args.gpus} but WORLD_SIZE=1; launch via "
              f"torchrun --nproc-per-node {args.gpus}", file=sys.stderr)
        sys.exit(2)

    import numpy as np
    import torch
    import ramba_amd as ra
    ra.init()  # HIP product backend; fails loudly without GPU/extension
    rt = ra._deferred.get_runtime()
    backend = rt.backend

    def barrier_sync():
        if world > 1:
            import torch.distributed as dist
            dist.barrier()
        torch.cuda.synchronize()

    if args.check:
        n = 1 << 20
        Ac = ra.arange(n) / 1000.0
        Dc = ra.sin(Ac) ** 2 + ra.cos(Ac) ** 2
        ref = np.sin(np.arange(n) * 0.001) ** 2 \
            + np.cos(np.arange(n) * 0.001) ** 2
        np.testing.assert_allclose(Dc.asarray(), ref, rtol=1e-12, atol=1e-12)
        del Ac, Dc
        if rank == 0:
            print("check ok", file=sys.stderr)

    # ---- workload definitions (BASELINE.json configs) ---------------------
    if args.workload == "flagship":
        N = args.elems
        alg_bytes_per_elem = 32
        A = ra.arange(N) / 1000.0
        ra.sync()

        def step():
            B = ra.sin(A)
            C = ra.cos(A)
            D = B * B + C ** 2
            ra.sync()
            return B, C, D
        metric = ("GElem/s, 1e9-elem fused sin²+cos² fp64 "
                  "(BASELINE configs[1])")
        wl_desc = ("1e9-elem fp64 arange→sin²+cos², one shard per GPU, "
                   "32 B/elem algorithmic")
    elif args.workload == "reduce":
        N = args.elems
        alg_bytes_per_elem = 8
        A0 = ra.arange(N) / 1000.0
        A = ra.sin(A0) ** 2 + ra.cos(A0) ** 2
        ra.sync()
        del A0

        def step():
            s = A.sum()
            assert abs(float(s) - N) < 1e-3 * N
            return s
        metric = ("GElem/s, 1e9-elem fp64 global sum via RCCL allreduce "
                  "(BASELINE configs[2])")
        wl_desc = "1e9-elem fp64 sum(), 8 B/elem read + one 8 B allreduce"
    elif args.workload == "mixed":
        # configs[4]: iota fill -> sin -> 5-pt stencil -> sum, 8192^2 fp64.
        # With cross-stage fusion (default, ramba_amd/staged.py) the sin
        # producer is recomputed into LDS inside the stencil kernel and
        # the dead intermediates never materialise: ONE tiled kernel
        # (A store, 8 B) + the fused sum read (8 B) = 16 B/elem
        # algorithmic — SURVEY §8d cfg5's "less if stencil fuses with sin
        # producer".  RAMBA_STAGE_FUSION=0 restores the 3-kernel
        # reference-order pipeline at 32 B/elem.
        from ramba_amd.common import stage_fusion
        S = 8192
        N = S * S
        # fully fused (round 2): ONE tiled kernel computes iota+sin in
        # LDS, writes A AND accumulates the sum in-kernel (+ a tiny rim
        # complement) — algorithmic traffic is just the A store, 8 B/elem
        alg_bytes_per_elem = 8 if stage_fusion else 32
        A = ra.zeros((S, S), dtype=np.float64)
        ra.sync()

        def step():
            src = ra.fromfunction(
                lambda x, y: (x * S + y) * 1e-6, (S, S), dtype=np.float64)
            ssin = ra.sin(src)
            A[1:-1, 1:-1] = (ssin[:-2, 1:-1] + ssin[2:, 1:-1]
                             + ssin[1:-1, :-2] + ssin[1:-1, 2:]
                             - 4.0 * ssin[1:-1, 1:-1])
            # the intermediates are dead from here on: dropping the refs
            # before the flush lets the engine demote them to registers/
            # LDS (the reference's live_gids rule, ramba.py:8123, applied
            # through the fused pair)
            del src, ssin
            # force the reduction like the reference does (its sum is
            # eager: internal_reduction2b gathers immediately).  Leaving
            # the scalar pending would also let the LAST step's group
            # escape the timed region (host-side flush happens after the
            # barrier) and triggers the WAR temp-dance on A every step.
            return float(A.sum())
        metric = ("GElem/s, 8192^2 fp64 iota→sin→stencil→sum pipeline "
                  "(BASELINE configs[4])")
        wl_desc = ("8192^2 fp64: iota+sin+stencil+sum in ONE fused "
                   "LDS-tiled kernel (+rim complement) + RCCL allreduce; "
                   f"{8 if stage_fusion else 32} B/elem algorithmic")
    else:  # stencil
        S = args.stencil_n
        N = S * S
        alg_bytes_per_elem = 8
        A = ra.fromfunction(lambda x, y: x + y, (S, S), dtype=np.float32)
        B = ra.zeros((S, S), dtype=np.float32)
        ra.sync()
        bufs = [A, B]

        def step():
            src, dst = bufs
            dst[1:-1, 1:-1] = (src[:-2, 1:-1] + src[2:, 1:-1]
                               + src[1:-1, :-2] + src[1:-1, 2:]
                               - 4.0 * src[1:-1, 1:-1])
            bufs.reverse()
            ra.sync()
            return dst
        metric = (f"GElem/s, {S}x{S} fp32 5-pt Laplacian stencil "
                  "(BASELINE configs[3])")
        wl_desc = (f"{S}^2 fp32 5-point stencil, 2-D blocks + halo "
                   "exchange, 8 B/elem algorithmic")

    trace = os.environ.get("RAMBA_BENCH_TRACE")
    keep = None
    # the box exhibits ONE ~36 ms runtime stall per process, 3-5 s after
    # start (observed at arbitrary step indices; independent of GC and of
    # our kernels — gpurun_out/s1err.log).  Warm up for at least 6 s of
    # wall time so it lands before the timed window; the JSON reports the
    # warmup steps actually done.
    warm_requested = args.warmup
    warm_done = 0
    # profiling runs shrink the floor (RAMBA_WARMUP_FLOOR=0) so PMC
    # databases stay small; the default 6 s stays for timing runs
    wfloor = float(os.environ.get("RAMBA_WARMUP_FLOOR", "6.0"))
    tw = time.perf_counter()
    if world == 1:
        while warm_done < args.warmup or time.perf_counter() - tw < wfloor:
            keep = step()
            warm_done += 1
            if warm_done > args.warmup + 100_000:
                break
    else:
        # N>1: a step may contain collectives (reduce's allreduce, halo
        # exchange), so every rank MUST run the same number of steps.
        # Agree in rounds: fixed step batches + an all_reduce(MIN) vote
        # on whether everyone has passed the 6 s floor.
        import torch.distributed as dist
        for _ in range(args.warmup):
            keep = step()
            warm_done += 1
        while True:
            flag = torch.tensor(
                [1.0 if time.perf_counter() - tw >= wfloor else 0.0],
                dtype=torch.float64, device="cuda")
            dist.all_reduce(flag, op=dist.ReduceOp.MIN)
            if float(flag.cpu()[0]) >= 1.0 or warm_done > 100_000:
                break
            for _ in range(4):
                keep = step()
                warm_done += 1
    args.warmup = warm_done
    barrier_sync()
    t0 = time.perf_counter()
    if trace:
        for i in range(args.steps):
            ts = time.perf_counter()
            keep = step()
            torch.cuda.synchronize()
            print(f"step {i}: {(time.perf_counter()-ts)*1e3:.2f} ms",
                  file=sys.stderr)
    else:
        for _ in range(args.steps):
            keep = step()
    barrier_sync()
    t1 = time.perf_counter()
    elapsed = t1 - t0

    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64, device="cuda")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.cpu()[0])

    ms_per_step = elapsed / args.steps * 1000.0
    value = N * args.steps / elapsed / 1e9     # GElem/s, whole job

    # ---- roofline leg: HIP-event time of the fused kernel itself ----------
    backend.time_kernels = True
    backend.kernel_times_ms = []
    backend.kernel_keys = []
    for _ in range(3):
        keep = step()
    backend.time_kernels = False
    local_elems = 0
    eb = rt.core_box(A.bdarray, rank)
    if eb is not None:
        local_elems = 1
        for i in range(eb.shape[1]):
            local_elems *= int(eb[1, i] - eb[0, i] + 1)
    kms = None
    kbreak = None
    kt = backend.kernel_times_ms
    if kt:
        # a step may launch several kernels (the mixed pipeline): sum the
        # kernel times within each measured step, take the fastest step
        per_step = len(kt) // 3 if len(kt) % 3 == 0 else len(kt)
        sums = [sum(kt[i:i + per_step]) for i in range(0, len(kt), per_step)]
        kms = min(sums)
        keys = backend.kernel_keys[:per_step]
        kbreak = [{"kernel": k, "ms": round(t, 4)}
                  for k, t in zip(keys, kt[:per_step])]
    roofline = None
    if kms:
        achieved = alg_bytes_per_elem * local_elems / (kms / 1e3)
        # PMC-measured per-launch HBM traffic is valid evidence only for
        # the EXACT kernel it was captured against: match the live
        # kernel's content-addressed key against the manifest written at
        # profiling time (tools/make_pmc_manifest.py); no match -> null.
        traffic = os.environ.get("RAMBA_BENCH_TRAFFIC")
        traffic_src = "RAMBA_BENCH_TRAFFIC env" if traffic else None
        if traffic is None:
            per_step_keys = backend.kernel_keys[:max(1, len(
                backend.kernel_keys) // 3)]
            dom = None
            if kt:
                di = max(range(len(per_step_keys)),
                         key=lambda i: kt[i]) if per_step_keys else None
                dom = per_step_keys[di] if di is not None else None
            mpath = os.path.join(ROOT, "profiles", "pmc_manifest.json")
            if dom and os.path.exists(mpath):
                try:
                    man = json.load(open(mpath))
                    ent = man.get(dom)
                    if ent and ent.get("workload") == args.workload \
                            and int(ent.get("elems", -1)) == N \
                            and int(ent.get("world", -1)) == world:
                        traffic = float(ent["bytes_per_launch"])
                        traffic_src = ent.get("source",
                                              "rocprofv3 PMC manifest")
                except Exception:  # noqa: BLE001
                    pass
        roofline = {
            "bound": "hbm",
            "achieved": achieved / 1e9,          # GB/s
            "peak": HBM_PEAK_BYTES / 1e9,
            "unit": "GB/s",
            "frac": achieved / HBM_PEAK_BYTES,
            "traffic": float(traffic) if traffic else None,
            "traffic_source": traffic_src if traffic else None,
            "kernel_ms": kms,
            "kernels": kbreak,
        }
        if args.workload == "mixed" and alg_bytes_per_elem == 8:
            # transparency: cross-stage fusion halves the algorithmic
            # bytes ("less if stencil fuses with sin
            # producer"), so `frac` above is the TRUE-traffic rate; the
            # round-1-comparable figure (the unfused 32 B/elem workload
            # unit over the fused execution time) is reported alongside
            roofline["equiv_unfused_frac"] =                 32 * local_elems / (kms / 1e3) / HBM_PEAK_BYTES

    # ---- cpu_baseline leg (rank 0, N=1 only): oracle/fused_cpu.c ----------
    cpu_baseline = None
    if rank == 0 and world == 1 and not args.no_cpu_baseline \
            and args.workload == "flagship":
        exe = os.path.join(ROOT, "oracle", "_build", "fused_cpu")
        if os.path.exists(exe):
            sample = min(N, 200_000_000)
            try:
                out = subprocess.run([exe, str(sample), "3"],
                                     capture_output=True, timeout=300,
                                     check=True)
                r = json.loads(out.stdout.decode().strip().splitlines()[-1])
                # BASELINE.md substitute (i): NumPy eager, single-threaded,
                # the reference README's methodology row 1
                ne = 20_000_000
                xs = np.arange(ne, dtype=np.float64) * 0.001
                t0 = time.perf_counter()
                bs = np.sin(xs)
                cs = np.cos(xs)
                ds = bs * bs + cs ** 2
                t1 = time.perf_counter()
                del bs, cs, ds
                cpu_baseline = {
                    "value": r["elems"] / r["last_iter_secs"] / 1e9,
                    "unit": "GElem/s",
                    "cores": r["threads"],
                    "kind": "port",
                    "sample": f"{sample} elems x 3 iters of the same fused "
                              f"loop (C/OpenMP restatement; the reference's "
                              f"Numba+MPI path is not runnable here — "
                              f"BASELINE.md)",
                    "numpy_eager_gelem_s": ne / (t1 - t0) / 1e9,
                }
            except Exception as e:  # noqa: BLE001
                print(f"cpu_baseline failed: {e}", file=sys.stderr)

    del keep

    if rank == 0:
        line = {
            "metric": metric,
            "value": value,
            "unit": "GElem/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": wl_desc,
                "elems": N,
                "parallelism": f"shard{world}",
                "warmup_requested": warm_requested,
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(line))


if __name__ == "__main__":
    main()
