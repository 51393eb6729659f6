"""Configuration and block-schedule computation.

Re-implements the scheduling heuristic of the reference's ramba/common.py:287-680
(`compute_regular_schedule` family): prime-factorise the worker count, enumerate the
per-dimension factorings, and pick the factoring that minimises the surface area
between blocks.  Re-targeted at one shard per GPU on a single 8-GPU node, so the
reference's "nodesurface" mode (common.py:521-560) degenerates to plain "surface"
(all workers share one node; the 0.1 intra-node weight scales every candidate
equally).

Env knobs (analog of ramba/common.py:26-235):
  RAMBA_DEBUG          - debug print level
  RAMBA_BORDER         - default shard border ring width (elements), default 4
  RAMBA_TIMING         - timing summary level
"""

import functools
import math
import os

import numpy as np

ndebug = int(os.environ.get("RAMBA_DEBUG", "0"))
default_border = int(os.environ.get("RAMBA_BORDER", "4"))
ntiming = int(os.environ.get("RAMBA_TIMING", "0"))
# debug sanitizer: poison fresh shard containers (NaN / 0xCC) so any read
# of an unfilled border or uninitialised cell corrupts results visibly —
# the analog of the reference build plan's hipMemsetD poisoning
debug_poison = int(os.environ.get("RAMBA_DEBUG_POISON", "0"))
# halo/compute overlap: split each rank's iteration box into interior
# (launched while the exchange is in flight) + rim slabs (after it lands)
# — BASELINE configs[4] "overlapped RCCL"; 0 restores the reference-style
# serial exchange-then-execute (ramba.py:3547-3693) for A/B measurement
overlap_exchange = int(os.environ.get("RAMBA_OVERLAP", "1"))
# cross-stage fusion (BASELINE configs[4] "fused-kernel"): a shifted read
# of an index-pure in-group-written array seals the group as a producer
# and the pair fuses into one LDS-tiled kernel on the HIP backend; 0
# restores the reference's flush-at-alias behaviour (ramba.py:8434-8443)
stage_fusion = int(os.environ.get("RAMBA_STAGE_FUSION", "1"))


def dprint(level, *args):
    if ndebug >= level:
        print(*args, flush=True)


# ---------------------------------------------------------------------------
# timing (analog of ramba/ramba.py:945-1022 add_time/get_timing)
# ---------------------------------------------------------------------------

timing_acc = {}


def add_time(tag, secs):
    c, s = timing_acc.get(tag, (0, 0.0))
    timing_acc[tag] = (c + 1, s + secs)


def get_timing():
    return dict(timing_acc)


def reset_timing():
    timing_acc.clear()


# ---------------------------------------------------------------------------
# divisions / schedule
# divisions: np.ndarray (nranks, 2, ndim) int64; [r,0]=start, [r,1]=end inclusive.
# An empty division has end < start.  Mirrors the reference's divisions layout
# (common.py:309 create_divisions, shardview_array.py divisions_to_distribution).
# ---------------------------------------------------------------------------


def gen_prime_factors(n):
    # reference: common.py gen_prime_factors
    factors, exps = [], []

    def one_prime(n, p):
        if n % p == 0:
            factors.append(p)
            exps.append(1)
            n //= p
            while n % p == 0:
                n //= p
                exps[-1] += 1
        return n

    n = one_prime(n, 2)
    for i in range(3, int(math.sqrt(n)) + 1, 2):
        n = one_prime(n, i)
    if n != 1:
        factors.append(n)
        exps.append(1)
    return factors, exps


@functools.lru_cache(maxsize=None)
def get_dim_factors(num_workers, num_dim):
    """All ways to write num_workers as an ordered product of num_dim factors.

    reference: common.py gen_ind_factors / get_dim_factors.
    """
    factors, exps = gen_prime_factors(num_workers)
    out = set()

    def rec(exps_left, remaining, thus_far):
        if remaining == 1:
            rest = 1
            for f, e in zip(factors, exps_left):
                rest *= f ** e
            out.add(tuple(thus_far + [rest]))
            return
        # choose exponents for this position
        def choose(idx, part):
            if idx >= len(exps_left):
                val = 1
                for f, e in zip(factors, part):
                    val *= f ** e
                rec([a - b for a, b in zip(exps_left, part)], remaining - 1,
                    thus_far + [val])
                return
            for i in range(exps_left[idx] + 1):
                choose(idx + 1, part + [i])
        choose(0, [])

    rec(list(exps), num_dim, [])
    return frozenset(out)


def get_div_sizes(dim_len, num_div):
    """reference: common.py:293 get_div_sizes -- (main, rem, largest, smallest)."""
    low = dim_len // num_div
    if dim_len % num_div == 0:
        return low, low, low, low
    rem = dim_len - (low * (num_div - 1))
    if rem >= num_div:
        main = low + 1
        rem = dim_len - (main * (num_div - 1))
    else:
        main = low
    if rem == 0:
        rem = main
    return main, rem, max(main, rem), min(main, rem)


def create_divisions(divisions, size, best):
    """Fill `divisions` with the block decomposition given per-dim factor counts.

    Port of the reference's recursive divider (common.py:309 create_divisions /
    crsi_div): split dim 0 into best[0] runs of workers, recurse.
    """
    num_dim = len(size)
    sizem1 = np.array(size, dtype=np.int64) - 1

    def crsi_div(index, min_worker, max_worker):
        if index >= num_dim:
            return
        total_workers = max_worker - min_worker + 1
        chunks_here = total_workers // best[index]
        last = -1
        for i in range(min_worker, max_worker + 1, chunks_here):
            num_left = sizem1[index] - last
            this_div = num_left // ((max_worker + 1 - i) // chunks_here)
            for j in range(chunks_here):
                divisions[i + j, 0, index] = last + 1
                divisions[i + j, 1, index] = min(last + this_div, sizem1[index])
            last += this_div
            crsi_div(index + 1, i, i + chunks_here - 1)

    crsi_div(0, 0, divisions.shape[0] - 1)


@functools.lru_cache(maxsize=None)
def compute_regular_schedule(num_workers, size, dims_do_not_distribute=()):
    """Pick the best factoring (surface-area heuristic) and build divisions.

    reference: common.py:570 compute_regular_schedule_internal +
    compute_regular_schedule_core (mode "surface"; "nodesurface" is equivalent
    on a single node).
    """
    num_dim = len(size)
    if any(s == 0 for s in size):
        # zero-size arrays: every rank empty
        divisions = np.zeros((num_workers, 2, num_dim), dtype=np.int64)
        divisions[:, 1, :] = -1
        return divisions
    best, best_value = None, math.inf
    for factored in get_dim_factors(num_workers, num_dim):
        ok = True
        for i in range(num_dim):
            if factored[i] != 1 and i in dims_do_not_distribute:
                ok = False
                break
            if factored[i] > size[i]:
                ok = False
                break
        if not ok:
            continue
        block = [size[i] / factored[i] for i in range(num_dim)]
        surface = 0.0
        for i in range(num_dim):
            t = block[i]
            block[i] = 1
            surface += float(np.prod(block))
            block[i] = t
        if surface < best_value:
            best_value = surface
            best = factored
    if best is None:
        # array too small to split over every worker (reference
        # make_uni_dist / do_not_distribute, common.py): rank 0 owns it
        divisions = np.zeros((num_workers, 2, num_dim), dtype=np.int64)
        divisions[:, 1, :] = -1
        divisions[0, 0, :] = 0
        divisions[0, 1, :] = np.array(size, dtype=np.int64) - 1
        return divisions
    divisions = np.empty((num_workers, 2, num_dim), dtype=np.int64)
    create_divisions(divisions, size, best)
    return divisions


_div_cache = {}


def default_divisions(num_workers, shape):
    """Divisions for a fresh array (reference: shardview_array.py:908
    default_distribution -> compute_regular_schedule).  Cached and shared
    per (world, shape) — divisions are never mutated after construction
    (flex adoption REBINDS bd.divisions), and the cached array is marked
    read-only to enforce that."""
    if len(shape) == 0:
        raise ValueError("0-d arrays are not distributed")
    key = (num_workers, tuple(int(s) for s in shape))
    d = _div_cache.get(key)
    if d is None:
        d = compute_regular_schedule(num_workers, key[1])
        d.flags.writeable = False
        if len(_div_cache) > 4096:
            _div_cache.clear()
        _div_cache[key] = d
    return d


def contiguous_divisions(world, shape):
    """Axis-0-only balanced split: every shard covers full trailing axes,
    so shards are C-contiguous flat intervals (reshape/compaction need
    this)."""
    import numpy as np
    nd = len(shape)
    divs = np.zeros((world, 2, nd), dtype=np.int64)
    if any(int(s) == 0 for s in shape):
        divs[:, 1, :] = -1
        return divs
    rows = int(shape[0])
    base, rem = divmod(rows, world)
    lo = 0
    for r in range(world):
        cnt = base + (1 if r < rem else 0)
        if cnt == 0:
            divs[r, 1, :] = -1
            continue
        divs[r, 0, 0] = lo
        divs[r, 1, 0] = lo + cnt - 1
        for d in range(1, nd):
            divs[r, 1, d] = int(shape[d]) - 1
        lo += cnt
    return divs
