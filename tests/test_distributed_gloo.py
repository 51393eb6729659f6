"""Multi-process SPMD tests on CPU: world_size 2 and 4 over gloo.

Covers the N>1 planner paths that the GPU bench exercises over RCCL:
halo exchange for shifted-slice stencils, part exchange for repartitioned
operands, allreduce reductions, gather.  The reference's analog is its CI
matrix running the suite under `mpiexec -n 2`
(/root/reference/.github/workflows/python-package.yml:41-45).
"""

import os
import pickle
import subprocess
import sys
import textwrap

import numpy as np
import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(HERE)

WORKER = r"""
import os, pickle, sys
sys.path.insert(0, {root!r})
import numpy as np
import ramba_amd as ra
from oracle.numpy_backend import NumpyBackend

ra.init(backend=NumpyBackend())

def impl(np_):
{body}

res = impl(ra)
if hasattr(res, "asarray"):
    res = res.asarray()
ref = impl(np)
tol = {tol!r}
if tol is None:
    assert np.array_equal(res, ref), f"rank {{os.environ['RANK']}}: {{res}} != {{ref}}"
else:
    np.testing.assert_allclose(res, ref, rtol=tol, atol=tol)
print("RANK", os.environ["RANK"], "OK")
"""


_PORT_COUNTER = [0]


def _next_port(base=29500):
    _PORT_COUNTER[0] += 1
    return str(base + (os.getpid() * 7 + _PORT_COUNTER[0] * 13) % 3000)


def run_spmd(body_src, world=2, tol=None):
    body = textwrap.indent(textwrap.dedent(body_src).strip(), "    ")
    script = WORKER.format(root=ROOT, body=body, tol=tol)
    port = _next_port()
    procs = []
    for r in range(world):
        env = dict(os.environ)
        env.update({"RANK": str(r), "WORLD_SIZE": str(world),
                    "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": port,
                    "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME", "lo")})
        procs.append(subprocess.Popen(
            [sys.executable, "-c", script], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    ok = True
    for p in procs:
        out, _ = p.communicate(timeout=180)
        outs.append(out.decode())
        ok = ok and p.returncode == 0
    assert ok, "\n==== rank outputs ====\n" + "\n----\n".join(outs)


@pytest.mark.parametrize("world", [2, 4])
def test_flagship_chain_spmd(world):
    run_spmd("""
        A = np_.arange(10000) / 1000.0
        B = np_.sin(A)
        C = np_.cos(A)
        D = B * B + C ** 2
        return D
    """, world=world, tol=1e-12)


@pytest.mark.parametrize("world", [2, 4])
def test_stencil_1d_halo(world):
    run_spmd("""
        A = np_.arange(4001) * 1.0
        B = np_.zeros(4001)
        B[2:-2] = (0.1 * A[:-4] + 0.2 * A[1:-3] + 0.4 * A[2:-2]
                   + 0.2 * A[3:-1] + 0.1 * A[4:])
        return B
    """, world=world, tol=1e-13)


@pytest.mark.parametrize("world", [2, 4])
def test_stencil_2d_halo_iterated(world):
    run_spmd("""
        A = np_.fromfunction(lambda x, y: x + y, (65, 67), dtype=np.float32)
        B = np_.zeros((65, 67), dtype=np.float32)
        for _ in range(3):
            B[1:-1, 1:-1] = (A[:-2, 1:-1] + A[2:, 1:-1] + A[1:-1, :-2]
                             + A[1:-1, 2:] - 4.0 * A[1:-1, 1:-1])
            A[1:-1, 1:-1] = 0.25 * B[1:-1, 1:-1]
        return A
    """, world=world, tol=1e-5)


@pytest.mark.parametrize("world", [2, 4])
def test_reduction_allreduce(world):
    run_spmd("""
        A = np_.arange(12345) / 1000.0
        D = np_.sin(A) ** 2 + np_.cos(A) ** 2
        s = D.sum()
        m = (np_.arange(1000) * 3).max()
        return np.array([s, m * 1.0])
    """, world=world, tol=1e-12)


def test_read_after_write_spmd():
    run_spmd("""
        B = np_.arange(3000) * 1.0
        if np_ is np:
            B[:-1] = B[:-1] + B[1:].copy()
        else:
            B[:-1] += B[1:]
        return B
    """, world=2)


def test_wide_shift_beyond_border():
    # shift of 100 exceeds the default border ring (4): exercises the
    # gathered temp-operand fallback path
    run_spmd("""
        A = np_.arange(2000) * 1.0
        B = A[:-100] + A[100:]
        return B
    """, world=2)


def test_repartition_differing_dists():
    # operands whose divisions differ from the exec partition
    run_spmd("""
        A = np_.arange(999) * 1.0        # divisions of 999
        C = np_.arange(1000) * 2.0       # divisions of 1000
        B = A + C[1:]                    # view of C repartitioned onto A's
        return B
    """, world=2, tol=0)


def test_gather_and_scatter():
    run_spmd("""
        src = np.arange(300, dtype=np.float64).reshape(20, 15) * 1.5
        if np_ is np:
            a = src.copy()
        else:
            a = np_.fromarray(src)
        return a * 2 + 1
    """, world=2)


def test_broadcast_2d_spmd():
    run_spmd("""
        a = np_.fromfunction(lambda x, y: x * 5 + y, (30, 50),
                             dtype=np.float64)
        b = np_.arange(50) * 1.0
        return a + b
    """, world=2)


@pytest.mark.parametrize("world", [3, 8])
def test_odd_and_full_world_spmd(world):
    """world sizes the 8-GPU node will see (8) and an odd split (3)."""
    run_spmd("""
        A = np_.arange(100001) / 1000.0
        D = np_.sin(A) ** 2 + np_.cos(A) ** 2
        s = D.sum()
        B = np_.zeros(100001)
        B[2:-2] = 0.25 * (A[:-4] + A[1:-3] + A[3:-1] + A[4:])
        c = np_.arange(10007).cumsum()
        a2 = np_.fromfunction(lambda x, y: x * 53 + y, (53, 71),
                              dtype=np.int64).sum(axis=0)
        if np_ is np:
            return np.concatenate([[s], B, c * 1.0, a2 * 1.0])
        import numpy as _np
        return _np.concatenate([[float(s)], B.asarray(),
                                c.asarray() * 1.0, a2.asarray() * 1.0])
    """, world=world, tol=1e-12)


@pytest.mark.parametrize("world", [2, 4])
def test_axis_reduction_spmd(world):
    run_spmd("""
        a = np_.fromfunction(lambda x, y: x * 101 + y, (101, 77),
                             dtype=np.int64)
        r0 = a.sum(axis=0)
        r1 = a.sum(axis=1)
        m = a[3:90, 5:70].max(axis=0)
        if np_ is np:
            return np.concatenate([r0, r1, m])
        import numpy as _np
        return _np.concatenate([r0.asarray(), r1.asarray(), m.asarray()])
    """, world=world)


@pytest.mark.parametrize("world", [2, 4])
def test_cumsum_spmd(world):
    run_spmd("""
        a = np_.arange(5001) * 1.0
        c = a.cumsum()
        d = np_.arange(4000).cumsum()
        if np_ is np:
            return np.concatenate([c, d * 1.0])
        import numpy as _np
        return _np.concatenate([c.asarray(), d.asarray() * 1.0])
    """, world=world, tol=1e-12)


def test_tiny_array_uni_dist():
    # array smaller than the worker count: rank 0 owns it (reference
    # make_uni_dist); arithmetic and gather still work on every rank
    run_spmd("""
        a = np_.arange(3)
        b = a * 2 + 1
        z = np_.arange(0)
        return b
    """, world=4)


def test_poison_sanitizer_halo_correct():
    """With RAMBA_DEBUG_POISON=1 every fresh container is junk-filled;
    correct results prove no unfilled border/cell leaks into outputs
    (the hipMemsetD-poisoning analog)."""
    old = os.environ.get("RAMBA_DEBUG_POISON")
    os.environ["RAMBA_DEBUG_POISON"] = "1"
    try:
        run_spmd("""
            A = np_.arange(2001) * 1.0
            B = np_.zeros(2001)
            B[2:-2] = (0.1 * A[:-4] + 0.2 * A[1:-3] + 0.4 * A[2:-2]
                       + 0.2 * A[3:-1] + 0.1 * A[4:])
            s = B.sum()
            c = np_.fromfunction(lambda x, y: x + y, (33, 35),
                                 dtype=np.float32)
            d = np_.zeros((33, 35), dtype=np.float32)
            d[1:-1, 1:-1] = (c[:-2, 1:-1] + c[2:, 1:-1] + c[1:-1, :-2]
                             + c[1:-1, 2:])
            if np_ is np:
                return np.concatenate([[s], B, d.reshape(-1)])
            import numpy as _np
            return _np.concatenate([[float(s)], B.asarray(),
                                    d.asarray().reshape(-1)])
        """, world=2, tol=1e-12)
    finally:
        if old is None:
            os.environ.pop("RAMBA_DEBUG_POISON", None)
        else:
            os.environ["RAMBA_DEBUG_POISON"] = old


def test_mask_getitem_spmd():
    """Compressing a[mask] across ranks: value+mask repartitioned onto a
    C-contiguous split, per-rank compaction, uneven result divisions
    (runtime.mask_compact_op).  world=4 puts a square 2-D input on a 2x2
    grid — the case the repartition makes legal."""
    for world in (2, 3, 4):
        run_spmd("""
            a = np_.arange(1000) * 1.0
            b = np_.sin(a * 0.01)
            sel = b[b > 0.3]
            t = np_.fromfunction(lambda x, y: x * 3 + y, (401, 3))
            sel2 = t[(t % 7.0) == 0.0]
            q = np_.fromfunction(lambda x, y: x * 37.0 + y, (36, 37))
            sel3 = q[(q % 5.0) == 0.0]
            if np_ is np:
                return np.concatenate([sel, sel2, sel3,
                                       [float(sel.size), float(sel2.size),
                                        float(sel3.size)]])
            import numpy as _np
            return _np.concatenate([sel.asarray(), sel2.asarray(),
                                    sel3.asarray(),
                                    [float(sel.shape[0]),
                                     float(sel2.shape[0]),
                                     float(sel3.shape[0])]])
        """, world=world, tol=1e-12)


def test_axis_cumsum_spmd():
    """N-D cumsum along an axis: local scan + pairwise line-totals slab
    exchange (runtime.cumsum_axis_op).  world=4 gives a 2x2 partition
    grid, exercising partial line overlaps and multi-predecessor sums."""
    for world in (2, 3, 4):
        run_spmd("""
            c = np_.fromfunction(lambda x, y: x * 97 + y, (37, 29))
            r0 = c.cumsum(axis=0)
            r1 = c.cumsum(axis=1)
            v = c[3:33:2, 1:25]
            r2 = v.cumsum(axis=1)
            e = np_.fromfunction(lambda x, y, z: x * 100 + y * 10 + z,
                                 (12, 10, 8))
            r3 = e.cumsum(axis=0)
            outs = [r0, r1, r2, r3]
            if np_ is np:
                return np.concatenate([o.reshape(-1) for o in outs])
            import numpy as _np
            return _np.concatenate([o.asarray().reshape(-1) for o in outs])
        """, world=world, tol=1e-12)


def test_reshape_spmd():
    """Distributed reshape: flat-interval intersections + gather/scatter
    (runtime.reshape_op); surface-split sources repartition through the
    fused engine first."""
    for world in (2, 3, 4):
        run_spmd("""
            a = np_.arange(840) * 1.0
            r1 = a.reshape(21, 40)
            r2 = a.reshape(8, 105)
            b = np_.fromfunction(lambda i, j: i * 3.0 + j, (281, 3))
            r3 = b.ravel()
            r4 = b.reshape(3, 281)
            # square 2-D: the surface heuristic splits BOTH axes at
            # world 4 -> exercises the repartition-first path
            c = np_.fromfunction(lambda i, j: i * 37.0 + j, (36, 37))
            r5 = c.reshape(37, 36)
            r6 = c.T.ravel()
            if np_ is np:
                return np.concatenate([r1.reshape(-1), r2.reshape(-1),
                                       r3, r4.reshape(-1), r5.reshape(-1),
                                       r6])
            import numpy as _np
            return _np.concatenate([r1.asarray().reshape(-1),
                                    r2.asarray().reshape(-1),
                                    r3.asarray(),
                                    r4.asarray().reshape(-1),
                                    r5.asarray().reshape(-1),
                                    r6.asarray()])
        """, world=world, tol=0.0)


def test_scalar_reads_and_fromarray_spmd():
    """Cross-rank scalar reads (owner broadcast) and numpy ingestion
    (scatter) at worlds 3/4."""
    for world in (3, 4):
        run_spmd("""
            a = np_.arange(997) * 3.0
            reads = [float(a[0]), float(a[500]), float(a[996]),
                     float(a[-1])]
            import numpy as _np
            src = _np.fromfunction(lambda i, j: i * 11.0 - j, (23, 17))
            b = np_.fromarray(src) if np_ is not np else src
            c = (b * 2.0).sum(axis=1)
            if np_ is np:
                return np.concatenate([reads, (src * 2).sum(axis=1)])
            return _np.concatenate([reads, c.asarray()])
        """, world=world, tol=1e-12)


@pytest.mark.parametrize("world", [2])
def test_minmax_nan_spmd(world):
    """Float min/max propagate a NaN living on only one rank (ADVICE r1:
    the backend allgathers float min/max partials instead of trusting
    collective MIN/MAX with NaN)."""
    run_spmd("""
        A = np_.arange(10000) * 1.0
        B = np_.where(A == 9999.0, (A - 20000.0) ** 0.5, A)  # one NaN
        return np.asarray([float(B.min()), float(B.max()),
                           float(A.min()), float(A.max())])
    """, world=world, tol=0.0)


@pytest.mark.parametrize("world", [2, 4])
def test_multi_axis_reduction_spmd(world):
    """Axis-TUPLE reductions across ranks: the combining box exchange
    projects several reduced axes at once."""
    run_spmd("""
        a = np_.fromfunction(lambda x, y, z: x * 1000 + y * 31 + z,
                             (32, 21, 45))
        s02 = a.sum(axis=(0, 2))
        s12 = a.sum(axis=(1, 2))
        m01 = a.max(axis=(0, 1))
        k = a.sum(axis=(0, 1), keepdims=True)
        if np_ is np:
            return np.concatenate([s02, s12, m01, k.reshape(-1)])
        import numpy as _np
        return _np.concatenate([s02.asarray(), s12.asarray(),
                                m01.asarray(), k.asarray().reshape(-1)])
    """, world=world, tol=1e-12)


WORKER_SPLIT = r"""
import os, sys
sys.path.insert(0, {root!r})
import numpy as np
import ramba_amd as ra
from oracle.numpy_backend import NumpyBackend
ra.init(backend=NumpyBackend())
rt = ra._deferred.get_runtime()
A = ra.fromfunction(lambda x, y: x + y, (257, 259), dtype=np.float32)
B = ra.zeros((257, 259), dtype=np.float32)
for it in range(3):
    B[1:-1, 1:-1] = (A[:-2, 1:-1] + A[2:, 1:-1] + A[1:-1, :-2]
                     + A[1:-1, 2:] - 4.0 * A[1:-1, 1:-1])
    A[1:-1, 1:-1] = 0.25 * B[1:-1, 1:-1]
ra.sync()
want = int(os.environ.get("RAMBA_OVERLAP", "1"))
cache = getattr(rt, "_recipe_cache", {{}})
split = [r for r in cache.values() if r.units is not None]
if want:
    # halo-bearing stencil groups must have split into interior+rim
    # (>=1: the first-iteration group signature differs from steady state)
    assert len(split) >= 1, [(r.units, r.comm_msgs) for r in cache.values()]
    for r in split:
        assert r.pre_wait_units == 1
        assert len(r.units) >= 2
else:
    assert not split, "RAMBA_OVERLAP=0 must stay sequential"
xs = np.fromfunction(lambda x, y: x + y, (257, 259), dtype=np.float32)
ys = np.zeros_like(xs)
for it in range(3):
    ys[1:-1, 1:-1] = (xs[:-2, 1:-1] + xs[2:, 1:-1] + xs[1:-1, :-2]
                      + xs[1:-1, 2:] - 4.0 * xs[1:-1, 1:-1])
    xs[1:-1, 1:-1] = 0.25 * ys[1:-1, 1:-1]
np.testing.assert_allclose(A.asarray(), xs, rtol=1e-5, atol=1e-5)
print("RANK", os.environ["RANK"], "OK")
"""


@pytest.mark.parametrize("world", [2, 4])
@pytest.mark.parametrize("overlap", ["1", "0"])
def test_overlap_split_stencil_spmd(world, overlap):
    """Halo/compute overlap (BASELINE configs[4] 'overlapped RCCL'):
    the stencil group splits into interior+rim with the exchange posted
    first; RAMBA_OVERLAP=0 restores the serial order.  Parity vs NumPy
    either way."""
    script = WORKER_SPLIT.format(root=ROOT)
    port = _next_port()
    procs = []
    for r in range(world):
        env = dict(os.environ)
        env.update({"RANK": str(r), "WORLD_SIZE": str(world),
                    "RAMBA_OVERLAP": overlap,
                    "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": port,
                    "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME",
                                                  "lo")})
        procs.append(subprocess.Popen(
            [sys.executable, "-c", script], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs, ok = [], True
    for p in procs:
        out, _ = p.communicate(timeout=180)
        outs.append(out.decode())
        ok = ok and p.returncode == 0
    assert ok, "\n==== rank outputs ====\n" + "\n----\n".join(outs)


@pytest.mark.parametrize("world", [2, 4])
def test_staged_sum_fusion_spmd(world):
    """The fused sum(A)-after-stencil pair at world>1 (sequential
    fallback on the oracle backend; partial = full-core reduce +
    allreduce)."""
    run_spmd("""
        S = 128
        A = np_.zeros((S, S), dtype=np.float64)
        out = []
        for it in range(3):
            src = np_.fromfunction(
                lambda x, y: (x * S + y + it) * 1e-4, (S, S),
                dtype=np.float64)
            ssin = np_.sin(src)
            A[1:-1, 1:-1] = (ssin[:-2, 1:-1] + ssin[2:, 1:-1]
                             + ssin[1:-1, :-2] + ssin[1:-1, 2:]
                             - 4.0 * ssin[1:-1, 1:-1])
            del src, ssin
            out.append(float(A.sum()))
        return np.asarray(out)
    """, world=world, tol=1e-9)
