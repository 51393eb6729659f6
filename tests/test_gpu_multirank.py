"""Two SPMD ranks sharing ONE MI355X: HIP compute path + gloo transport
(RAMBA_PG_BACKEND=gloo with host staging).

This exercises the multi-rank GPU code — shard partitioning, halo
exchange with device pack/unpack kernels, allreduce, axis-reduction
combine, cumsum prefix fixup — without needing a multi-GPU box (gpurun
exposes one GPU; the driver's round-end 8-GPU bench uses the same code
over RCCL)."""

import os
import subprocess
import sys
import textwrap

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(HERE)

_PORT_COUNTER = [0]

WORKER = r"""
import os, sys
sys.path.insert(0, {root!r})
import numpy as np
import ramba_amd as ra
ra.init()   # HIP product backend

def impl(np_):
{body}

res = impl(ra)
if hasattr(res, "asarray"):
    res = res.asarray()
ref = impl(np)
tol = {tol!r}
if tol is None:
    assert np.array_equal(res, ref), f"rank {{os.environ['RANK']}}: mismatch"
else:
    np.testing.assert_allclose(res, ref, rtol=tol, atol=tol)
print("RANK", os.environ["RANK"], "OK")
"""


def _device_count():
    import torch
    return torch.cuda.device_count() if torch.cuda.is_available() else 0


# RCCL (like NCCL) refuses two ranks on one device: "Duplicate GPU
# detected : rank 1 and rank 0 both on CUDA device ..." — measured on a
# leased MI355X (this repo, round 2; librccl 2.26.6 has no bypass env).
# The nccl-backend variants therefore need >= 2 devices and run for real
# on any multi-GPU box; world-1 RCCL transport coverage that runs on ONE
# GPU lives in test_gpu_rccl_world1.py.
needs_multi_gpu = pytest.mark.skipif(
    "RAMBA_FORCE_RCCL_TESTS" not in os.environ and _device_count() < 2,
    reason="RCCL forbids 2 ranks on 1 device (duplicate-GPU check)")


def run_spmd_gpu(body_src, world=2, tol=None, backend="gloo"):
    body = textwrap.indent(textwrap.dedent(body_src).strip(), "    ")
    script = WORKER.format(root=ROOT, body=body, tol=tol)
    _PORT_COUNTER[0] += 1
    port = str(26000 + (os.getpid() * 7 + _PORT_COUNTER[0] * 13) % 3000)
    ndev = max(1, _device_count())
    procs = []
    for r in range(world):
        env = dict(os.environ)
        env.update({"RANK": str(r), "WORLD_SIZE": str(world),
                    "LOCAL_RANK": "0" if backend == "gloo"
                    else str(r % ndev),
                    "RAMBA_PG_BACKEND": backend,
                    "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": port,
                    "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME",
                                                  "lo")})
        if backend == "nccl":
            # dmabuf IPC is the only mode the host driver supports
            env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        procs.append(subprocess.Popen(
            [sys.executable, "-c", script], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    ok = True
    for p in procs:
        try:
            out, _ = p.communicate(timeout=240)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
            ok = False
        outs.append(out.decode())
        ok = ok and p.returncode == 0
    assert ok, "\n==== rank outputs ====\n" + "\n----\n".join(outs)


def test_flagship_2rank_gpu():
    run_spmd_gpu("""
        A = np_.arange(200000) / 1000.0
        D = np_.sin(A) ** 2 + np_.cos(A) ** 2
        return D
    """, tol=1e-12)


def test_stencil_halo_2rank_gpu():
    run_spmd_gpu("""
        A = np_.fromfunction(lambda x, y: x + y, (257, 259),
                             dtype=np.float32)
        B = np_.zeros((257, 259), dtype=np.float32)
        for _ in range(3):
            B[1:-1, 1:-1] = (A[:-2, 1:-1] + A[2:, 1:-1] + A[1:-1, :-2]
                             + A[1:-1, 2:] - 4.0 * A[1:-1, 1:-1])
            A[1:-1, 1:-1] = 0.25 * B[1:-1, 1:-1]
        return A
    """, tol=1e-4)


def test_reductions_2rank_gpu():
    run_spmd_gpu("""
        A = np_.arange(100001) / 1000.0
        s = (np_.sin(A) ** 2 + np_.cos(A) ** 2).sum()
        a2 = np_.fromfunction(lambda x, y: x * 53 + y, (53, 71),
                              dtype=np.int64)
        ax = a2.sum(axis=0)
        cs = np_.arange(5000).cumsum()
        if np_ is np:
            return np.concatenate([[s], ax * 1.0, cs * 1.0])
        import numpy as _np
        return _np.concatenate([[float(s)], ax.asarray() * 1.0,
                                cs.asarray() * 1.0])
    """, tol=1e-12)


def test_wide_shift_temp_path_2rank_gpu():
    run_spmd_gpu("""
        A = np_.arange(3000) * 1.0
        return A[:-300] + A[300:]
    """)


def test_axis_cumsum_and_mask_2rank_gpu():
    """Cross-rank slab-exchange fixup (cumsum_axis_op) and mask-getitem
    count allgather + uneven divisions, on the real HIP backend."""
    run_spmd_gpu("""
        c = np_.fromfunction(lambda x, y: x * 97 + y, (401, 37))
        r0 = c.cumsum(axis=0)
        r1 = c.cumsum(axis=1)
        a = np_.arange(10000) * 1.0
        sel = a[(a % 7.0) == 0.0]
        if np_ is np:
            return np.concatenate([r0.reshape(-1), r1.reshape(-1), sel])
        import numpy as _np
        return _np.concatenate([r0.asarray().reshape(-1),
                                r1.asarray().reshape(-1), sel.asarray()])
    """, tol=1e-12)


def test_mixed_fused_sum_2rank_gpu():
    """configs[4] at world 2 with DEAD intermediates: the staged/tiled
    kernel runs on both ranks (rank-uniform decision), each folding its
    interior+rim partial, combined by allreduce."""
    run_spmd_gpu("""
        S = 512
        A = np_.zeros((S, S), dtype=np.float64)
        out = []
        for it in range(3):
            src = np_.fromfunction(
                lambda x, y: (x * S + y + it) * 1e-6, (S, S),
                dtype=np.float64)
            ssin = np_.sin(src)
            A[1:-1, 1:-1] = (ssin[:-2, 1:-1] + ssin[2:, 1:-1]
                             + ssin[1:-1, :-2] + ssin[1:-1, 2:]
                             - 4.0 * ssin[1:-1, 1:-1])
            del src, ssin
            out.append(float(A.sum()))
        return np.asarray(out)
    """, tol=1e-9)


def test_mixed_pipeline_2rank_gpu():
    """The full configs[4] shape at world 2: fused fill, stencil with halo
    exchange, fused reduction + allreduce, eager force each step."""
    run_spmd_gpu("""
        S = 512
        A = np_.zeros((S, S), dtype=np.float64)
        out = []
        for _ in range(3):
            src = np_.fromfunction(
                lambda x, y: (x * S + y) * 1e-6, (S, S), dtype=np.float64)
            ssin = np_.sin(src)
            A[1:-1, 1:-1] = (ssin[:-2, 1:-1] + ssin[2:, 1:-1]
                             + ssin[1:-1, :-2] + ssin[1:-1, 2:]
                             - 4.0 * ssin[1:-1, 1:-1])
            out.append(float(A.sum()))
        return np.asarray(out)
    """, tol=1e-9)


# ---------------------------------------------------------------------------
# RCCL transport proper: the SAME code paths the 8-GPU
# driver bench exercises — nccl(=RCCL) allreduce, batch_isend_irecv halo /
# part exchange, all_gather — executed for real with 2 ranks sharing the
# leased GPU (reference analog: the mpiexec -n 2 CI matrix,
# .github/workflows/python-package.yml:41-45).
# ---------------------------------------------------------------------------


@needs_multi_gpu
def test_flagship_2rank_rccl_gpu():
    run_spmd_gpu("""
        A = np_.arange(200000) / 1000.0
        D = np_.sin(A) ** 2 + np_.cos(A) ** 2
        return D
    """, tol=1e-12, backend="nccl")


@needs_multi_gpu
def test_stencil_halo_2rank_rccl_gpu():
    """Halo exchange over RCCL batch_isend_irecv (device buffers, no host
    staging)."""
    run_spmd_gpu("""
        A = np_.fromfunction(lambda x, y: x + y, (257, 259),
                             dtype=np.float32)
        B = np_.zeros((257, 259), dtype=np.float32)
        for _ in range(3):
            B[1:-1, 1:-1] = (A[:-2, 1:-1] + A[2:, 1:-1] + A[1:-1, :-2]
                             + A[1:-1, 2:] - 4.0 * A[1:-1, 1:-1])
            A[1:-1, 1:-1] = 0.25 * B[1:-1, 1:-1]
        return A
    """, tol=1e-4, backend="nccl")


@needs_multi_gpu
def test_reductions_2rank_rccl_gpu():
    """RCCL allreduce (sum), axis-reduction combining exchange, cumsum
    prefix all_gather."""
    run_spmd_gpu("""
        A = np_.arange(100001) / 1000.0
        s = (np_.sin(A) ** 2 + np_.cos(A) ** 2).sum()
        a2 = np_.fromfunction(lambda x, y: x * 53 + y, (53, 71),
                              dtype=np.int64)
        ax = a2.sum(axis=0)
        cs = np_.arange(5000).cumsum()
        if np_ is np:
            return np.concatenate([[s], ax * 1.0, cs * 1.0])
        import numpy as _np
        return _np.concatenate([[float(s)], ax.asarray() * 1.0,
                                cs.asarray() * 1.0])
    """, tol=1e-12, backend="nccl")


@needs_multi_gpu
def test_minmax_nan_2rank_rccl_gpu():
    """Cross-rank float min/max must propagate a NaN that lives on only
    one rank (NCCL MIN/MAX would drop it; we allgather)."""
    run_spmd_gpu("""
        A = np_.arange(10000) * 1.0
        B = np_.where(A == 9999.0, (A - 20000.0) ** 0.5, A)  # one NaN
        return np.asarray([float(B.min()), float(B.max()),
                           float(A.min()), float(A.max())])
    """, tol=0.0, backend="nccl")  # assert_allclose: NaN==NaN, else exact


@needs_multi_gpu
def test_mixed_pipeline_2rank_rccl_gpu():
    """configs[4] at world 2 over RCCL: fused fill, stencil halo exchange
    via batch_isend_irecv, reduction allreduce, every step."""
    run_spmd_gpu("""
        S = 512
        A = np_.zeros((S, S), dtype=np.float64)
        out = []
        for _ in range(3):
            src = np_.fromfunction(
                lambda x, y: (x * S + y) * 1e-6, (S, S), dtype=np.float64)
            ssin = np_.sin(src)
            A[1:-1, 1:-1] = (ssin[:-2, 1:-1] + ssin[2:, 1:-1]
                             + ssin[1:-1, :-2] + ssin[1:-1, 2:]
                             - 4.0 * ssin[1:-1, 1:-1])
            out.append(float(A.sum()))
        return np.asarray(out)
    """, tol=1e-9, backend="nccl")
