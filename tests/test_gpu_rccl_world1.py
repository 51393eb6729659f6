"""World-1 RCCL transport coverage on ONE MI355X.

RCCL (librccl 2.26.6, like NCCL) refuses two ranks on one device
("Duplicate GPU detected", measured round 2 — see test_gpu_multirank.py),
so a single leased GPU cannot run a true 2-rank RCCL job.  What CAN run,
and what this file covers, is the real RCCL library on the real device
path: communicator creation against our backend's init path, all_reduce
and all_gather on device tensors through the same backend methods the
8-GPU driver bench uses, and a self-P2P batch_isend_irecv probe for the
halo-exchange call shape.  The 2-rank variants run unchanged on any
multi-GPU box (test_gpu_multirank.py, @needs_multi_gpu).
"""

import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(HERE)

SCRIPT = r"""
import os, sys
sys.path.insert(0, %(root)r)
import numpy as np
import torch
import ramba_amd as ra

ra.init()          # world 1 + RAMBA_FORCE_PG=1 -> nccl(=RCCL) comm init
rt = ra._deferred.get_runtime()
backend = rt.backend
import torch.distributed as dist
assert dist.is_initialized() and dist.get_backend() == "nccl"

# product path still green with the PG live
A = ra.arange(100000) / 1000.0
D = ra.sin(A) ** 2 + ra.cos(A) ** 2
ref = np.sin(np.arange(100000) * 0.001) ** 2 \
    + np.cos(np.arange(100000) * 0.001) ** 2
np.testing.assert_allclose(D.asarray(), ref, rtol=1e-12, atol=1e-12)

# backend.allreduce through RCCL on a device tensor (the cross-rank
# reduction finish the 8-GPU bench uses)
v = backend.allreduce(np.float64(41.5), "sum")
assert v == 41.5, v
v = backend.allreduce(np.int64(7), "max")
assert v == 7, v

# all_gather on device tensors (allgather_scalars' transport)
t = torch.full((4,), 3.25, dtype=torch.float64, device="cuda")
outs = [torch.empty_like(t)]
dist.all_gather(outs, t)
assert float(outs[0].sum().cpu()) == 13.0

# batch_isend_irecv self-exchange: the halo-exchange call shape.
a = torch.arange(1024, dtype=torch.float64, device="cuda")
b = torch.empty_like(a)
try:
    ops = [dist.P2POp(dist.isend, a, 0), dist.P2POp(dist.irecv, b, 0)]
    for req in dist.batch_isend_irecv(ops):
        req.wait()
    torch.cuda.synchronize()
    assert torch.equal(a, b)
    print("SELF_P2P OK")
except Exception as e:  # noqa: BLE001
    # capability note only: real P2P runs rank-to-rank on the 8-GPU box
    print("SELF_P2P UNSUPPORTED:", type(e).__name__, str(e)[:200])

dist.destroy_process_group()
print("WORLD1 RCCL OK")
"""


def test_world1_rccl_transport():
    env = dict(os.environ)
    env.update({"RANK": "0", "WORLD_SIZE": "1", "LOCAL_RANK": "0",
                "RAMBA_FORCE_PG": "1", "RAMBA_PG_BACKEND": "nccl",
                "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29627"})
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    p = subprocess.run([sys.executable, "-c", SCRIPT % {"root": ROOT}],
                       env=env, capture_output=True, timeout=240)
    out = p.stdout.decode() + p.stderr.decode()
    assert p.returncode == 0, out
    assert "WORLD1 RCCL OK" in out, out
