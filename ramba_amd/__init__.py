"""ramba_amd — MI355X-native backend for Ramba's deferred/fused hot path.

Drop-in surface for the in-scope path:  `import ramba_amd as np` then use
arange / sin / cos / slicing / reductions / sync exactly as with the
reference (`import ramba as np`, /root/reference/README.md:24-36).

SPMD: launch one process per GPU (torch.distributed env: RANK / WORLD_SIZE /
MASTER_ADDR / MASTER_PORT).  Single-process single-GPU needs no env.

The product backend is HIP (hand-written gfx950 kernels via the C-ABI runtime
`libramba_rt.so`); it FAILS LOUDLY if the extension or a GPU is missing.
Tests inject the CPU oracle backend explicitly (oracle/numpy_backend.py).
"""

import os

from . import deferred as _deferred
from .runtime import Runtime
from .ndarray import *          # noqa: F401,F403  (the API surface)
from .ndarray import ndarray    # noqa: F401
from .ndarray import _MOD_DELEGATES as _mdel, _module_delegate as _mkdel
for _n in _mdel:
    globals()[_n] = _mkdel(_n)
del _mdel, _mkdel
from .common import dprint, get_timing, reset_timing  # noqa: F401

__version__ = "0.1.0"

_initialized = {"done": False}


def init(backend=None, world_size=None, rank=None):
    """Initialise the SPMD runtime.

    backend=None selects the HIP product backend (requires a visible GPU and
    the built `libramba_rt.so`).  Tests pass an explicit backend object.
    """
    if _initialized["done"]:
        return
    rank = int(os.environ.get("RANK", "0")) if rank is None else rank
    world = int(os.environ.get("WORLD_SIZE", "1")) if world_size is None \
        else world_size

    if backend is None:
        from .hip_backend import HipBackend
        backend = HipBackend()

    if world > 1 or int(os.environ.get("RAMBA_FORCE_PG", "0")):
        # RAMBA_FORCE_PG=1 initialises the process group even at world 1
        # (world-1 RCCL transport tests on a single-GPU box)
        backend.init_process_group(rank, world)

    rt = Runtime(backend, rank=rank, world=world)
    backend.attach(rt)
    _deferred.set_runtime(rt)
    _initialized["done"] = True


def _auto_init():
    if not _initialized["done"]:
        init()


# auto-init on first runtime access (the reference initialises at import,
# ramba/ramba.py:10646; we defer to first use so tests can inject a backend)
_orig_get_runtime = _deferred.get_runtime


def _get_runtime_auto():
    if not _initialized["done"]:
        init()
    return _orig_get_runtime()


_deferred.get_runtime = _get_runtime_auto


def shutdown():
    _deferred.flush()
    from .common import ntiming, get_timing
    if ntiming >= 1:
        print("[ramba_amd timing] tag: (count, seconds)")
        for tag, (c, secs) in sorted(get_timing().items()):
            print(f"  {tag}: ({c}, {secs:.6f})")
    _initialized["done"] = False
    _deferred.set_runtime(None)
