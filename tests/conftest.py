import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on an MI355X box)")


@pytest.fixture(scope="session")
def ra():
    """ramba_amd initialised with the CPU oracle backend (test-only)."""
    import ramba_amd
    from oracle.numpy_backend import NumpyBackend
    ramba_amd.init(backend=NumpyBackend())
    return ramba_amd


@pytest.fixture(scope="session")
def ra_gpu():
    """ramba_amd initialised with the HIP product backend."""
    import ramba_amd
    from ramba_amd import deferred
    if not ramba_amd._initialized["done"]:
        ramba_amd.init()  # product default: HIP, fails loudly without GPU
    else:
        rt = deferred.get_runtime()
        if getattr(rt.backend, "name", "") != "hip":
            # a CPU-backend test ran first in this process; swap in HIP
            from ramba_amd.hip_backend import HipBackend
            deferred.flush()
            be = HipBackend()
            rt.backend = be
            be.attach(rt)
    return ramba_amd


def run_both(impl, ra, comparator=None, tol=None):
    """The reference's parity harness (run_both,
    /root/reference/ramba/tests/test_distributed_array.py:240-259): run the
    same program under ramba_amd and under NumPy, compare."""
    res_r = impl(ra)
    res_n = impl(np)
    if hasattr(res_r, "asarray"):
        res_r = res_r.asarray()
    if comparator is not None:
        assert comparator(res_r, res_n)
    elif tol is not None:
        np.testing.assert_allclose(res_r, res_n, rtol=tol, atol=tol)
    else:
        assert np.array_equal(res_r, res_n), f"{res_r} != {res_n}"
    return res_r, res_n
