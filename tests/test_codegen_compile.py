"""Validate the HIP code generator on CPU: every fused-group kernel the
in-scope workloads produce must compile cleanly for gfx950 via hiprtc
(rt_compile_check needs no GPU), and the oracle's numpy evaluation of the
same plan provides the values (checked against plain NumPy by run_both).
"""

import ctypes
import os

import numpy as np
import pytest

import ramba_amd
from ramba_amd import codegen
from ramba_amd.hip_backend import LIBPATH
from oracle.numpy_backend import NumpyBackend

from conftest import run_both


class CompileCheckBackend(NumpyBackend):
    """Oracle backend that additionally feeds every generated kernel
    through hiprtc for gfx950 (compile-only; also seeds the on-disk
    kernel cache that ships to GPU boxes)."""

    def __init__(self):
        super().__init__()
        assert os.path.exists(LIBPATH), \
            f"{LIBPATH} missing — run __graft_entry__.build() first"
        kc = os.path.join(os.path.dirname(LIBPATH), "..", "_kcache")
        os.environ.setdefault("RAMBA_KCACHE", os.path.abspath(kc))
        os.makedirs(os.environ["RAMBA_KCACHE"], exist_ok=True)
        self.lib = ctypes.CDLL(LIBPATH)
        self.lib.rt_compile_check.argtypes = [ctypes.c_char_p]
        self.lib.rt_last_error.restype = ctypes.c_char_p
        self.compiled = {}

    def launch(self, plan, recipe=None):
        gk = codegen.generate(plan)
        if gk.key not in self.compiled:
            rc = self.lib.rt_compile_check(gk.source.encode())
            assert rc == 0, (
                "hiprtc rejected generated kernel:\n"
                + self.lib.rt_last_error().decode() + "\n" + gk.source)
            if gk.finish_source:
                rc = self.lib.rt_compile_check(gk.finish_source.encode())
                assert rc == 0, (
                    "hiprtc rejected finish kernel:\n"
                    + self.lib.rt_last_error().decode())
            self.compiled[gk.key] = gk
        return super().launch(plan, recipe)


@pytest.fixture(scope="module")
def rac():
    """Swap the compile-check backend into the live runtime (restored
    afterwards so other test modules keep their NumpyBackend state)."""
    import ramba_amd as ra
    from ramba_amd import deferred
    if not ra._initialized["done"]:
        ra.init(backend=CompileCheckBackend())
        yield ra
        return
    deferred.flush()
    rt = deferred.get_runtime()
    old = rt.backend
    new = CompileCheckBackend()
    rt.backend = new
    new.attach(rt)
    yield ra
    deferred.flush()
    rt.backend = old


def test_flagship_chain_kernel_compiles(rac):
    def impl(np_):
        A = np_.arange(2048) / 1000.0
        B = np_.sin(A)
        C = np_.cos(A)
        D = B * B + C ** 2
        return D
    run_both(impl, rac, tol=1e-12)
    be = rac._deferred.get_runtime().backend
    assert len(be.compiled) >= 1
    # the flagship kernel must contain a fused sincos and 4 vector stores
    src = next(iter(be.compiled.values())).source
    assert "sincos" in src, src


def test_reduction_kernel_compiles(rac):
    def impl(np_):
        A = np_.arange(4096) / 64.0
        return (np_.sin(A) ** 2 + np_.cos(A) ** 2).sum()
    run_both(impl, rac, tol=1e-12)


def test_stencil_kernel_compiles(rac):
    def impl(np_):
        A = np_.fromfunction(lambda x, y: x + y, (64, 64), dtype=np.float32)
        B = np_.zeros((64, 64), dtype=np.float32)
        B[1:-1, 1:-1] = (A[:-2, 1:-1] + A[2:, 1:-1] + A[1:-1, :-2]
                         + A[1:-1, 2:] - 4.0 * A[1:-1, 1:-1])
        return B
    run_both(impl, rac)


def test_int_ops_kernel_compiles(rac):
    def impl(np_):
        a = np_.arange(1000)
        return (a * 3 + 7) % 11 - (a // 13) + np_.where(a % 2 == 0, a, -a)
    run_both(impl, rac)


def test_minmax_reduction_compiles(rac):
    def impl(np_):
        a = (np_.arange(3000) * 7919) % 104729
        return a.min() + a.max()
    run_both(impl, rac)


def test_misc_unops_compile(rac):
    def impl(np_):
        a = np_.arange(500) * 0.01 + 0.001
        return (np_.sqrt(a) + np_.exp(-a) + np_.log(a) + np_.tanh(a)
                + np_.arctan(a))
    run_both(impl, rac, tol=1e-13)


def test_axis_reduce_kernels_compile():
    """Every axis-reduce kernel shape compiles for gfx950 (hiprtc)."""
    import ctypes
    import itertools
    from ramba_amd.codegen import generate_axis_reduce
    lib = ctypes.CDLL(LIBPATH)
    lib.rt_compile_check.argtypes = [ctypes.c_char_p]
    lib.rt_last_error.restype = ctypes.c_char_p
    cases = [
        (2, (0,), np.int64, np.int64, "sum"),
        (2, (1,), np.float64, np.float64, "sum"),     # lane-split
        (2, (0,), np.float32, np.float32, "max"),
        (2, (1,), np.int32, np.int32, "min"),
        (3, (1,), np.float64, np.float64, "sum"),
        (3, (0, 2), np.int64, np.int64, "prod"),      # mixed + lane-split
        (2, (1,), np.int64, np.bool_, "any"),
        (2, (0,), np.float64, np.bool_, "all"),
    ]
    for nd, axes, idt, odt, kind in cases:
        key, src, kname, fields, ls = generate_axis_reduce(
            nd, axes, np.dtype(idt), np.dtype(odt), kind)
        rc = lib.rt_compile_check(src.encode())
        assert rc == 0, (f"{nd}d axes={axes} {kind}: "
                         + lib.rt_last_error().decode() + "\n" + src)
    # chunked stage-1 variants (small-nout parallel split)
    for nd, axes, idt, odt, kind in [
            (2, (0,), np.float64, np.float64, "sum"),
            (2, (0,), np.int64, np.int64, "max"),
            (3, (0,), np.float32, np.float32, "sum")]:
        key, src, kname, fields, ls = generate_axis_reduce(
            nd, axes, np.dtype(idt), np.dtype(odt), kind, chunked=True)
        rc = lib.rt_compile_check(src.encode())
        assert rc == 0, lib.rt_last_error().decode() + "\n" + src


def test_4d_kernel_compiles(rac):
    def impl(np_):
        a = np_.fromfunction(
            lambda w, x, y, z: w + x * 2 + y * 3 + z, (3, 4, 5, 8),
            dtype=np.int64)
        return a * 3 + a[:, :, :, ::-1]
    run_both(impl, rac)
