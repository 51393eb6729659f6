"""Golden-vector case definitions, shared by the generator
(tests/golden/generate.py — runs them under NumPy, the executable oracle,
and commits the outputs) and the GPU parity tests (which run them under
ramba_amd's HIP path).  Cases mirror the reference's KATs
(/root/reference/ramba/tests/test_distributed_array.py — see SURVEY §8c).
"""

import numpy as np


def _flagship(np_):
    A = np_.arange(100_000) / 1000.0
    B = np_.sin(A)
    C = np_.cos(A)
    return B * B + C ** 2


def _stencil1d(np_):
    A = np_.arange(10_001) * 1.0
    B = np_.zeros(10_001)
    B[2:-2] = (0.1 * A[:-4] + 0.2 * A[1:-3] + 0.4 * A[2:-2]
               + 0.2 * A[3:-1] + 0.1 * A[4:])
    return B


def _stencil2d(np_):
    A = np_.fromfunction(lambda x, y: x + y, (129, 131), dtype=np.float32)
    B = np_.zeros((129, 131), dtype=np.float32)
    B[1:-1, 1:-1] = (A[:-2, 1:-1] + A[2:, 1:-1] + A[1:-1, :-2]
                     + A[1:-1, 2:] - 4.0 * A[1:-1, 1:-1])
    return B


def _raw_antifusion(np_):
    # read-after-write KAT (reference test_distributed_array.py:34-45)
    B = np_.arange(5_000) * 1.0
    if np_ is np:
        B[:-1] = B[:-1] + B[1:].copy()
    else:
        B[:-1] += B[1:]
    return B


def _int_ops(np_):
    a = np_.arange(50_000) - 25_000
    return (a * 3 + 7) % 11 - (a // 13) + a ** 2


def _arange_exact(np_):
    return np_.arange(5, 300_001, 7)


def _reductions(np_):
    a = np_.arange(1_000_000) / 1000.0
    d = np_.sin(a) ** 2 + np_.cos(a) ** 2
    s = d.sum() if hasattr(d, "asarray") else d.sum()
    m = ((np_.arange(100_000) * 7919) % 104729)
    return np.array([float(s), float(m.max()), float(m.min())])


def _pi(np_):
    # pi-integration KAT (reference test_distributed_array.py:89-98)
    n = 1_000_000
    h = 1.0 / n
    x = h * (np_.arange(n) + 0.5)
    return np.array([float(4.0 * h * (1.0 / (1.0 + x * x)).sum())])


def _axis_cumsum(np_):
    c = np_.fromfunction(lambda x, y: x * 97 + y, (211, 67))
    r0 = c.cumsum(axis=0)
    r1 = c.cumsum(axis=1)
    i = np_.fromfunction(lambda x, y: x + y, (50, 33),
                         dtype=np.int32).cumsum(axis=0)
    out = [r0, r1, i]
    if hasattr(r0, "asarray"):
        out = [o.asarray() for o in out]
    return np.concatenate([np.asarray(o, dtype=np.float64).reshape(-1)
                           for o in out])


def _mask_getitem(np_):
    a = np_.arange(100_000) * 1.0
    sel = a[(a % 7.0) == 0.0]
    t = np_.fromfunction(lambda x, y: x * 31 + y, (300, 31))
    sel2 = t[(t % 13.0) == 0.0]
    if hasattr(sel, "asarray"):
        sel, sel2 = sel.asarray(), sel2.asarray()
    return np.concatenate([sel, sel2, [float(sel.size), float(sel2.size)]])


def _reshape(np_):
    a = np_.arange(60_060) * 1.0
    r1 = a.reshape(231, 260)
    b = np_.fromfunction(lambda i, j: i * 77.0 + j, (91, 77))
    r2 = b.T.reshape(7007)
    if hasattr(r1, "asarray"):
        r1, r2 = r1.asarray(), r2.asarray()
    return np.concatenate([np.asarray(r1).reshape(-1), np.asarray(r2)])


def _matmul(np_):
    A = np_.fromfunction(lambda i, j: i * 7.0 + j, (65, 33))
    B = np_.fromfunction(lambda i, j: i - 2.0 * j, (33, 49))
    v = np_.arange(33) * 1.0
    M = A @ B
    w = A @ v
    if hasattr(M, "asarray"):
        M, w = M.asarray(), w.asarray()
    return np.concatenate([np.asarray(M).reshape(-1), np.asarray(w)])


# name -> (fn, tolerance; 0 = bit-exact)
CASES_TOL = {
    "flagship": (_flagship, 1e-12),
    "stencil1d": (_stencil1d, 1e-12),
    "stencil2d": (_stencil2d, 1e-5),
    "raw_antifusion": (_raw_antifusion, 0),
    "int_ops": (_int_ops, 0),
    "arange_exact": (_arange_exact, 0),
    "reductions": (_reductions, 1e-12),
    "pi": (_pi, 1e-12),
    "axis_cumsum": (_axis_cumsum, 1e-9),
    "mask_getitem": (_mask_getitem, 0),
    "reshape": (_reshape, 0),
    "matmul": (_matmul, 1e-9),
}

CASES = {k: v[0] for k, v in CASES_TOL.items()}
