"""Targeted fuzzer for the cross-stage fusion path (ramba_amd/staged.py).

The general program fuzzer almost never seals a producer (its chains read
materialised arrays, which breaks index-purity), so this generator builds
exactly the staged shape space: a random INDEX-PURE producer chain
(iota-based arithmetic + transcendentals), a consumer combining several
randomly-shifted reads of the producer outputs (plus optional independent
HBM operands and compound assignment), random liveness (intermediates
deleted or kept), odd array shapes, and an optional trailing fused
`sum()`.  Compared against direct NumPy; on the HIP backend this
exercises the tiled kernel (rolling ring, halo recompute, fused-sum rim
fold), on other backends the sequential fallback."""

import numpy as np


def build_staged_program(seed):
    rng = np.random.default_rng(seed)
    S0 = int(rng.integers(34, 500))
    S1 = int(rng.integers(34, 500))
    r0 = int(rng.integers(1, 4))       # max shift radius per axis
    r1 = int(rng.integers(1, 4))
    nreads = int(rng.integers(2, 6))
    offs = [(int(rng.integers(0, 2 * r0 + 1)),
             int(rng.integers(0, 2 * r1 + 1))) for _ in range(nreads)]
    coefs = [round(float(rng.uniform(-2, 2)), 3) for _ in range(nreads)]
    producer_kind = int(rng.integers(0, 4))
    use_second = bool(rng.integers(0, 2))
    use_w = bool(rng.integers(0, 2))
    del_mid = bool(rng.integers(0, 2))
    do_sum = bool(rng.integers(0, 2))
    compound = bool(rng.integers(0, 2))
    scale = float(rng.uniform(1e-6, 1e-3))
    h0 = S0 - 2 * r0
    h1 = S1 - 2 * r1

    def impl(np_):
        is_np = np_ is np
        A = np_.zeros((S0, S1), dtype=np.float64)
        if use_w:
            W = np_.fromfunction(lambda x, y: (x % 11) * 0.3 + y * 0.01,
                                 (S0, S1), dtype=np.float64)
        if not is_np:
            # materialise A/W first so the producer group holds only the
            # index-pure chain (otherwise the pair correctly falls back:
            # the consumer would write an array the producer also wrote)
            np_.sync()
        src = np_.fromfunction(
            lambda x, y: (x * S1 + y) * scale, (S0, S1), dtype=np.float64)
        if producer_kind == 0:
            mid = np_.sin(src)
        elif producer_kind == 1:
            mid = np_.cos(src) + src * 0.5
        elif producer_kind == 2:
            mid = np_.sqrt(src + 1.0) - src
        else:
            mid = np_.tanh(src) * 2.0
        expr = None
        for (o0, o1), c in zip(offs, coefs):
            term = c * mid[o0:o0 + h0, o1:o1 + h1]
            expr = term if expr is None else expr + term
        if use_w:
            expr = expr * W[r0:r0 + h0, r1:r1 + h1]
        tgt = (slice(r0, r0 + h0), slice(r1, r1 + h1))
        if compound:
            if is_np:
                A[tgt] = A[tgt] + expr
            else:
                A[tgt] += expr
        else:
            A[tgt] = expr
        if del_mid:
            del src, mid
        outs = []
        if do_sum:
            outs.append(float(A.sum()))
        a = A if is_np else A.asarray()
        m = None
        if not del_mid:
            m = mid if is_np else mid.asarray()
        return (np.asarray(a, dtype=np.float64),
                None if m is None else np.asarray(m, dtype=np.float64),
                np.asarray(outs, dtype=np.float64))

    return impl


def check_staged_seed(ra_module, seed):
    impl = build_staged_program(seed)
    with np.errstate(all="ignore"):
        ga, gm, gs = impl(ra_module)
        na, nm, ns = impl(np)
    np.testing.assert_allclose(ga, na, rtol=1e-11, atol=1e-11,
                               err_msg=f"staged fuzz seed {seed} (A)")
    if nm is not None:
        np.testing.assert_allclose(gm, nm, rtol=1e-11, atol=1e-11,
                                   err_msg=f"staged fuzz seed {seed} (mid)")
    np.testing.assert_allclose(gs, ns, rtol=1e-9, atol=1e-9,
                               err_msg=f"staged fuzz seed {seed} (sum)")
