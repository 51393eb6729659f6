# ============================================================================
# ORACLE — TEST INFRASTRUCTURE ONLY.
#
# This package is the CPU restatement of the reference's fused execution path
# (Python-for-HPC/ramba, mounted read-only at /root/reference during
# development).  It exists to PIN SEMANTICS and CHECK the HIP product path.
#
# Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
# import, call, link or execute anything under oracle/.  The product package
# (ramba_amd) never imports it; on a GPU box the product ops fail loudly if
# the HIP extension is missing instead of falling back here.
#
# Parity pinning: the reference itself cannot run in this environment
# (ramba/ramba.py:23 `import numba`; numba/mpi4py/ray absent, no network).
# The reference's own CI oracle is NumPy (`run_both` harness,
# ramba/tests/test_distributed_array.py:240-259), so NumPy 2.2.6 is the
# executable oracle here; golden fixtures generated from it live in
# tests/golden/ with the generating script.
# ============================================================================
