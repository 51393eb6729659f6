"""Generate the committed golden fixtures from the NumPy oracle.

NumPy is the same executable oracle the reference's own CI compares
against (run_both, /root/reference/ramba/tests/test_distributed_array.py:
240-259).  Run from the repo root:  python tests/golden/generate.py
"""

import os
import sys

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, os.path.dirname(HERE))

from golden_cases import CASES_TOL  # noqa: E402


def main():
    for name, (fn, tol) in CASES_TOL.items():
        out = fn(np)
        path = os.path.join(HERE, f"{name}.npz")
        np.savez_compressed(path, out=np.asarray(out), tol=np.float64(tol))
        print(f"{name}: shape={np.asarray(out).shape} tol={tol} -> {path}")


if __name__ == "__main__":
    main()
