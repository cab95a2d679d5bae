#!/usr/bin/env python3
"""Generate the frozen golden fixtures under tests/golden/.

Run ONCE in the survey/build container where /root/reference is mounted;
the outputs are committed so GPU-box runs never read /root/reference.

  c100.npy       — product of the reference's own data/a.100.100 x
                   data/b.100.100 text fixtures (format MTUtils.scala:292-298),
                   computed by the oracle restatement's broadcast route
                   (a single fp64 dgemm) and cross-checked against every
                   blocked split mode.
  a100.npy/b100.npy — the parsed inputs (so GPU tests don't re-parse text
                   or touch /root/reference).
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
from oracle import load_matrix_file, blocked_multiply  # noqa: E402

REF = "/root/reference"
HERE = os.path.dirname(os.path.abspath(__file__))


def main():
    a = load_matrix_file(os.path.join(REF, "data", "a.100.100"))
    b = load_matrix_file(os.path.join(REF, "data", "b.100.100"))
    assert a.shape == (100, 100) and b.shape == (100, 100), (a.shape, b.shape)
    c = a @ b
    # pin the blocked path against the plain product before freezing
    for mkn in [(2, 2, 1), (2, 1, 2), (2, 2, 2), (3, 3, 3), (7, 5, 3)]:
        cb = blocked_multiply(a, b, mkn)
        rel = np.max(np.abs(cb - c)) / np.max(np.abs(c))
        assert rel < 1e-12, (mkn, rel)
    np.save(os.path.join(HERE, "a100.npy"), a)
    np.save(os.path.join(HERE, "b100.npy"), b)
    np.save(os.path.join(HERE, "c100.npy"), c)
    print("frozen:", a.shape, b.shape, c.shape, "max|c| =", np.max(np.abs(c)))


if __name__ == "__main__":
    main()
