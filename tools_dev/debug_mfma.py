# Diagnose the MFMA fragment mapping: run small multiplies and compare
# against plain numpy, printing the mismatch pattern.
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
from marlin_amd import Engine

eng = Engine(0)

# 1) 4x4 golden
m4 = np.array([[0., 1, 2, 3], [2, 3, 4, 5], [3, 2, 1, 0], [1, 1, 1, 1]])
c4 = np.array([[11., 10, 9, 8], [23, 24, 25, 26], [7, 11, 15, 19], [6, 7, 8, 9]])
got = eng.dgemm(m4, m4)
print("4x4 ok:", np.allclose(got, c4), "\n", got)

# 2) identity x asymmetric at one-tile size (128^3): C should equal B
n = 128
I = np.eye(n)
B = np.arange(n * n, dtype=np.float64).reshape(n, n) / (n * n)
C = eng.dgemm(I, B)
print("I@B == B:", np.allclose(C, B))
print("I@B == B.T:", np.allclose(C, B.T))
if not np.allclose(C, B):
    idx = np.argmax(np.abs(C - B) > 1e-12)
    r, c = np.unravel_index(idx, C.shape)
    print("first mismatch at", r, c, "got", C[r, c], "want", B[r, c])
    np.set_printoptions(precision=3, suppress=True, linewidth=200)
    print("C corner:\n", C[:18, :6])
    print("B corner:\n", B[:18, :6])

# 3) asymmetric x identity: C should equal A
A = np.arange(n * n, dtype=np.float64).reshape(n, n) / (n * n)
C2 = eng.dgemm(A, I)
print("A@I == A:", np.allclose(C2, A), " == A.T:", np.allclose(C2, A.T))

# 4) random 100x100 (the failing case)
from oracle import gen_matrix
a = gen_matrix(100, 100, seed=1)
b = gen_matrix(100, 100, seed=2)
got = eng.dgemm(a, b)
ref = a @ b
rel = np.max(np.abs(got - ref)) / np.max(np.abs(ref))
print("100x100 rel:", rel)
# 128-aligned random
a = gen_matrix(128, 128, seed=3)
b = gen_matrix(128, 128, seed=4)
got = eng.dgemm(a, b)
ref = a @ b
print("128x128 rel:", np.max(np.abs(got - ref)) / np.max(np.abs(ref)))
# 256-aligned
a = gen_matrix(256, 256, seed=5)
b = gen_matrix(256, 256, seed=6)
got = eng.dgemm(a, b)
ref = a @ b
print("256x256 rel:", np.max(np.abs(got - ref)) / np.max(np.abs(ref)))
eng.close()

# --- f32 probes ---
eng = Engine(0)
n = 128
I32 = np.eye(n, dtype=np.float32)
B32 = (np.arange(n * n, dtype=np.float32).reshape(n, n)) / (n * n)
C32 = eng.sgemm(I32, B32)
print("f32 I@B == B:", np.allclose(C32, B32, atol=1e-6))
if not np.allclose(C32, B32, atol=1e-6):
    np.set_printoptions(precision=3, suppress=True, linewidth=200)
    print("f32 C col0[:18]:", C32[:18, 0])
    print("f32 B col0[:18]:", B32[:18, 0])
a = gen_matrix(256, 256, seed=7, dtype=np.float32)
b = gen_matrix(256, 256, seed=8, dtype=np.float32)
ref = a.astype(np.float64) @ b.astype(np.float64)
got = eng.sgemm(a, b).astype(np.float64)
print("f32 256 rel:", np.max(np.abs(got - ref)) / np.max(np.abs(ref)))
# tn epilogue probe
add = gen_matrix(256, 256, seed=9, dtype=np.float32)
got_t = eng.sgemm_transpose_add(a, b, add).astype(np.float64)
ref_t = ref.T + add
print("f32 tn+add rel:", np.max(np.abs(got_t - ref_t)) / np.max(np.abs(ref_t)))
eng.close()
