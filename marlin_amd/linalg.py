# linalg.py — the decomposition tier built on the GEMM engine:
# blocked LU and blocked inverse, restating the reference's distributed
# algorithms (DenseVecMatrix.luDecompose, DenseVecMatrix.scala:283-464;
# DenseVecMatrix.inverse, DenseVecMatrix.scala:565-764).
#
# Structure mirrors the reference exactly: the diagonal base block is
# factored on the HOST (the reference factors it on the Spark driver with
# Breeze LU/inv — a deliberate host step, not a fallback), and every
# panel scale and trailing-matrix update is a dense GEMM on the engine
# (the reference's executor-side SubMatrix.multiply chain).
#
# luDecompose: block pairwise pivoting (pivoting INSIDE each diagonal
# block only — the reference's scheme). Returns (blocks, p_array) with
# the property  P_blockdiag @ A == L @ U  (L unit-lower / U upper packed
# in the returned blocks; verified in tests).
# inverse: recursive block Gauss-Jordan; forward sweep stores
#   S1=A11^-1, S2=-A11^-1 A12, S3=-A21 A11^-1 and the Schur trailing
#   update; backward sweep recombines (DenseVecMatrix.scala:677-764).
import math

import numpy as np


def _ceil_block(total, parts):
    return int(math.ceil(total / parts))


def _split(n, base):
    nb = int(math.ceil(n / base))
    bl = _ceil_block(n, nb)
    offs = [i * bl for i in range(nb)]
    lens = [min(bl, n - o) for o in offs]
    return nb, offs, lens


def lu_decompose(dvm, mode="auto", base_size=1000):
    """DenseVecMatrix.luDecompose (DenseVecMatrix.scala:283-464).
    Returns (BlockMatrix with packed L\\U, p_array) where p_array[g] is
    the source row of output row g within its block row."""
    from .api import BlockMatrix
    import scipy.linalg

    a = dvm.toBreeze()
    n = a.shape[0]
    if a.shape[0] != a.shape[1]:
        raise ValueError(
            f"LU decompose only support square matrix: {a.shape[0]} v.s "
            f"{a.shape[1]}")
    if mode == "auto":
        mode = "dist" if n > 6000 else "local"
    if mode in ("local", "breeze"):
        lu, piv = scipy.linalg.lu_factor(a)
        p_array = np.arange(n)
        for i, p in enumerate(piv):
            p_array[i], p_array[p] = p_array[p], p_array[i]
        return (BlockMatrix({(0, 0): lu}, n, n, engine=dvm._eng), p_array)

    eng = dvm._engine()
    nb, offs, lens = _split(n, base_size)
    blk = {(i, j): a[offs[i]:offs[i] + lens[i],
                     offs[j]:offs[j] + lens[j]].copy()
           for i in range(nb) for j in range(nb)}
    p_array = np.arange(n)

    for i in range(nb):
        lu, piv = scipy.linalg.lu_factor(blk[(i, i)])
        perm = np.arange(lens[i])
        for r, p in enumerate(piv):
            perm[r], perm[p] = perm[p], perm[r]
        p_array[offs[i]:offs[i] + lens[i]] = offs[i] + perm
        blk[(i, i)] = lu                      # packed L\U of the diagonal
        if i == nb - 1:
            break
        L = np.tril(lu, -1) + np.eye(lens[i])
        U = np.triu(lu)
        P = np.zeros((lens[i], lens[i]))
        P[np.arange(lens[i]), perm] = 1.0     # row g of P·X is X[perm[g]]
        Minv_l = np.linalg.solve(L, P)        # L^-1 P   (host, base-sized)
        Uinv = np.linalg.inv(U)               # U^-1     (host, base-sized)
        # panel scales + trailing update — engine GEMMs
        for j in range(i + 1, nb):
            blk[(i, j)] = eng.dgemm(Minv_l, blk[(i, j)])   # U12 panel
        for r in range(i + 1, nb):
            blk[(r, i)] = eng.dgemm(blk[(r, i)], Uinv)     # L21 panel
        for r in range(i + 1, nb):
            neg_l = -blk[(r, i)]
            for j in range(i + 1, nb):
                # A22 -= L21 U12  (reference: A4 - A3 (A11 \\ A2))
                blk[(r, j)] = eng.tile_dgemm_acc(neg_l, blk[(i, j)],
                                                 blk[(r, j)])
    # sub-diagonal permutation fix-up (DenseVecMatrix.scala:444-462):
    # L21 block rows permuted by their OWN block row's perm
    for r in range(1, nb):
        perm = p_array[offs[r]:offs[r] + lens[r]] - offs[r]
        P = np.zeros((lens[r], lens[r]))
        P[np.arange(lens[r]), perm] = 1.0
        for c in range(r):
            blk[(r, c)] = P @ blk[(r, c)]
    return (BlockMatrix(blk, n, n, engine=dvm._eng), p_array)


def inverse(dvm, mode="auto", base_size=1000):
    """DenseVecMatrix.inverse (DenseVecMatrix.scala:565-764)."""
    from .api import BlockMatrix

    a = dvm.toBreeze()
    n = a.shape[0]
    if a.shape[0] != a.shape[1]:
        raise ValueError("inverse only supports square matrices")
    if mode == "auto":
        mode = "dist" if n > 6000 else "local"
    if mode in ("local", "breeze"):
        # the reference's LocalBreeze route: driver-side inverse
        # (DenseVecMatrix.scala:585-590)
        inv = np.linalg.inv(a)
        return BlockMatrix({(0, 0): inv}, n, n, engine=dvm._eng)

    eng = dvm._engine()
    nb, offs, lens = _split(n, base_size)
    blk = {(i, j): a[offs[i]:offs[i] + lens[i],
                     offs[j]:offs[j] + lens[j]].copy()
           for i in range(nb) for j in range(nb)}
    S1, S2, S3 = {}, {}, {}

    # forward sweep (DenseVecMatrix.scala:604-675)
    for i in range(nb - 1):
        inv = np.linalg.inv(blk[(i, i)])      # host base block (driver inv)
        S1[i] = inv
        for j in range(i + 1, nb):
            S2[(i, j)] = -eng.dgemm(inv, blk[(i, j)])      # -A11^-1 A12
        for r in range(i + 1, nb):
            S3[(r, i)] = -eng.dgemm(blk[(r, i)], inv)      # -A21 A11^-1
        for j in range(i + 1, nb):
            t = eng.dgemm(inv, blk[(i, j)])                # A11^-1 A12
            for r in range(i + 1, nb):
                neg = -blk[(r, i)]
                blk[(r, j)] = eng.tile_dgemm_acc(neg, t, blk[(r, j)])
    # last trailing block
    T = {(nb - 1, nb - 1): np.linalg.inv(blk[(nb - 1, nb - 1)])}

    # backward sweep (DenseVecMatrix.scala:677-764)
    for i in range(nb - 2, -1, -1):
        col = {}
        for r in range(i + 1, nb):
            acc = None
            for c in range(i + 1, nb):
                acc = eng.tile_dgemm_acc(T[(r, c)], S3[(c, i)], acc)
            col[r] = acc                                    # inv[r, i]
        row = {}
        for c in range(i + 1, nb):
            acc = None
            for j in range(i + 1, nb):
                acc = eng.tile_dgemm_acc(S2[(i, j)], T[(j, c)], acc)
            row[c] = acc                                    # inv[i, c]
        corner = S1[i].copy()
        for j in range(i + 1, nb):
            corner = eng.tile_dgemm_acc(S2[(i, j)], col[j], corner)
        newT = {(i, i): corner}
        for r in range(i + 1, nb):
            newT[(r, i)] = col[r]
        for c in range(i + 1, nb):
            newT[(i, c)] = row[c]
        for r in range(i + 1, nb):
            for c in range(i + 1, nb):
                newT[(r, c)] = T[(r, c)]
        T = newT
    return BlockMatrix(T, n, n, engine=dvm._eng)
