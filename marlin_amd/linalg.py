# linalg.py — the decomposition tier built on the GEMM engine:
# blocked LU and blocked inverse, restating the reference's distributed
# algorithms (DenseVecMatrix.luDecompose, DenseVecMatrix.scala:283-464;
# DenseVecMatrix.inverse, DenseVecMatrix.scala:565-764).
#
# Structure mirrors the reference exactly: the diagonal base block is
# factored on the HOST (the reference factors it on the Spark driver with
# Breeze LU/inv — a deliberate host step, not a fallback), and every
# panel scale and trailing-matrix update is a dense GEMM on the engine.
# All blocks stay DEVICE-RESIDENT (DeviceMatrix) across the sweeps —
# the RDD.cache() analog — so only the base-sized diagonal factors cross
# PCIe; sign flips are folded into the small host factors so no device
# elementwise pass is needed.
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


def _split(n, base):
    nb = int(math.ceil(n / base))
    bl = int(math.ceil(n / nb))
    offs = [i * bl for i in range(nb)]
    lens = [min(bl, n - o) for o in offs]
    return nb, offs, lens


def _upload_blocks(eng, a, nb, offs, lens):
    return {(i, j): eng.upload_matrix(a[offs[i]:offs[i] + lens[i],
                                        offs[j]:offs[j] + lens[j]])
            for i in range(nb) for j in range(nb)}


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
    blk = _upload_blocks(eng, a, nb, offs, lens)
    host = {}                       # finished blocks, host-side
    p_array = np.arange(n)

    for i in range(nb):
        diag = eng.download_matrix(blk[(i, i)])
        lu, piv = scipy.linalg.lu_factor(diag)
        perm = np.arange(lens[i])
        for r, p in enumerate(piv):
            perm[r], perm[p] = perm[p], perm[r]
        p_array[offs[i]:offs[i] + lens[i]] = offs[i] + perm
        host[(i, i)] = lu                     # packed L\U of the diagonal
        if i == nb - 1:
            break
        L = np.tril(lu, -1) + np.eye(lens[i])
        U = np.triu(lu)
        P = np.zeros((lens[i], lens[i]))
        P[np.arange(lens[i]), perm] = 1.0     # row g of P·X is X[perm[g]]
        d_Minv = eng.upload_matrix(np.linalg.solve(L, P))     # L^-1 P
        d_negMinv = eng.upload_matrix(-np.linalg.solve(L, P))
        d_Uinv = eng.upload_matrix(np.linalg.inv(U))          # U^-1
        # panel scales + trailing update — device-resident GEMMs
        negA2 = {}
        for j in range(i + 1, nb):
            negA2[j] = eng.gemm_dd(d_negMinv, blk[(i, j)])    # -L^-1 P A12
            new = eng.gemm_dd(d_Minv, blk[(i, j)])            # U12 panel
            blk[(i, j)].free()
            blk[(i, j)] = new
        for r in range(i + 1, nb):
            new = eng.gemm_dd(blk[(r, i)], d_Uinv)            # L21 panel
            blk[(r, i)].free()
            blk[(r, i)] = new
        for r in range(i + 1, nb):
            for j in range(i + 1, nb):
                # A22 += L21 · (-U12)  (reference: A4 - A3 (A11 \\ A2))
                eng.gemm_dd(blk[(r, i)], negA2[j], blk[(r, j)],
                            accumulate=True)
        for d in negA2.values():
            d.free()
        d_Minv.free(); d_negMinv.free(); d_Uinv.free()

    # materialise + sub-diagonal permutation fix-up
    # (DenseVecMatrix.scala:444-462): L21 block rows permuted by their
    # OWN block row's perm
    for (i, j), d in blk.items():
        if (i, j) not in host:
            host[(i, j)] = eng.download_matrix(d)
        d.free()
    for r in range(1, nb):
        perm = p_array[offs[r]:offs[r] + lens[r]] - offs[r]
        P = np.zeros((lens[r], lens[r]))
        P[np.arange(lens[r]), perm] = 1.0
        for c in range(r):
            host[(r, c)] = P @ host[(r, c)]
    return (BlockMatrix(host, n, n, engine=dvm._eng), p_array)


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
    blk = _upload_blocks(eng, a, nb, offs, lens)
    S1, S2, S3 = {}, {}, {}

    # forward sweep (DenseVecMatrix.scala:604-675)
    for i in range(nb - 1):
        inv = np.linalg.inv(eng.download_matrix(blk[(i, i)]))
        S1[i] = inv
        d_inv = eng.upload_matrix(inv)
        d_neg = eng.upload_matrix(-inv)
        for j in range(i + 1, nb):
            S2[(i, j)] = eng.gemm_dd(d_neg, blk[(i, j)])      # -A11^-1 A12
        for r in range(i + 1, nb):
            S3[(r, i)] = eng.gemm_dd(blk[(r, i)], d_neg)      # -A21 A11^-1
        for r in range(i + 1, nb):
            for j in range(i + 1, nb):
                # A22 += A21 · (-A11^-1 A12)
                eng.gemm_dd(blk[(r, i)], S2[(i, j)], blk[(r, j)],
                            accumulate=True)
        d_inv.free(); d_neg.free()
    last = eng.download_matrix(blk[(nb - 1, nb - 1)])
    T = {(nb - 1, nb - 1): eng.upload_matrix(np.linalg.inv(last))}

    # backward sweep (DenseVecMatrix.scala:677-764) — device-resident
    for i in range(nb - 2, -1, -1):
        col = {}
        for r in range(i + 1, nb):
            acc = None
            for c in range(i + 1, nb):
                if acc is None:
                    acc = eng.gemm_dd(T[(r, c)], S3[(c, i)])
                else:
                    eng.gemm_dd(T[(r, c)], S3[(c, i)], acc, accumulate=True)
            col[r] = acc                                      # inv[r, i]
        row = {}
        for c in range(i + 1, nb):
            acc = None
            for j in range(i + 1, nb):
                if acc is None:
                    acc = eng.gemm_dd(S2[(i, j)], T[(j, c)])
                else:
                    eng.gemm_dd(S2[(i, j)], T[(j, c)], acc, accumulate=True)
            row[c] = acc                                      # inv[i, c]
        corner = eng.upload_matrix(S1[i])
        for j in range(i + 1, nb):
            eng.gemm_dd(S2[(i, j)], col[j], corner, accumulate=True)
        T[(i, i)] = corner
        for r in range(i + 1, nb):
            T[(r, i)] = col[r]
        for c in range(i + 1, nb):
            T[(i, c)] = row[c]

    out = {k: eng.download_matrix(d) for k, d in T.items()}
    for d in T.values():
        d.free()
    for d in list(S2.values()) + list(S3.values()) + list(blk.values()):
        d.free()
    return BlockMatrix(out, n, n, engine=dvm._eng)


def cholesky_decompose(dvm, mode="auto", base_size=1000):
    """DenseVecMatrix.choleskyDecompose (DenseVecMatrix.scala:475-566):
    A = L L^T, lower-triangular L. Right-looking blocked: host chol of
    the (symmetrized, as the reference does) diagonal block, panel scale
    L21 = A21 L11^-T and trailing A22 -= L21 L21^T as device-resident
    engine GEMMs; signs folded into the host factor."""
    from .api import BlockMatrix

    a = dvm.toBreeze()
    n = a.shape[0]
    if a.shape[0] != a.shape[1]:
        raise ValueError(
            f"LU decompose only support square matrix: {n} v.s {n}")
    if mode == "auto":
        mode = "dist" if n > 6000 else "local"
    if mode in ("local", "breeze"):
        L = np.linalg.cholesky(np.tril(a) + np.tril(a, -1).T)
        return BlockMatrix({(0, 0): L}, n, n, engine=dvm._eng)

    eng = dvm._engine()
    nb, offs, lens = _split(n, base_size)
    blk = {(i, j): eng.upload_matrix(a[offs[i]:offs[i] + lens[i],
                                       offs[j]:offs[j] + lens[j]])
           for i in range(nb) for j in range(nb) if i >= j}
    host = {}
    for i in range(nb):
        diag = eng.download_matrix(blk[(i, i)])
        diag = np.tril(diag) + np.tril(diag, -1).T   # symmetrize (ref does)
        L11 = np.linalg.cholesky(diag)
        host[(i, i)] = L11
        if i == nb - 1:
            break
        linvT = np.linalg.inv(L11).T                 # L11^-T (host, base)
        d_pos = eng.upload_matrix(linvT)
        d_neg = eng.upload_matrix(-linvT)
        negT = {}
        for r in range(i + 1, nb):
            old = blk[(r, i)]
            l21 = eng.gemm_dd(old, d_pos)            # L21 panel
            negp = eng.gemm_dd(old, d_neg)           # -L21
            negT[r] = eng.transpose_dd(negp)         # -L21^T
            negp.free()
            old.free()
            blk[(r, i)] = l21
        for c in range(i + 1, nb):
            for r in range(c, nb):
                # A22 += L21[r] @ (-L21[c]^T)
                eng.gemm_dd(blk[(r, i)], negT[c], blk[(r, c)],
                            accumulate=True)
        for d in negT.values():
            d.free()
        d_pos.free()
        d_neg.free()

    for (i, j), d in blk.items():
        if (i, j) not in host:
            host[(i, j)] = eng.download_matrix(d)
        d.free()
    # upper-triangle blocks are zero in the L result
    for i in range(nb):
        for j in range(i + 1, nb):
            host[(i, j)] = np.zeros((lens[i], lens[j]))
    # keep strictly-lower of diagonal factor only (chol returns lower)
    return BlockMatrix(host, n, n, engine=dvm._eng)
