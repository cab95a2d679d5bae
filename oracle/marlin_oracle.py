# marlin_oracle — numpy restatement of the reference's (PasaLab/marlin)
# BlockMatrix.multiply hot path, followed function by function from the
# Scala sources under /root/reference (cited per function).
#
# TEST INFRASTRUCTURE ONLY — see oracle/__init__.py header. The reference
# itself (Scala 2.10 / Spark 1.4) cannot be compiled or executed in this
# container (no JVM/Maven — probed), so this restatement is pinned against
# the reference's OWN golden vectors instead:
#   - the 4x4 multiply literals of DistributedMatrixSuite.scala:15-24,
#     225-297 (all split modes + per-block expected tiles) -> tests/test_oracle.py
#   - the 100x100 text fixtures data/a.100.100 / b.100.100
#     (format MTUtils.scala:292-298); the product is frozen once under
#     tests/golden/ by tests/golden/make_golden.py.
# Parity is therefore PINNED by reference-authored literals + frozen fixture.
#
# Third-party arithmetic not under /root/reference: the per-tile dgemm is
# Breeze 0.11.2 -> netlib-java (pom.xml:114-118). Its published algorithm is
# IEEE-754 fp64 BLAS dgemm; numpy's OpenBLAS dgemm is the same arithmetic
# class. fp64 differences are << the 1e-10 relative parity bar at every
# config size (error growth ~ sqrt(K)*eps ~ 1e-13 at K=40000).
import math
import re

import numpy as np

__all__ = [
    "split_method", "slab_len", "slab_off", "to_blocks",
    "blocked_multiply", "block_matrix_multiply", "multiply_dispatch",
    "load_matrix_file", "gen_matrix", "gen_uniform_u64",
]


# ---------------------------------------------------------------------------
# CARMA-style split planner — MTUtils.scala:150-175 (splitMethod) and
# :204-213 (dimToSplit). Recursively halves the largest of (m, k, n)
# until cores are exhausted or any live dimension reaches 1.
# dimToSplit ties: n wins over m wins over k (>= comparisons).
def split_method(m, k, n, cores):
    m_split = k_split = n_split = 1
    _m, _k, _n, _cores = int(m), int(k), int(n), int(cores)
    while _cores > 1 and _m > 1 and _k > 1 and _n > 1:
        if _n >= _k and _n >= _m:          # dimToSplit == 1
            n_split *= 2
            _n //= 2
        elif _m >= _k and _m >= _n:        # dimToSplit == 2
            m_split *= 2
            _m //= 2
        else:                              # dimToSplit == 3
            k_split *= 2
            _k //= 2
        _cores //= 2
    return (m_split, k_split, n_split)


# ---------------------------------------------------------------------------
# Ceil-based blocking — DenseVecMatrix.scala:1091-1094 / 1262-1265:
# block length = ceil(total/parts); the number of EFFECTIVE blocks is
# ceil(total/block_len) (can be < parts); the last block is ragged.
def _block_len(total, parts):
    return int(math.ceil(total / parts))


def effective_blocks(total, parts):
    bl = _block_len(total, parts)
    return int(math.ceil(total / bl))


def slab_len(total, parts, idx):
    """Length of ceil-split block idx (mirrors toBlocks's smRows/smCols,
    DenseVecMatrix.scala:1137-1146)."""
    bl = _block_len(total, parts)
    start = idx * bl
    if start >= total:
        return 0
    return min(bl, total - start)


def slab_off(total, parts, idx):
    return idx * _block_len(total, parts)


def to_blocks(mat, row_parts, col_parts):
    """Partition a 2-D array into the reference's ceil-sized tile grid
    (DenseVecMatrix.scala:1084-1223 'right'/'left' tiling without the
    emit replication). Returns dict {(bi, bj): tile}."""
    rows, cols = mat.shape
    nbr = effective_blocks(rows, row_parts)
    nbc = effective_blocks(cols, col_parts)
    brl = _block_len(rows, row_parts)
    bcl = _block_len(cols, col_parts)
    out = {}
    for bi in range(nbr):
        for bj in range(nbc):
            out[(bi, bj)] = mat[bi * brl: min((bi + 1) * brl, rows),
                                bj * bcl: min((bj + 1) * bcl, cols)]
    return out


# ---------------------------------------------------------------------------
# The hot multiply.
def block_matrix_multiply(a_blocks, b_blocks, mkn):
    """RMM core — BlockMatrix.multiply (BlockMatrix.scala:149-220):
    emit A(i,l) keyed BlockID(i,j,seq=i*n*k+j*k+l), B(l,j) the same key,
    join -> SubMatrix.multiply (Breeze dgemm, SubMatrix.scala:87-105),
    k>1 -> reduceByKey add (SubMatrix.scala:41-50). The reduce order is
    restated as ascending l (Spark's order is nondeterministic; the
    parity bar absorbs reorder error). Returns dict {(i,j): C_tile}."""
    m_split, k_split, n_split = mkn
    out = {}
    for (i, j) in [(i, j) for i in range(m_split) for j in range(n_split)]:
        acc = None
        for l in range(k_split):
            if (i, l) not in a_blocks or (l, j) not in b_blocks:
                continue
            prod = a_blocks[(i, l)] @ b_blocks[(l, j)]
            acc = prod if acc is None else acc + prod
        if acc is not None:
            out[(i, j)] = acc
    return out


def assemble(blocks):
    """BlockMatrix -> dense (toBreeze, BlockMatrix.scala:70-85)."""
    nbr = 1 + max(i for i, _ in blocks)
    nbc = 1 + max(j for _, j in blocks)
    rows = [np.hstack([blocks[(i, j)] for j in range(nbc)]) for i in range(nbr)]
    return np.vstack(rows)


def blocked_multiply(A, B, mkn):
    """DenseVecMatrix.multiply(that, (m,k,n)) — DenseVecMatrix.scala:109-141:
    toBlocks both sides with ceil sizes, run the RMM core, assemble."""
    m_split, k_split, n_split = mkn
    if A.shape[1] != B.shape[0]:
        raise ValueError(
            f"Dimension mismatch during matrix-matrix multiplication: "
            f"{A.shape[1]} vs {B.shape[0]}")
    a_blocks = to_blocks(A, m_split, k_split)
    b_blocks = to_blocks(B, k_split, n_split)
    # effective split counts (ceil blocking may collapse small dims)
    eff = (effective_blocks(A.shape[0], m_split),
           effective_blocks(A.shape[1], k_split),
           effective_blocks(B.shape[1], n_split))
    return assemble(block_matrix_multiply(a_blocks, b_blocks, eff))


def multiply_dispatch(A, B, cores, broadcast_threshold_mb=300):
    """Strategy dispatch — DenseVecMatrix.scala:196-231.
    Returns (route, result): route in {'broadcast', 'near_square', 'carma'}."""
    if A.shape[1] != B.shape[0]:
        raise ValueError("Dimension mismatch during matrix-matrix multiplication")
    m, k = A.shape
    n = B.shape[1]
    bsize = broadcast_threshold_mb * 1024 * 1024 // 8
    if k * n <= bsize:
        # broadcast route: one local dgemm per row partition — numerically
        # a single dgemm (DenseVecMatrix.scala:1660-1680)
        return "broadcast", A @ B
    if (0.8 < (m * n) / (k * k) < 1.2) and (0.8 < m / k < 1.2):
        s = int(math.floor((3 * cores) ** (1.0 / 3.0)))
        return "near_square", blocked_multiply(A, B, (s, s, s))
    return "carma", blocked_multiply(A, B, split_method(m, k, n, cores))


# ---------------------------------------------------------------------------
# Text matrix format — MTUtils.scala:286-300 ("row:v1,v2,...",
# value separator regex ",\s?|\s+"); writer: tools/generateMatrix.cpp.
_SEP = re.compile(r",\s?|\s+")


def load_matrix_file(path):
    rows = {}
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            idx_s, data = line.split(":", 1)
            rows[int(idx_s)] = np.array(
                [float(v) for v in _SEP.split(data.strip()) if v],
                dtype=np.float64)
    n = 1 + max(rows)
    return np.vstack([rows[i] for i in range(n)])


# ---------------------------------------------------------------------------
# Deterministic synthetic inputs — the randomDenVecMatrix stand-in
# (MTUtils.scala:63-73; RandomRDD.scala:161-182 semantics: seeded,
# per-element reproducible U[0,1) fp64). The exact generator is OUR spec
# (splitmix64 per element — parallel on GPU and vectorized here); the
# distribution matches the reference's U[0,1) and is timing-irrelevant
# to GEMM. The engine kernel (mx_fill_random) implements the SAME stream;
# tests/test_gpu_parity.py asserts bit-equality.
_SM_GAMMA = np.uint64(0x9E3779B97F4A7C15)
_SM_M1 = np.uint64(0xBF58476D1CE4E5B9)
_SM_M2 = np.uint64(0x94D049BB133111EB)


def gen_uniform_u64(seed, start, count):
    """splitmix64 of (seed + (index+1)*gamma) for index in [start, start+count)."""
    with np.errstate(over="ignore"):
        idx = np.arange(start + 1, start + count + 1, dtype=np.uint64)
        z = np.uint64(seed) + idx * _SM_GAMMA
        z = (z ^ (z >> np.uint64(30))) * _SM_M1
        z = (z ^ (z >> np.uint64(27))) * _SM_M2
        z = z ^ (z >> np.uint64(31))
    return z


def gen_matrix(rows, cols, seed, dtype=np.float64):
    """Column-major U[0,1) matrix: element (r, c) has linear index
    c*rows + r (the col-major convention of the whole engine)."""
    z = gen_uniform_u64(seed, 0, rows * cols)
    vals = (z >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)
    return np.asfortranarray(vals.reshape((cols, rows)).T.astype(dtype))
