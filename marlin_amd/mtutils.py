# mtutils.py — MTUtils-surface helpers (MTUtils.scala): matrix repeat
# utilities and the seeded synthetic-matrix generator (host mirror of the
# engine's device fill_random stream; bit-identical by construction).
import numpy as np

from .api import BlockMatrix, DenseVecMatrix

_SM_GAMMA = np.uint64(0x9E3779B97F4A7C15)
_SM_M1 = np.uint64(0xBF58476D1CE4E5B9)
_SM_M2 = np.uint64(0x94D049BB133111EB)


def random_den_vec_matrix(rows, cols, seed=0xA11CE, engine=None):
    """randomDenVecMatrix (MTUtils.scala:63-73 semantics: seeded U[0,1)
    fp64). Same splitmix64-per-element stream as the engine's
    mx_fill_random (col-major linear index)."""
    with np.errstate(over="ignore"):
        idx = np.arange(1, rows * cols + 1, dtype=np.uint64)
        z = np.uint64(seed) + idx * _SM_GAMMA
        z = (z ^ (z >> np.uint64(30))) * _SM_M1
        z = (z ^ (z >> np.uint64(27))) * _SM_M2
        z = z ^ (z >> np.uint64(31))
    vals = (z >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)
    a = np.asfortranarray(vals.reshape((cols, rows)).T)
    return DenseVecMatrix(a, engine=engine)


def repeat_by_row(matrix, times):
    """MTUtils.repeatByRow (MTUtils.scala:446-464): each row repeated
    `times` -> numCols * times."""
    if times <= 0:
        raise ValueError(f"repeat times: {times} illegal")
    if times == 1:
        return matrix
    if isinstance(matrix, BlockMatrix):
        nbc = matrix.numBlksByCol()
        blocks = {}
        for (i, j), b in matrix._blocks.items():
            for t in range(times):
                # independent copies: repeated blocks must not alias one
                # ndarray (in-place ops on one copy would corrupt all)
                blocks[(i, j + t * nbc)] = b if t == 0 else b.copy()
        return BlockMatrix(blocks, matrix.numRows(),
                           matrix.numCols() * times, engine=matrix._eng)
    a = matrix.toBreeze()
    return DenseVecMatrix(np.tile(a, (1, times)), engine=matrix._eng)


def repeat_by_column(matrix, times):
    """MTUtils.repeatByColumn: rows repeated -> numRows * times."""
    if times <= 0:
        raise ValueError(f"repeat times: {times} illegal")
    if times == 1:
        return matrix
    if isinstance(matrix, BlockMatrix):
        nbr = matrix.numBlksByRow()
        blocks = {}
        for (i, j), b in matrix._blocks.items():
            for t in range(times):
                blocks[(i + t * nbr, j)] = b if t == 0 else b.copy()
        return BlockMatrix(blocks, matrix.numRows() * times,
                           matrix.numCols(), engine=matrix._eng)
    a = matrix.toBreeze()
    return DenseVecMatrix(np.tile(a, (times, 1)), engine=matrix._eng)
