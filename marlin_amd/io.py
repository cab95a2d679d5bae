# io.py — the reference's on-disk text formats (product-path
# implementation; independent of oracle/):
#
#   DenseVecMatrix: one line per row, "rowIndex:v1,v2,..."
#     (loader MTUtils.loadMatrixFile, MTUtils.scala:286-300; value
#      separator regex ",\s?|\s+"; writer DenseVecMatrix.saveToFileSystem,
#      DenseVecMatrix.scala:1042-1064; generator tools/generateMatrix.cpp)
#   BlockMatrix: one line per block,
#     "row-col-rows-cols:v1,v2,..." with the data COLUMN-MAJOR
#     (loader MTUtils.loadBlockMatrixFile, MTUtils.scala:324-340; writer
#      BlockMatrix.saveToFileSystem(path, "blockmatrix"),
#      BlockMatrix.scala:550-559)
import re

import numpy as np

from .api import DenseVecMatrix, BlockMatrix

_SEP = re.compile(r",\s?|\s+")


def load_matrix_file(path, engine=None):
    """MTUtils.loadMatrixFile — text rows -> DenseVecMatrix."""
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
    return DenseVecMatrix(rows, engine=engine)


def save_matrix_file(mat, path):
    """DenseVecMatrix.saveToFileSystem — one "row:csv" line per row."""
    if isinstance(mat, BlockMatrix):
        mat = mat.toDenseVecMatrix()       # BlockMatrix.scala:556-558
    a = mat.toBreeze()
    with open(path, "w") as f:
        for i in range(a.shape[0]):
            f.write(f"{i}:" + ",".join(repr(float(v)) for v in a[i]) + "\n")


def load_block_matrix_file(path, engine=None):
    """MTUtils.loadBlockMatrixFile — "r-c-rows-cols:colmajor" lines."""
    blocks = {}
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            head, data = line.split(":", 1)
            r, c, rows, cols = (int(x) for x in head.split("-"))
            arr = np.array([float(v) for v in _SEP.split(data.strip()) if v],
                           dtype=np.float64)
            blocks[(r, c)] = arr.reshape((cols, rows)).T  # col-major data
    return BlockMatrix(blocks, engine=engine)


def save_block_matrix_file(mat, path):
    """BlockMatrix.saveToFileSystem(path, "blockmatrix")."""
    with open(path, "w") as f:
        for (r, c), blk in sorted(mat._blocks.items()):
            data = ",".join(repr(float(v)) for v in blk.flatten(order="F"))
            f.write(f"{r}-{c}-{blk.shape[0]}-{blk.shape[1]}:{data}\n")
