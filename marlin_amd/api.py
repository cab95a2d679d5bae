# api.py — host-side mirror of the reference's matrix operator surface
# (the DistributedMatrix trait, DistributedMatrix.scala:9-76, and the
# concrete DenseVecMatrix / BlockMatrix multiply overloads,
# DenseVecMatrix.scala:103-231 / BlockMatrix.scala:87-335), with the same
# names, argument meaning and error behaviour — compute delegated to the
# MI355X engine through the C ABI. The reference host is compiled (Scala);
# the first-class host above the ABI is C++ (src/host/marlinx.cpp) plus
# the JNI stub in INTEGRATION.md; this Python mirror is the in-container
# driver for tests and the bench.
#
# NO CPU fallback: every multiply goes through libmarlin_gpu.so. On a
# machine without a GPU these raise EngineUnavailable/EngineError loudly.
import math

import numpy as np

from .engine import Engine


def _ceil_block(total, parts):
    return int(math.ceil(total / parts))


def split_method(m, k, n, cores):
    """CARMA-style split — MTUtils.splitMethod (MTUtils.scala:150-175)."""
    ms = ks = ns = 1
    _m, _k, _n, _c = int(m), int(k), int(n), int(cores)
    while _c > 1 and _m > 1 and _k > 1 and _n > 1:
        if _n >= _k and _n >= _m:
            ns *= 2; _n //= 2
        elif _m >= _k and _m >= _n:
            ms *= 2; _m //= 2
        else:
            ks *= 2; _k //= 2
        _c //= 2
    return ms, ks, ns


class BlockID:
    """Block.scala:37-49."""
    __slots__ = ("row", "column", "seq")

    def __init__(self, row, column, seq=0):
        self.row, self.column, self.seq = row, column, seq

    def __hash__(self):
        return self.row * 31 + self.column + self.seq

    def __eq__(self, o):
        return (self.row, self.column, self.seq) == (o.row, o.column, o.seq)


class DenseVecMatrix:
    """Row-major distributed matrix facade (DenseVecMatrix.scala).
    Backed by one ndarray here (single-node host); the engine shards
    on-device. Multiply semantics mirror the reference dispatch."""

    def __init__(self, rows, engine=None, _dev=None):
        # rows: ndarray (m x n) or dict {row_index: 1-D array}; _dev: a
        # DeviceMatrix for results materialised lazily (host copy made
        # only at toBreeze / IO time — the RDD laziness analog)
        if _dev is not None:
            self._a = None
            self._dev = _dev
            self._eng = engine
            return
        if isinstance(rows, dict):
            n = 1 + max(rows)
            rows = np.vstack([rows[i] for i in range(n)])
        self._a = np.asarray(rows, dtype=np.float64)
        if self._a.size == 0:
            raise RuntimeError("empty rows")
        self._eng = engine
        self._dev = None

    def _engine(self):
        if self._eng is None:
            self._eng = Engine()
        return self._eng

    def numRows(self):
        return self._dev.m if self._a is None else self._a.shape[0]

    def numCols(self):
        return self._dev.n if self._a is None else self._a.shape[1]

    def cache(self):
        """Keep this matrix resident in HBM (RDD.cache(),
        DenseVecMatrix.scala:321,505): subsequent multiplies skip the
        host->device copy."""
        if self._dev is None:
            self._dev = self._engine().upload_matrix(self._host())
        return self

    def unpersist(self):
        if self._dev is not None and self._a is not None:
            self._dev.free()
            self._dev = None
        return self

    def _host(self):
        if self._a is None:
            self._a = self._eng.download_matrix(self._dev)
        return self._a

    def multiply(self, other, cores=None, broadcast_threshold=300):
        """DenseVecMatrix.multiply(other, cores, broadcastThreshold)
        (DenseVecMatrix.scala:196-231). The strategy dispatch decided
        Spark data movement in the reference; on MI355X every route is
        the same device GEMM, so dispatch only shapes the result type:
        a local ndarray route returns DenseVecMatrix, block routes
        return BlockMatrix with the split the reference would pick."""
        if isinstance(other, (int, float)):
            # multiply(b: Double) scalar overload (reference test
            # `mat.multiply(2)`, DistributedMatrixSuite.scala:194)
            return DenseVecMatrix(self._engine().map_op("muls", self._host(),
                                                        scalar=other),
                                  self._eng)
        if isinstance(other, np.ndarray) and other.ndim == 1:
            # multiply(v: BDV) matrix-vector route (BlockMatrix.scala:265)
            return self._engine().dgemv(self._host(), other)
        if isinstance(other, np.ndarray):
            # multiply(local Breeze matrix) -> DenseVecMatrix
            if self.numCols() != other.shape[0]:
                raise ValueError(
                    f"Dimension mismatch during matrix-matrix multiplication: "
                    f"{self.numCols()} vs {other.shape[0]}")
            return DenseVecMatrix(self._engine().dgemm(self._host(), other),
                                  self._eng)
        if isinstance(other, BlockMatrix):
            other = other.toDenseVecMatrix()
        if self.numCols() != other.numRows():
            raise ValueError(
                f"Dimension mismatch during matrix-matrix multiplication: "
                f"{self.numCols()} vs {other.numRows()}")
        m, k, n = self.numRows(), self.numCols(), other.numCols()
        bsize = broadcast_threshold * 1024 * 1024 // 8
        if self._dev is not None and other._dev is not None:
            # Deliberate divergence from the reference dispatch: cached
            # (device-resident) operands return a device-resident
            # DenseVecMatrix instead of a BlockMatrix — re-blocking the
            # result would force a D2H round trip, defeating cache().
            # Call .toBlockMatrix(r, c) on the result to get the
            # reference's block layout when needed.
            cdev = self._engine().gemm_dd(self._dev, other._dev)
            return DenseVecMatrix(None, self._eng, _dev=cdev)
        c = self._engine().dgemm(self._host(), other._host())
        if k * n <= bsize or m * k <= bsize:
            return DenseVecMatrix(c, self._eng)
        if (0.8 < (m * n) / (k * k) < 1.2) and (0.8 < m / k < 1.2):
            s = int(math.floor((3 * (cores or 8)) ** (1.0 / 3.0)))
            mkn = (s, s, s)
        else:
            mkn = split_method(m, k, n, cores or 8)
        return _to_block(c, mkn[0], mkn[2], self._eng)

    # -- elementwise family (DenseVecMatrix add/subtract/multiply/divide,
    #    dotProduct, sum; same names/overloads as the reference) ----------
    def add(self, other):
        return self._ew2("add", "adds", other)

    def subtract(self, other):
        return self._ew2("sub", "subs", other)

    def subtractBy(self, b):
        return DenseVecMatrix(self._engine().map_op("rsubs", self._host(),
                                                    scalar=b), self._eng)

    def multiply_scalar(self, b):
        return DenseVecMatrix(self._engine().map_op("muls", self._host(),
                                                    scalar=b), self._eng)

    def divide(self, b):
        return DenseVecMatrix(self._engine().map_op("divs", self._host(),
                                                    scalar=b), self._eng)

    def divideBy(self, b):
        return DenseVecMatrix(self._engine().map_op("rdivs", self._host(),
                                                    scalar=b), self._eng)

    def dotProduct(self, other):
        if isinstance(other, BlockMatrix):
            other = other.toDenseVecMatrix()
        if (self.numRows() != other.numRows()
                or self.numCols() != other.numCols()):
            raise ValueError("matrix dimension mismatch")
        return DenseVecMatrix(self._engine().map_op("emul", self._host(),
                                                    other._host()), self._eng)

    def sum(self):
        return self._engine().sum(self._host())

    def transpose(self):
        return DenseVecMatrix(self._engine().transpose(self._host()), self._eng)

    def _ew2(self, op2, op1, other):
        if isinstance(other, (int, float)):
            return DenseVecMatrix(self._engine().map_op(op1, self._host(),
                                                        scalar=other),
                                  self._eng)
        if isinstance(other, BlockMatrix):
            other = other.toDenseVecMatrix()
        if (self.numRows() != other.numRows()
                or self.numCols() != other.numCols()):
            raise ValueError("matrix dimension mismatch")
        return DenseVecMatrix(self._engine().map_op(op2, self._host(),
                                                    other._host()),
                              self._eng)

    # -- slicing (DenseVecMatrix.sliceByRow/sliceByColumn/getSubMatrix;
    #    inclusive index ranges, DistributedMatrixSuite.scala:207-224) ----
    def sliceByRow(self, start, end):
        return DenseVecMatrix(self._host()[start:end + 1, :], self._eng)

    def sliceByColumn(self, start, end):
        return DenseVecMatrix(self._host()[:, start:end + 1], self._eng)

    def getSubMatrix(self, r0, r1, c0, c1):
        return DenseVecMatrix(self._host()[r0:r1 + 1, c0:c1 + 1], self._eng)

    def elementsCount(self):
        """DistributedMatrix.elementsCount: row count for DenseVecMatrix."""
        return self.numRows()

    def cBind(self, other):
        """DistributedMatrix.cBind: column bind."""
        if isinstance(other, BlockMatrix):
            other = other.toDenseVecMatrix()
        if self.numRows() != other.numRows():
            raise ValueError("matrix dimension mismatch")
        return DenseVecMatrix(np.hstack([self._host(), other._host()]),
                              self._eng)

    def print(self, max_rows=10, max_cols=10):
        a = self._host()
        print(a[:max_rows, :max_cols])

    def printAll(self):
        print(self._host())

    def toBlockMatrix(self, blks_by_row, blks_by_col):
        """DenseVecMatrix.toBlockMatrix (DenseVecMatrix.scala:1259-1328)."""
        return _to_block(self._host(), blks_by_row, blks_by_col, self._eng)

    def saveToFileSystem(self, path):
        """DenseVecMatrix.saveToFileSystem (DenseVecMatrix.scala:1042)."""
        from .io import save_matrix_file
        save_matrix_file(self, path)

    def luDecompose(self, mode="auto", base_size=1000):
        """DenseVecMatrix.luDecompose (DenseVecMatrix.scala:283-464)."""
        from .linalg import lu_decompose
        return lu_decompose(self, mode, base_size)

    def inverse(self, mode="auto", base_size=1000):
        """DenseVecMatrix.inverse (DenseVecMatrix.scala:565-764)."""
        from .linalg import inverse
        return inverse(self, mode, base_size)

    def lr(self, step_size, iters):
        """DenseVecMatrix.lr (DenseVecMatrix.scala:1005-1035): SGD
        logistic-regression gradient sums. Row format (label, features);
        the first feature element is overwritten with the intercept 1,
        exactly as the reference does. The per-iteration reduce of
        per-row gradients equals X'^T (sigmoid(X' w) - labels) — two
        gemv passes on the cached device matrix per iteration."""
        a = self._host()
        labels = a[:, 0].copy()
        X = a.copy()
        X[:, 0] = 1.0                      # intercept
        m, f = X.shape
        eng = self._engine()
        dX = eng.upload_matrix(X)
        dXT = eng.transpose_dd(dX)
        w = np.zeros(f)
        for i in range(1, iters + 1):
            margin = -eng.dgemv_dd(dX, w)
            gmul = 1.0 / (1.0 + np.exp(margin)) - labels
            delta = eng.dgemv_dd(dXT, gmul)
            w = w - delta * (step_size / m / np.sqrt(i))
        dX.free()
        dXT.free()
        return w

    def computeGramianMatrix(self):
        """DenseVecMatrix.computeGramianMatrix (DenseVecMatrix.scala:1464):
        the local Gram matrix A^T A (device-resident compute)."""
        n = self.numCols()
        if n > 65535:
            raise ValueError(f"Argument with more than 65535 cols: {n}")
        eng = self._engine()
        dA = eng.upload_matrix(self._host())
        dAT = eng.transpose_dd(dA)
        G = eng.gemm_dd(dAT, dA)
        out = eng.download_matrix(G)
        for d in (dA, dAT, G):
            d.free()
        return out

    def choleskyDecompose(self, mode="auto", base_size=1000):
        """DenseVecMatrix.choleskyDecompose (DenseVecMatrix.scala:475-566)."""
        from .linalg import cholesky_decompose
        return cholesky_decompose(self, mode, base_size)

    def toBreeze(self):
        return self._host().copy()


def _to_block(a, r, c, engine):
    rows, cols = a.shape
    brl, bcl = _ceil_block(rows, r), _ceil_block(cols, c)
    nbr = int(math.ceil(rows / brl))
    nbc = int(math.ceil(cols / bcl))
    blocks = {}
    for bi in range(nbr):
        for bj in range(nbc):
            blocks[(bi, bj)] = a[bi * brl: min((bi + 1) * brl, rows),
                                 bj * bcl: min((bj + 1) * bcl, cols)]
    return BlockMatrix(blocks, rows, cols, engine=engine)


class BlockMatrix:
    """Distributed blocked matrix facade (BlockMatrix.scala)."""

    def __init__(self, blocks, num_rows=None, num_cols=None, engine=None):
        # blocks: dict {(bi, bj): ndarray tile}
        self._blocks = {k: np.asarray(v, dtype=np.float64)
                        for k, v in blocks.items()}
        self._nbr = 1 + max(i for i, _ in self._blocks)
        self._nbc = 1 + max(j for _, j in self._blocks)
        self._rows = num_rows if num_rows is not None else sum(
            self._blocks[(i, 0)].shape[0] for i in range(self._nbr))
        self._cols = num_cols if num_cols is not None else sum(
            self._blocks[(0, j)].shape[1] for j in range(self._nbc))
        self._eng = engine

    def _engine(self):
        if self._eng is None:
            self._eng = Engine()
        return self._eng

    def numRows(self):
        return self._rows

    def numCols(self):
        return self._cols

    def numBlksByRow(self):
        return self._nbr

    def numBlksByCol(self):
        return self._nbc

    def multiply(self, other):
        """BlockMatrix.multiply (BlockMatrix.scala:149-220). The reference
        re-shuffles (emit x n / x m, join, reduceByKey); the engine computes
        the same per-C-tile sums with one owner per tile, so only the
        per-tile GEMM-accumulate chain remains (mx_tile_dgemm_acc)."""
        if isinstance(other, (int, float)):
            return self._ew1("muls", other)
        if isinstance(other, np.ndarray) and other.ndim == 1:
            # multiply(v: BDV) (BlockMatrix.scala:265-274)
            if self.numCols() != other.shape[0]:
                raise ValueError(
                    f"matrix columns size {self.numCols()} not support "
                    f"vector length {other.shape[0]}")
            return self._engine().dgemv(self.toBreeze(), other)
        if isinstance(other, np.ndarray):
            # multiply(B: BDM) broadcast small-B route
            # (BlockMatrix.scala:280-303): one engine GEMM, result kept
            # in this matrix's block grid
            if self.numCols() != other.shape[0]:
                raise ValueError(
                    f"Dimension mismatch during matrix-matrix "
                    f"multiplication: {self.numCols()} vs {other.shape[0]}")
            c = self._engine().dgemm(self.toBreeze(), other)
            return _to_block(c, self.numBlksByRow(), self.numBlksByCol(),
                             self._eng)
        if isinstance(other, DenseVecMatrix):
            if self.numCols() != other.numRows():
                raise ValueError(
                    f"Dimension mismatch during matrix-matrix multiplication: "
                    f"{self.numCols()} vs {other.numRows()}")
            # BlockMatrix x DenseVecMatrix (BlockMatrix.scala:305-335)
            return DenseVecMatrix(
                self._engine().dgemm(self.toBreeze(), other._host()), self._eng)
        if self.numCols() != other.numRows():
            raise ValueError(
                f"Dimension mismatch during matrix-matrix multiplication: "
                f"{self.numCols()} vs {other.numRows()}")
        if self.numBlksByCol() != other.numBlksByRow():
            # reference re-slices (BlockMatrix.scala:187-216); we re-block
            other = _to_block(other.toBreeze(), self.numBlksByCol(),
                              other.numBlksByCol(), self._eng)
        eng = self._engine()
        ks = self.numBlksByCol()
        out = {}
        for i in range(self.numBlksByRow()):
            for j in range(other.numBlksByCol()):
                acc = None
                for l in range(ks):
                    a = self._blocks[(i, l)]
                    b = other._blocks[(l, j)]
                    acc = eng.tile_dgemm_acc(a, b, acc)
                out[(i, j)] = acc
        return BlockMatrix(out, self.numRows(), other.numCols(),
                           engine=self._eng)

    # -- elementwise family (BlockMatrix.scala:344-523 names) ------------
    def add(self, other):
        return self._ew2("add", "adds", other)

    def subtract(self, other):
        return self._ew2("sub", "subs", other)

    def subtractBy(self, b):
        return self._ew1("rsubs", b)

    def multiply_scalar(self, b):
        return self._ew1("muls", b)

    def divide(self, b):
        return self._ew1("divs", b)

    def divideBy(self, b):
        return self._ew1("rdivs", b)

    def dotProduct(self, other):
        if isinstance(other, DenseVecMatrix):
            other = _to_block(other._host(), self._nbr, self._nbc, self._eng)
        if (self.numRows() != other.numRows()
                or self.numCols() != other.numCols()):
            raise ValueError("matrix dimension mismatch")
        eng = self._engine()
        out = {k: eng.map_op("emul", v, other._blocks[k])
               for k, v in self._blocks.items()}
        return BlockMatrix(out, self._rows, self._cols, engine=self._eng)

    def sum(self):
        eng = self._engine()
        return float(sum(eng.sum(v) for v in self._blocks.values()))

    def transpose(self):
        """BlockMatrix.transpose (BlockMatrix.scala:514-523): per-block
        device transpose + block-ID swap."""
        eng = self._engine()
        out = {(j, i): eng.transpose(v) for (i, j), v in self._blocks.items()}
        return BlockMatrix(out, self._cols, self._rows, engine=self._eng)

    def _ew1(self, op, b):
        eng = self._engine()
        out = {k: eng.map_op(op, v, scalar=b) for k, v in self._blocks.items()}
        return BlockMatrix(out, self._rows, self._cols, engine=self._eng)

    def _ew2(self, op2, op1, other):
        if isinstance(other, (int, float)):
            return self._ew1(op1, other)
        if isinstance(other, DenseVecMatrix):
            other = _to_block(other._host(), self._nbr, self._nbc, self._eng)
        if (self.numRows() != other.numRows()
                or self.numCols() != other.numCols()):
            raise ValueError("matrix dimension mismatch")
        if (self.numBlksByRow() != other.numBlksByRow()
                or self.numBlksByCol() != other.numBlksByCol()):
            other = _to_block(other.toBreeze(), self._nbr, self._nbc,
                              self._eng)
        eng = self._engine()
        out = {k: eng.map_op(op2, v, other._blocks[k])
               for k, v in self._blocks.items()}
        return BlockMatrix(out, self._rows, self._cols, engine=self._eng)

    def saveToFileSystem(self, path, format=" "):
        """BlockMatrix.saveToFileSystem (BlockMatrix.scala:538-559)."""
        from .io import save_block_matrix_file, save_matrix_file
        if format.lower() == "blockmatrix":
            save_block_matrix_file(self, path)
        else:
            save_matrix_file(self, path)

    def inverse(self, mode="auto", base_size=1000):
        """BlockMatrix.inverse (BlockMatrix.scala:527-530):
        delegates via toDenseVecMatrix."""
        return self.toDenseVecMatrix().inverse(mode, base_size)

    def elementsCount(self):
        """DistributedMatrix.elementsCount: sub-block count."""
        return len(self._blocks)

    def cBind(self, other):
        return self.toDenseVecMatrix().cBind(other)

    def print(self, max_rows=10, max_cols=10):
        print(self.toBreeze()[:max_rows, :max_cols])

    def printAll(self):
        print(self.toBreeze())

    def toDenseVecMatrix(self):
        """BlockMatrix.toDenseVecMatrix (BlockMatrix.scala:575-594)."""
        return DenseVecMatrix(self.toBreeze(), self._eng)

    def toBreeze(self):
        """BlockMatrix.toBreeze (BlockMatrix.scala:70-85)."""
        rows = [np.hstack([self._blocks[(i, j)] for j in range(self._nbc)])
                for i in range(self._nbr)]
        return np.vstack(rows)
