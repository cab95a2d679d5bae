# Pins the oracle restatement against the reference's OWN golden vectors:
# the 4x4 multiply literals of DistributedMatrixSuite.scala (restated below
# as data, not code) and the frozen 100x100 fixture from the reference's
# data/ files. Everything here runs on CPU.
import os

import numpy as np
import pytest

from oracle import (
    split_method, blocked_multiply, block_matrix_multiply, to_blocks,
    multiply_dispatch, slab_len, slab_off, gen_matrix,
)
from oracle.marlin_oracle import assemble, effective_blocks

# The reference's fixed 4x4 matrix — DistributedMatrixSuite.scala:15-24
# (rows 0..3 of `data`).
M4 = np.array([
    [0.0, 1.0, 2.0, 3.0],
    [2.0, 3.0, 4.0, 5.0],
    [3.0, 2.0, 1.0, 0.0],
    [1.0, 1.0, 1.0, 1.0],
])
# Expected M4 @ M4 — DistributedMatrixSuite.scala:228-233 (and repeated at
# :241-248, :258-262, :293-297 for the other routes).
C4 = np.array([
    [11.0, 10.0, 9.0, 8.0],
    [23.0, 24.0, 25.0, 26.0],
    [7.0, 11.0, 15.0, 19.0],
    [6.0, 7.0, 8.0, 9.0],
])


def test_golden_4x4_broadcast_route():
    route, c = multiply_dispatch(M4, M4, cores=2)
    assert route == "broadcast"
    np.testing.assert_array_equal(c, C4)


@pytest.mark.parametrize("mkn", [(2, 2, 1), (2, 1, 2), (2, 2, 2)])
def test_golden_4x4_split_modes(mkn):
    # DistributedMatrixSuite.scala:236-249 ("new matrix multiplication")
    np.testing.assert_array_equal(blocked_multiply(M4, M4, mkn), C4)


def test_golden_4x4_blockmatrix_tiles():
    # Block x Block per-tile expected values —
    # DistributedMatrixSuite.scala:283-286: result blocks of the 2x2-blocked
    # product, e.g. BlockID(0,0) -> [[11,10],[23,24]].
    a_blocks = to_blocks(M4, 2, 2)
    c_blocks = block_matrix_multiply(a_blocks, a_blocks, (2, 2, 2))
    np.testing.assert_array_equal(c_blocks[(0, 0)], [[11.0, 10.0], [23.0, 24.0]])
    np.testing.assert_array_equal(c_blocks[(0, 1)], [[9.0, 8.0], [25.0, 26.0]])
    np.testing.assert_array_equal(c_blocks[(1, 0)], [[7.0, 11.0], [6.0, 7.0]])
    np.testing.assert_array_equal(c_blocks[(1, 1)], [[15.0, 19.0], [8.0, 9.0]])


def test_golden_100x100_fixture(golden_dir):
    a = np.load(os.path.join(golden_dir, "a100.npy"))
    b = np.load(os.path.join(golden_dir, "b100.npy"))
    c = np.load(os.path.join(golden_dir, "c100.npy"))
    assert a.shape == (100, 100)
    got = a @ b
    np.testing.assert_allclose(got, c, rtol=0, atol=1e-12)
    for mkn in [(2, 2, 2), (4, 4, 4), (7, 3, 5)]:
        rel = np.max(np.abs(blocked_multiply(a, b, mkn) - c)) / np.max(np.abs(c))
        assert rel < 1e-12


# ---------------------------------------------------------------------------
# Planner — MTUtils.scala:150-175 semantics.
@pytest.mark.parametrize("mkn_cores,expected", [
    # near-cube halving: n wins ties, then m, then k
    ((8, 8, 8, 8), (2, 2, 2)),
    ((8, 8, 8, 2), (1, 1, 2)),
    ((8, 8, 8, 4), (2, 1, 2)),
    # cores exhaust by integer halving: cores=6 -> 6//2=3 -> 3//2=1 (2 rounds)
    ((100, 100, 100, 6), (1, 1, 4)),   # round1: n(tie) 100->50; round2: m>=k,n? m=100 largest -> m... see note
])
def test_split_method_shapes(mkn_cores, expected):
    m, k, n, cores = mkn_cores
    got = split_method(m, k, n, cores)
    # structural checks always hold:
    assert got[0] * got[1] * got[2] >= 1
    # exact expectations where hand-derived
    if mkn_cores != (100, 100, 100, 6):
        assert got == expected


def test_split_method_hand_trace():
    # hand trace (100,100,100,6): r1 n>=k,n>=m -> n:100->50 cores 3;
    # r2 m=100 is largest (n=50) -> m:100->50 cores 1; stop. -> (2,1,2)
    assert split_method(100, 100, 100, 6) == (2, 1, 2)
    # tall-skinny config 4 shape: (50000, 4096, 50000, 8):
    # r1 n(tie with m? n>=m yes) n->25000 c4; r2 m=50000 largest m->25000 c2;
    # r3 n=25000 tie n>=m(25000)>=k yes n->12500 c1 -> (2,1,4)
    assert split_method(50000, 4096, 50000, 8) == (2, 1, 4)
    # degenerate dims stop the loop
    assert split_method(1, 64, 64, 8) == (1, 1, 1)
    assert split_method(64, 64, 64, 1) == (1, 1, 1)


def test_near_square_route_split():
    # dispatch: 4000^2 exceeds 300MB? 4000*4000*8 = 128MB <= 300MB -> broadcast.
    route, _ = multiply_dispatch(np.zeros((4000, 10)), np.zeros((10, 4000)), 8)
    assert route == "broadcast"


# ---------------------------------------------------------------------------
# Ceil blocking semantics — DenseVecMatrix.scala:1091-1094, 1262-1265.
def test_ceil_blocking_ragged():
    # total=5, parts=4: block_len=2 -> 3 effective blocks [2,2,1]
    assert effective_blocks(5, 4) == 3
    assert [slab_len(5, 4, i) for i in range(4)] == [2, 2, 1, 0]
    assert [slab_off(5, 4, i) for i in range(3)] == [0, 2, 4]


@pytest.mark.parametrize("shape_mkn", [
    ((5, 7, 3), (2, 3, 2)),
    ((1, 9, 4), (1, 4, 2)),
    ((13, 1, 6), (4, 1, 3)),
    ((257, 129, 63), (8, 8, 8)),
])
def test_blocked_equals_plain_on_ragged(shape_mkn):
    (m, k, n), mkn = shape_mkn
    a = gen_matrix(m, k, seed=0xA11CE)
    b = gen_matrix(k, n, seed=0xB0B)
    np.testing.assert_allclose(
        blocked_multiply(a, b, mkn), a @ b, rtol=1e-13, atol=1e-13)


def test_dimension_mismatch_raises():
    with pytest.raises(ValueError):
        blocked_multiply(np.zeros((4, 5)), np.zeros((4, 5)), (2, 2, 2))


def test_assemble_roundtrip():
    a = gen_matrix(11, 7, seed=42)
    assert np.array_equal(assemble(to_blocks(a, 3, 2)), a)


def test_gen_matrix_deterministic_colmajor():
    a = gen_matrix(3, 2, seed=1)
    b = gen_matrix(3, 2, seed=1)
    np.testing.assert_array_equal(a, b)
    assert a.flags.f_contiguous
    assert np.all((a >= 0) & (a < 1))
    # element (r,c) is a pure function of linear index c*rows+r
    big = gen_matrix(6, 1, seed=1)
    np.testing.assert_array_equal(a.T.reshape(-1)[:3], big[:3, 0])


def test_api_slices_golden():
    # DistributedMatrixSuite.scala:207-224 (host-side slicing; no GPU)
    from marlin_amd import DenseVecMatrix
    mat = DenseVecMatrix(M4)
    np.testing.assert_array_equal(mat.sliceByRow(1, 2).toBreeze(),
                                  [[2.0, 3, 4, 5], [3, 2, 1, 0]])
    np.testing.assert_array_equal(mat.sliceByColumn(1, 2).toBreeze(),
                                  [[1.0, 2], [3, 4], [2, 1], [1, 1]])
    np.testing.assert_array_equal(mat.getSubMatrix(1, 2, 1, 2).toBreeze(),
                                  [[3.0, 4], [2, 1]])


def test_api_empty_rows_raises():
    # "empty rows" behaviour (DistributedMatrixSuite.scala:54-63)
    from marlin_amd import DenseVecMatrix
    with pytest.raises(RuntimeError):
        DenseVecMatrix(np.zeros((0, 0)))


def test_blockid_hash_eq():
    # Block.scala:37-49: hash = row*31 + column + seq
    from marlin_amd import BlockID
    assert hash(BlockID(2, 3, 4)) == 2 * 31 + 3 + 4
    assert BlockID(1, 2, 3) == BlockID(1, 2, 3)
    assert BlockID(1, 2, 3) != BlockID(2, 1, 3)


def test_repeat_by_row_and_column():
    # DistributedMatrixSuite.scala:355-374 ("repeat by row and column")
    from marlin_amd import DenseVecMatrix, repeat_by_row, repeat_by_column
    mat = DenseVecMatrix(M4)
    np.testing.assert_array_equal(repeat_by_row(mat, 2).toBreeze(),
                                  np.tile(M4, (1, 2)))
    np.testing.assert_array_equal(repeat_by_column(mat, 2).toBreeze(),
                                  np.tile(M4, (2, 1)))
    blk = mat.toBlockMatrix(2, 2)
    np.testing.assert_array_equal(repeat_by_row(blk, 2).toBreeze(),
                                  np.tile(M4, (1, 2)))
    np.testing.assert_array_equal(repeat_by_column(blk, 3).toBreeze(),
                                  np.tile(M4, (3, 1)))
    with pytest.raises(ValueError):
        repeat_by_row(mat, 0)


def test_random_den_vec_matrix_matches_engine_spec():
    from marlin_amd import random_den_vec_matrix
    got = random_den_vec_matrix(5, 3, seed=0xA11CE).toBreeze()
    np.testing.assert_array_equal(got, gen_matrix(5, 3, seed=0xA11CE))


def test_trait_surface_complete():
    # every public method of the reference's DistributedMatrix trait
    # (DistributedMatrix.scala:9-76) exists on both mirrors
    from marlin_amd import DenseVecMatrix, BlockMatrix
    trait = ["numRows", "numCols", "toBreeze", "add", "subtract",
             "subtractBy", "multiply", "divide", "divideBy",
             "elementsCount", "sum", "dotProduct", "transpose", "inverse",
             "cBind", "saveToFileSystem", "print", "printAll"]
    dvm = DenseVecMatrix(M4)
    blk = dvm.toBlockMatrix(2, 2)
    for name in trait:
        assert hasattr(dvm, name), f"DenseVecMatrix missing {name}"
        assert hasattr(blk, name), f"BlockMatrix missing {name}"
    assert dvm.elementsCount() == 4
    assert blk.elementsCount() == 4
    both = dvm.cBind(DenseVecMatrix(M4))
    np.testing.assert_array_equal(both.toBreeze(), np.hstack([M4, M4]))
