# GPU parity for the elementwise/scalar op family — the reference's own
# golden literals (DistributedMatrixSuite.scala:164-205 "element-wise
# addition/subtract ...", :319-324 "sum", :326-338 "dot product",
# :302-316 transpose), restated as data, run through the engine's
# mx_map/mx_sum/mx_transpose via the operator mirror.
import numpy as np
import pytest

from marlin_amd import Engine, DenseVecMatrix
from oracle import gen_matrix

pytestmark = pytest.mark.gpu

M4 = np.array([[0., 1, 2, 3], [2, 3, 4, 5], [3, 2, 1, 0], [1, 1, 1, 1]])


@pytest.fixture(scope="module")
def eng():
    e = Engine(0)
    yield e
    e.close()


def dvm(eng):
    return DenseVecMatrix(M4, engine=eng)


def blk(eng):
    return DenseVecMatrix(M4, engine=eng).toBlockMatrix(2, 2)


def test_elementwise_golden(eng):
    ele_add1 = M4 + 1
    add_self = M4 + M4
    ele_sub1 = M4 - 1
    divide2 = M4 / 2
    for mat in (dvm(eng), blk(eng)):
        np.testing.assert_array_equal(mat.add(1).toBreeze(), ele_add1)
        np.testing.assert_array_equal(mat.add(mat).toBreeze(), add_self)
        np.testing.assert_array_equal(mat.subtract(1).toBreeze(), ele_sub1)
        np.testing.assert_array_equal(mat.subtract(mat).toBreeze(),
                                      np.zeros((4, 4)))
        np.testing.assert_array_equal(mat.multiply(2).toBreeze(), add_self)
        np.testing.assert_array_equal(mat.divide(2).toBreeze(), divide2)
    # cross-type: BlockMatrix op DenseVecMatrix
    np.testing.assert_array_equal(blk(eng).add(dvm(eng)).toBreeze(), add_self)
    np.testing.assert_array_equal(blk(eng).subtract(dvm(eng)).toBreeze(),
                                  np.zeros((4, 4)))


def test_subtract_by_divide_by(eng):
    np.testing.assert_array_equal(dvm(eng).subtractBy(10).toBreeze(), 10 - M4)
    m = M4 + 1
    got = DenseVecMatrix(m, engine=eng).divideBy(2).toBreeze()
    np.testing.assert_array_equal(got, 2 / m)


def test_sum_golden(eng):
    # DistributedMatrixSuite.scala:319-324: sum == 30.0
    assert dvm(eng).sum() == 30.0
    assert blk(eng).sum() == 30.0


def test_dot_product_golden(eng):
    # DistributedMatrixSuite.scala:326-338 ("dot product" = elementwise mul)
    expected = M4 * M4
    np.testing.assert_array_equal(dvm(eng).dotProduct(dvm(eng)).toBreeze(),
                                  expected)
    np.testing.assert_array_equal(blk(eng).dotProduct(blk(eng)).toBreeze(),
                                  expected)
    np.testing.assert_array_equal(dvm(eng).dotProduct(blk(eng)).toBreeze(),
                                  expected)


def test_transpose(eng):
    # DistributedMatrixSuite.scala:302-316
    np.testing.assert_array_equal(dvm(eng).transpose().toBreeze(), M4.T)
    np.testing.assert_array_equal(blk(eng).transpose().toBreeze(), M4.T)
    # ragged + larger
    a = gen_matrix(517, 301, seed=99)
    got = DenseVecMatrix(a, engine=eng).transpose().toBreeze()
    np.testing.assert_array_equal(got, a.T)


def test_elementwise_large_random(eng):
    a = gen_matrix(1000, 700, seed=1)
    b = gen_matrix(1000, 700, seed=2)
    e = eng
    np.testing.assert_array_equal(e.map_op("add", a, b), a + b)
    np.testing.assert_array_equal(e.map_op("sub", a, b), a - b)
    np.testing.assert_array_equal(e.map_op("emul", a, b), a * b)
    np.testing.assert_array_equal(e.map_op("muls", a, scalar=3.5), a * 3.5)
    np.testing.assert_array_equal(e.map_op("rdivs", a, scalar=1.0), 1.0 / a)
    assert abs(e.sum(a) - a.sum()) / abs(a.sum()) < 1e-12


def test_dimension_mismatch(eng):
    with pytest.raises(ValueError):
        dvm(eng).add(DenseVecMatrix(np.ones((3, 4)), engine=eng))
