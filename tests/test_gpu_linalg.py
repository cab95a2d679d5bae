# GPU tests for the decomposition tier (blocked LU + inverse on engine
# GEMMs — DenseVecMatrix.scala:283-464 / 565-764 restatement).
import numpy as np
import pytest

from marlin_amd import Engine, DenseVecMatrix
from oracle import gen_matrix

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    e = Engine(0)
    yield e
    e.close()


def well_conditioned(n, seed):
    a = gen_matrix(n, n, seed=seed)
    return a + n * np.eye(n)


def test_inverse_golden_permutation(eng):
    # DistributedMatrixSuite.scala:340-352: inverse of the flip
    # permutation is itself (local route, n <= 6000 -> "auto" = local)
    p3 = np.array([[0., 0, 1], [0, 1, 0], [1, 0, 0]])
    got = DenseVecMatrix(p3, engine=eng).inverse()
    np.testing.assert_allclose(got.toBreeze(), p3, atol=1e-12)


@pytest.mark.parametrize("n,base", [(300, 96), (517, 128), (64, 16)])
def test_inverse_dist_route(eng, n, base):
    a = well_conditioned(n, seed=n)
    inv = DenseVecMatrix(a, engine=eng).inverse(mode="dist",
                                                base_size=base).toBreeze()
    err = np.max(np.abs(a @ inv - np.eye(n)))
    assert err < 1e-9, err


@pytest.mark.parametrize("n,base", [(300, 96), (250, 64)])
def test_lu_dist_route(eng, n, base):
    a = well_conditioned(n, seed=1000 + n)
    dvm = DenseVecMatrix(a, engine=eng)
    blk, p_array = dvm.luDecompose(mode="dist", base_size=base)
    lu = blk.toBreeze()
    L = np.tril(lu, -1) + np.eye(n)
    U = np.triu(lu)
    # block pairwise pivoting property: P_blockdiag A == L U
    PA = a[p_array, :]
    rel = np.max(np.abs(PA - L @ U)) / np.max(np.abs(a))
    assert rel < 1e-10, rel


def test_lu_local_route(eng):
    a = well_conditioned(100, seed=77)
    blk, p_array = DenseVecMatrix(a, engine=eng).luDecompose(mode="breeze")
    lu = blk.toBreeze()
    L = np.tril(lu, -1) + np.eye(100)
    U = np.triu(lu)
    rel = np.max(np.abs(a[p_array, :] - L @ U)) / np.max(np.abs(a))
    assert rel < 1e-12, rel


@pytest.mark.parametrize("n,base", [(300, 96), (250, 64)])
def test_cholesky_dist_route(eng, n, base):
    # SPD input: L L^T must reproduce the symmetrized matrix
    r = gen_matrix(n, n, seed=2000 + n)
    a = (r + r.T) / 2 + n * np.eye(n)
    L = DenseVecMatrix(a, engine=eng).choleskyDecompose(
        mode="dist", base_size=base).toBreeze()
    assert np.allclose(np.triu(L, 1), 0)
    rel = np.max(np.abs(L @ L.T - a)) / np.max(np.abs(a))
    assert rel < 1e-12, rel


def test_cholesky_local_route(eng):
    r = gen_matrix(60, 60, seed=3000)
    a = (r + r.T) / 2 + 60 * np.eye(60)
    L = DenseVecMatrix(a, engine=eng).choleskyDecompose().toBreeze()
    rel = np.max(np.abs(L @ L.T - a)) / np.max(np.abs(a))
    assert rel < 1e-12, rel


def test_transpose_dd(eng):
    a = gen_matrix(130, 70, seed=4000)
    A = eng.upload_matrix(a)
    At = eng.transpose_dd(A)
    got = eng.download_matrix(At)
    np.testing.assert_array_equal(got, a.T)
    # transposed handle is GEMM-ready: A^T @ A
    prod = eng.gemm_dd(At, A)
    np.testing.assert_allclose(eng.download_matrix(prod), a.T @ a,
                               rtol=1e-12, atol=1e-12)
    for d in (A, At, prod):
        d.free()


def test_gramian(eng):
    a = gen_matrix(500, 120, seed=5000)
    g = DenseVecMatrix(a, engine=eng).computeGramianMatrix()
    np.testing.assert_allclose(g, a.T @ a, rtol=1e-12, atol=1e-12)


def test_dgemv_dd(eng):
    a = gen_matrix(300, 200, seed=5100)
    x = gen_matrix(200, 1, seed=5101)[:, 0]
    dA = eng.upload_matrix(a)
    np.testing.assert_allclose(eng.dgemv_dd(dA, x), a @ x,
                               rtol=1e-12, atol=1e-12)
    dA.free()


def test_lr_matches_reference_sgd(eng):
    # reproduce the reference's full-batch gradient loop on the host and
    # compare weights after a few iterations
    rows = gen_matrix(200, 6, seed=5200)
    labels = (rows[:, 1] > 0.5).astype(np.float64)
    data = rows.copy()
    data[:, 0] = labels
    got = DenseVecMatrix(data, engine=eng).lr(0.5, 5)
    # host restatement (DenseVecMatrix.scala:1005-1035)
    X = data.copy()
    X[:, 0] = 1.0
    w = np.zeros(6)
    m = X.shape[0]
    for i in range(1, 6):
        margin = -(X @ w)
        gmul = 1.0 / (1.0 + np.exp(margin)) - labels
        delta = X.T @ gmul
        w = w - delta * (0.5 / m / np.sqrt(i))
    np.testing.assert_allclose(got, w, rtol=1e-10, atol=1e-12)
