# On-disk text format round-trips (reference formats; host-side, no GPU):
#   DenseVecMatrix "row:v1,v2,..." (MTUtils.scala:286-300 loader,
#     DenseVecMatrix.saveToFileSystem writer, tools/generateMatrix.cpp)
#   BlockMatrix "r-c-rows-cols:colmajor-csv" (MTUtils.scala:324-340,
#     BlockMatrix.scala:550-559)
import os

import numpy as np

from marlin_amd import (DenseVecMatrix, load_matrix_file, save_matrix_file,
                        load_block_matrix_file, save_block_matrix_file)
from oracle import gen_matrix
from oracle import load_matrix_file as oracle_load

GOLD = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


def test_load_matrix_file_matches_oracle_parser(tmp_path):
    # both parsers read the reference's own data file identically
    a_ref = np.load(os.path.join(GOLD, "a100.npy"))
    # write in the reference format and re-load with the product parser
    dvm = DenseVecMatrix(a_ref)
    p = tmp_path / "a.txt"
    save_matrix_file(dvm, str(p))
    np.testing.assert_array_equal(load_matrix_file(str(p)).toBreeze(), a_ref)
    np.testing.assert_array_equal(oracle_load(str(p)), a_ref)


def test_densevec_roundtrip_exact(tmp_path):
    a = gen_matrix(37, 19, seed=5)
    p = tmp_path / "m.txt"
    save_matrix_file(DenseVecMatrix(a), str(p))
    got = load_matrix_file(str(p)).toBreeze()
    np.testing.assert_array_equal(got, a)  # repr() round-trips fp64 exactly


def test_separator_variants(tmp_path):
    # loader accepts ", " and whitespace separators (regex ",\s?|\s+")
    p = tmp_path / "v.txt"
    p.write_text("0:1.0, 2.0,3.0\n1:4.0 5.0 6.0\n")
    got = load_matrix_file(str(p)).toBreeze()
    np.testing.assert_array_equal(got, [[1.0, 2.0, 3.0], [4.0, 5.0, 6.0]])


def test_blockmatrix_roundtrip(tmp_path):
    a = gen_matrix(10, 8, seed=6)
    blk = DenseVecMatrix(a).toBlockMatrix(3, 2)
    p = tmp_path / "b.txt"
    save_block_matrix_file(blk, str(p))
    got = load_block_matrix_file(str(p))
    assert got.numBlksByRow() == blk.numBlksByRow()
    np.testing.assert_array_equal(got.toBreeze(), a)
    # line format spot check: "r-c-rows-cols:colmajor"
    line = p.read_text().splitlines()[0]
    head, data = line.split(":")
    assert head == "0-0-4-4"
    vals = [float(v) for v in data.split(",")]
    np.testing.assert_array_equal(
        np.array(vals).reshape((4, 4), order="F"), a[:4, :4])


def test_block_save_via_api_method(tmp_path):
    a = gen_matrix(9, 9, seed=7)
    blk = DenseVecMatrix(a).toBlockMatrix(2, 2)
    p1 = tmp_path / "bm.txt"
    blk.saveToFileSystem(str(p1), "blockmatrix")
    np.testing.assert_array_equal(load_block_matrix_file(str(p1)).toBreeze(), a)
    p2 = tmp_path / "dv.txt"
    blk.saveToFileSystem(str(p2))  # DenseVec format route
    np.testing.assert_array_equal(load_matrix_file(str(p2)).toBreeze(), a)


def test_load_reference_generated_file(tmp_path):
    """Parity anchor against the reference's OWN native generator:
    oracle/_ref/generateMatrix is tools/generateMatrix.cpp compiled
    unmodified from /root/reference (recipe: oracle/Makefile). Our
    loader (MTUtils.loadMatrixFile semantics) must parse its output
    exactly; values are U[0,5) floats (generateMatrix.cpp:14-24)."""
    import subprocess
    import pytest
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ref_bin = os.path.join(root, "oracle", "_ref", "generateMatrix")
    if not os.path.exists(ref_bin):
        if not os.path.exists("/root/reference/tools/generateMatrix.cpp"):
            pytest.skip("no reference tree and no prebuilt _ref binary")
        subprocess.run(["make", "-C", os.path.join(root, "oracle"), "-s"],
                       check=True)
    out = subprocess.run([ref_bin, "37", "11"], capture_output=True,
                         text=True, check=True).stdout
    p = tmp_path / "ref_gen.txt"
    p.write_text(out)
    m = load_matrix_file(str(p))
    a = m.toBreeze()
    assert a.shape == (37, 11)
    assert ((a >= 0) & (a < 5)).all()
    # cross-parse with an independent minimal parser: exact float match
    for line in out.strip().splitlines():
        idx, data = line.split(":", 1)
        vals = [float(v) for v in data.split(",")]
        assert (a[int(idx)] == vals).all()
