# Property-based tests (hypothesis) for the oracle restatement's blocking
# semantics — the ceil partition and emit/join/reduce algebra must equal
# the plain product for EVERY shape and split, not just the golden cases.
# CPU-only; sizes bounded so the whole file runs in seconds.
import numpy as np
import pytest

try:
    from hypothesis import given, settings, strategies as st
except ImportError:  # pragma: no cover
    pytest.skip("hypothesis not installed", allow_module_level=True)

from oracle import (blocked_multiply, gen_matrix, slab_len, slab_off,
                    split_method, to_blocks)
from oracle.marlin_oracle import assemble, effective_blocks

dims = st.integers(min_value=1, max_value=40)
parts = st.integers(min_value=1, max_value=9)


@settings(max_examples=60, deadline=None)
@given(m=dims, k=dims, n=dims, sm=parts, sk=parts, sn=parts)
def test_blocked_multiply_equals_plain(m, k, n, sm, sk, sn):
    a = gen_matrix(m, k, seed=m * 1000 + k)
    b = gen_matrix(k, n, seed=k * 1000 + n)
    got = blocked_multiply(a, b, (sm, sk, sn))
    assert got.shape == (m, n)
    np.testing.assert_allclose(got, a @ b, rtol=1e-12, atol=1e-12)


@settings(max_examples=100, deadline=None)
@given(total=st.integers(1, 10 ** 6), p=st.integers(1, 64))
def test_slabs_tile_exactly(total, p):
    lens = [slab_len(total, p, i) for i in range(p)]
    offs = [slab_off(total, p, i) for i in range(p)]
    assert sum(lens) == total
    # non-empty slabs are contiguous from 0
    pos = 0
    for o, l in zip(offs, lens):
        if l:
            assert o == pos
            pos += l
    # every non-terminal non-empty slab has the ceil length
    ceil_len = -(-total // p)
    nonzero = [l for l in lens if l]
    assert all(l == ceil_len for l in nonzero[:-1])


@settings(max_examples=60, deadline=None)
@given(m=dims, n=dims, r=parts, c=parts)
def test_to_blocks_assemble_roundtrip(m, n, r, c):
    a = gen_matrix(m, n, seed=m + 77 * n)
    blocks = to_blocks(a, r, c)
    assert len(blocks) == effective_blocks(m, r) * effective_blocks(n, c)
    assert np.array_equal(assemble(blocks), a)


@settings(max_examples=100, deadline=None)
@given(m=st.integers(1, 10 ** 5), k=st.integers(1, 10 ** 5),
       n=st.integers(1, 10 ** 5), cores=st.integers(1, 1024))
def test_split_method_invariants(m, k, n, cores):
    sm, sk, sn = split_method(m, k, n, cores)
    # splits are powers of two and never exceed what halving allows
    for s, d in ((sm, m), (sk, k), (sn, n)):
        assert s & (s - 1) == 0
        assert 1 <= s <= max(1, d)
    # total split factor bounded by the core budget's halving rounds
    rounds = 0
    c = cores
    while c > 1:
        c //= 2
        rounds += 1
    assert sm * sk * sn <= 2 ** rounds
