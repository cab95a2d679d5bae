# Virtual-rank simulation of the engine's SUMMA dataflow for EVERY grid
# the engine can run (1x1, 2x1, 2x2, 4x2) — single process, numpy, using
# the .so's own plan (mx_plan_panels) and slab helpers. Complements the
# 2-process gloo test: this pins the 8-GPU (4x2) plan that cannot be run
# on the 1-GPU dev box. Mirrors summa_device (marlin_gpu.cpp) step for
# step: pack k-padded panels, "broadcast" (copy from the owning rank's
# shard), GEMM-accumulate per panel in plan order.
import numpy as np
import pytest

from marlin_amd import engine as E
from oracle import gen_matrix


def roundup(x, a):
    return (x + a - 1) // a * a


@pytest.mark.parametrize("nranks", [1, 2, 4, 8])
@pytest.mark.parametrize("mkn", [(300, 500, 260), (129, 4097, 65),
                                 (5, 300, 70)])  # m < pr: empty slabs
def test_summa_virtual_ranks(nranks, mkn):
    m, k, n = mkn
    pr, pc = E.grid_shape(nranks)
    A = gen_matrix(m, k, seed=0xA11CE)
    B = gen_matrix(k, n, seed=0xB0B)
    panels = E.plan_panels(k, pr, pc, kb_max=64)

    C = np.full((m, n), np.nan)
    for rank in range(nranks):
        prow, pcol = rank // pc, rank % pc
        mi, mo = E.slab_len(m, pr, prow), E.slab_off(m, pr, prow)
        nj, no = E.slab_len(n, pc, pcol), E.slab_off(n, pc, pcol)
        ka_off = E.slab_off(k, pc, pcol)
        kb_off = E.slab_off(k, pr, prow)
        if mi == 0 or nj == 0:
            continue
        C_local = None
        for (k0, k1, rootA, rootB) in panels:
            kb = k1 - k0
            kbp = roundup(kb, 16)
            # A panel: owner column rootA's shard columns, k-padded
            a_owner_off = E.slab_off(k, pc, rootA)
            pa = np.zeros((mi, kbp))
            pa[:, :kb] = A[mo:mo + mi, k0:k1]
            # correctness of ownership: the slice must lie inside rootA's
            # k-col slab (what the real code reads from its local shard)
            assert a_owner_off <= k0 and k1 <= a_owner_off + \
                E.slab_len(k, pc, rootA)
            b_owner_off = E.slab_off(k, pr, rootB)
            assert b_owner_off <= k0 and k1 <= b_owner_off + \
                E.slab_len(k, pr, rootB)
            pb = np.zeros((kbp, nj))
            pb[:kb, :] = B[k0:k1, no:no + nj]
            part = pa @ pb
            C_local = part if C_local is None else C_local + part
        C[mo:mo + mi, no:no + nj] = C_local

    ref = A @ B
    assert not np.isnan(C).any()
    rel = np.max(np.abs(C - ref)) / np.max(np.abs(ref))
    assert rel < 1e-13, rel
