# Multi-process CPU validation of the SUMMA dataflow (world_size 2, gloo):
# each rank holds exactly the shards the engine's mx_dgemm_summa expects
# (ceil slabs on the pr x pc grid), panels are broadcast per the engine's
# mx_plan_panels roots over torch.distributed, each rank accumulates its
# C shard locally, and the assembled result must equal A @ B. This pins
# the distributed algorithm (ownership, panel plan, accumulate order)
# that the GPU RCCL path executes — summa_device in marlin_gpu.cpp is the
# same step sequence with ncclBroadcast on row/col sub-communicators.
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from marlin_amd import engine as E
from oracle import gen_matrix

def _summa_rank(rank, world, m, k, n, q, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        pr, pc = E.grid_shape(world)
        prow, pcol = rank // pc, rank % pc
        # global inputs are deterministic -> every rank can slice its shards
        A = gen_matrix(m, k, seed=0xA11CE)
        B = gen_matrix(k, n, seed=0xB0B)
        mi_off = E.slab_off(m, pr, prow)
        mi = E.slab_len(m, pr, prow)
        nj_off = E.slab_off(n, pc, pcol)
        nj = E.slab_len(n, pc, pcol)
        ka_off = E.slab_off(k, pc, pcol)
        kaj = E.slab_len(k, pc, pcol)
        kb_off = E.slab_off(k, pr, prow)
        kbi = E.slab_len(k, pr, prow)
        A_local = A[mi_off:mi_off + mi, ka_off:ka_off + kaj]
        B_local = B[kb_off:kb_off + kbi, nj_off:nj_off + nj]
        C_local = np.zeros((mi, nj))

        # row group: ranks with same prow; col group: same pcol
        row_groups = [dist.new_group([r for r in range(world)
                                      if r // pc == i]) for i in range(pr)]
        col_groups = [dist.new_group([r for r in range(world)
                                      if r % pc == j]) for j in range(pc)]

        for (k0, k1, rootA, rootB) in E.plan_panels(k, pr, pc, kb_max=64):
            kb = k1 - k0
            # A panel: owner column rootA broadcasts its mi x kb slice in-row
            if pcol == rootA:
                pa = torch.from_numpy(
                    np.ascontiguousarray(A_local[:, k0 - ka_off:k1 - ka_off]))
            else:
                pa = torch.zeros((mi, kb), dtype=torch.float64)
            dist.broadcast(pa, src=prow * pc + rootA, group=row_groups[prow])
            # B panel: owner row rootB broadcasts its kb x nj slice in-column
            if prow == rootB:
                pb = torch.from_numpy(
                    np.ascontiguousarray(B_local[k0 - kb_off:k1 - kb_off, :]))
            else:
                pb = torch.zeros((kb, nj), dtype=torch.float64)
            dist.broadcast(pb, src=rootB * pc + pcol, group=col_groups[pcol])
            C_local += pa.numpy() @ pb.numpy()

        # assemble on rank 0 and check against the plain product
        out = [torch.zeros(1)] * world
        mine = torch.from_numpy(C_local.copy())
        gathered = [torch.zeros_like(mine) if True else None
                    for _ in range(world)]
        # shards differ in shape across ranks -> gather shapes via all_gather
        # of flattened padded? keep simple: send to rank 0 pairwise
        if rank == 0:
            shards = {0: C_local}
            for r in range(1, world):
                ri, rj = r // pc, r % pc
                shp = (E.slab_len(m, pr, ri), E.slab_len(n, pc, rj))
                t = torch.zeros(shp, dtype=torch.float64)
                dist.recv(t, src=r)
                shards[r] = t.numpy()
            C = np.zeros((m, n))
            for r, sh in shards.items():
                ri, rj = r // pc, r % pc
                ro, co = E.slab_off(m, pr, ri), E.slab_off(n, pc, rj)
                C[ro:ro + sh.shape[0], co:co + sh.shape[1]] = sh
            ref = A @ B
            rel = np.max(np.abs(C - ref)) / np.max(np.abs(ref))
            q.put(("ok", rel) if rel < 1e-13 else ("fail", rel))
        else:
            dist.send(mine, dst=0)
        dist.barrier()
    finally:
        dist.destroy_process_group()


def _free_port():
    # ephemeral port from the kernel: immune to TIME_WAIT collisions
    # between parametrised cases / parallel runs
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_ranks(target, world, args):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [ctx.Process(target=target, args=(r, world) + args + (q, port))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    return q.get(timeout=10)


@pytest.mark.parametrize("world,mkn", [
    (2, (96, 130, 64)), (2, (257, 99, 121)), (4, (150, 200, 120)),
])
def test_summa_dataflow_gloo(world, mkn):
    m, k, n = mkn
    status, rel = _run_ranks(_summa_rank, world, mkn)
    assert status == "ok", f"SUMMA mismatch rel={rel}"


def _kres_rank(rank, world, m, k, n, q, port):
    # k-resident dataflow (config 4): each rank multiplies
    # ONLY its resident shards — no broadcast is even issued; the
    # assembled result must equal A @ B. Gloo is used solely to assemble.
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        assert E.summa_kresident(m, k, n, world)
        pr, pc = E.grid_shape(world)
        prow, pcol = rank // pc, rank % pc
        A = gen_matrix(m, k, seed=0xA11CE)
        B = gen_matrix(k, n, seed=0xB0B)
        mi, mo = E.slab_len(m, pr, prow), E.slab_off(m, pr, prow)
        nj, no = E.slab_len(n, pc, pcol), E.slab_off(n, pc, pcol)
        A_local = A[mo:mo + mi, :]          # resident: ALL k columns
        B_local = B[:, no:no + nj]          # resident: ALL k rows
        C_local = A_local @ B_local          # zero inter-rank exchange
        if rank == 0:
            C = np.zeros((m, n))
            C[mo:mo + mi, no:no + nj] = C_local
            for r in range(1, world):
                ri, rj = r // pc, r % pc
                shp = (E.slab_len(m, pr, ri), E.slab_len(n, pc, rj))
                t = torch.zeros(shp, dtype=torch.float64)
                dist.recv(t, src=r)
                ro, co = E.slab_off(m, pr, ri), E.slab_off(n, pc, rj)
                C[ro:ro + shp[0], co:co + shp[1]] = t.numpy()
            ref = A @ B
            rel = np.max(np.abs(C - ref)) / np.max(np.abs(ref))
            q.put(("ok", rel) if rel < 1e-13 else ("fail", rel))
        else:
            dist.send(torch.from_numpy(C_local.copy()), dst=0)
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world,mkn", [
    # config-4 shape scaled (k never split by splitMethod); ragged slabs
    (2, (500, 41, 460)), (4, (500, 41, 460)), (4, (501, 96, 463)),
])
def test_kresident_dataflow_gloo(world, mkn):
    m, k, n = mkn
    status, rel = _run_ranks(_kres_rank, world, mkn)
    assert status == "ok", f"k-resident mismatch rel={rel}"
