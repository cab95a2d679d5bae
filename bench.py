#!/usr/bin/env python3
# bench.py — headline benchmark of the MI355X block-matrix multiply engine.
#
# Metric (BASELINE.json): dense C=A*B TFLOP/s (fp64) at N=20000.
# A "step" = one full C = A x B over device-resident inputs (inputs are in
# HBM before the timed region starts; H2D is NOT in the timed region and
# the PCIe-inclusive rate is reported in DESIGN.md).
#
#   python bench.py --gpus N --steps K --warmup W
#
# N > 1 is launched by the driver via torch.distributed.run (one rank per
# GPU); ranks use gloo ONLY as the control plane (RCCL unique-id exchange,
# barriers) — the data path is the engine's own RCCL-over-xGMI SUMMA.
# Rank 0 prints ONE JSON line.
import argparse
import json
import os
import sys
import time

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, HERE)

from marlin_amd import Engine              # noqa: E402
from marlin_amd import engine as eng_mod   # noqa: E402

FP64_PEAK_TF = 78.6   # gfx950: 256 CU x 4 SIMD x 32 FLOP/clk x 2.4 GHz
FP32_PEAK_TF = 157.3


def roundup(x, a):
    return (x + a - 1) // a * a


def cpu_baseline_leg(args):
    """Oracle restatement (the reference's blocked algorithm, OpenBLAS
    per-tile dgemm) timed on the host cores — BASELINE.md plan. Bounded
    sample: nb^3 multiply (~seconds of CPU), scaled to TFLOP/s."""
    from oracle import gen_matrix, blocked_multiply, split_method
    ncpu = os.cpu_count() or 1
    try:  # report the BLAS threads actually used, not just nproc
        import threadpoolctl
        infos = [i for i in threadpoolctl.threadpool_info()
                 if i.get("user_api") == "blas"]
        cores = max((i["num_threads"] for i in infos), default=1)
    except Exception:
        cores = ncpu
    nb = int(args.cpu_sample)
    a = gen_matrix(nb, nb, seed=0xA11CE)
    b = gen_matrix(nb, nb, seed=0xB0B)
    mkn = split_method(nb, nb, nb, ncpu)
    t0 = time.perf_counter()
    blocked_multiply(a, b, mkn)
    dt = time.perf_counter() - t0
    tf = 2.0 * nb ** 3 / dt / 1e12
    return {
        "value": round(tf, 4), "unit": "TFLOP/s", "cores": cores,
        "kind": "port",
        "sample": f"{nb}^3 fp64 blocked multiply (CARMA split {mkn} for "
                  f"{ncpu} host cores, OpenBLAS tiles x{cores} threads, "
                  f"{dt:.2f}s)",
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    # size flags also readable from env (MARLIN_BENCH_*): torchrun's arg
    # parser steals abbreviated flags like --n from the script args, so
    # under torchrun sizes are passed via env instead
    env = os.environ
    p.add_argument("--n", type=int,
                   default=int(env.get("MARLIN_BENCH_N", "20000")),
                   help="square problem size (m=k=n)")
    p.add_argument("--m", type=int, default=int(env.get("MARLIN_BENCH_M", "0")))
    p.add_argument("--k", type=int, default=int(env.get("MARLIN_BENCH_K", "0")))
    p.add_argument("--nn", type=int,
                   default=int(env.get("MARLIN_BENCH_NN", "0")))
    p.add_argument("--dtype", choices=["f64", "f32"],
                   default=env.get("MARLIN_BENCH_DTYPE", "f64"))
    # workload "epilogue": config 5's fused (A*B)^T + addC leg as its own
    # timed line (fp32, device-resident), never the default
    p.add_argument("--workload", choices=["gemm", "epilogue"],
                   default=env.get("MARLIN_BENCH_WORKLOAD", "gemm"))
    p.add_argument("--cpu-sample", type=int, default=12000)
    p.add_argument("--no-cpu-baseline", action="store_true")
    # run the SUMMA panel-loop path even at world 1 (1x1 grid, RCCL comm
    # of size 1): measures the distributed pipeline's overhead (panel
    # packs + beta-chained GEMMs) without xGMI traffic — the pre-SCALE
    # machinery check
    p.add_argument("--force-summa", action="store_true",
                   default=bool(env.get("MARLIN_BENCH_FORCE_SUMMA")))
    args = p.parse_args()

    if args.workload == "epilogue":
        args.dtype = "f32"
        if args.n == 20000 and not (args.m or args.k or args.nn):
            args.n = 40000          # config 5's quoted size
    m = args.m or args.n
    k = args.k or args.n
    n = args.nn or args.n

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist = None
    if world > 1:
        import torch.distributed as tdist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        tdist.init_process_group(backend="gloo", rank=rank, world_size=world)
        dist = tdist

    fp32 = args.dtype == "f32"
    elem = 4 if fp32 else 8
    # MARLIN_FORCE_DEV0: place every rank on device 0 (multi-rank RCCL
    # smoke on a single-GPU box; never set by the driver)
    dev = 0 if os.environ.get("MARLIN_FORCE_DEV0") else local_rank
    eng = Engine(dev)

    # --- distributed setup + device-resident inputs ----------------------
    if world > 1:
        import torch
        if rank == 0:
            uid = Engine.comm_id()
            t = torch.tensor(list(uid), dtype=torch.uint8)
        else:
            t = torch.zeros(eng_mod.UNIQUE_ID_BYTES, dtype=torch.uint8)
        dist.broadcast(t, src=0)
        eng.comm_init(rank, world, bytes(t.tolist()))
        pr, pc, prow, pcol = eng.grid()
    else:
        pr = pc = 1
        prow = pcol = 0

    mi = eng_mod.slab_len(m, pr, prow)
    nj = eng_mod.slab_len(n, pc, pcol)
    kaj = eng_mod.slab_len(k, pc, pcol)
    kbi = eng_mod.slab_len(k, pr, prow)
    mip, njp = roundup(mi, 128), roundup(nj, 128)
    kbi_p = roundup(kbi, 16)

    kres = False
    if world > 1:
        # layout per CARMA splitMethod semantics (MTUtils.scala:150-175):
        # kSplit==1 -> k-resident shards, zero steady-state xGMI traffic
        # (config 4); else k-slabbed panel-broadcast SUMMA (config 3)
        kres = eng_mod.summa_kresident(m, k, n, world)
        if kres:
            kp = roundup(k, 16)
            dA = eng.alloc(mip * kp * elem)
            dB = eng.alloc(kp * njp * elem)
            dC = eng.alloc(mip * njp * elem)
            eng.fill_random(dA, mip * kp, 0xA11CE + rank, fp32)
            eng.fill_random(dB, kp * njp, 0xB0B + rank, fp32)
            # zero-pad invariant: pad rows/cols must be zero
            eng.zero_pad(dA, mip, kp, mip, mi, k, fp32)
            eng.zero_pad(dB, kp, njp, kp, k, nj, fp32)

            def step():
                eng.gemm_summa_kres_device(m, k, n, dA, dB, dC, fp32)
        else:
            dA = eng.alloc(mip * kaj * elem)
            dB = eng.alloc(kbi_p * nj * elem)
            dC = eng.alloc(mip * njp * elem)
            eng.fill_random(dA, mip * kaj, 0xA11CE + rank, fp32)
            eng.fill_random(dB, kbi_p * nj, 0xB0B + rank, fp32)
            eng.zero_pad(dA, mip, kaj, mip, mi, kaj, fp32)
            eng.zero_pad(dB, kbi_p, nj, kbi_p, kbi, nj, fp32)

            summa = eng.sgemm_summa_device if fp32 else eng.dgemm_summa_device

            def step():
                summa(m, k, n, dA, dB, dC)
    elif args.force_summa:
        # 1x1-grid SUMMA: the full panel-loop machinery, zero comm
        eng.comm_init(0, 1, Engine.comm_id())
        kbi_p1 = roundup(k, 16)
        dA = eng.alloc(mip * kaj * elem)       # mi=m, kaj=k at 1x1
        dB = eng.alloc(kbi_p1 * nj * elem)
        dC = eng.alloc(mip * njp * elem)
        eng.fill_random(dA, mip * kaj, 0xA11CE, fp32)
        eng.fill_random(dB, kbi_p1 * nj, 0xB0B, fp32)
        eng.zero_pad(dA, mip, kaj, mip, mi, kaj, fp32)
        eng.zero_pad(dB, kbi_p1, nj, kbi_p1, kbi, nj, fp32)
        summa = eng.sgemm_summa_device if fp32 else eng.dgemm_summa_device

        def step():
            summa(m, k, n, dA, dB, dC)
    else:
        mp, kp, np_ = roundup(m, 128), roundup(k, 16), roundup(n, 128)
        dA = eng.alloc(mp * kp * elem)
        dB = eng.alloc(kp * np_ * elem)
        eng.fill_random(dA, mp * kp, 0xA11CE, fp32)
        eng.fill_random(dB, kp * np_, 0xB0B, fp32)
        eng.zero_pad(dA, mp, kp, mp, m, k, fp32)
        eng.zero_pad(dB, kp, np_, kp, k, n, fp32)
        if args.workload == "epilogue":
            # C is (A*B)^T + addC: n x m col-major, pitch np_
            dC = eng.alloc(np_ * mp * elem)
            dAdd = eng.alloc(np_ * mp * elem)
            eng.fill_random(dAdd, np_ * mp, 0xADD, fp32)

            def step():
                eng.sgemm_epilogue_device(mp, kp, np_, dA, mp, dB, kp,
                                          dC, np_, dAdd)
        else:
            dC = eng.alloc(mp * np_ * elem)
            gem = eng.sgemm_device if fp32 else eng.dgemm_device

            def step():
                gem(mp, kp, np_, dA, mp, dB, kp, dC, mp)

    # --- warmup / timed region ------------------------------------------
    for _ in range(args.warmup):
        step()
    if dist:
        dist.barrier()
    t0 = time.perf_counter()
    gemm_ms_acc = 0.0
    launches = 0
    for _ in range(args.steps):
        step()
        st = eng.stats()
        gemm_ms_acc += st["gemm_ms"]
        launches += st["gemm_launches"]
    # engine entries synchronize internally; no extra device sync needed
    dt = time.perf_counter() - t0
    if os.environ.get("MARLIN_SUMMA_DEBUG") and world > 1:
        st = eng.stats()
        print(f"[rank {rank}] layout {'kres' if kres else 'slab'}, wall "
              f"{dt*1e3:.1f} ms, last-step gemm {st['gemm_ms']:.1f} ms, "
              f"comm {st['comm_ms']:.1f} ms, launches "
              f"{st['gemm_launches']}", file=sys.stderr, flush=True)
    if dist:
        import torch
        tmax = torch.tensor([dt])
        dist.all_reduce(tmax, op=dist.ReduceOp.MAX)
        dt = float(tmax[0])

    if rank == 0:
        flops = 2.0 * m * k * n * args.steps
        tf = flops / dt / 1e12
        ms_per_step = dt * 1000.0 / args.steps
        peak = FP32_PEAK_TF if fp32 else FP64_PEAK_TF
        # confirm the peak from the box's own CU count x max clock
        # (SURVEY 8d): fp64 MFMA = 32 FLOP/clk/SIMD x 4 SIMD/CU;
        # fp32 = 2x that rate
        try:
            cus, clk = eng.device_info()
            flop_per_clk = (64 if fp32 else 32) * 4
            measured_peak = cus * flop_per_clk * clk * 1e3 / 1e12
            if 0.5 * peak < measured_peak < 1.5 * peak:
                peak = round(measured_peak, 2)
        except Exception:
            pass
        # dominant-kernel roofline: HIP-event time of the MFMA GEMM kernel
        # launches (stats from the engine's gemm-stream events)
        if launches and gemm_ms_acc > 0:
            ach = (2.0 * m * k * n * args.steps) / (gemm_ms_acc / 1e3) / 1e12
        else:
            ach = None
        wl = (f"sgemm_tn_epilogue_{m}x{k}x{n}_f32"
              if args.workload == "epilogue" else
              f"dgemm_{m}x{k}x{n}_{args.dtype}")
        # HBM traffic per launch: measured offline by rocprofv3 --pmc
        # FETCH_SIZE passes (tools_dev/rocpd_stats.py) and committed under
        # profiles/hbm_traffic.json keyed by workload; null when this
        # workload has no committed measurement.
        traffic = None
        try:
            tj = json.load(open(os.path.join(HERE, "profiles",
                                             "hbm_traffic.json")))
            rec = tj.get(wl)
            if rec and world == 1:
                traffic = rec["reads_bytes_per_launch"]
        except Exception:
            pass
        roofline = {
            "bound": "mfma",
            "achieved": round(ach, 3) if ach else None,
            "peak": peak * world,
            "unit": "TFLOP/s",
            "frac": round(ach / (peak * world), 4) if ach else None,
            "traffic": traffic,
        }
        if args.workload == "epilogue":
            metric = "dense C=(AxB)^T+D TFLOP/s (fp32)"
        else:
            metric = ("dense C=AxB TFLOP/s (fp64)" if not fp32 else
                      "dense C=AxB TFLOP/s (fp32)")
        if world > 1:
            par = (f"kres_grid_{pr}x{pc}" if kres else
                   f"summa_grid_{pr}x{pc}")
        elif args.force_summa:
            par = "summa_1x1_machinery"
        else:
            par = "single_gpu"
        out = {
            "metric": metric,
            "value": round(tf, 3),
            "unit": "TFLOP/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "workload": wl,
                "m": m, "k": k, "n": n,
                "parallelism": par,
            },
            "roofline": roofline,
        }
        if not args.no_cpu_baseline and world == 1:
            out["cpu_baseline"] = cpu_baseline_leg(args)
        print(json.dumps(out), flush=True)
    if dist:
        dist.destroy_process_group()
    eng.close()


if __name__ == "__main__":
    main()
