#!/usr/bin/env python3
"""Device-resident bandwidth runs of the auxiliary HBM-bound kernels
(map/transpose/sum/gemv) at sizes far past the 256 MiB L3, so rocprofv3
FETCH_SIZE/WRITE_SIZE passes over this script attribute real HBM
traffic per kernel (profiles/hbm_traffic.json aux entries).

Each kernel runs PASSES times on a 16384^2 fp64 image (2.1 GB, 8x L3).
Prints achieved GB/s from wall time (device-resident, no PCIe in the
loop) against the 6.3 TB/s achievable HBM roofline.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402

from marlin_amd import Engine  # noqa: E402
from marlin_amd import engine as E  # noqa: E402

PASSES = 5
N = 16384


def main():
    eng = Engine(0)
    lib = E.lib()
    nb = N * N * 8
    dA = eng.alloc(nb)
    dB = eng.alloc(nb)
    dC = eng.alloc(nb)
    eng.fill_random(dA, N * N, 0xA11CE)
    eng.fill_random(dB, N * N, 0xB0B)

    # transpose (device-resident): out = in^T, 2*nb bytes per pass
    t0 = time.perf_counter()
    for _ in range(PASSES):
        rc = lib.mx_transpose_device(eng._ctx, 0, N, N, dA, dC)
        assert rc == 0
    dt = (time.perf_counter() - t0) / PASSES
    print(f"transpose_device {N}^2 fp64: {dt*1e3:8.2f} ms  "
          f"{2*nb/dt/1e12:5.2f} TB/s (algorithmic 2x{nb>>20} MiB)")

    # gemv (device-resident): reads nb + writes small, nb bytes per pass
    x = np.random.RandomState(1).rand(N)
    t0 = time.perf_counter()
    for _ in range(PASSES):
        eng.dgemv_device_raw(N, N, dA, N, x)
    dt = (time.perf_counter() - t0) / PASSES
    print(f"gemv_device      {N}^2 fp64: {dt*1e3:8.2f} ms  "
          f"{nb/dt/1e12:5.2f} TB/s (algorithmic {nb>>20} MiB reads"
          f" + PCIe x/y vectors)")

    # map add / sum: ABI entries are host-buffer; time the whole entry
    # but report the kernel leg via mx_stats-less estimate -- the PMC
    # pass attributes per-kernel traffic regardless of the PCIe legs.
    a = np.asfortranarray(np.random.RandomState(2).rand(4096, 4096))
    b = np.asfortranarray(np.random.RandomState(3).rand(4096, 4096))
    for _ in range(PASSES):
        eng.map_op("add", a, b)
    print("map_kernel: 5 passes of 4096^2 add (PCIe-inclusive entry; "
          "PMC attributes the kernel leg)")
    for _ in range(PASSES):
        eng.sum(a)
    print("sum kernels: 5 passes of 4096^2")

    for d in (dA, dB, dC):
        eng.free(d)
    eng.close()


if __name__ == "__main__":
    main()
