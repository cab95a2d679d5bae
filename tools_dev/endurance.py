# Endurance / stability checks (GPU):
#   1. 50-step 20000^3 run — per-step timing drift (thermal/DVFS, leaks)
#   2. repeated Engine open/close with RCCL comm_init — lifecycle leaks
#   3. big-footprint cycle: config-4 shape alloc/compute/free x3
import sys
import time

import numpy as np

sys.path.insert(0, ".")
from marlin_amd import Engine
from oracle import gen_matrix

# 1. timing drift
eng = Engine(0)
n, mp, kp = 20000, 20096, 20000
dA = eng.alloc(mp * kp * 8)
dB = eng.alloc(kp * mp * 8)
dC = eng.alloc(mp * mp * 8)
eng.fill_random(dA, mp * kp, 1)
eng.fill_random(dB, kp * mp, 2)
times = []
for i in range(52):
    t0 = time.perf_counter()
    eng.dgemm_device(mp, kp, mp, dA, mp, dB, kp, dC, mp)
    times.append(time.perf_counter() - t0)
t = np.array(times[2:])
print(f"50-step drift: mean {t.mean()*1e3:.1f} ms, min {t.min()*1e3:.1f}, "
      f"max {t.max()*1e3:.1f}, last10/first10 = "
      f"{t[-10:].mean()/t[:10].mean():.3f}")
for d in (dA, dB, dC):
    eng.free(d)
eng.close()

# 2. engine lifecycle
for i in range(5):
    e = Engine(0)
    e.comm_init(0, 1, Engine.comm_id())
    a = gen_matrix(100, 100, seed=i)
    c = e.dgemm(a, a)
    assert np.allclose(c, a @ a)
    e.close()
print("5x engine open/comm_init/dgemm/close: ok")

# 3. big-footprint cycles (config 4: C alone 20 GB)
e = Engine(0)
for i in range(3):
    m, k, nn = 50048, 4096, 50176
    dA = e.alloc(m * k * 8)
    dB = e.alloc(k * nn * 8)
    dC = e.alloc(m * nn * 8)
    e.fill_random(dA, m * k, 10 + i)
    e.fill_random(dB, k * nn, 20 + i)
    t0 = time.perf_counter()
    e.dgemm_device(m, k, nn, dA, m, dB, k, dC, m)
    print(f"cycle {i}: config-4 GEMM {(time.perf_counter()-t0)*1e3:.0f} ms")
    for d in (dA, dB, dC):
        e.free(d)
e.close()
print("endurance: ok")
