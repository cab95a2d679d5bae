# 112 GiB working-set GEMM demonstration + column-0 identity check.
import sys, time
import numpy as np
sys.path.insert(0, ".")
from marlin_amd import Engine
from oracle import gen_uniform_u64

eng = Engine(0)
m, k, n = 100000, 20000, 100000
mp, kp, np_ = 100096, 20000, 100096
print(f"working set: {(mp*kp + kp*np_ + mp*np_)*8/2**30:.1f} GiB of 288 GB HBM3E")
dA = eng.alloc(mp * kp * 8)
dB = eng.alloc(kp * np_ * 8)
dC = eng.alloc(mp * np_ * 8)
eng.fill_random(dA, mp * kp, 0xA11CE)
eng.fill_random(dB, kp * np_, 0xB0B)
t0 = time.perf_counter()
eng.dgemm_device(mp, kp, np_, dA, mp, dB, kp, dC, mp)
dt = time.perf_counter() - t0
print(f"{m}x{k}x{n} fp64: {2.0*m*k*n/dt/1e12:.1f} TF/s ({dt:.1f} s)")
col = np.empty(mp)
eng.download(col, dC, mp * 8)
bcol = np.empty(kp)
eng.download(bcol, dB, kp * 8)
acc = np.zeros(m)
for c0 in range(0, k, 512):
    c1 = min(c0 + 512, k)
    z = gen_uniform_u64(0xA11CE, c0 * mp, (c1 - c0) * mp)
    blockA = ((z >> np.uint64(11)).astype(np.float64) * 2.0**-53
              ).reshape((c1 - c0, mp)).T[:m]
    acc += blockA @ bcol[c0:c1]
rel = np.max(np.abs(col[:m] - acc)) / np.max(np.abs(acc))
verdict = "OK" if rel < 1e-10 else "FAIL"
print(f"column-0 identity check: rel = {rel:.2e} {verdict}")
eng.close()
