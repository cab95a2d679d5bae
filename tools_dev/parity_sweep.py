# Extensive randomized parity sweep (GPU): many shapes/seeds vs numpy.
import sys
import numpy as np
sys.path.insert(0, ".")
from marlin_amd import Engine
from oracle import gen_matrix

eng = Engine(0)
seed = int(sys.argv[2]) if len(sys.argv) > 2 else 20260915
rng = np.random.RandomState(seed)
worst = 0.0
fails = 0
N = int(sys.argv[1]) if len(sys.argv) > 1 else 200
for t in range(N):
    m = int(rng.randint(1, 4097))
    k = int(rng.randint(1, 4097))
    n = int(rng.randint(1, 4097))
    a = gen_matrix(m, k, seed=seed + 100000 + 2 * t)
    b = gen_matrix(k, n, seed=seed + 100001 + 2 * t)
    got = eng.dgemm(a, b)
    ref = a @ b
    rel = np.max(np.abs(got - ref)) / max(np.max(np.abs(ref)), 1e-300)
    worst = max(worst, rel)
    if rel > 1e-10:
        fails += 1
        print(f"FAIL ({m},{k},{n}) rel={rel}")
print(f"sweep(seed={seed}): {N} shapes, worst rel = {worst:.3e}, fails = {fails}")
eng.close()
