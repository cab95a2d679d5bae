# Bandwidth record for the auxiliary (HBM-bound) kernels.
import sys, time
import numpy as np
sys.path.insert(0, ".")
from marlin_amd import Engine
from oracle import gen_matrix

eng = Engine(0)
n = 4000
a = gen_matrix(n, n, seed=1)
b = gen_matrix(n, n, seed=2)

def timeit(fn, passes, bytes_moved, label):
    fn()
    t0 = time.perf_counter()
    for _ in range(passes):
        fn()
    dt = (time.perf_counter() - t0) / passes
    print(f"{label:24s} {dt*1e3:8.2f} ms  {bytes_moved/dt/1e12:6.2f} TB/s (incl. PCIe)")

nb = n * n * 8
timeit(lambda: eng.map_op("add", a, b), 3, 3 * nb, "map add (H2D+k+D2H)")
timeit(lambda: eng.transpose(a), 3, 2 * nb, "transpose")
timeit(lambda: eng.sum(a), 3, nb, "sum")
x = gen_matrix(n, 1, seed=3)[:, 0]
timeit(lambda: eng.dgemv(a, x), 3, nb, "gemv")

# device-only legs (exclude PCIe): time via stats? use DeviceMatrix path
A = eng.upload_matrix(a)
B = eng.upload_matrix(b)
timeit(lambda: eng.gemm_dd(A, B).free(), 3, 2 * n**3 * 1e-12 and 0 or 0, "gemm_dd 4000^3 (alloc+k)")
print("(gemm_dd bytes column is n/a; wall includes alloc)")
eng.close()
