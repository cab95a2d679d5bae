# Time dgemm_device against an arbitrary engine .so (diagnostics only).
import ctypes, sys, time
so, n = sys.argv[1], int(sys.argv[2])
lib = ctypes.CDLL(so)
i64 = ctypes.c_int64
lib.mx_init.argtypes = [ctypes.POINTER(ctypes.c_void_p), ctypes.c_int]
ctx = ctypes.c_void_p()
assert lib.mx_init(ctypes.byref(ctx), 0) == 0
lib.mx_alloc.argtypes = [ctypes.c_void_p, i64, ctypes.POINTER(ctypes.c_void_p)]
lib.mx_fill_random.argtypes = [ctypes.c_void_p, ctypes.c_void_p, i64, ctypes.c_uint64, ctypes.c_int]
lib.mx_dgemm_device.argtypes = [ctypes.c_void_p, i64, i64, i64, ctypes.c_void_p, i64, ctypes.c_void_p, i64, ctypes.c_void_p, i64]
mp = (n + 127) // 128 * 128
kp = (n + 15) // 16 * 16
bufs = []
for sz in (mp * kp, kp * mp, mp * mp):
    b = ctypes.c_void_p(); assert lib.mx_alloc(ctx, sz * 8, ctypes.byref(b)) == 0
    bufs.append(b)
dA, dB, dC = bufs
lib.mx_fill_random(ctx, dA, mp * kp, 1, 0)
lib.mx_fill_random(ctx, dB, kp * mp, 2, 0)
assert lib.mx_dgemm_device(ctx, mp, kp, mp, dA, mp, dB, kp, dC, mp) == 0
t0 = time.perf_counter()
for _ in range(3):
    assert lib.mx_dgemm_device(ctx, mp, kp, mp, dA, mp, dB, kp, dC, mp) == 0
dt = (time.perf_counter() - t0) / 3
print(f"{so}: {2*n**3/dt/1e12:.2f} TF/s ({dt*1e3:.1f} ms)")
