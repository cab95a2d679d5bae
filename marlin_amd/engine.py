# engine.py — ctypes binding to libmarlin_gpu.so (the C ABI of
# include/marlin_gpu.h). This is the product compute path: it NEVER falls
# back to CPU. If the extension or a GPU is missing, every compute entry
# raises EngineUnavailable loudly.
import ctypes
import os

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_HERE, "libmarlin_gpu.so")

UNIQUE_ID_BYTES = 128

_ERRNAMES = {
    0: "MX_OK", -1: "MX_EDIM", -2: "MX_EHIP", -3: "MX_ENOMEM",
    -4: "MX_EINVAL", -5: "MX_ENOCOMM", -6: "MX_ERCCL", -7: "MX_ENODEV",
}


class EngineUnavailable(RuntimeError):
    pass


class EngineError(RuntimeError):
    def __init__(self, code, what=""):
        super().__init__(f"marlin_gpu error {_ERRNAMES.get(code, code)} {what}")
        self.code = code


class MxStats(ctypes.Structure):
    _fields_ = [
        ("h2d_ms", ctypes.c_double), ("d2h_ms", ctypes.c_double),
        ("pack_ms", ctypes.c_double), ("gemm_ms", ctypes.c_double),
        ("comm_ms", ctypes.c_double), ("total_ms", ctypes.c_double),
        ("gemm_launches", ctypes.c_int64), ("flops", ctypes.c_double),
        ("bytes_moved", ctypes.c_double),
    ]


def _load():
    if not os.path.exists(_SO):
        raise EngineUnavailable(
            f"HIP extension not built: {_SO} missing. Run "
            f"python -c 'import __graft_entry__; __graft_entry__.build()'")
    try:
        lib = ctypes.CDLL(_SO)
    except OSError as e:
        raise EngineUnavailable(f"cannot load {_SO}: {e}")
    P = ctypes.POINTER
    i64, u64, dbl, flt = (ctypes.c_int64, ctypes.c_uint64, ctypes.c_double,
                          ctypes.c_float)
    vp = ctypes.c_void_p
    sigs = {
        "mx_strerror": (ctypes.c_char_p, [ctypes.c_int]),
        "mx_init": (ctypes.c_int, [P(vp), ctypes.c_int]),
        "mx_shutdown": (ctypes.c_int, [vp]),
        "mx_comm_id": (ctypes.c_int, [ctypes.c_char_p]),
        "mx_comm_init": (ctypes.c_int, [vp, ctypes.c_int, ctypes.c_int,
                                        ctypes.c_char_p]),
        "mx_grid": (ctypes.c_int, [vp] + [P(ctypes.c_int)] * 4),
        "mx_device_info": (ctypes.c_int, [vp, P(ctypes.c_int),
                                          P(ctypes.c_int)]),
        "mx_dgemm": (ctypes.c_int, [vp, i64, i64, i64, P(dbl), P(dbl), P(dbl)]),
        "mx_sgemm": (ctypes.c_int, [vp, i64, i64, i64, P(flt), P(flt), P(flt)]),
        "mx_sgemm_epilogue": (ctypes.c_int, [vp, i64, i64, i64, P(flt), P(flt),
                                             P(flt), ctypes.c_int, P(flt)]),
        "mx_tile_dgemm_acc": (ctypes.c_int, [vp, i64, i64, i64, P(dbl), P(dbl),
                                             P(dbl), ctypes.c_int]),
        "mx_dgemm_summa": (ctypes.c_int, [vp, i64, i64, i64, P(dbl), P(dbl),
                                          P(dbl)]),
        "mx_sgemm_summa": (ctypes.c_int, [vp, i64, i64, i64, P(flt), P(flt),
                                          P(flt)]),
        "mx_slab_len": (i64, [i64, ctypes.c_int, ctypes.c_int]),
        "mx_slab_off": (i64, [i64, ctypes.c_int, ctypes.c_int]),
        "mx_plan_panels": (ctypes.c_int, [i64, ctypes.c_int, ctypes.c_int, i64,
                                          P(i64), P(i64), P(ctypes.c_int),
                                          P(ctypes.c_int), ctypes.c_int]),
        "mx_alloc": (ctypes.c_int, [vp, i64, P(vp)]),
        "mx_free": (ctypes.c_int, [vp, vp]),
        "mx_upload": (ctypes.c_int, [vp, vp, vp, i64]),
        "mx_download": (ctypes.c_int, [vp, vp, vp, i64]),
        "mx_fill_random": (ctypes.c_int, [vp, vp, i64, u64, ctypes.c_int]),
        "mx_dgemm_device": (ctypes.c_int, [vp, i64, i64, i64, vp, i64, vp, i64,
                                           vp, i64]),
        "mx_sgemm_device": (ctypes.c_int, [vp, i64, i64, i64, vp, i64, vp, i64,
                                           vp, i64]),
        "mx_dgemm_summa_device": (ctypes.c_int, [vp, i64, i64, i64, vp, vp, vp]),
        "mx_sgemm_summa_device": (ctypes.c_int, [vp, i64, i64, i64, vp, vp, vp]),
        "mx_summa_kresident": (ctypes.c_int, [i64, i64, i64, ctypes.c_int]),
        "mx_dgemm_summa_kres": (ctypes.c_int, [vp, i64, i64, i64, P(dbl),
                                               P(dbl), P(dbl)]),
        "mx_sgemm_summa_kres": (ctypes.c_int, [vp, i64, i64, i64, P(flt),
                                               P(flt), P(flt)]),
        "mx_gemm_summa_kres_device": (ctypes.c_int, [vp, ctypes.c_int, i64,
                                                     i64, i64, vp, vp, vp]),
        "mx_zero_pad": (ctypes.c_int, [vp, vp, i64, i64, i64, i64, i64,
                                       ctypes.c_int]),
        "mx_download_off": (ctypes.c_int, [vp, vp, vp, i64, i64]),
        "mx_download2d_off": (ctypes.c_int, [vp, vp, vp, i64, i64, i64, i64,
                                             ctypes.c_int]),
        "mx_sgemm_epilogue_device": (ctypes.c_int, [vp, i64, i64, i64, vp,
                                                    i64, vp, i64, vp, i64,
                                                    vp]),
        "mx_test_rccl_error": (ctypes.c_int, [vp]),
        "mx_map": (ctypes.c_int, [vp, ctypes.c_int, ctypes.c_int, i64, vp,
                                  vp, dbl, vp]),
        "mx_sum": (ctypes.c_int, [vp, ctypes.c_int, i64, vp, P(dbl)]),
        "mx_transpose": (ctypes.c_int, [vp, ctypes.c_int, i64, i64, vp, vp]),
        "mx_dgemv": (ctypes.c_int, [vp, i64, i64, P(dbl), P(dbl), P(dbl)]),
        "mx_upload2d": (ctypes.c_int, [vp, vp, i64, vp, i64, i64,
                                       ctypes.c_int]),
        "mx_download2d": (ctypes.c_int, [vp, vp, vp, i64, i64, i64,
                                         ctypes.c_int]),
        "mx_memset": (ctypes.c_int, [vp, vp, i64]),
        "mx_gemm_device_ex": (ctypes.c_int, [vp, ctypes.c_int, ctypes.c_int,
                                             i64, i64, i64, vp, i64, vp, i64,
                                             vp, i64]),
        "mx_transpose_device": (ctypes.c_int, [vp, ctypes.c_int, i64, i64,
                                               vp, vp]),
        "mx_dgemv_device": (ctypes.c_int, [vp, i64, i64, vp, i64, P(dbl),
                                           P(dbl)]),
        "mx_stats": (ctypes.c_int, [vp, P(MxStats)]),
    }
    for name, (res, args) in sigs.items():
        fn = getattr(lib, name)
        fn.restype = res
        fn.argtypes = args
    return lib


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = _load()
    return _lib


_LOG = os.environ.get("MARLIN_LOG")


def _log(msg):
    """stderr logger (SURVEY §5 observability plan): MARLIN_LOG=1 enables
    per-call reports, =2 adds stage timings."""
    if _LOG:
        import sys
        print(f"[marlin_amd] {msg}", file=sys.stderr, flush=True)


def _ck(code, what=""):
    if code != 0:
        raise EngineError(code, what)


# --- pure planning helpers (no GPU needed) ---------------------------------
def slab_len(total, parts, idx):
    return lib().mx_slab_len(total, parts, idx)


def slab_off(total, parts, idx):
    return lib().mx_slab_off(total, parts, idx)


def plan_panels(K, pr, pc, kb_max=4096):
    cap = 4096
    k0 = (ctypes.c_int64 * cap)()
    k1 = (ctypes.c_int64 * cap)()
    ra = (ctypes.c_int * cap)()
    rb = (ctypes.c_int * cap)()
    npan = lib().mx_plan_panels(K, pr, pc, kb_max, k0, k1, ra, rb, cap)
    if npan < 0:
        raise EngineError(-4, f"panel overflow {npan}")
    return [(k0[i], k1[i], ra[i], rb[i]) for i in range(npan)]


def grid_shape(nranks):
    return {8: (4, 2), 4: (2, 2), 2: (2, 1), 1: (1, 1)}.get(nranks,
                                                            (nranks, 1))


def summa_kresident(m, k, n, nranks):
    """True when CARMA splitMethod (MTUtils.scala:150-175) leaves k
    unsplit for this job size -> the k-resident distributed layout
    applies (zero steady-state xGMI traffic; BASELINE config 4)."""
    return bool(lib().mx_summa_kresident(m, k, n, nranks))


def _fbuf(a, dtype):
    assert a.dtype == dtype and a.flags.f_contiguous
    return a.ctypes.data_as(ctypes.POINTER(
        ctypes.c_double if dtype == np.float64 else ctypes.c_float))


class DeviceMatrix:
    """A matrix resident in HBM (the RDD.cache() analog): logical m x n,
    col-major with GEMM-ready padded pitch (rows padded to 128, pad
    region zeroed). Owned by an Engine; freed via .free() or engine
    close."""

    __slots__ = ("eng", "buf", "m", "n", "pitch", "fp32")

    def __init__(self, eng, buf, m, n, pitch, fp32=False):
        self.eng, self.buf = eng, buf
        self.m, self.n, self.pitch, self.fp32 = m, n, pitch, fp32

    @property
    def elem(self):
        return 4 if self.fp32 else 8

    def free(self):
        if self.buf is not None:
            self.eng.free(self.buf)
            self.buf = None


class Engine:
    """One GPU per process. Owns an mx_ctx."""

    def __init__(self, device=-1):
        self._ctx = ctypes.c_void_p()
        rc = lib().mx_init(ctypes.byref(self._ctx), device)
        if rc == -7:
            raise EngineUnavailable("no MI355X visible (mx_init -> MX_ENODEV)")
        _ck(rc, "mx_init")

    def close(self):
        if self._ctx:
            lib().mx_shutdown(self._ctx)
            self._ctx = None

    # -- distributed setup --------------------------------------------------
    @staticmethod
    def comm_id():
        buf = ctypes.create_string_buffer(UNIQUE_ID_BYTES)
        _ck(lib().mx_comm_id(buf), "mx_comm_id")
        return buf.raw

    def comm_init(self, rank, nranks, uid_bytes):
        assert len(uid_bytes) == UNIQUE_ID_BYTES
        _ck(lib().mx_comm_init(self._ctx, rank, nranks, uid_bytes),
            "mx_comm_init")

    def device_info(self):
        """(CU count, max clock kHz) of the bound device."""
        cus, clk = ctypes.c_int(), ctypes.c_int()
        _ck(lib().mx_device_info(self._ctx, ctypes.byref(cus),
                                 ctypes.byref(clk)))
        return cus.value, clk.value

    def grid(self):
        vals = [ctypes.c_int() for _ in range(4)]
        _ck(lib().mx_grid(self._ctx, *[ctypes.byref(v) for v in vals]))
        return tuple(v.value for v in vals)  # pr, pc, prow, pcol

    # -- whole multiplies ----------------------------------------------------
    def dgemm(self, A, B):
        A = np.asfortranarray(A, dtype=np.float64)
        B = np.asfortranarray(B, dtype=np.float64)
        m, k = A.shape
        k2, n = B.shape
        if k != k2:
            raise ValueError(
                f"Dimension mismatch during matrix-matrix multiplication: "
                f"{k} vs {k2}")
        C = np.empty((m, n), dtype=np.float64, order="F")
        _ck(lib().mx_dgemm(self._ctx, m, k, n, _fbuf(A, np.float64),
                           _fbuf(B, np.float64), _fbuf(C, np.float64)),
            "mx_dgemm")
        if _LOG:
            st = self.stats()
            _log(f"dgemm {m}x{k}x{n}: gemm {st['gemm_ms']:.2f} ms "
                 f"(h2d {st['h2d_ms']:.2f} ms, "
                 f"{st['flops'] / max(st['gemm_ms'], 1e-9) / 1e9:.1f} TF/s)")
        return C

    def sgemm(self, A, B):
        A = np.asfortranarray(A, dtype=np.float32)
        B = np.asfortranarray(B, dtype=np.float32)
        m, k = A.shape
        _, n = B.shape
        if A.shape[1] != B.shape[0]:
            raise ValueError("Dimension mismatch during matrix-matrix multiplication")
        C = np.empty((m, n), dtype=np.float32, order="F")
        _ck(lib().mx_sgemm(self._ctx, m, k, n, _fbuf(A, np.float32),
                           _fbuf(B, np.float32), _fbuf(C, np.float32)),
            "mx_sgemm")
        return C

    def sgemm_transpose_add(self, A, B, add_c=None):
        """C = (A*B)^T (+ add_c) with the epilogue fused on-device."""
        A = np.asfortranarray(A, dtype=np.float32)
        B = np.asfortranarray(B, dtype=np.float32)
        m, k = A.shape
        _, n = B.shape
        C = np.empty((n, m), dtype=np.float32, order="F")
        addp = None
        if add_c is not None:
            add_c = np.asfortranarray(add_c, dtype=np.float32)
            assert add_c.shape == (n, m)
            addp = _fbuf(add_c, np.float32)
        _ck(lib().mx_sgemm_epilogue(self._ctx, m, k, n, _fbuf(A, np.float32),
                                    _fbuf(B, np.float32), _fbuf(C, np.float32),
                                    1, addp), "mx_sgemm_epilogue")
        return C

    def tile_dgemm_acc(self, A, B, C=None):
        """C (+)= A*B per tile — the SubMatrix.multiply/add replacement."""
        A = np.asfortranarray(A, dtype=np.float64)
        B = np.asfortranarray(B, dtype=np.float64)
        m, k = A.shape
        _, n = B.shape
        beta = 1 if C is not None else 0
        if C is None:
            C = np.empty((m, n), dtype=np.float64, order="F")
        else:
            C = np.asfortranarray(C, dtype=np.float64)
        _ck(lib().mx_tile_dgemm_acc(self._ctx, m, k, n, _fbuf(A, np.float64),
                                    _fbuf(B, np.float64), _fbuf(C, np.float64),
                                    beta), "mx_tile_dgemm_acc")
        return C

    def dgemm_summa(self, m, k, n, A_local, B_local):
        """Distributed multiply on this rank's shards (after comm_init)."""
        A_local = np.asfortranarray(A_local, dtype=np.float64)
        B_local = np.asfortranarray(B_local, dtype=np.float64)
        pr, pc, prow, pcol = self.grid()
        mi = slab_len(m, pr, prow)
        nj = slab_len(n, pc, pcol)
        C = np.empty((mi, nj), dtype=np.float64, order="F")
        _ck(lib().mx_dgemm_summa(self._ctx, m, k, n, _fbuf(A_local, np.float64),
                                 _fbuf(B_local, np.float64),
                                 _fbuf(C, np.float64)), "mx_dgemm_summa")
        return C

    # -- elementwise / reduction / transpose (epilogue-op family) ----------
    OPS = {"add": 0, "sub": 1, "emul": 2, "adds": 3, "subs": 4, "rsubs": 5,
           "muls": 6, "divs": 7, "rdivs": 8}

    def map_op(self, op, A, B=None, scalar=0.0):
        """Elementwise op on same-shape col-major arrays (flat)."""
        A = np.asfortranarray(A, dtype=np.float64)
        opc = self.OPS[op]
        bp = None
        if opc <= 2:
            B = np.asfortranarray(B, dtype=np.float64)
            if A.shape != B.shape:
                raise ValueError("matrix dimension mismatch")
            bp = B.ctypes.data_as(ctypes.c_void_p)
        C = np.empty_like(A)
        _ck(lib().mx_map(self._ctx, opc, 0, A.size,
                         A.ctypes.data_as(ctypes.c_void_p), bp,
                         float(scalar), C.ctypes.data_as(ctypes.c_void_p)),
            "mx_map")
        return C

    def dgemv(self, A, x):
        """y = A x (BlockMatrix.multiply(DistributedVector) replacement)."""
        A = np.asfortranarray(A, dtype=np.float64)
        x = np.ascontiguousarray(x, dtype=np.float64)
        m, n = A.shape
        if n != x.shape[0]:
            raise ValueError(
                f"matrix columns size {n} not support vector length "
                f"{x.shape[0]}")
        y = np.empty(m, dtype=np.float64)
        _ck(lib().mx_dgemv(self._ctx, m, n, _fbuf(A, np.float64),
                           x.ctypes.data_as(ctypes.POINTER(ctypes.c_double)),
                           y.ctypes.data_as(ctypes.POINTER(ctypes.c_double))),
            "mx_dgemv")
        return y

    def sum(self, A):
        A = np.asfortranarray(A, dtype=np.float64)
        out = ctypes.c_double()
        _ck(lib().mx_sum(self._ctx, 0, A.size,
                         A.ctypes.data_as(ctypes.c_void_p),
                         ctypes.byref(out)), "mx_sum")
        return out.value

    def transpose(self, A):
        A = np.asfortranarray(A, dtype=np.float64)
        m, n = A.shape
        C = np.empty((n, m), dtype=np.float64, order="F")
        _ck(lib().mx_transpose(self._ctx, 0, m, n,
                               A.ctypes.data_as(ctypes.c_void_p),
                               C.ctypes.data_as(ctypes.c_void_p)),
            "mx_transpose")
        return C

    def sgemm_summa(self, m, k, n, A_local, B_local):
        """fp32 distributed multiply on this rank's shards."""
        A_local = np.asfortranarray(A_local, dtype=np.float32)
        B_local = np.asfortranarray(B_local, dtype=np.float32)
        pr, pc, prow, pcol = self.grid()
        mi = slab_len(m, pr, prow)
        nj = slab_len(n, pc, pcol)
        C = np.empty((mi, nj), dtype=np.float32, order="F")
        _ck(lib().mx_sgemm_summa(self._ctx, m, k, n,
                                 _fbuf(A_local, np.float32),
                                 _fbuf(B_local, np.float32),
                                 _fbuf(C, np.float32)), "mx_sgemm_summa")
        return C

    def dgemm_summa_kres(self, m, k, n, A_local, B_local):
        """k-resident distributed multiply (config-4 layout): A_local is
        this rank's row slab with ALL k columns, B_local its col slab
        with ALL k rows; one local GEMM, zero collectives."""
        A_local = np.asfortranarray(A_local, dtype=np.float64)
        B_local = np.asfortranarray(B_local, dtype=np.float64)
        pr, pc, prow, pcol = self.grid()
        mi = slab_len(m, pr, prow)
        nj = slab_len(n, pc, pcol)
        C = np.empty((mi, nj), dtype=np.float64, order="F")
        _ck(lib().mx_dgemm_summa_kres(self._ctx, m, k, n,
                                      _fbuf(A_local, np.float64),
                                      _fbuf(B_local, np.float64),
                                      _fbuf(C, np.float64)),
            "mx_dgemm_summa_kres")
        return C

    def gemm_summa_kres_device(self, m, k, n, dA, dB, dC, fp32=False):
        _ck(lib().mx_gemm_summa_kres_device(self._ctx, 1 if fp32 else 0,
                                            m, k, n, dA, dB, dC),
            "mx_gemm_summa_kres_device")

    def zero_pad(self, dbuf, rows_total, cols_total, ld, m, n, fp32=False):
        _ck(lib().mx_zero_pad(self._ctx, dbuf, rows_total, cols_total, ld,
                              m, n, 1 if fp32 else 0), "mx_zero_pad")

    def download_off(self, host_arr, dbuf, off_bytes, nbytes):
        _ck(lib().mx_download_off(self._ctx,
                                  host_arr.ctypes.data_as(ctypes.c_void_p),
                                  dbuf, off_bytes, nbytes), "mx_download_off")

    def download_row(self, dbuf, row, pitch_elems, ncols, fp32=False):
        """One row of a padded col-major device image."""
        dt = np.float32 if fp32 else np.float64
        elem = 4 if fp32 else 8
        out = np.empty(ncols, dtype=dt)
        _ck(lib().mx_download2d_off(self._ctx,
                                    out.ctypes.data_as(ctypes.c_void_p),
                                    dbuf, row * elem, pitch_elems, 1, ncols,
                                    elem), "mx_download2d_off")
        return out

    def sgemm_epilogue_device(self, m, k, n, dA, lda, dB, ldb, dC, ldc,
                              dAdd=None):
        _ck(lib().mx_sgemm_epilogue_device(self._ctx, m, k, n, dA, lda, dB,
                                           ldb, dC, ldc, dAdd),
            "mx_sgemm_epilogue_device")

    def test_rccl_error(self):
        """Failure-injection probe: returns the raw code (expected
        MX_ERCCL = -6 for the injected invalid collective)."""
        return lib().mx_test_rccl_error(self._ctx)

    def stats(self):
        st = MxStats()
        _ck(lib().mx_stats(self._ctx, ctypes.byref(st)))
        return {f: getattr(st, f) for f, _ in MxStats._fields_}

    # -- device-resident matrices (RDD.cache() analog) ----------------------
    def upload_matrix(self, a, fp32=False):
        dt = np.float32 if fp32 else np.float64
        a = np.asfortranarray(a, dtype=dt)
        m, n = a.shape
        pitch = (m + 127) // 128 * 128
        # pad columns to 128 so the matrix can stand as the GEMM rhs
        # (n-dim padded to the 128 tile) as well as lhs (k padded to 16);
        # pad region zeroed -> padded-dim GEMM stays exact
        ncap = (n + 127) // 128 * 128
        elem = 4 if fp32 else 8
        buf = self.alloc(pitch * ncap * elem)
        _ck(lib().mx_memset(self._ctx, buf, pitch * ncap * elem))
        _ck(lib().mx_upload2d(self._ctx, buf, pitch,
                              a.ctypes.data_as(ctypes.c_void_p), m, n, elem),
            "mx_upload2d")
        return DeviceMatrix(self, buf, m, n, pitch, fp32)

    def download_matrix(self, dm):
        dt = np.float32 if dm.fp32 else np.float64
        out = np.empty((dm.m, dm.n), dtype=dt, order="F")
        _ck(lib().mx_download2d(self._ctx, out.ctypes.data_as(ctypes.c_void_p),
                                dm.buf, dm.pitch, dm.m, dm.n, dm.elem),
            "mx_download2d")
        return out

    def dgemv_dd(self, dm, x):
        """y = A x with A device-resident (pads are zero -> exact)."""
        x = np.ascontiguousarray(x, dtype=np.float64)
        if dm.n != x.shape[0]:
            raise ValueError("dimension mismatch")
        # run over padded rows/cols; x padded with zeros to the pitch cols
        ncap = ((dm.n + 127) // 128 * 128)
        xp = np.zeros(ncap, dtype=np.float64)
        xp[:dm.n] = x
        y = np.empty(dm.pitch, dtype=np.float64)
        _ck(lib().mx_dgemv_device(self._ctx, dm.pitch, ncap, dm.buf,
                                  dm.pitch,
                                  xp.ctypes.data_as(ctypes.POINTER(
                                      ctypes.c_double)),
                                  y.ctypes.data_as(ctypes.POINTER(
                                      ctypes.c_double))),
            "mx_dgemv_device")
        return y[:dm.m].copy()

    def dgemv_device_raw(self, m, n, dbuf, lda, x):
        """y = A x on a raw device buffer (fp64, pitch lda): the
        whole-matrix probe primitive for full-size parity tests."""
        x = np.ascontiguousarray(x, dtype=np.float64)
        assert x.shape[0] == n
        y = np.empty(m, dtype=np.float64)
        _ck(lib().mx_dgemv_device(self._ctx, m, n, dbuf, lda,
                                  x.ctypes.data_as(ctypes.POINTER(
                                      ctypes.c_double)),
                                  y.ctypes.data_as(ctypes.POINTER(
                                      ctypes.c_double))),
            "mx_dgemv_device")
        return y

    def transpose_dd(self, dm):
        """Device-resident transpose: returns a GEMM-ready DeviceMatrix
        (the padded image transposes wholesale; zero pads stay zero)."""
        ncap = ((dm.n + 127) // 128 * 128)
        out = self.alloc(ncap * dm.pitch * dm.elem)
        _ck(lib().mx_transpose_device(self._ctx, 1 if dm.fp32 else 0,
                                      dm.pitch, ncap, dm.buf, out),
            "mx_transpose_device")
        return DeviceMatrix(self, out, dm.n, dm.m, ncap, dm.fp32)

    def gemm_dd(self, A, B, C=None, accumulate=False):
        """Device-resident C (+)= A @ B on cached matrices. Pads are
        zero, so padded-dim GEMM is exact; result C is a DeviceMatrix
        with logical (A.m, B.n)."""
        assert A.fp32 == B.fp32
        if A.n != B.m:
            raise ValueError(
                f"Dimension mismatch during matrix-matrix multiplication: "
                f"{A.n} vs {B.m}")
        kp = (A.n + 15) // 16 * 16
        np_ = (B.n + 127) // 128 * 128
        if C is None:
            assert not accumulate
            cbuf = self.alloc(A.pitch * np_ * A.elem)
            C = DeviceMatrix(self, cbuf, A.m, B.n, A.pitch, A.fp32)
        assert C.pitch == A.pitch and C.m == A.m and C.n == B.n
        # B must expose kp padded rows: its pitch is >= kp iff B.m pads ok
        assert B.pitch >= kp
        _ck(lib().mx_gemm_device_ex(self._ctx, 1 if A.fp32 else 0,
                                    1 if accumulate else 0,
                                    A.pitch, kp, np_, A.buf, A.pitch,
                                    B.buf, B.pitch, C.buf, C.pitch),
            "mx_gemm_device_ex")
        return C

    # -- device-resident bench path ------------------------------------------
    def alloc(self, nbytes):
        b = ctypes.c_void_p()
        _ck(lib().mx_alloc(self._ctx, nbytes, ctypes.byref(b)), "mx_alloc")
        return b

    def free(self, b):
        _ck(lib().mx_free(self._ctx, b))

    def fill_random(self, dbuf, n_elems, seed, fp32=False):
        _ck(lib().mx_fill_random(self._ctx, dbuf, n_elems, seed,
                                 1 if fp32 else 0), "mx_fill_random")

    def dgemm_device(self, m, k, n, dA, lda, dB, ldb, dC, ldc):
        _ck(lib().mx_dgemm_device(self._ctx, m, k, n, dA, lda, dB, ldb, dC,
                                  ldc), "mx_dgemm_device")

    def sgemm_device(self, m, k, n, dA, lda, dB, ldb, dC, ldc):
        _ck(lib().mx_sgemm_device(self._ctx, m, k, n, dA, lda, dB, ldb, dC,
                                  ldc), "mx_sgemm_device")

    def dgemm_summa_device(self, m, k, n, dA, dB, dC):
        _ck(lib().mx_dgemm_summa_device(self._ctx, m, k, n, dA, dB, dC),
            "mx_dgemm_summa_device")

    def sgemm_summa_device(self, m, k, n, dA, dB, dC):
        _ck(lib().mx_sgemm_summa_device(self._ctx, m, k, n, dA, dB, dC),
            "mx_sgemm_summa_device")

    def download(self, host_arr, dbuf, nbytes):
        _ck(lib().mx_download(self._ctx,
                              host_arr.ctypes.data_as(ctypes.c_void_p), dbuf,
                              nbytes), "mx_download")

    def upload(self, dbuf, host_arr, nbytes):
        _ck(lib().mx_upload(self._ctx, dbuf,
                            host_arr.ctypes.data_as(ctypes.c_void_p), nbytes),
            "mx_upload")
