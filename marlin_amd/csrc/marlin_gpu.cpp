// marlin_gpu.cpp — host runtime of the MI355X-native block-matrix multiply
// engine behind the C ABI of include/marlin_gpu.h.
//
// Replaces, MI355X-first, the reference's distributed machinery:
//   - Spark shuffle emit/join/reduce (BlockMatrix.scala:161-186,
//     DenseVecMatrix.scala:109-141) -> 2-D grid SUMMA with RCCL panel
//     broadcasts over xGMI, double-buffered against the MFMA stream.
//     Each C shard has ONE owner, so the reference's reduceByKey add
//     (SubMatrix.scala:41-50) disappears into the GEMM accumulator.
//   - MatrixMultPartitioner (MatrixMultPartitioner.scala:6-33) -> the
//     pr x pc rank grid + ceil slab ownership (mx_slab_*).
//   - MTUtils.evaluate timing (MTUtils.scala:218-220) -> hipEvent stage
//     timers surfaced through mx_stats.
//
// One process per GPU; mx_ctx is externally synchronized.

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <vector>
#include <algorithm>
#include <utility>

#include "../../include/marlin_gpu.h"

// kernels.hip launchers
extern "C" {
int mxk_gemm(int is_fp32, int beta_one, int64_t M, int64_t N, int64_t K,
             const void* A, int64_t lda, const void* B, int64_t ldb,
             void* C, int64_t ldc, hipStream_t stream);
int mxk_sgemm_tn_epilogue(int64_t M, int64_t N, int64_t K, const float* A,
                          int64_t lda, const float* B, int64_t ldb, float* C,
                          int64_t ldc, const float* addC, hipStream_t stream);
int mxk_fill_random(int is_fp32, void* buf, int64_t m, int64_t n, int64_t ld,
                    uint64_t seed, hipStream_t stream);
int mxk_zero_pad(int is_fp32, void* buf, int64_t rows_total,
                 int64_t cols_total, int64_t ld, int64_t m, int64_t n,
                 hipStream_t stream);
int mxk_map(int is_fp32, int op, int64_t n, const void* a, const void* b,
            double scalar, void* c, hipStream_t stream);
int mxk_sum(int is_fp32, int64_t n, const void* a, double* partials,
            double* out, hipStream_t stream);
int mxk_transpose(int is_fp32, int64_t m, int64_t n, const void* in,
                  void* out, hipStream_t stream);
int mxk_gemv(int is_fp32, int64_t m, int64_t n, int64_t lda, const void* A,
             const void* x, double* partial, void* y, hipStream_t stream);
}

#define HIP_OK(x)                                                        \
  do {                                                                   \
    hipError_t _e = (x);                                                 \
    if (_e != hipSuccess) {                                              \
      fprintf(stderr, "[marlin_gpu] HIP error %s at %s:%d\n",            \
              hipGetErrorString(_e), __FILE__, __LINE__);                \
      return MX_EHIP;                                                    \
    }                                                                    \
  } while (0)

#define RCCL_OK(x)                                                       \
  do {                                                                   \
    ncclResult_t _e = (x);                                               \
    if (_e != ncclSuccess) {                                             \
      fprintf(stderr, "[marlin_gpu] RCCL error %s at %s:%d\n",           \
              ncclGetErrorString(_e), __FILE__, __LINE__);               \
      return MX_ERCCL;                                                   \
    }                                                                    \
  } while (0)

static inline int64_t round_up(int64_t x, int64_t a) {
  return (x + a - 1) / a * a;
}

// ceil slab split — DenseVecMatrix.scala:1262-1265 blocking semantics
int64_t mx_slab_len(int64_t total, int parts, int idx) {
  int64_t bl = (total + parts - 1) / parts;
  int64_t start = (int64_t)idx * bl;
  if (start >= total) return 0;
  return bl < total - start ? bl : total - start;
}
int64_t mx_slab_off(int64_t total, int parts, int idx) {
  int64_t bl = (total + parts - 1) / parts;
  return (int64_t)idx * bl;
}

struct mx_dbuf {
  void* ptr = nullptr;
  int64_t bytes = 0;
};

struct mx_ctx {
  int device = 0;
  hipStream_t s_gemm = nullptr;
  hipStream_t s_copy = nullptr;
  hipStream_t s_comm = nullptr;
  hipEvent_t ev[16] = {};
  // distributed
  int rank = 0, nranks = 1;
  int pr = 1, pc = 1, prow = 0, pcol = 0;
  ncclComm_t world = nullptr, rowc = nullptr, colc = nullptr;
  bool have_comm = false;
  // cached workspaces for the host-buffer entries
  mx_dbuf wsA, wsB, wsC, wsPA[2], wsPB[2];
  mx_stats_t st = {};
};

static int ensure(mx_ctx* c, mx_dbuf* b, int64_t bytes) {
  if (b->bytes >= bytes) return MX_OK;
  if (b->ptr) (void)hipFree(b->ptr);
  b->ptr = nullptr;
  b->bytes = 0;
  hipError_t e = hipMalloc(&b->ptr, (size_t)bytes);
  if (e != hipSuccess) return MX_ENOMEM;
  b->bytes = bytes;
  return MX_OK;
}

const char* mx_strerror(int code) {
  switch (code) {
    case MX_OK: return "ok";
    case MX_EDIM: return "dimension mismatch during matrix-matrix multiplication";
    case MX_EHIP: return "HIP runtime failure";
    case MX_ENOMEM: return "device allocation failure";
    case MX_EINVAL: return "invalid argument";
    case MX_ENOCOMM: return "distributed entry before mx_comm_init";
    case MX_ERCCL: return "RCCL failure";
    case MX_ENODEV: return "no GPU device visible";
    default: return "unknown error";
  }
}

// destroys whatever streams/events a partially-initialised ctx created
static void ctx_teardown(mx_ctx* c) {
  for (int i = 0; i < 16; i++)
    if (c->ev[i]) (void)hipEventDestroy(c->ev[i]);
  if (c->s_gemm) (void)hipStreamDestroy(c->s_gemm);
  if (c->s_copy) (void)hipStreamDestroy(c->s_copy);
  if (c->s_comm) (void)hipStreamDestroy(c->s_comm);
  delete c;
}

int mx_init(mx_ctx** out, int device) {
  if (!out) return MX_EINVAL;
  int ndev = 0;
  if (hipGetDeviceCount(&ndev) != hipSuccess || ndev <= 0) return MX_ENODEV;
  mx_ctx* c = new mx_ctx();
  c->device = device < 0 ? 0 : device;
  if (hipSetDevice(c->device) != hipSuccess) { delete c; return MX_ENODEV; }
  if (hipStreamCreate(&c->s_gemm) != hipSuccess ||
      hipStreamCreate(&c->s_copy) != hipSuccess ||
      hipStreamCreate(&c->s_comm) != hipSuccess) {
    ctx_teardown(c);
    return MX_EHIP;
  }
  for (int i = 0; i < 16; i++)
    if (hipEventCreate(&c->ev[i]) != hipSuccess) { ctx_teardown(c); return MX_EHIP; }
  *out = c;
  return MX_OK;
}

int mx_shutdown(mx_ctx* c) {
  if (!c) return MX_EINVAL;
  (void)hipSetDevice(c->device);
  (void)hipDeviceSynchronize();
  if (c->rowc && c->rowc != c->world) ncclCommDestroy(c->rowc);
  if (c->colc && c->colc != c->world) ncclCommDestroy(c->colc);
  if (c->world) ncclCommDestroy(c->world);
  for (mx_dbuf* b : {&c->wsA, &c->wsB, &c->wsC,
                     &c->wsPA[0], &c->wsPA[1], &c->wsPB[0], &c->wsPB[1]})
    if (b->ptr) (void)hipFree(b->ptr);
  for (int i = 0; i < 16; i++)
    if (c->ev[i]) (void)hipEventDestroy(c->ev[i]);
  if (c->s_gemm) (void)hipStreamDestroy(c->s_gemm);
  if (c->s_copy) (void)hipStreamDestroy(c->s_copy);
  if (c->s_comm) (void)hipStreamDestroy(c->s_comm);
  delete c;
  return MX_OK;
}

int mx_comm_id(char unique_id[MX_UNIQUE_ID_BYTES]) {
  static_assert(sizeof(ncclUniqueId) <= MX_UNIQUE_ID_BYTES, "id size");
  ncclUniqueId id;
  RCCL_OK(ncclGetUniqueId(&id));
  memset(unique_id, 0, MX_UNIQUE_ID_BYTES);
  memcpy(unique_id, &id, sizeof(id));
  return MX_OK;
}

static void grid_shape(int nranks, int* pr, int* pc) {
  // MARLIN_GRID=RxC overrides (e.g. 2x4 for the 8-GPU grid sweep)
  const char* g = getenv("MARLIN_GRID");
  if (g) {
    int r = 0, cc = 0;
    if (sscanf(g, "%dx%d", &r, &cc) == 2 && r > 0 && cc > 0 &&
        r * cc == nranks) {
      *pr = r;
      *pc = cc;
      return;
    }
  }
  switch (nranks) {
    case 8: *pr = 4; *pc = 2; break;
    case 4: *pr = 2; *pc = 2; break;
    case 2: *pr = 2; *pc = 1; break;
    default: *pr = nranks; *pc = 1; break;
  }
}

int mx_comm_init(mx_ctx* c, int rank, int nranks,
                 const char unique_id[MX_UNIQUE_ID_BYTES]) {
  if (!c || rank < 0 || nranks <= 0 || rank >= nranks) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  ncclUniqueId id;
  memcpy(&id, unique_id, sizeof(id));
  RCCL_OK(ncclCommInitRank(&c->world, nranks, id, rank));
  c->rank = rank;
  c->nranks = nranks;
  grid_shape(nranks, &c->pr, &c->pc);
  c->prow = rank / c->pc;
  c->pcol = rank % c->pc;
  if (nranks > 1) {
    RCCL_OK(ncclCommSplit(c->world, c->prow, c->pcol, &c->rowc, nullptr));
    RCCL_OK(ncclCommSplit(c->world, c->pcol, c->prow, &c->colc, nullptr));
  } else {
    c->rowc = c->world;
    c->colc = c->world;
  }
  c->have_comm = true;
  return MX_OK;
}

// Device facts for peak computation (confirm the fp64 MFMA
// peak from CU count x clock on the actual box, not just the 78.6 spec).
int mx_device_info(mx_ctx* c, int* cus, int* clock_khz) {
  if (!c) return MX_EINVAL;
  hipDeviceProp_t prop;
  HIP_OK(hipGetDeviceProperties(&prop, c->device));
  if (cus) *cus = prop.multiProcessorCount;
  if (clock_khz) *clock_khz = prop.clockRate;
  return MX_OK;
}

int mx_grid(mx_ctx* c, int* pr, int* pc, int* prow, int* pcol) {
  if (!c) return MX_EINVAL;
  if (pr) *pr = c->pr;
  if (pc) *pc = c->pc;
  if (prow) *prow = c->prow;
  if (pcol) *pcol = c->pcol;
  return MX_OK;
}

// ---------------------------------------------------------------------------
// device buffer helpers
int mx_alloc(mx_ctx* c, int64_t bytes, mx_dbuf** out) {
  if (!c || !out || bytes <= 0) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  mx_dbuf* b = new mx_dbuf();
  int rc = ensure(c, b, bytes);
  if (rc != MX_OK) { delete b; return rc; }
  *out = b;
  return MX_OK;
}
int mx_free(mx_ctx* c, mx_dbuf* b) {
  if (!c || !b) return MX_EINVAL;
  if (b->ptr) (void)hipFree(b->ptr);
  delete b;
  return MX_OK;
}
int mx_upload(mx_ctx* c, mx_dbuf* dst, const void* src, int64_t bytes) {
  if (!c || !dst || !src || bytes > dst->bytes) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  HIP_OK(hipMemcpy(dst->ptr, src, (size_t)bytes, hipMemcpyHostToDevice));
  return MX_OK;
}
int mx_download(mx_ctx* c, void* dst, const mx_dbuf* src, int64_t bytes) {
  if (!c || !dst || !src || bytes > src->bytes) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  HIP_OK(hipMemcpy(dst, src->ptr, (size_t)bytes, hipMemcpyDeviceToHost));
  return MX_OK;
}
int mx_fill_random(mx_ctx* c, mx_dbuf* buf, int64_t n_elems, uint64_t seed,
                   int is_fp32) {
  if (!c || !buf) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  int rc = mxk_fill_random(is_fp32, buf->ptr, n_elems, 1, n_elems, seed,
                           c->s_gemm);
  if (rc) return rc;
  HIP_OK(hipStreamSynchronize(c->s_gemm));
  return MX_OK;
}

// Zero the pad region of an m x n logical image inside a rows_total x
// cols_total padded buffer (pitch ld) — restores the zero-pad invariant
// after a whole-buffer fill.
int mx_zero_pad(mx_ctx* c, mx_dbuf* buf, int64_t rows_total,
                int64_t cols_total, int64_t ld, int64_t m, int64_t n,
                int is_fp32) {
  if (!c || !buf) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  if (mxk_zero_pad(is_fp32, buf->ptr, rows_total, cols_total, ld, m, n,
                   c->s_gemm))
    return MX_EHIP;
  HIP_OK(hipStreamSynchronize(c->s_gemm));
  return MX_OK;
}

// Download starting at a byte offset into the device buffer (e.g. one
// column of a padded col-major image: offset = j * pitch * elem).
int mx_download_off(mx_ctx* c, void* dst, const mx_dbuf* src,
                    int64_t off_bytes, int64_t bytes) {
  if (!c || !dst || !src || off_bytes < 0 || bytes <= 0 ||
      off_bytes + bytes > src->bytes)
    return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  HIP_OK(hipMemcpy(dst, (const char*)src->ptr + off_bytes, (size_t)bytes,
                   hipMemcpyDeviceToHost));
  return MX_OK;
}

// 2D pitched download starting at a byte offset (e.g. one ROW of a
// col-major image: offset = i * elem, m = 1, n = cols, pitch = ld).
int mx_download2d_off(mx_ctx* c, void* dst, const mx_dbuf* src,
                      int64_t off_bytes, int64_t pitch_elems, int64_t m,
                      int64_t n, int elem) {
  if (!c || !dst || !src || m <= 0 || n <= 0 || off_bytes < 0)
    return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  HIP_OK(hipMemcpy2D(dst, (size_t)(m * elem),
                     (const char*)src->ptr + off_bytes,
                     (size_t)(pitch_elems * elem), (size_t)(m * elem),
                     (size_t)n, hipMemcpyDeviceToHost));
  return MX_OK;
}

// ---------------------------------------------------------------------------
// device-resident GEMM (padded pitches required)
static int gemm_device(mx_ctx* c, int is_fp32, int beta_one, int64_t m,
                       int64_t k, int64_t n, const void* dA, int64_t lda,
                       const void* dB, int64_t ldb, void* dC, int64_t ldc) {
  if (m % 128 || n % 128 || k % 16) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  HIP_OK(hipEventRecord(c->ev[0], c->s_gemm));
  int rc = mxk_gemm(is_fp32, beta_one, m, n, k, dA, lda, dB, ldb, dC, ldc,
                    c->s_gemm);
  if (rc) return rc == -4 ? MX_EINVAL : MX_EHIP;
  HIP_OK(hipEventRecord(c->ev[1], c->s_gemm));
  HIP_OK(hipEventSynchronize(c->ev[1]));
  float ms = 0;
  HIP_OK(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
  c->st.gemm_ms += ms;
  c->st.gemm_launches += 1;
  return MX_OK;
}

int mx_dgemm_device(mx_ctx* c, int64_t m, int64_t k, int64_t n,
                    const mx_dbuf* dA, int64_t lda, const mx_dbuf* dB,
                    int64_t ldb, mx_dbuf* dC, int64_t ldc) {
  if (!c || !dA || !dB || !dC) return MX_EINVAL;
  c->st = {};
  c->st.flops = 2.0 * m * k * n;
  c->st.bytes_moved = 8.0 * (m * k + k * n + m * n);
  return gemm_device(c, 0, 0, m, k, n, dA->ptr, lda, dB->ptr, ldb, dC->ptr,
                     ldc);
}
int mx_sgemm_device(mx_ctx* c, int64_t m, int64_t k, int64_t n,
                    const mx_dbuf* dA, int64_t lda, const mx_dbuf* dB,
                    int64_t ldb, mx_dbuf* dC, int64_t ldc) {
  if (!c || !dA || !dB || !dC) return MX_EINVAL;
  c->st = {};
  c->st.flops = 2.0 * m * k * n;
  c->st.bytes_moved = 4.0 * (m * k + k * n + m * n);
  return gemm_device(c, 1, 0, m, k, n, dA->ptr, lda, dB->ptr, ldb, dC->ptr,
                     ldc);
}

// 2D pitched transfers for device-resident matrices (the RDD.cache()
// analog: DenseVecMatrix.cache(), DenseVecMatrix.scala:321,505 usage).
int mx_upload2d(mx_ctx* c, mx_dbuf* dst, int64_t pitch_elems,
                const void* src, int64_t m, int64_t n, int elem) {
  if (!c || !dst || !src || m <= 0 || n <= 0) return MX_EINVAL;
  if (pitch_elems * (n - 1) + m > dst->bytes / elem) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  HIP_OK(hipMemcpy2D(dst->ptr, (size_t)(pitch_elems * elem), src,
                     (size_t)(m * elem), (size_t)(m * elem), (size_t)n,
                     hipMemcpyHostToDevice));
  return MX_OK;
}
int mx_download2d(mx_ctx* c, void* dst, const mx_dbuf* src,
                  int64_t pitch_elems, int64_t m, int64_t n, int elem) {
  if (!c || !dst || !src || m <= 0 || n <= 0) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  HIP_OK(hipMemcpy2D(dst, (size_t)(m * elem), src->ptr,
                     (size_t)(pitch_elems * elem), (size_t)(m * elem),
                     (size_t)n, hipMemcpyDeviceToHost));
  return MX_OK;
}
int mx_memset(mx_ctx* c, mx_dbuf* b, int64_t bytes) {
  if (!c || !b || bytes > b->bytes) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  HIP_OK(hipMemset(b->ptr, 0, (size_t)bytes));
  return MX_OK;
}
// device-resident transpose (operates on the full padded image; pads
// are zero so the transposed pads stay zero)
int mx_transpose_device(mx_ctx* c, int is_fp32, int64_t m, int64_t n,
                        const mx_dbuf* in, mx_dbuf* out) {
  if (!c || !in || !out) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  if (mxk_transpose(is_fp32, m, n, in->ptr, out->ptr, c->s_gemm))
    return MX_EHIP;
  HIP_OK(hipStreamSynchronize(c->s_gemm));
  return MX_OK;
}

// beta-capable device-resident GEMM (C += A*B when beta_one)
int mx_gemm_device_ex(mx_ctx* c, int is_fp32, int beta_one, int64_t m,
                      int64_t k, int64_t n, const mx_dbuf* dA, int64_t lda,
                      const mx_dbuf* dB, int64_t ldb, mx_dbuf* dC,
                      int64_t ldc) {
  if (!c || !dA || !dB || !dC) return MX_EINVAL;
  c->st = {};
  c->st.flops = 2.0 * m * k * n;
  return gemm_device(c, is_fp32, beta_one, m, k, n, dA->ptr, lda, dB->ptr,
                     ldb, dC->ptr, ldc);
}

// ---------------------------------------------------------------------------
// host-buffer whole-multiply entries: pad -> H2D -> kernel -> D2H.
// elem = 8 (fp64) or 4 (fp32).
static int gemm_host(mx_ctx* c, int is_fp32, int beta_one, int64_t m,
                     int64_t k, int64_t n, const void* A, const void* B,
                     void* C, int transpose_c, const void* addC) {
  if (!c || !A || !B || !C) return MX_EINVAL;
  if (m <= 0 || k <= 0 || n <= 0) return MX_EDIM;
  const int64_t elem = is_fp32 ? 4 : 8;
  const int64_t mp = round_up(m, 128), np = round_up(n, 128),
                kp = round_up(k, 16);
  c->st = {};
  c->st.flops = 2.0 * m * k * n;
  c->st.bytes_moved = (double)elem * (m * k + k * n + m * n);

  HIP_OK(hipSetDevice(c->device));
  int rc;
  if ((rc = ensure(c, &c->wsA, mp * kp * elem))) return rc;
  if ((rc = ensure(c, &c->wsB, kp * np * elem))) return rc;
  if ((rc = ensure(c, &c->wsC, mp * np * elem))) return rc;

  HIP_OK(hipEventRecord(c->ev[2], c->s_copy));
  if (m != mp || k != kp) HIP_OK(hipMemsetAsync(c->wsA.ptr, 0, mp * kp * elem, c->s_copy));
  if (k != kp || n != np) HIP_OK(hipMemsetAsync(c->wsB.ptr, 0, kp * np * elem, c->s_copy));
  HIP_OK(hipMemcpy2DAsync(c->wsA.ptr, mp * elem, A, m * elem, m * elem, k,
                          hipMemcpyHostToDevice, c->s_copy));
  HIP_OK(hipMemcpy2DAsync(c->wsB.ptr, kp * elem, B, k * elem, k * elem, n,
                          hipMemcpyHostToDevice, c->s_copy));
  if (beta_one) {
    if (m != mp || n != np) HIP_OK(hipMemsetAsync(c->wsC.ptr, 0, mp * np * elem, c->s_copy));
    HIP_OK(hipMemcpy2DAsync(c->wsC.ptr, mp * elem, C, m * elem, m * elem, n,
                            hipMemcpyHostToDevice, c->s_copy));
  }
  HIP_OK(hipEventRecord(c->ev[3], c->s_copy));
  HIP_OK(hipStreamWaitEvent(c->s_gemm, c->ev[3], 0));

  if (transpose_c) {
    // fused transpose(+add) epilogue path (fp32 only): addC is N x M on
    // host; stage it padded on device (reuse wsC tail? keep simple: own buf)
    if (!is_fp32) return MX_EINVAL;
    const float* dAdd = nullptr;
    if (addC) {
      if ((rc = ensure(c, &c->wsPA[0], np * mp * elem))) return rc;
      if (n != np || m != mp)
        HIP_OK(hipMemsetAsync(c->wsPA[0].ptr, 0, np * mp * elem, c->s_gemm));
      HIP_OK(hipMemcpy2DAsync(c->wsPA[0].ptr, np * elem, addC, n * elem,
                              n * elem, m, hipMemcpyHostToDevice, c->s_gemm));
      dAdd = (const float*)c->wsPA[0].ptr;
    }
    HIP_OK(hipEventRecord(c->ev[0], c->s_gemm));
    rc = mxk_sgemm_tn_epilogue(mp, np, kp, (const float*)c->wsA.ptr, mp,
                               (const float*)c->wsB.ptr, kp,
                               (float*)c->wsC.ptr, np, dAdd, c->s_gemm);
    if (rc) return rc == -4 ? MX_EINVAL : MX_EHIP;
    HIP_OK(hipEventRecord(c->ev[1], c->s_gemm));
    HIP_OK(hipEventSynchronize(c->ev[1]));
    float ms = 0;
    HIP_OK(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
    c->st.gemm_ms += ms;
    c->st.gemm_launches += 1;
    // C_out is n x m
    HIP_OK(hipMemcpy2DAsync(C, n * elem, c->wsC.ptr, np * elem, n * elem, m,
                            hipMemcpyDeviceToHost, c->s_gemm));
    HIP_OK(hipStreamSynchronize(c->s_gemm));
  } else {
    if ((rc = gemm_device(c, is_fp32, beta_one, mp, kp, np, c->wsA.ptr, mp,
                          c->wsB.ptr, kp, c->wsC.ptr, mp)))
      return rc;
    HIP_OK(hipEventRecord(c->ev[4], c->s_gemm));
    HIP_OK(hipStreamWaitEvent(c->s_copy, c->ev[4], 0));
    HIP_OK(hipMemcpy2DAsync(C, m * elem, c->wsC.ptr, mp * elem, m * elem, n,
                            hipMemcpyDeviceToHost, c->s_copy));
    HIP_OK(hipStreamSynchronize(c->s_copy));
  }
  float h2d = 0;
  HIP_OK(hipEventElapsedTime(&h2d, c->ev[2], c->ev[3]));
  c->st.h2d_ms = h2d;
  c->st.total_ms = c->st.h2d_ms + c->st.gemm_ms;  // d2h folded into total sync
  return MX_OK;
}

int mx_dgemm(mx_ctx* c, int64_t m, int64_t k, int64_t n, const double* A,
             const double* B, double* C) {
  return gemm_host(c, 0, 0, m, k, n, A, B, C, 0, nullptr);
}
int mx_sgemm(mx_ctx* c, int64_t m, int64_t k, int64_t n, const float* A,
             const float* B, float* C) {
  return gemm_host(c, 1, 0, m, k, n, A, B, C, 0, nullptr);
}
int mx_sgemm_epilogue(mx_ctx* c, int64_t m, int64_t k, int64_t n,
                      const float* A, const float* B, float* C,
                      int transpose_c, const float* add_c) {
  if (!transpose_c && add_c) return MX_EINVAL;  // plain add: use mx_sgemm+axpy later
  if (!transpose_c) return gemm_host(c, 1, 0, m, k, n, A, B, C, 0, nullptr);
  return gemm_host(c, 1, 0, m, k, n, A, B, C, 1, add_c);
}
int mx_tile_dgemm_acc(mx_ctx* c, int64_t tm, int64_t tk, int64_t tn,
                      const double* hA, const double* hB, double* hC,
                      int beta_one) {
  return gemm_host(c, 0, beta_one, tm, tk, tn, hA, hB, hC, 0, nullptr);
}

// ---------------------------------------------------------------------------
// SUMMA — the Spark-shuffle replacement (BlockMatrix.scala:161-186).
//
// Layout on the pr x pc grid (ceil slabs, reference blocking semantics):
//   A_local at rank (i,j): rows slab i of M  x  k-cols slab j of K
//   B_local at rank (i,j): k-rows slab i of K x  n-cols slab j of N
//   C_local at rank (i,j): rows slab i of M  x  n-cols slab j of N
// Per k-panel: the owning column broadcasts its A panel within each grid
// row; the owning row broadcasts its B panel within each grid column;
// every rank runs a local MFMA GEMM accumulate. One C owner per shard ->
// no reduce. Panels are packed (and k-padded to 16) on the comm stream,
// double-buffered against the GEMM stream.

struct panel_t {
  int64_t k0, k1;  // global k range
  int rootA;       // grid column owning the A panel (key in row comm)
  int rootB;       // grid row owning the B panel (key in col comm)
};

static std::vector<panel_t> plan_panels(int64_t K, int pr, int pc,
                                        int64_t kb_max) {
  std::vector<int64_t> cuts = {0, K};
  for (int j = 1; j < pc; j++) {
    int64_t o = mx_slab_off(K, pc, j);
    if (o < K) cuts.push_back(o);
  }
  for (int i = 1; i < pr; i++) {
    int64_t o = mx_slab_off(K, pr, i);
    if (o < K) cuts.push_back(o);
  }
  std::sort(cuts.begin(), cuts.end());
  cuts.erase(std::unique(cuts.begin(), cuts.end()), cuts.end());
  std::vector<panel_t> out;
  for (size_t s = 0; s + 1 < cuts.size(); s++) {
    for (int64_t k0 = cuts[s]; k0 < cuts[s + 1]; k0 += kb_max) {
      panel_t p;
      p.k0 = k0;
      p.k1 = std::min(k0 + kb_max, cuts[s + 1]);
      int64_t blA = (K + pc - 1) / pc;
      int64_t blB = (K + pr - 1) / pr;
      p.rootA = (int)(k0 / blA);
      p.rootB = (int)(k0 / blB);
      out.push_back(p);
    }
  }
  return out;
}

// C-visible for CPU tests (panel plan correctness without a GPU).
extern "C" int mx_plan_panels(int64_t K, int pr, int pc, int64_t kb_max,
                              int64_t* k0s, int64_t* k1s, int* rootsA,
                              int* rootsB, int cap) {
  auto v = plan_panels(K, pr, pc, kb_max);
  if ((int)v.size() > cap) return -(int)v.size();
  for (size_t i = 0; i < v.size(); i++) {
    k0s[i] = v[i].k0;
    k1s[i] = v[i].k1;
    rootsA[i] = v[i].rootA;
    rootsB[i] = v[i].rootB;
  }
  return (int)v.size();
}

// RAII pool for the per-panel timing events: destroyed on EVERY exit path
// (early error returns included), so failed SUMMA calls leak nothing.
struct ev_pool {
  std::vector<hipEvent_t> v;
  hipEvent_t mk() {
    hipEvent_t e = nullptr;
    (void)hipEventCreate(&e);
    v.push_back(e);
    return e;
  }
  ~ev_pool() {
    for (hipEvent_t e : v)
      if (e) (void)hipEventDestroy(e);
  }
};

static int summa_device(mx_ctx* c, int is_fp32, int64_t m, int64_t k,
                        int64_t n, const void* dA, const void* dB, void* dC) {
  if (!c->have_comm) return MX_ENOCOMM;
  const int64_t elem = is_fp32 ? 4 : 8;
  const ncclDataType_t nty = is_fp32 ? ncclFloat32 : ncclFloat64;
  const int64_t mi = mx_slab_len(m, c->pr, c->prow);
  const int64_t nj = mx_slab_len(n, c->pc, c->pcol);
  const int64_t kbi = mx_slab_len(k, c->pr, c->prow);   // B k-rows here
  const int64_t ka_off = mx_slab_off(k, c->pc, c->pcol);
  const int64_t kb_off = mx_slab_off(k, c->pr, c->prow);
  const int64_t mip = round_up(mi, 128), njp = round_up(nj, 128);
  // Empty local shard (e.g. m < pr on tiny inputs): the reference path
  // handles any size, so this rank must still take part in every panel
  // broadcast (counts are uniform per row/col comm: mip is shared by the
  // whole grid row, njp by the whole grid column) but launches no GEMM.
  const bool has_tile = mip > 0 && njp > 0;
  // local shard pitches ARE the padded sizes (bench fills them that way;
  // host entry packs them that way)
  // panel width: overlap granularity of the comm/MFMA pipeline.
  // Default 2048 (measured: the 1-rank panel-loop machinery runs at
  // 98.9% of the monolithic GEMM rate at KB=2048 vs 97.2% at 4096 —
  // profiles/r02_experiments.md; broadcasts still hide under the
  // ~3 ms per-panel GEMM at 8 GPUs). MARLIN_SUMMA_KB overrides.
  static const char* kbenv = getenv("MARLIN_SUMMA_KB");
  const int64_t kb_max = kbenv && atoll(kbenv) > 0 ? atoll(kbenv) : 2048;
  auto panels = plan_panels(k, c->pr, c->pc, kb_max);

  c->st = {};
  c->st.flops = 2.0 * m * k * n;  // whole-job flops (rank-aggregate metric)
  c->st.bytes_moved = (double)elem * (m * k + k * n + m * n);

  HIP_OK(hipSetDevice(c->device));
  int rc;
  const int64_t kbp_max = round_up(kb_max, 16);
  for (int b = 0; b < 2; b++) {
    if (mip > 0 && (rc = ensure(c, &c->wsPA[b], mip * kbp_max * elem)))
      return rc;
    if (njp > 0 && (rc = ensure(c, &c->wsPB[b], kbp_max * njp * elem)))
      return rc;
  }

  // ev[8+b]: gemm done reading panel buffer b; ev[12+b]: panel b ready
  HIP_OK(hipEventRecord(c->ev[8], c->s_gemm));
  HIP_OK(hipEventRecord(c->ev[9], c->s_gemm));

  // deferred per-panel timing: events recorded in-loop, synchronised
  // only AFTER both streams drain (an in-loop sync would serialise the
  // comm/GEMM double-buffer pipeline)
  ev_pool tev;
  std::vector<std::pair<hipEvent_t, hipEvent_t>> gemm_tv, comm_tv;

  for (size_t p = 0; p < panels.size(); p++) {
    const panel_t& pan = panels[p];
    const int buf = (int)(p & 1);
    const int64_t kb = pan.k1 - pan.k0;
    const int64_t kbp = round_up(kb, 16);
    void* pa = c->wsPA[buf].ptr;
    void* pb = c->wsPB[buf].ptr;

    // comm stream: wait until the GEMM that read this buffer 2 panels ago
    // is done, then pack (root) and broadcast.
    HIP_OK(hipStreamWaitEvent(c->s_comm, c->ev[8 + buf], 0));
    if (kbp != kb) {
      if (mip > 0)
        HIP_OK(hipMemsetAsync(pa, 0, (size_t)(mip * kbp * elem), c->s_comm));
      if (njp > 0)
        HIP_OK(hipMemsetAsync(pb, 0, (size_t)(kbp * njp * elem), c->s_comm));
    }
    if (c->pcol == pan.rootA && mip > 0) {
      // A panel: k-cols [k0-ka_off, k1-ka_off) of A_local (pitch mip) are
      // contiguous -> strided copy into packed panel (pitch mip, kb cols)
      const char* src = (const char*)dA + (pan.k0 - ka_off) * mip * elem;
      HIP_OK(hipMemcpyAsync(pa, src, (size_t)(mip * kb * elem),
                            hipMemcpyDeviceToDevice, c->s_comm));
    }
    if (c->prow == pan.rootB && njp > 0 && kbi > 0) {
      // B panel: k-rows [k0-kb_off, k1-kb_off) of B_local (pitch kbi_p):
      // strided 2D copy into packed pitch kbp
      const int64_t kbi_p = round_up(kbi, 16);
      const char* src = (const char*)dB + (pan.k0 - kb_off) * elem;
      HIP_OK(hipMemcpy2DAsync(pb, kbp * elem, src, kbi_p * elem, kb * elem,
                              nj, hipMemcpyDeviceToDevice, c->s_comm));
    }
    if (c->nranks > 1) {
      hipEvent_t c0 = tev.mk(), c1 = tev.mk();
      HIP_OK(hipEventRecord(c0, c->s_comm));
      RCCL_OK(ncclGroupStart());
      RCCL_OK(ncclBroadcast(pa, pa, (size_t)(mip * kbp), nty, pan.rootA,
                            c->rowc, c->s_comm));
      RCCL_OK(ncclBroadcast(pb, pb, (size_t)(kbp * njp), nty, pan.rootB,
                            c->colc, c->s_comm));
      RCCL_OK(ncclGroupEnd());
      HIP_OK(hipEventRecord(c1, c->s_comm));
      comm_tv.push_back({c0, c1});
    }
    HIP_OK(hipEventRecord(c->ev[12 + buf], c->s_comm));

    // gemm stream: wait for the panel, accumulate
    HIP_OK(hipStreamWaitEvent(c->s_gemm, c->ev[12 + buf], 0));
    if (has_tile) {
      int beta = p == 0 ? 0 : 1;
      hipEvent_t g0 = tev.mk(), g1 = tev.mk();
      HIP_OK(hipEventRecord(g0, c->s_gemm));
      rc = mxk_gemm(is_fp32, beta, mip, njp, kbp, pa, mip, pb, kbp, dC, mip,
                    c->s_gemm);
      if (rc) return rc == -4 ? MX_EINVAL : MX_EHIP;
      c->st.gemm_launches += 1;
      HIP_OK(hipEventRecord(g1, c->s_gemm));
      gemm_tv.push_back({g0, g1});
    }
    HIP_OK(hipEventRecord(c->ev[8 + buf], c->s_gemm));
  }
  HIP_OK(hipStreamSynchronize(c->s_gemm));
  HIP_OK(hipStreamSynchronize(c->s_comm));
  for (auto& pr2 : gemm_tv) {
    float ms = 0;
    (void)hipEventElapsedTime(&ms, pr2.first, pr2.second);
    c->st.gemm_ms += ms;
  }
  for (auto& pr2 : comm_tv) {
    float ms = 0;
    (void)hipEventElapsedTime(&ms, pr2.first, pr2.second);
    c->st.comm_ms += ms;
  }
  return MX_OK;
}

// ---------------------------------------------------------------------------
// k-resident distributed layout (BASELINE config 4).
//
// CARMA splitMethod (MTUtils.scala:150-175) halves the LARGEST of m,k,n
// per step; for tall-skinny x short-fat shapes (50000x4096 · 4096x50000 on
// 8 GPUs) it never splits k -> kSplit == 1. The MI355X layout mirrors that
// choice: each rank keeps its A row-slab with ALL K columns and its B
// col-slab with ALL K rows resident in HBM (replicated across the other
// grid dimension), so the multiply is ONE local MFMA GEMM with ZERO
// steady-state xGMI traffic — no panel broadcasts at all.
static void split_method_c(int64_t m, int64_t k, int64_t n, int cores,
                           int* ms, int* ks, int* ns) {
  // exact MTUtils.scala:150-175 semantics (n tested first, then m, else k)
  *ms = *ks = *ns = 1;
  int64_t _m = m, _k = k, _n = n;
  int c = cores;
  while (c > 1 && _m > 1 && _k > 1 && _n > 1) {
    if (_n >= _k && _n >= _m) { *ns *= 2; _n /= 2; }
    else if (_m >= _k && _m >= _n) { *ms *= 2; _m /= 2; }
    else { *ks *= 2; _k /= 2; }
    c /= 2;
  }
}

int mx_summa_kresident(int64_t m, int64_t k, int64_t n, int nranks) {
  if (nranks <= 1) return 0;
  int ms, ks, ns;
  split_method_c(m, k, n, nranks, &ms, &ks, &ns);
  return ks == 1 ? 1 : 0;
}

static int summa_kres_device(mx_ctx* c, int is_fp32, int64_t m, int64_t k,
                             int64_t n, const void* dA, const void* dB,
                             void* dC) {
  if (!c->have_comm) return MX_ENOCOMM;
  const int64_t elem = is_fp32 ? 4 : 8;
  const int64_t mi = mx_slab_len(m, c->pr, c->prow);
  const int64_t nj = mx_slab_len(n, c->pc, c->pcol);
  const int64_t mip = round_up(mi, 128), njp = round_up(nj, 128);
  const int64_t kp = round_up(k, 16);
  c->st = {};
  c->st.flops = 2.0 * m * k * n;
  c->st.bytes_moved = (double)elem * (m * k + k * n + m * n);
  if (mip == 0 || njp == 0) return MX_OK;  // empty shard: nothing to do
  HIP_OK(hipSetDevice(c->device));
  HIP_OK(hipEventRecord(c->ev[0], c->s_gemm));
  int rc = mxk_gemm(is_fp32, 0, mip, njp, kp, dA, mip, dB, kp, dC, mip,
                    c->s_gemm);
  if (rc) return rc == -4 ? MX_EINVAL : MX_EHIP;
  HIP_OK(hipEventRecord(c->ev[1], c->s_gemm));
  HIP_OK(hipEventSynchronize(c->ev[1]));
  float ms = 0;
  HIP_OK(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
  c->st.gemm_ms += ms;
  c->st.gemm_launches += 1;
  return MX_OK;
}

int mx_gemm_summa_kres_device(mx_ctx* c, int is_fp32, int64_t m, int64_t k,
                              int64_t n, const mx_dbuf* dA, const mx_dbuf* dB,
                              mx_dbuf* dC) {
  if (!c || !dA || !dB || !dC) return MX_EINVAL;
  return summa_kres_device(c, is_fp32, m, k, n, dA->ptr, dB->ptr, dC->ptr);
}

// host-buffer k-resident entry: shards are A_local mi x K, B_local K x nj
// (tight col-major); engine pads and runs the single local GEMM.
static int summa_kres_host(mx_ctx* c, int is_fp32, int64_t m, int64_t k,
                           int64_t n, const void* A, const void* B, void* C) {
  if (!c || !A || !B || !C) return MX_EINVAL;
  if (!c->have_comm) return MX_ENOCOMM;
  const int64_t elem = is_fp32 ? 4 : 8;
  const int64_t mi = mx_slab_len(m, c->pr, c->prow);
  const int64_t nj = mx_slab_len(n, c->pc, c->pcol);
  const int64_t mip = round_up(mi, 128), njp = round_up(nj, 128);
  const int64_t kp = round_up(k, 16);
  if (mi == 0 || nj == 0) return MX_OK;
  int rc;
  HIP_OK(hipSetDevice(c->device));
  if ((rc = ensure(c, &c->wsA, mip * kp * elem))) return rc;
  if ((rc = ensure(c, &c->wsB, kp * njp * elem))) return rc;
  if ((rc = ensure(c, &c->wsC, mip * njp * elem))) return rc;
  if (mi != mip || k != kp)
    HIP_OK(hipMemset(c->wsA.ptr, 0, (size_t)(mip * kp * elem)));
  if (k != kp || nj != njp)
    HIP_OK(hipMemset(c->wsB.ptr, 0, (size_t)(kp * njp * elem)));
  HIP_OK(hipMemcpy2D(c->wsA.ptr, mip * elem, A, mi * elem, mi * elem, k,
                     hipMemcpyHostToDevice));
  HIP_OK(hipMemcpy2D(c->wsB.ptr, kp * elem, B, k * elem, k * elem, nj,
                     hipMemcpyHostToDevice));
  if ((rc = summa_kres_device(c, is_fp32, m, k, n, c->wsA.ptr, c->wsB.ptr,
                              c->wsC.ptr)))
    return rc;
  HIP_OK(hipMemcpy2D(C, mi * elem, c->wsC.ptr, mip * elem, mi * elem, nj,
                     hipMemcpyDeviceToHost));
  return MX_OK;
}

int mx_dgemm_summa_kres(mx_ctx* c, int64_t m, int64_t k, int64_t n,
                        const double* A, const double* B, double* C) {
  return summa_kres_host(c, 0, m, k, n, A, B, C);
}
int mx_sgemm_summa_kres(mx_ctx* c, int64_t m, int64_t k, int64_t n,
                        const float* A, const float* B, float* C) {
  return summa_kres_host(c, 1, m, k, n, A, B, C);
}

int mx_dgemm_summa_device(mx_ctx* c, int64_t m, int64_t k, int64_t n,
                          const mx_dbuf* dA, const mx_dbuf* dB, mx_dbuf* dC) {
  if (!c || !dA || !dB || !dC) return MX_EINVAL;
  return summa_device(c, 0, m, k, n, dA->ptr, dB->ptr, dC->ptr);
}
int mx_sgemm_summa_device(mx_ctx* c, int64_t m, int64_t k, int64_t n,
                          const mx_dbuf* dA, const mx_dbuf* dB, mx_dbuf* dC) {
  if (!c || !dA || !dB || !dC) return MX_EINVAL;
  return summa_device(c, 1, m, k, n, dA->ptr, dB->ptr, dC->ptr);
}

// host-buffer SUMMA entries: pack local shards padded, H2D, run, D2H
static int summa_host(mx_ctx* c, int is_fp32, int64_t m, int64_t k, int64_t n,
                      const void* A, const void* B, void* C) {
  if (!c || !A || !B || !C) return MX_EINVAL;
  if (!c->have_comm) return MX_ENOCOMM;
  const int64_t elem = is_fp32 ? 4 : 8;
  const int64_t mi = mx_slab_len(m, c->pr, c->prow);
  const int64_t nj = mx_slab_len(n, c->pc, c->pcol);
  const int64_t kaj = mx_slab_len(k, c->pc, c->pcol);
  const int64_t kbi = mx_slab_len(k, c->pr, c->prow);
  const int64_t mip = round_up(mi, 128), njp = round_up(nj, 128);
  const int64_t kbi_p = round_up(kbi, 16);
  int rc;
  HIP_OK(hipSetDevice(c->device));
  if ((rc = ensure(c, &c->wsA, mip * (kaj > 0 ? kaj : 1) * elem))) return rc;
  if ((rc = ensure(c, &c->wsB, kbi_p * (nj > 0 ? nj : 1) * elem))) return rc;
  if ((rc = ensure(c, &c->wsC, mip * njp * elem))) return rc;
  if (mi != mip && kaj > 0)
    HIP_OK(hipMemset(c->wsA.ptr, 0, (size_t)(mip * kaj * elem)));
  if (kbi != kbi_p && nj > 0)
    HIP_OK(hipMemset(c->wsB.ptr, 0, (size_t)(kbi_p * nj * elem)));
  if (mi > 0 && kaj > 0)
    HIP_OK(hipMemcpy2D(c->wsA.ptr, mip * elem, A, mi * elem, mi * elem, kaj,
                       hipMemcpyHostToDevice));
  if (kbi > 0 && nj > 0)
    HIP_OK(hipMemcpy2D(c->wsB.ptr, kbi_p * elem, B, kbi * elem, kbi * elem,
                       nj, hipMemcpyHostToDevice));
  if ((rc = summa_device(c, is_fp32, m, k, n, c->wsA.ptr, c->wsB.ptr,
                         c->wsC.ptr)))
    return rc;
  if (mi > 0 && nj > 0)
    HIP_OK(hipMemcpy2D(C, mi * elem, c->wsC.ptr, mip * elem, mi * elem, nj,
                       hipMemcpyDeviceToHost));
  return MX_OK;
}

int mx_dgemm_summa(mx_ctx* c, int64_t m, int64_t k, int64_t n,
                   const double* A, const double* B, double* C) {
  return summa_host(c, 0, m, k, n, A, B, C);
}
int mx_sgemm_summa(mx_ctx* c, int64_t m, int64_t k, int64_t n, const float* A,
                   const float* B, float* C) {
  return summa_host(c, 1, m, k, n, A, B, C);
}

// ---------------------------------------------------------------------------
// Elementwise / reduction / transpose host entries (the BlockMatrix
// epilogue-op family, BlockMatrix.scala:344-523). Host col-major buffers;
// elementwise ops are layout-agnostic (flat n elements).
int mx_map(mx_ctx* c, int op, int is_fp32, int64_t n, const void* A,
           const void* B, double scalar, void* C) {
  if (!c || !A || !C || n <= 0) return MX_EINVAL;
  const int binary = op <= 2;
  if (binary && !B) return MX_EINVAL;
  const int64_t elem = is_fp32 ? 4 : 8;
  HIP_OK(hipSetDevice(c->device));
  int rc;
  if ((rc = ensure(c, &c->wsA, n * elem))) return rc;
  if (binary && (rc = ensure(c, &c->wsB, n * elem))) return rc;
  HIP_OK(hipMemcpyAsync(c->wsA.ptr, A, n * elem, hipMemcpyHostToDevice,
                        c->s_gemm));
  if (binary)
    HIP_OK(hipMemcpyAsync(c->wsB.ptr, B, n * elem, hipMemcpyHostToDevice,
                          c->s_gemm));
  if ((rc = mxk_map(is_fp32, op, n, c->wsA.ptr, binary ? c->wsB.ptr : nullptr,
                    scalar, c->wsA.ptr, c->s_gemm)))
    return MX_EHIP;
  HIP_OK(hipMemcpyAsync(C, c->wsA.ptr, n * elem, hipMemcpyDeviceToHost,
                        c->s_gemm));
  HIP_OK(hipStreamSynchronize(c->s_gemm));
  return MX_OK;
}

int mx_sum(mx_ctx* c, int is_fp32, int64_t n, const void* A, double* out) {
  if (!c || !A || !out || n <= 0) return MX_EINVAL;
  const int64_t elem = is_fp32 ? 4 : 8;
  HIP_OK(hipSetDevice(c->device));
  int rc;
  if ((rc = ensure(c, &c->wsA, n * elem))) return rc;
  if ((rc = ensure(c, &c->wsB, (1024 + 1) * 8))) return rc;
  HIP_OK(hipMemcpyAsync(c->wsA.ptr, A, n * elem, hipMemcpyHostToDevice,
                        c->s_gemm));
  double* parts = (double*)c->wsB.ptr;
  if ((rc = mxk_sum(is_fp32, n, c->wsA.ptr, parts, parts + 1024, c->s_gemm)))
    return MX_EHIP;
  HIP_OK(hipMemcpyAsync(out, parts + 1024, 8, hipMemcpyDeviceToHost,
                        c->s_gemm));
  HIP_OK(hipStreamSynchronize(c->s_gemm));
  return MX_OK;
}

int mx_transpose(mx_ctx* c, int is_fp32, int64_t m, int64_t n, const void* A,
                 void* C) {
  if (!c || !A || !C || m <= 0 || n <= 0) return MX_EINVAL;
  const int64_t elem = is_fp32 ? 4 : 8;
  HIP_OK(hipSetDevice(c->device));
  int rc;
  if ((rc = ensure(c, &c->wsA, m * n * elem))) return rc;
  if ((rc = ensure(c, &c->wsB, m * n * elem))) return rc;
  HIP_OK(hipMemcpyAsync(c->wsA.ptr, A, m * n * elem, hipMemcpyHostToDevice,
                        c->s_gemm));
  if ((rc = mxk_transpose(is_fp32, m, n, c->wsA.ptr, c->wsB.ptr, c->s_gemm)))
    return MX_EHIP;
  HIP_OK(hipMemcpyAsync(C, c->wsB.ptr, m * n * elem, hipMemcpyDeviceToHost,
                        c->s_gemm));
  HIP_OK(hipStreamSynchronize(c->s_gemm));
  return MX_OK;
}

// Matrix-vector multiply (BlockMatrix.scala:240-274): y = A x.
int mx_dgemv(mx_ctx* c, int64_t m, int64_t n, const double* A,
             const double* x, double* y) {
  if (!c || !A || !x || !y || m <= 0 || n <= 0) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  int rc;
  if ((rc = ensure(c, &c->wsA, m * n * 8))) return rc;
  if ((rc = ensure(c, &c->wsB, (n + m) * 8))) return rc;
  if ((rc = ensure(c, &c->wsC, 32 * m * 8))) return rc;  // 32 chunk partials
  HIP_OK(hipMemcpyAsync(c->wsA.ptr, A, m * n * 8, hipMemcpyHostToDevice,
                        c->s_gemm));
  HIP_OK(hipMemcpyAsync(c->wsB.ptr, x, n * 8, hipMemcpyHostToDevice,
                        c->s_gemm));
  double* dy = (double*)c->wsB.ptr + n;
  if ((rc = mxk_gemv(0, m, n, m, c->wsA.ptr, c->wsB.ptr, (double*)c->wsC.ptr,
                     dy, c->s_gemm)))
    return MX_EHIP;
  HIP_OK(hipMemcpyAsync(y, dy, m * 8, hipMemcpyDeviceToHost, c->s_gemm));
  HIP_OK(hipStreamSynchronize(c->s_gemm));
  return MX_OK;
}

// Device-resident gemv on a cached matrix (x, y host vectors; the
// matrix never crosses PCIe). lda = the DeviceMatrix pitch.
int mx_dgemv_device(mx_ctx* c, int64_t m, int64_t n, const mx_dbuf* dA,
                    int64_t lda, const double* x, double* y) {
  if (!c || !dA || !x || !y || m <= 0 || n <= 0) return MX_EINVAL;
  HIP_OK(hipSetDevice(c->device));
  int rc;
  if ((rc = ensure(c, &c->wsB, (n + m + 32 * m) * 8))) return rc;
  double* dx = (double*)c->wsB.ptr;
  double* dy = dx + n;
  double* parts = dy + m;
  HIP_OK(hipMemcpyAsync(dx, x, n * 8, hipMemcpyHostToDevice, c->s_gemm));
  if ((rc = mxk_gemv(0, m, n, lda, dA->ptr, dx, parts, dy, c->s_gemm)))
    return MX_EHIP;
  HIP_OK(hipMemcpyAsync(y, dy, m * 8, hipMemcpyDeviceToHost, c->s_gemm));
  HIP_OK(hipStreamSynchronize(c->s_gemm));
  return MX_OK;
}

int mx_stats(mx_ctx* c, mx_stats_t* out) {
  if (!c || !out) return MX_EINVAL;
  *out = c->st;
  return MX_OK;
}

// Device-resident fused-epilogue GEMM (BASELINE config 5 timed leg):
// C[n x m] = (A*B)^T (+ addC), all buffers already in HBM, padded dims
// (m,n mult of 128, k of 16). ldc is the padded n pitch of C/addC.
int mx_sgemm_epilogue_device(mx_ctx* c, int64_t m, int64_t k, int64_t n,
                             const mx_dbuf* dA, int64_t lda,
                             const mx_dbuf* dB, int64_t ldb, mx_dbuf* dC,
                             int64_t ldc, const mx_dbuf* dAdd) {
  if (!c || !dA || !dB || !dC) return MX_EINVAL;
  if (m % 128 || n % 128 || k % 16) return MX_EINVAL;
  c->st = {};
  c->st.flops = 2.0 * m * k * n;
  c->st.bytes_moved = 4.0 * (m * k + k * n + m * n + (dAdd ? m * n : 0));
  HIP_OK(hipSetDevice(c->device));
  HIP_OK(hipEventRecord(c->ev[0], c->s_gemm));
  int rc = mxk_sgemm_tn_epilogue(m, n, k, (const float*)dA->ptr, lda,
                                 (const float*)dB->ptr, ldb, (float*)dC->ptr,
                                 ldc, dAdd ? (const float*)dAdd->ptr : nullptr,
                                 c->s_gemm);
  if (rc) return rc == -4 ? MX_EINVAL : MX_EHIP;
  HIP_OK(hipEventRecord(c->ev[1], c->s_gemm));
  HIP_OK(hipEventSynchronize(c->ev[1]));
  float ms = 0;
  HIP_OK(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
  c->st.gemm_ms += ms;
  c->st.gemm_launches += 1;
  return MX_OK;
}

// Failure-injection probe (RCCL error codes surfaced through
// the ABI, never aborts): issues an INVALID collective (null buffer) on
// the world communicator and returns the surfaced error. A healthy
// engine returns MX_ERCCL here and keeps working afterwards.
int mx_test_rccl_error(mx_ctx* c) {
  if (!c) return MX_EINVAL;
  if (!c->have_comm) return MX_ENOCOMM;
  HIP_OK(hipSetDevice(c->device));
  // out-of-range root: validated for ANY comm size (a null-buffer
  // broadcast is short-circuited on a size-1 comm and never checked)
  ncclResult_t e = ncclBroadcast(nullptr, nullptr, 16, ncclFloat64,
                                 c->nranks + 7, c->world, c->s_comm);
  if (e != ncclSuccess) return MX_ERCCL;  // expected: invalid argument
  HIP_OK(hipStreamSynchronize(c->s_comm));
  return MX_OK;
}
