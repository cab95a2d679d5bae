/* marlin_gpu.h — C ABI of the MI355X-native block-matrix multiply engine.
 *
 * This is the drop-in boundary replacing the reference's (PasaLab/marlin)
 * native hand-off point: Breeze `*` -> netlib-java JNI `dgemm` invoked at
 *   SubMatrix.scala:87-105  (per-tile dense multiply)
 *   SubMatrix.scala:41-50   (per-tile partial-sum add, reduceByKey combiner)
 * and the whole-multiply operator surface
 *   BlockMatrix.scala:149-220  (BlockMatrix.multiply)
 *   DenseVecMatrix.scala:109-141, 196-231 (split-mode multiply + dispatch).
 *
 * A JVM host (Marlin's Scala DenseVecMatrix/BlockMatrix) binds these entry
 * points over JNI (see INTEGRATION.md for the binding stub); the in-container
 * hosts are src/host/marlinx.cpp (C++ CLI) and marlin_amd/engine.py (ctypes).
 *
 * Conventions (match the reference):
 *   - All matrices are COLUMN-MAJOR fp64/fp32 (Breeze's layout,
 *     Matrices.scala:34-48), caller-owned host buffers, fully materialised.
 *   - All functions return int: 0 = ok, < 0 = error (mx_strerror).
 *   - Dimension checks mirror `require(numCols == other.numRows)`
 *     (BlockMatrix.scala:150-151) -> MX_EDIM.
 *   - An mx_ctx is externally synchronised (one call at a time — matches
 *     Spark's one-action-at-a-time driver); internally multi-stream.
 */
#ifndef MARLIN_GPU_H
#define MARLIN_GPU_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- error codes ------------------------------------------------------- */
#define MX_OK            0
#define MX_EDIM         -1   /* dimension mismatch (require(...) analogue)  */
#define MX_EHIP         -2   /* HIP runtime failure                          */
#define MX_ENOMEM       -3   /* device allocation failure                    */
#define MX_EINVAL       -4   /* bad argument (null ptr, size<=0, bad grid)   */
#define MX_ENOCOMM      -5   /* distributed entry without mx_comm_init       */
#define MX_ERCCL        -6   /* RCCL failure                                 */
#define MX_ENODEV       -7   /* no MI355X device visible                     */

const char* mx_strerror(int code);

/* ---- context ----------------------------------------------------------- */
typedef struct mx_ctx mx_ctx;

/* Create a context bound to one GPU (one process per GPU).
 * device < 0 means "current HIP device". */
int mx_init(mx_ctx** out, int device);
int mx_shutdown(mx_ctx* ctx);

/* ---- distributed setup (SUMMA over RCCL/xGMI) --------------------------- */
/* 128-byte opaque RCCL unique id, exchanged out-of-band by the launcher
 * (bench.py uses torch.distributed/gloo as the control plane). */
#define MX_UNIQUE_ID_BYTES 128
int mx_comm_id(char unique_id[MX_UNIQUE_ID_BYTES]);  /* rank 0 calls this */

/* Collective: every rank of the job calls with the same unique_id.
 * Builds the world communicator plus the pr x pc grid row/col
 * sub-communicators (grids: 8=4x2, 4=2x2, 2=2x1, 1=1x1). */
int mx_comm_init(mx_ctx* ctx, int rank, int nranks,
                 const char unique_id[MX_UNIQUE_ID_BYTES]);

/* Device facts (CU count, max clock kHz) for on-box peak computation. */
int mx_device_info(mx_ctx* ctx, int* cus, int* clock_khz);

/* Grid geometry of this rank after mx_comm_init (pr,pc,row,col). */
int mx_grid(mx_ctx* ctx, int* pr, int* pc, int* prow, int* pcol);

/* ---- whole-multiply entries (the BlockMatrix.multiply replacement) ------ */
/* Single-GPU: C = A * B, host col-major buffers; the engine does
 * pad/H2D/tiled MFMA GEMM/D2H internally. */
int mx_dgemm(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
             const double* A, const double* B, double* C);
int mx_sgemm(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
             const float* A, const float* B, float* C);

/* fp32 multiply with fused epilogue (BASELINE config 5):
 * C_out = op(A*B) [+ add_c], op = transpose if transpose_c != 0.
 * If transpose_c: C is n x m col-major, add_c (optional, may be NULL)
 * n x m; else C is m x n. Epilogue runs on-device, fused into the
 * result store — replaces BlockMatrix transpose (BlockMatrix.scala:514-523)
 * + elementwise add (BlockMatrix.scala:344-452) composed after multiply. */
int mx_sgemm_epilogue(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
                      const float* A, const float* B, float* C,
                      int transpose_c, const float* add_c);

/* Distributed SUMMA (replaces the Spark shuffle route,
 * BlockMatrix.scala:161-186): every rank passes ITS OWN shards.
 * Layout at rank (prow, pcol) of the pr x pc grid (see DESIGN.md §3;
 * ceil splits, last slab ragged — DenseVecMatrix.scala:1262-1265
 * semantics; my_x = mx_slab_len(x, parts, idx)):
 *   A_local: [my_m x my_ka]  rows slab prow of m,  k-cols slab pcol of k
 *   B_local: [my_kb x my_n]  k-rows slab prow of k, n-cols slab pcol of n
 *   C_local: [my_m x my_n]
 * Host-buffer entries take tight column-major shards; the engine pads
 * internally. Per k-panel the owning column broadcasts its A panel in
 * each grid row and the owning row its B panel in each grid column over
 * RCCL/xGMI, double-buffered against the MFMA stream; each C shard has
 * one owner, so no reduce exists. */
int mx_dgemm_summa(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
                   const double* A_local, const double* B_local,
                   double* C_local);
int mx_sgemm_summa(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
                   const float* A_local, const float* B_local,
                   float* C_local);

/* Distributed layout selector (CARMA semantics, MTUtils.scala:150-175):
 * returns 1 when splitMethod(m,k,n,nranks) leaves k unsplit (kSplit==1),
 * i.e. the k-RESIDENT layout applies — each rank keeps its A row-slab
 * with ALL K columns and its B col-slab with ALL K rows in HBM
 * (replicated across the other grid dimension) and the multiply is one
 * local GEMM with ZERO steady-state xGMI traffic (BASELINE config 4:
 * 50000x4096 · 4096x50000 on 8 GPUs). Returns 0 when the k-slabbed
 * panel-broadcast SUMMA applies. */
int mx_summa_kresident(int64_t m, int64_t k, int64_t n, int nranks);

/* k-resident distributed multiply. Shard layout at rank (prow, pcol):
 *   A_local: [my_m x K]   rows slab prow of m, ALL k columns
 *   B_local: [K x my_n]   ALL k rows, n-cols slab pcol of n
 *   C_local: [my_m x my_n]
 * Host entries take tight col-major shards; the device entry (declared
 * with the device-resident group below) takes padded pitches
 * (A roundup(my_m,128) x K cols, B roundup(K,16) x my_n,
 * C roundup(my_m,128)). No collective is issued. */
int mx_dgemm_summa_kres(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
                        const double* A_local, const double* B_local,
                        double* C_local);
int mx_sgemm_summa_kres(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
                        const float* A_local, const float* B_local,
                        float* C_local);

/* Ceil slab split helper (exact reference blocking semantics). */
int64_t mx_slab_len(int64_t total, int parts, int idx);
int64_t mx_slab_off(int64_t total, int parts, int idx);

/* ---- per-tile entries (the SubMatrix.multiply/add replacement) ----------
 * For a host that still does its own blocking (Marlin's Scala layer):
 * device-resident accumulate-GEMM per tile. beta_one=0: C=A*B;
 * beta_one=1: C+=A*B (the SubMatrix.add combiner folded in). */
int mx_tile_dgemm_acc(mx_ctx* ctx, int64_t tm, int64_t tk, int64_t tn,
                      const double* hA_tile, const double* hB_tile,
                      double* hC_tile, int beta_one);

/* ---- device-resident entries (buffers already in HBM; used by bench) ---- */
typedef struct mx_dbuf mx_dbuf;          /* opaque device buffer            */
int mx_alloc(mx_ctx* ctx, int64_t bytes, mx_dbuf** out);
int mx_free(mx_ctx* ctx, mx_dbuf* buf);
int mx_upload(mx_ctx* ctx, mx_dbuf* dst, const void* src, int64_t bytes);
int mx_download(mx_ctx* ctx, void* dst, const mx_dbuf* src, int64_t bytes);
/* Fill a device fp64 buffer with the deterministic xorshift64* U[0,1)
 * stream (the randomDenVecMatrix stand-in, MTUtils.scala:63-73 semantics:
 * seeded, per-element reproducible). */
int mx_fill_random(mx_ctx* ctx, mx_dbuf* buf, int64_t n_elems,
                   uint64_t seed, int is_fp32);
/* Device-resident GEMM on padded-pitch device buffers created by the
 * helpers above. lda/ldb/ldc are element pitches. */
int mx_dgemm_device(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
                    const mx_dbuf* dA, int64_t lda,
                    const mx_dbuf* dB, int64_t ldb,
                    mx_dbuf* dC, int64_t ldc);
int mx_sgemm_device(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
                    const mx_dbuf* dA, int64_t lda,
                    const mx_dbuf* dB, int64_t ldb,
                    mx_dbuf* dC, int64_t ldc);
/* 2D pitched transfers + zeroing for device-resident matrices (the
 * RDD.cache() analog) and a beta-capable device GEMM. */
int mx_upload2d(mx_ctx* ctx, mx_dbuf* dst, int64_t pitch_elems,
                const void* src, int64_t m, int64_t n, int elem);
int mx_download2d(mx_ctx* ctx, void* dst, const mx_dbuf* src,
                  int64_t pitch_elems, int64_t m, int64_t n, int elem);
int mx_memset(mx_ctx* ctx, mx_dbuf* buf, int64_t bytes);
int mx_transpose_device(mx_ctx* ctx, int is_fp32, int64_t m, int64_t n,
                        const mx_dbuf* in, mx_dbuf* out);
int mx_gemm_device_ex(mx_ctx* ctx, int is_fp32, int beta_one, int64_t m,
                      int64_t k, int64_t n, const mx_dbuf* dA, int64_t lda,
                      const mx_dbuf* dB, int64_t ldb, mx_dbuf* dC,
                      int64_t ldc);

/* SUMMA on device-resident local shards (bench hot loop: inputs already
 * in HBM when the timed region starts). Shard pitches are the PADDED
 * sizes: A_local pitch roundup(my_m,128) x my_ka cols; B_local pitch
 * roundup(my_kb,16) x my_n cols; C_local pitch roundup(my_m,128). */
int mx_dgemm_summa_device(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
                          const mx_dbuf* dA_local, const mx_dbuf* dB_local,
                          mx_dbuf* dC_local);
int mx_sgemm_summa_device(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
                          const mx_dbuf* dA_local, const mx_dbuf* dB_local,
                          mx_dbuf* dC_local);
/* k-resident layout on device-resident shards (see mx_summa_kresident). */
int mx_gemm_summa_kres_device(mx_ctx* ctx, int is_fp32, int64_t m, int64_t k,
                              int64_t n, const mx_dbuf* dA_local,
                              const mx_dbuf* dB_local, mx_dbuf* dC_local);

/* Restore the zero-pad invariant of a rows_total x cols_total padded
 * image (pitch ld) whose logical content is m x n: pads outside m x n
 * are zeroed (used after mx_fill_random of a whole padded buffer). */
int mx_zero_pad(mx_ctx* ctx, mx_dbuf* buf, int64_t rows_total,
                int64_t cols_total, int64_t ld, int64_t m, int64_t n,
                int is_fp32);

/* Download starting at a byte offset (one COLUMN of a padded col-major
 * image: off = j * pitch * elem). */
int mx_download_off(mx_ctx* ctx, void* dst, const mx_dbuf* src,
                    int64_t off_bytes, int64_t bytes);
/* Pitched download starting at a byte offset (one ROW of a col-major
 * image: off = i * elem, m = 1, n = cols, pitch = ld). */
int mx_download2d_off(mx_ctx* ctx, void* dst, const mx_dbuf* src,
                      int64_t off_bytes, int64_t pitch_elems, int64_t m,
                      int64_t n, int elem);

/* Device-resident fused-epilogue GEMM (config 5 timed leg): C[n x m] =
 * (A*B)^T (+ addC), padded dims, ldc = padded-n pitch of C/addC;
 * dAdd may be NULL. */
int mx_sgemm_epilogue_device(mx_ctx* ctx, int64_t m, int64_t k, int64_t n,
                             const mx_dbuf* dA, int64_t lda,
                             const mx_dbuf* dB, int64_t ldb, mx_dbuf* dC,
                             int64_t ldc, const mx_dbuf* dAdd);

/* Failure-injection probe: issues an invalid RCCL collective
 * on the world communicator; a healthy engine surfaces MX_ERCCL (no
 * abort) and keeps serving calls afterwards. */
int mx_test_rccl_error(mx_ctx* ctx);

/* ---- elementwise / reduction / transpose (BlockMatrix epilogue ops,
 * BlockMatrix.scala:344-523; DenseVecMatrix.scala scalar ops) ---------- */
#define MX_OP_ADD   0   /* C = A + B          (add(other))              */
#define MX_OP_SUB   1   /* C = A - B          (subtract(other))         */
#define MX_OP_EMUL  2   /* C = A .* B         (dotProduct(other))       */
#define MX_OP_ADDS  3   /* C = A + s          (add(b))                  */
#define MX_OP_SUBS  4   /* C = A - s          (subtract(b))             */
#define MX_OP_RSUBS 5   /* C = s - A          (subtractBy(b))           */
#define MX_OP_MULS  6   /* C = A * s          (multiply(b))             */
#define MX_OP_DIVS  7   /* C = A / s          (divide(b))               */
#define MX_OP_RDIVS 8   /* C = s / A          (divideBy(b))             */
int mx_map(mx_ctx* ctx, int op, int is_fp32, int64_t n, const void* A,
           const void* B /* null for scalar ops */, double scalar, void* C);
int mx_sum(mx_ctx* ctx, int is_fp32, int64_t n, const void* A, double* out);
int mx_transpose(mx_ctx* ctx, int is_fp32, int64_t m, int64_t n,
                 const void* A, void* C /* n x m col-major */);

/* Matrix-vector multiply (BlockMatrix.multiply(DistributedVector/BDV),
 * BlockMatrix.scala:240-274): y = A x, A col-major m x n. */
int mx_dgemv(mx_ctx* ctx, int64_t m, int64_t n, const double* A,
             const double* x, double* y);

/* Device-resident gemv on a cached matrix (mx_dbuf + pitch). */
int mx_dgemv_device(mx_ctx* ctx, int64_t m, int64_t n, const mx_dbuf* dA,
                    int64_t lda, const double* x, double* y);

/* ---- timing / stats (MTUtils.evaluate + RMMcompare.scala:47-51 analog) -- */
typedef struct {
  double h2d_ms;          /* host->device copies of the last call           */
  double d2h_ms;          /* device->host copies                            */
  double pack_ms;         /* padding/pack kernels                           */
  double gemm_ms;         /* sum of dgemm kernel time (hipEvent, gemm strm) */
  double comm_ms;         /* RCCL panel broadcast time (comm stream)        */
  double total_ms;        /* wall of the whole entry                        */
  int64_t gemm_launches;  /* number of GEMM kernel launches                 */
  double flops;           /* algorithmic flops of the last call (2mkn)      */
  double bytes_moved;     /* algorithmic HBM bytes (inputs+outputs, padded) */
} mx_stats_t;
int mx_stats(mx_ctx* ctx, mx_stats_t* out);

#ifdef __cplusplus
}
#endif
#endif /* MARLIN_GPU_H */
