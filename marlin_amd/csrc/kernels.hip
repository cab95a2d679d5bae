// kernels.hip — hand-written CDNA4 (gfx950) kernels for the MI355X-native
// block-matrix multiply engine.
//
// This is the MI355X replacement of the reference's per-block compute:
// SubMatrix.multiply (SubMatrix.scala:87-105, Breeze * -> netlib JNI dgemm)
// and the reduceByKey partial-sum add (SubMatrix.scala:41-50) — the add is
// folded into the MFMA accumulator (BETA template parameter), so the
// reference's k-partial C tiles never materialise.
//
// Design:
//   - v_mfma_f64_16x16x4_f64 / v_mfma_f32_16x16x4_f32 tiles, 64-wide waves.
//   - 128x128 block tile, BK=16 K-step, 256 threads = 4 waves in a 2x2
//     wave grid, each wave owns a 64x64 sub-tile = 4x4 MFMA fragments.
//   - A/B staged through LDS with 16-byte global_load_lds (direct HBM->LDS
//     DMA), double-buffered: raw s_barrier + counted vmcnt keeps the next
//     tile's DMA in flight across the barrier.
//   - All matrices COLUMN-MAJOR with padded leading dims (multiples of the
//     tile) — the host pads, so the kernel has no edge paths.
//   - Grid is remapped into column bands so the ~512 resident blocks share
//     A-row and B-col panels in L2 (XCD-friendly; blocks land on XCD b%8).
//
// fp64 peak on gfx950: 256 CU x 4 SIMD x 32 FLOP/clk x 2.4 GHz = 78.6 TF
// (v_mfma_f64_16x16x4_f64 = 2048 flop / 64 cyc / SIMD).

#include <hip/hip_runtime.h>
#include <stdint.h>
#include <stdlib.h>

typedef double v4d __attribute__((ext_vector_type(4)));
typedef float v4f __attribute__((ext_vector_type(4)));

#define DEVFN __device__ __forceinline__

// ---------------------------------------------------------------------------
// MFMA wrappers (16x16x4 shape; one A/B element per lane):
//   a: A[row = lane&15][k = lane>>4]      (A is m x k)
//   b: B[k = lane>>4][col = lane&15]      (B is k x n)
//   acc (4 per lane): D[row = 4*j + (lane>>4)][col = lane&15]
//     (measured on MI355X — tools_dev/debug_mfma.py; the f64/f32 16x16x4
//      "SGEMM-class" row map is reg-major, unlike the bf16 16x16x32 map)
DEVFN v4d mfma_16x16x4(double a, double b, v4d c) {
    return __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c, 0, 0, 0);
}
DEVFN v4f mfma_16x16x4(float a, float b, v4f c) {
    return __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, c, 0, 0, 0);
}

template <typename T> struct acc_t;
template <> struct acc_t<double> { using type = v4d; };
template <> struct acc_t<float>  { using type = v4f; };

DEVFN void glds16(const void* g, void* lds) {
    // 16-byte direct-to-LDS DMA; LDS dest = wave-uniform base + lane*16.
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)g,
        (__attribute__((address_space(3))) uint32_t*)lds, 16, 0, 0);
}

// ---------------------------------------------------------------------------
// Main GEMM kernel. C[MxN] (+)= A[MxK] * B[KxN], col-major, padded pitches:
// M a multiple of BM, N of 128, K of BK (the host pads to 128/16; the
// launcher only selects BK=32 / BM=256 configs when the padded dims
// divide). BETA == 1 accumulates into C (the SubMatrix.add combiner,
// folded into the MFMA accumulator chain).
//
// Geometry template: BN=128 fixed; wave grid WM x WN, wave tile
// (BM/WM) x (128/WN). Production config (measured winner): BK=16,
// BM=128, 2x2 waves = two 256-thread blocks per CU at 64 KB LDS each,
// 2 waves/SIMD. Alternates kept for re-evaluation behind MARLIN_GEMM_CFG:
// bk32 (512-thread, half barrier rate, 58.6 TF) and bm256 (256x128 tile,
// half DMA-per-flop, 63.6 TF) — both measured slower than the default's
// 67.9 TF at 20000^3.
template <typename T, int BETA, int BK, int BM, int WM, int WN>
__launch_bounds__(WM * WN * 64, 2)
__global__ void gemm_mfma_kernel(int64_t M, int64_t N, int64_t K,
                                 const T* __restrict__ A, int64_t lda,
                                 const T* __restrict__ B, int64_t ldb,
                                 T* __restrict__ C, int64_t ldc,
                                 int nbm /* grid rows = M/BM */,
                                 int band /* column-band width in blocks */,
                                 int phase_mask /* k-phase stagger mask */) {
    constexpr int BN = 128;
    constexpr int NWAVES = WM * WN;
    constexpr int E = 16 / sizeof(T);       // elems per 16B glds chunk
    constexpr int MT = (BM / WM) / 16;      // 16x16 frags per wave, m
    constexpr int NT = (BN / WN) / 16;      // 16x16 frags per wave, n
    using ACC = typename acc_t<T>::type;

    // --- block remap: column bands of `band` block-cols, split into 8x8
    // block supertiles (band == 8 normally). Within a supertile the 8
    // blocks landing on one XCD (dispatch places block b on XCD b%8, and
    // all the stride terms are multiples of 8) form one COLUMN: each
    // XCD's L2 re-reads one B tile 8x and a contiguous 8-row A panel.
    // band < 0: plain bm-fastest bands (A/B comparison fallback).
    int id = blockIdx.x;
    int bn, bm;
    {
        int abands = band < 0 ? -band : band;
        int nbn = (int)(N / BN);
        int per_band = nbm * abands;
        int b0 = id / per_band;
        int w = id - b0 * per_band;
        int first_bn = b0 * abands;
        int bw = min(abands, nbn - first_bn);
        if (band < 0) {                      // plain band order
            bn = first_bn + w / nbm;
            bm = w % nbm;
        } else {                             // supertiled band order
            int sh = 8 * bw;
            int t = w / sh, l = w - t * sh;
            bm = t * 8 + l / bw;
            bn = first_bn + l % bw;
        }
    }

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wm = wid / WN, wn = wid % WN;  // wave grid coords (WM x WN)
    const int l15 = lane & 15, l4 = lane >> 4;   // MFMA fragment coords

    // --- LDS: A as [BK][BM] (k-col major), B as [BN][BK] (n-col major) ---
    __shared__ T sAB[2 * BK * BM + 2 * BN * BK];
    T* As = sAB;                              // [2][BK][BM]
    T* Bs = sAB + 2 * BK * BM;                // [2][BN][BK]

    const int64_t row0 = (int64_t)bm * BM;    // global row of tile
    const int64_t col0 = (int64_t)bn * BN;    // global col of tile

    constexpr int A_COLS_PER_GLDS = (64 * E) / BM > 0 ? (64 * E) / BM : 1;
    constexpr int A_GLDS = (BM * sizeof(T) >= 1024)
        ? (BK * (int)(BM * sizeof(T) / 1024) / NWAVES)
        : (BK / A_COLS_PER_GLDS / NWAVES);
    constexpr int B_COLS_PER_GLDS = (64 * E) / BK;
    constexpr int B_GLDS = BN / B_COLS_PER_GLDS / NWAVES;
    // B swizzle parameters for fp64: chunk index within a column is
    // XORed with (c >> BSH) & BMASK (fp32 uses its own masks; see below)
    constexpr int BMASK = BK / 2 - 1;
    constexpr int BSH = (BK == 16) ? 1 : 0;

    // --- LDS bank-conflict swizzles (measured 8-way on the B-fragment
    // reads without them — profiles/r01 PMC: conflict cycles were 88% of
    // LDS cycles). glds forces a lane-linear LDS image, so the
    // swizzle is applied to the per-lane GLOBAL source address (guide
    // idiom); both XORs permute 16-byte chunks within one 128-byte global
    // run, so HBM coalescing is unchanged.
    //   fp64 A image: column kk stores row-pair i at chunk i ^ (8*(kk&1))
    //     -> a-read banks: lanes 0-15 distinct, lanes 16-31 shifted by 32.
    //   fp64 B image: column c stores k-pair p at chunk p ^ ((c>>BSH)&BMASK)
    //     -> b-read banks: all 32 lanes of a ds_read_b64 group distinct
    //        (BK=16: column stride 128B; BK=32: column stride 256B).
    //   fp32 (banks are mod 32 for b32 reads): A chunk ^ 4*(kk&1);
    //     B chunk ^ (c>>1)&3 -> 2-way worst case (was 8-way).
    auto issue_tile = [&](int kt, int buf) {
        const int64_t kbase = (int64_t)kt * BK;
        if constexpr (BM * sizeof(T) >= 1024) {
            // >=1 glds per column: chunk q covers column q/CPC, piece q%CPC
            constexpr int CPC = BM * sizeof(T) / 1024;
            constexpr int A_CH = BK * CPC / NWAVES;
            #pragma unroll
            for (int i = 0; i < A_CH; i++) {
                int q = wid * A_CH + i;
                int c = q / CPC, half = q - (q / CPC) * CPC;
                int lane_row;
                if constexpr (E == 2)
                    lane_row = (half * 64 + (lane ^ ((c & 1) << 3))) * E;
                else
                    lane_row = (half * (1024 / sizeof(T) / E) * E)
                               + ((lane ^ ((c & 1) << 2)) * E);
                const T* g = A + (kbase + c) * lda + row0 + lane_row;
                glds16(g, &As[buf * BK * BM + c * BM
                              + half * (int)(1024 / sizeof(T))]);
            }
        } else {
            #pragma unroll
            for (int i = 0; i < A_GLDS; i++) {
                int c = (wid * A_GLDS + i) * A_COLS_PER_GLDS;
                int lane_col = lane / (BM / E);        // 0 or extra col
                int lane_row;
                if constexpr (E == 2)
                    lane_row = (lane ^ (((c + lane_col) & 1) << 3)) * E;
                else  // f32: 16B chunk XOR 4 by k parity (banks mod 32)
                    lane_row = ((lane % (BM / E)) ^ (((c + lane_col) & 1) << 2)) * E;
                const T* g = A + (kbase + c + lane_col) * lda + row0 + lane_row;
                glds16(g, &As[buf * BK * BM + c * BM]);
            }
        }
        #pragma unroll
        for (int i = 0; i < B_GLDS; i++) {
            int c = (wid * B_GLDS + i) * B_COLS_PER_GLDS;
            int lane_col = lane / (BK / E);
            int lane_row;
            if constexpr (E == 2)
                lane_row = ((lane % (BK / E)) ^ (((c + lane_col) >> BSH) & BMASK)) * E;
            else if constexpr (E == 4 && BK == 16)
                // f32: 4 chunks/col; XOR by (c>>1)&3 -> 2-way worst case
                lane_row = ((lane % 4) ^ (((c + lane_col) >> 1) & 3)) * E;
            else
                lane_row = (lane % (BK / E)) * E;
            const T* g = B + (col0 + c + lane_col) * ldb + kbase + lane_row;
            glds16(g, &Bs[buf * BN * BK + c * BK]);
        }
    };

    // C/D row of accumulator element j for this lane (measured per type:
    // tools_dev/debug_mfma.py): f64 16x16x4 is reg-major (l4 + 4j),
    // f32 16x16x4 is contiguous (4*l4 + j).
    constexpr bool REGMAJOR = sizeof(T) == 8;
    const int rbase = REGMAJOR ? l4 : l4 * 4;
    constexpr int rstep = REGMAJOR ? 4 : 1;
    ACC acc[MT][NT];
    if (BETA) {
        #pragma unroll
        for (int mt = 0; mt < MT; mt++)
            #pragma unroll
            for (int nt = 0; nt < NT; nt++) {
                int64_t r = row0 + wm * (MT * 16) + mt * 16 + rbase;
                int64_t cc = col0 + wn * (NT * 16) + nt * 16 + l15;
                const T* cp = C + cc * ldc + r;
                #pragma unroll
                for (int j = 0; j < 4; j++) acc[mt][nt][j] = cp[rstep * j];
            }
    } else {
        #pragma unroll
        for (int mt = 0; mt < MT; mt++)
            #pragma unroll
            for (int nt = 0; nt < NT; nt++) acc[mt][nt] = ACC{0};
    }

    const int ntiles = (int)(K / BK);
    // k-phase stagger experiment (MARLIN_GEMM_PHASE): blocks matching
    // the mask start their k-loop half-way round (accumulation order
    // rotated per tile — still deterministic run-to-run), so the two
    // blocks sharing a CU do not hit their DMA-wait/barrier windows in
    // sync. Default off (phase_mask 0 = ascending k, reference order).
    const int ph = (phase_mask && (id & phase_mask)) ? (ntiles >> 1) : 0;
    issue_tile(ph, 0);

    // Two barriers per k-step with the next tile's DMA issued EARLY and
    // kept in flight across the first barrier via a counted vmcnt (a
    // single-barrier variant with vmcnt(0)+late issue measured ~1% slower
    // fp64 / 4% slower fp32 - less DMA lead time).
    for (int s = 0; s < ntiles; s++) {
        const int buf = s & 1;
        if (s + 1 < ntiles) {
            int nxt = s + 1 + ph;
            if (nxt >= ntiles) nxt -= ntiles;
            issue_tile(nxt, buf ^ 1);
            // our own current-tile DMAs landed; next tile's stay in flight
            asm volatile("s_waitcnt vmcnt(%0)" :: "n"(A_GLDS + B_GLDS) : "memory");
        } else {
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();        // all waves' tile visible

        const T* At = &As[buf * BK * BM];
        const T* Bt = &Bs[buf * BN * BK];
        #pragma unroll
        for (int q = 0; q < BK / 4; q++) {
            const int kk = q * 4 + l4;
            T a[MT], b[NT];
            #pragma unroll
            for (int mt = 0; mt < MT; mt++) {
                int rr = wm * (MT * 16) + mt * 16 + l15;
                if constexpr (E == 2)
                    a[mt] = At[kk * BM + (((rr >> 1) ^ ((kk & 1) << 3)) << 1)
                               + (rr & 1)];
                else
                    a[mt] = At[kk * BM + (((rr >> 2) ^ ((kk & 1) << 2)) << 2)
                               + (rr & 3)];
            }
            #pragma unroll
            for (int nt = 0; nt < NT; nt++) {
                int cb = wn * (NT * 16) + nt * 16 + l15;
                if constexpr (E == 2)
                    b[nt] = Bt[cb * BK
                               + ((((kk >> 1) ^ ((cb >> BSH) & BMASK)) << 1))
                               + (kk & 1)];
                else if constexpr (E == 4 && BK == 16)
                    b[nt] = Bt[cb * BK
                               + (((kk >> 2) ^ ((cb >> 1) & 3)) << 2)
                               + (kk & 3)];
                else
                    b[nt] = Bt[cb * BK + kk];
            }
            // (s_setprio(1) around this block measured -5% — not used)
            #pragma unroll
            for (int mt = 0; mt < MT; mt++)
                #pragma unroll
                for (int nt = 0; nt < NT; nt++)
                    acc[mt][nt] = mfma_16x16x4(a[mt], b[nt], acc[mt][nt]);
#ifdef MARLIN_SCHED_HINT
            // scheduler hint: interleave next quad's DS reads between
            // MFMAs, one read per two MFMAs (A/B experiment flag)
            #pragma unroll
            for (int g = 0; g < (MT + NT); g++) {
                __builtin_amdgcn_sched_group_barrier(0x100, 1, 0);  // 1 DS read
                __builtin_amdgcn_sched_group_barrier(0x008, (MT * NT) / (MT + NT) + 1, 0);  // MFMAs
            }
#endif
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();        // readers done before overwrite
    }

    // --- epilogue (row map per type, see above) --------------------------
    #pragma unroll
    for (int mt = 0; mt < MT; mt++)
        #pragma unroll
        for (int nt = 0; nt < NT; nt++) {
            int64_t r = row0 + wm * (MT * 16) + mt * 16 + rbase;
            int64_t cc = col0 + wn * (NT * 16) + nt * 16 + l15;
            T* cp = C + cc * ldc + r;
            #pragma unroll
            for (int j = 0; j < 4; j++) cp[rstep * j] = acc[mt][nt][j];
        }
}

// fp32 GEMM with fused transpose/add epilogue (BASELINE config 5):
// C_out[n x m] = (A*B)^T (+ addC). Same main loop as the sgemm config
// (incl. the f32 XOR swizzles and the supertiled band remap); the
// transposed store is staged through LDS in two 64-row half-tiles so
// every global write is a coalesced float4 run (the naive per-element
// store was 4-byte scattered and cost ~8% at 40000^2).
__launch_bounds__(256, 2)
__global__ void sgemm_mfma_tn_epilogue_kernel(
        int64_t M, int64_t N, int64_t K,
        const float* __restrict__ A, int64_t lda,
        const float* __restrict__ B, int64_t ldb,
        float* __restrict__ C, int64_t ldc,      // C is N x M col-major
        const float* __restrict__ addC,          // N x M or nullptr
        int nbm, int band) {
    constexpr int BM = 128, BN = 128, BK = 16;
    // supertiled band remap (same as gemm_mfma_kernel: the 8 blocks on
    // one XCD share a B tile column and a contiguous A panel in L2)
    int id = blockIdx.x;
    int bn, bm;
    {
        int nbn = (int)(N / BN);
        int per_band = nbm * band;
        int b0 = id / per_band;
        int w = id - b0 * per_band;
        int first_bn = b0 * band;
        int bw = min(band, nbn - first_bn);
        int sh = 8 * bw;
        int t = w / sh, l = w - t * sh;
        bm = t * 8 + l / bw;
        bn = first_bn + l % bw;
    }

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wm = wid >> 1, wn = wid & 1;
    const int l15 = lane & 15, l4 = lane >> 4;

    __shared__ __align__(16) float sAB[2 * BK * BM + 2 * BN * BK];
    float* As = sAB;
    float* Bs = sAB + 2 * BK * BM;
    const int64_t row0 = (int64_t)bm * BM;
    const int64_t col0 = (int64_t)bn * BN;

    constexpr int E = 4;
    constexpr int A_COLS_PER_GLDS = (64 * E) / BM;
    constexpr int A_GLDS = BK / A_COLS_PER_GLDS / 4;
    constexpr int B_COLS_PER_GLDS = (64 * E) / BK;
    constexpr int B_GLDS = BN / B_COLS_PER_GLDS / 4;

    // f32 XOR swizzles on the GLOBAL source address (same as the sgemm
    // config of gemm_mfma_kernel: A chunk ^4 by k parity, B chunk
    // ^(c>>1)&3 — banks are mod 32 for b32 reads)
    auto issue_tile = [&](int kt, int buf) {
        const int64_t kbase = (int64_t)kt * BK;
        #pragma unroll
        for (int i = 0; i < A_GLDS; i++) {
            int c = (wid * A_GLDS + i) * A_COLS_PER_GLDS;
            int lane_col = lane / (BM / E);
            int lane_row = ((lane % (BM / E)) ^ (((c + lane_col) & 1) << 2))
                           * E;
            glds16(A + (kbase + c + lane_col) * lda + row0 + lane_row,
                   &As[buf * BK * BM + c * BM]);
        }
        #pragma unroll
        for (int i = 0; i < B_GLDS; i++) {
            int c = (wid * B_GLDS + i) * B_COLS_PER_GLDS;
            int lane_col = lane / (BK / E);
            int lane_row = ((lane % 4) ^ (((c + lane_col) >> 1) & 3)) * E;
            glds16(B + (col0 + c + lane_col) * ldb + kbase + lane_row,
                   &Bs[buf * BN * BK + c * BK]);
        }
    };

    v4f acc[4][4];
    #pragma unroll
    for (int mt = 0; mt < 4; mt++)
        #pragma unroll
        for (int nt = 0; nt < 4; nt++) acc[mt][nt] = v4f{0};

    const int ntiles = (int)(K / BK);
    issue_tile(0, 0);
    for (int kt = 0; kt < ntiles; kt++) {
        const int buf = kt & 1;
        if (kt + 1 < ntiles) {
            issue_tile(kt + 1, buf ^ 1);
            asm volatile("s_waitcnt vmcnt(%0)" :: "n"(A_GLDS + B_GLDS) : "memory");
        } else {
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
        const float* At = &As[buf * BK * BM];
        const float* Bt = &Bs[buf * BN * BK];
        #pragma unroll
        for (int q = 0; q < BK / 4; q++) {
            const int kk = q * 4 + l4;
            float a[4], b[4];
            #pragma unroll
            for (int mt = 0; mt < 4; mt++) {
                int rr = wm * 64 + mt * 16 + l15;
                a[mt] = At[kk * BM + (((rr >> 2) ^ ((kk & 1) << 2)) << 2)
                           + (rr & 3)];
            }
            #pragma unroll
            for (int nt = 0; nt < 4; nt++) {
                int cb = wn * 64 + nt * 16 + l15;
                b[nt] = Bt[cb * BK + (((kk >> 2) ^ ((cb >> 1) & 3)) << 2)
                           + (kk & 3)];
            }
            #pragma unroll
            for (int mt = 0; mt < 4; mt++)
                #pragma unroll
                for (int nt = 0; nt < 4; nt++)
                    acc[mt][nt] = mfma_16x16x4(a[mt], b[nt], acc[mt][nt]);
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
    }

    // Transposed store via LDS: result element (r, cc) of A*B goes to
    // C[cc + r*ldc] — contiguous output runs are (fixed r, varying cc).
    // Stage 64 result rows at a time into sAB (exactly 64x128 floats),
    // laid out stage[r'*128 + cc], then write float4 runs cooperatively.
    float* stage = sAB;                 // k-loop done; reuse the buffers
    v4f* stage4 = (v4f*)sAB;
    #pragma unroll
    for (int half = 0; half < 2; half++) {
        __builtin_amdgcn_s_barrier();   // prior phase's reads complete
        if (wm == half) {
            // this wave pair owns result rows half*64 .. half*64+63
            #pragma unroll
            for (int mt = 0; mt < 4; mt++)
                #pragma unroll
                for (int nt = 0; nt < 4; nt++) {
                    int rloc = mt * 16 + l4 * 4;      // 0..63
                    int cc = wn * 64 + nt * 16 + l15;
                    #pragma unroll
                    for (int j = 0; j < 4; j++)
                        stage[(rloc + j) * BN + cc] = acc[mt][nt][j];
                }
        }
        __builtin_amdgcn_s_barrier();
        // 64 rows x 32 float4 = 2048 stores over 256 threads
        #pragma unroll
        for (int i = 0; i < 8; i++) {
            int idx = tid + i * 256;
            int rloc = idx >> 5;                      // 0..63
            int q = idx & 31;                         // float4 index in row
            int64_t r = row0 + half * 64 + rloc;
            int64_t off = col0 + 4 * q + r * ldc;
            v4f v = stage4[rloc * (BN / 4) + q];
            if (addC) {
                const v4f* ap = (const v4f*)(addC + off);
                v4f av = *ap;
                v.x += av.x; v.y += av.y; v.z += av.z; v.w += av.w;
            }
            *(v4f*)(C + off) = v;
        }
    }
}

// ---------------------------------------------------------------------------
// Deterministic synthetic input fill: splitmix64 of (seed + (idx+1)*gamma),
// idx = logical col-major linear index c*m + r; matches
// oracle.marlin_oracle.gen_matrix bit-for-bit (tests assert it). The
// randomDenVecMatrix stand-in (MTUtils.scala:63-73 semantics).
__device__ __forceinline__ uint64_t splitmix64(uint64_t z) {
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
    return z ^ (z >> 31);
}

template <typename T>
__global__ void fill_random_kernel(T* __restrict__ buf, int64_t m, int64_t n,
                                   int64_t ld, uint64_t seed) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t total = m * n;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        int64_t c = i / m, r = i - c * m;
        uint64_t z = splitmix64(seed + (uint64_t)(i + 1) * 0x9E3779B97F4A7C15ULL);
        buf[c * ld + r] = (T)((double)(z >> 11) * (1.0 / 9007199254740992.0));
    }
}

// Zero an m x n padded region (pitch ld) — pad columns/rows stay zero.
template <typename T>
__global__ void zero_pad_kernel(T* __restrict__ buf, int64_t rows_total,
                                int64_t cols_total, int64_t ld,
                                int64_t m, int64_t n) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t total = rows_total * cols_total;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        int64_t c = i / rows_total, r = i - c * rows_total;
        if (r >= m || c >= n) buf[c * ld + r] = (T)0;
    }
}

// ---------------------------------------------------------------------------
// Elementwise / scalar op family — the BlockMatrix epilogue ops
// (BlockMatrix.scala:344-507: add/subtract matrix+scalar, subtractBy,
// divide, divideBy, dotProduct=element-wise multiply; DenseVecMatrix
// multiply(scalar)). HBM-bound map kernels; layout-agnostic (flat).
enum MxMapOp {
    MX_OP_ADD = 0,    // C = A + B
    MX_OP_SUB = 1,    // C = A - B
    MX_OP_EMUL = 2,   // C = A .* B   (reference "dotProduct")
    MX_OP_ADDS = 3,   // C = A + s
    MX_OP_SUBS = 4,   // C = A - s
    MX_OP_RSUBS = 5,  // C = s - A    (subtractBy)
    MX_OP_MULS = 6,   // C = A * s
    MX_OP_DIVS = 7,   // C = A / s
    MX_OP_RDIVS = 8,  // C = s / A    (divideBy)
};

template <typename T>
__global__ void map_kernel(int op, int64_t n, const T* __restrict__ a,
                           const T* __restrict__ b, double scalar,
                           T* __restrict__ c) {
    const T s = (T)scalar;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        T x = a[i];
        T y;
        switch (op) {
            case MX_OP_ADD:   y = x + b[i]; break;
            case MX_OP_SUB:   y = x - b[i]; break;
            case MX_OP_EMUL:  y = x * b[i]; break;
            case MX_OP_ADDS:  y = x + s; break;
            case MX_OP_SUBS:  y = x - s; break;
            case MX_OP_RSUBS: y = s - x; break;
            case MX_OP_MULS:  y = x * s; break;
            case MX_OP_DIVS:  y = x / s; break;
            default:          y = s / x; break;
        }
        c[i] = y;
    }
}

// sum reduction (DistributedMatrix.sum, BlockMatrix.scala sum tests):
// stage 1: per-block tree sums -> partials; stage 2: one block combines
// in fixed order (deterministic given the fixed grid).
template <typename T>
__global__ void sum_stage1_kernel(int64_t n, const T* __restrict__ a,
                                  double* __restrict__ partials) {
    __shared__ double sm[256];
    double acc = 0;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) acc += (double)a[i];
    sm[threadIdx.x] = acc;
    __syncthreads();
    for (int w = 128; w > 0; w >>= 1) {
        if (threadIdx.x < w) sm[threadIdx.x] += sm[threadIdx.x + w];
        __syncthreads();
    }
    if (threadIdx.x == 0) partials[blockIdx.x] = sm[0];
}

__global__ void sum_stage2_kernel(int nparts, const double* __restrict__ p,
                                  double* __restrict__ out) {
    __shared__ double sm[256];
    double acc = 0;
    for (int i = threadIdx.x; i < nparts; i += 256) acc += p[i];
    sm[threadIdx.x] = acc;
    __syncthreads();
    for (int w = 128; w > 0; w >>= 1) {
        if (threadIdx.x < w) sm[threadIdx.x] += sm[threadIdx.x + w];
        __syncthreads();
    }
    if (threadIdx.x == 0) *out = sm[0];
}

// Tiled transpose (BlockMatrix.transpose, BlockMatrix.scala:514-523):
// out[n x m] = in[m x n]^T, col-major, 32x32 LDS tiles (+1 pad), both
// sides coalesced.
template <typename T>
__global__ void transpose_kernel(int64_t m, int64_t n,
                                 const T* __restrict__ in,
                                 T* __restrict__ out) {
    __shared__ T tile[32][33];
    int64_t tm = (m + 31) / 32;
    int64_t bi = blockIdx.x % tm;           // tile row of input
    int64_t bj = blockIdx.x / tm;           // tile col of input
    int tx = threadIdx.x % 32, ty = threadIdx.x / 32;  // 32x8 threads
    int64_t r0 = bi * 32, c0 = bj * 32;
    #pragma unroll
    for (int k = 0; k < 4; k++) {
        int64_t r = r0 + tx, c = c0 + ty + k * 8;
        if (r < m && c < n) tile[ty + k * 8][tx] = in[c * m + r];
    }
    __syncthreads();
    #pragma unroll
    for (int k = 0; k < 4; k++) {
        int64_t r = c0 + tx, c = r0 + ty + k * 8;   // output coords
        if (r < n && c < m) out[c * n + r] = tile[tx][ty + k * 8];
    }
}

// ---------------------------------------------------------------------------
// Matrix-vector multiply (BlockMatrix.multiply(DistributedVector/BDV),
// BlockMatrix.scala:240-274): y = A x, col-major A. HBM-bound; row-slab
// blocks x column chunks with fp64 partials, reduced in ascending chunk
// order (deterministic). Reads of A are fully coalesced per column.
template <typename T>
__global__ void gemv_partial_kernel(int64_t m, int64_t n, int64_t lda,
                                    const T* __restrict__ A,
                                    const T* __restrict__ x,
                                    double* __restrict__ partial,
                                    int nchunks) {
    int64_t nrb = (m + 255) / 256;
    int64_t rb = blockIdx.x % nrb;
    int chunk = (int)(blockIdx.x / nrb);
    int64_t r = rb * 256 + threadIdx.x;
    int64_t clen = (n + nchunks - 1) / nchunks;
    int64_t c0 = chunk * clen;
    int64_t c1 = c0 + clen < n ? c0 + clen : n;
    if (r >= m) return;
    // 4 independent partial sums: keeps >=4 column loads in flight per
    // lane (the serial 1-acc loop measured latency-bound at 0.5-0.6
    // TB/s); summed in fixed order below -> still deterministic.
    double a0 = 0, a1 = 0, a2 = 0, a3 = 0;
    int64_t j = c0;
    for (; j + 4 <= c1; j += 4) {
        a0 += (double)A[(j + 0) * lda + r] * (double)x[j + 0];
        a1 += (double)A[(j + 1) * lda + r] * (double)x[j + 1];
        a2 += (double)A[(j + 2) * lda + r] * (double)x[j + 2];
        a3 += (double)A[(j + 3) * lda + r] * (double)x[j + 3];
    }
    for (; j < c1; j++)
        a0 += (double)A[j * lda + r] * (double)x[j];
    partial[(int64_t)chunk * m + r] = ((a0 + a1) + (a2 + a3));
}

template <typename T>
__global__ void gemv_reduce_kernel(int64_t m, int nchunks,
                                   const double* __restrict__ partial,
                                   T* __restrict__ y) {
    int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (r >= m) return;
    double acc = 0;
    for (int c = 0; c < nchunks; c++) acc += partial[(int64_t)c * m + r];
    y[r] = (T)acc;
}

// ---------------------------------------------------------------------------
// C-visible launchers (called from marlin_gpu.cpp).
extern "C" {

int mxk_gemm(int is_fp32, int beta_one,
             int64_t M, int64_t N, int64_t K,
             const void* A, int64_t lda, const void* B, int64_t ldb,
             void* C, int64_t ldc, hipStream_t stream) {
    if (M < 0 || N < 0 || K < 0) return -4;
    if (M == 0 || N == 0) return 0;            // empty tile: nothing to do
    if (K == 0) {                              // C = 0 (or unchanged if +=)
        if (!beta_one) {
            size_t es = is_fp32 ? 4 : 8;
            hipError_t e = hipMemset2DAsync(C, (size_t)ldc * es, 0,
                                            (size_t)M * es, (size_t)N, stream);
            if (e != hipSuccess) return -2;
        }
        return 0;
    }
    if (M % 128 || N % 128 || K % 16) return -4;
    int nbn = (int)(N / 128);
    static const char* bandenv = getenv("MARLIN_GEMM_BAND");
    int bandw = bandenv ? atoi(bandenv) : 8;
    if (bandw < 1) bandw = 8;
    int band = nbn < bandw ? nbn : bandw;
    // default: supertiled bands; MARLIN_GEMM_REMAP=bands -> plain bands
    static const char* remap = getenv("MARLIN_GEMM_REMAP");
    if (remap && remap[0] == 'b') band = -band;
    // configs: default BK=16/BM=128 2x256-thread blocks per CU (measured
    // winner); MARLIN_GEMM_CFG=bk32 -> 512-thread BK=32; =bm256 -> 256x128
    // tile / 512-thread (requires M % 256 == 0; halves DMA-per-flop and
    // barrier rate at 1 block/CU).
    static const char* cfg = getenv("MARLIN_GEMM_CFG");
    bool bk32 = (K % 32 == 0) && cfg && cfg[0] == 'b' && cfg[1] == 'k';
    bool bm256 = (M % 256 == 0) && !is_fp32 && cfg && cfg[0] == 'b' && cfg[1] == 'm';
    // k-phase stagger mask (experiment; see kernel comment). 0 = off.
    static const char* phenv = getenv("MARLIN_GEMM_PHASE");
    int phase_mask = phenv ? atoi(phenv) : 0;
    #define LAUNCH(TY, BETA, BK, BMv, WMv, WNv)                            \
        hipLaunchKernelGGL((gemm_mfma_kernel<TY, BETA, BK, BMv, WMv, WNv>),\
            dim3((unsigned)((M / BMv) * nbn)), dim3(WMv * WNv * 64), 0,    \
            stream, M, N, K, (const TY*)A, lda, (const TY*)B, ldb, (TY*)C, \
            ldc, (int)(M / BMv), band, phase_mask)
    if (is_fp32) {
        if (beta_one) { if (bk32) LAUNCH(float, 1, 32, 128, 2, 4); else LAUNCH(float, 1, 16, 128, 2, 2); }
        else          { if (bk32) LAUNCH(float, 0, 32, 128, 2, 4); else LAUNCH(float, 0, 16, 128, 2, 2); }
    } else if (bm256) {
        if (beta_one) LAUNCH(double, 1, 16, 256, 4, 2);
        else          LAUNCH(double, 0, 16, 256, 4, 2);
    } else {
        if (beta_one) { if (bk32) LAUNCH(double, 1, 32, 128, 2, 4); else LAUNCH(double, 1, 16, 128, 2, 2); }
        else          { if (bk32) LAUNCH(double, 0, 32, 128, 2, 4); else LAUNCH(double, 0, 16, 128, 2, 2); }
    }
    #undef LAUNCH
    return (int)hipGetLastError() == 0 ? 0 : -2;
}

int mxk_sgemm_tn_epilogue(int64_t M, int64_t N, int64_t K,
                          const float* A, int64_t lda,
                          const float* B, int64_t ldb,
                          float* C, int64_t ldc, const float* addC,
                          hipStream_t stream) {
    if (M % 128 || N % 128 || K % 16) return -4;
    int nbm = (int)(M / 128), nbn = (int)(N / 128);
    int band = nbn < 8 ? nbn : 8;
    dim3 grid((unsigned)(nbm * nbn)), block(256);
    hipLaunchKernelGGL(sgemm_mfma_tn_epilogue_kernel, grid, block, 0, stream,
                       M, N, K, A, lda, B, ldb, C, ldc, addC, nbm, band);
    return (int)hipGetLastError() == 0 ? 0 : -2;
}

int mxk_fill_random(int is_fp32, void* buf, int64_t m, int64_t n, int64_t ld,
                    uint64_t seed, hipStream_t stream) {
    dim3 grid(2048), block(256);
    if (is_fp32)
        hipLaunchKernelGGL(fill_random_kernel<float>, grid, block, 0, stream,
                           (float*)buf, m, n, ld, seed);
    else
        hipLaunchKernelGGL(fill_random_kernel<double>, grid, block, 0, stream,
                           (double*)buf, m, n, ld, seed);
    return (int)hipGetLastError() == 0 ? 0 : -2;
}

int mxk_zero_pad(int is_fp32, void* buf, int64_t rows_total, int64_t cols_total,
                 int64_t ld, int64_t m, int64_t n, hipStream_t stream) {
    dim3 grid(2048), block(256);
    if (is_fp32)
        hipLaunchKernelGGL(zero_pad_kernel<float>, grid, block, 0, stream,
                           (float*)buf, rows_total, cols_total, ld, m, n);
    else
        hipLaunchKernelGGL(zero_pad_kernel<double>, grid, block, 0, stream,
                           (double*)buf, rows_total, cols_total, ld, m, n);
    return (int)hipGetLastError() == 0 ? 0 : -2;
}

int mxk_map(int is_fp32, int op, int64_t n, const void* a, const void* b,
            double scalar, void* c, hipStream_t stream) {
    dim3 grid(2048), block(256);
    if (is_fp32)
        hipLaunchKernelGGL(map_kernel<float>, grid, block, 0, stream, op, n,
                           (const float*)a, (const float*)b, scalar, (float*)c);
    else
        hipLaunchKernelGGL(map_kernel<double>, grid, block, 0, stream, op, n,
                           (const double*)a, (const double*)b, scalar,
                           (double*)c);
    return (int)hipGetLastError() == 0 ? 0 : -2;
}

int mxk_sum(int is_fp32, int64_t n, const void* a, double* partials,
            double* out, hipStream_t stream) {
    dim3 grid(1024), block(256);
    if (is_fp32)
        hipLaunchKernelGGL(sum_stage1_kernel<float>, grid, block, 0, stream,
                           n, (const float*)a, partials);
    else
        hipLaunchKernelGGL(sum_stage1_kernel<double>, grid, block, 0, stream,
                           n, (const double*)a, partials);
    hipLaunchKernelGGL(sum_stage2_kernel, dim3(1), block, 0, stream, 1024,
                       partials, out);
    return (int)hipGetLastError() == 0 ? 0 : -2;
}

int mxk_transpose(int is_fp32, int64_t m, int64_t n, const void* in,
                  void* out, hipStream_t stream) {
    int64_t tm = (m + 31) / 32, tn = (n + 31) / 32;
    dim3 grid((unsigned)(tm * tn)), block(256);
    if (is_fp32)
        hipLaunchKernelGGL(transpose_kernel<float>, grid, block, 0, stream,
                           m, n, (const float*)in, (float*)out);
    else
        hipLaunchKernelGGL(transpose_kernel<double>, grid, block, 0, stream,
                           m, n, (const double*)in, (double*)out);
    return (int)hipGetLastError() == 0 ? 0 : -2;
}

int mxk_gemv(int is_fp32, int64_t m, int64_t n, int64_t lda, const void* A,
             const void* x, double* partial, void* y, hipStream_t stream) {
    int64_t nrb = (m + 255) / 256;
    // 32 column chunks: at m=16384 that is 2048 workgroups (8 per CU),
    // enough in-flight loads to cover HBM latency (8 chunks measured
    // 0.53 TB/s -- only 2 blocks/CU; see profiles/r02_experiments.md)
    int nchunks = 32;
    dim3 grid((unsigned)(nrb * nchunks)), block(256);
    if (is_fp32) {
        hipLaunchKernelGGL(gemv_partial_kernel<float>, grid, block, 0, stream,
                           m, n, lda, (const float*)A, (const float*)x,
                           partial, nchunks);
        hipLaunchKernelGGL(gemv_reduce_kernel<float>, dim3((unsigned)nrb),
                           block, 0, stream, m, nchunks, partial, (float*)y);
    } else {
        hipLaunchKernelGGL(gemv_partial_kernel<double>, grid, block, 0, stream,
                           m, n, lda, (const double*)A, (const double*)x,
                           partial, nchunks);
        hipLaunchKernelGGL(gemv_reduce_kernel<double>, dim3((unsigned)nrb),
                           block, 0, stream, m, nchunks, partial, (double*)y);
    }
    return (int)hipGetLastError() == 0 ? 0 : -2;
}

}  // extern "C"
