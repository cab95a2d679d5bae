// marlinx — C++ host CLI above the C ABI (the stand-in for the reference's
// spark-submit example drivers: examples/MatrixMultiply.scala:16-48 and
// examples/BLAS3.scala:29-57). Args mirror MatrixMultiply.main:
//
//   marlinx bench  <m> <k> <n> [steps=3] [warmup=1]   random fp64 multiply
//   marlinx verify <m> <k> <n>                        engine vs CPU check
//   marlinx epilogue <m> <k> <n>                      fp32 (A*B)^T + D check
//     (config 5's fused transpose/add — BlockMatrix.scala:514-523 +
//      :344-452 composed — through mx_sgemm_epilogue)
//
// The CPU check in `verify` is a naive triple loop on small sizes only —
// a smoke-level verifier for the CLI; authoritative parity runs in
// tests/ against the oracle.
#include <chrono>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#include "marlin_gpu.h"

static double now_ms() {
  using namespace std::chrono;
  return duration<double, std::milli>(
             steady_clock::now().time_since_epoch()).count();
}

// splitmix64 U[0,1) — the engine's synthetic-input spec (kernels.hip)
static double gen(uint64_t seed, uint64_t idx) {
  uint64_t z = seed + (idx + 1) * 0x9E3779B97F4A7C15ULL;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  z = z ^ (z >> 31);
  return (double)(z >> 11) * (1.0 / 9007199254740992.0);
}

int main(int argc, char** argv) {
  if (argc < 5) {
    fprintf(stderr, "usage: %s bench|verify m k n [steps] [warmup]\n",
            argv[0]);
    return 2;
  }
  const char* mode = argv[1];
  int64_t m = atoll(argv[2]), k = atoll(argv[3]), n = atoll(argv[4]);
  int steps = argc > 5 ? atoi(argv[5]) : 3;
  int warmup = argc > 6 ? atoi(argv[6]) : 1;

  std::vector<double> A((size_t)(m * k)), B((size_t)(k * n)),
      C((size_t)(m * n));
  for (int64_t i = 0; i < m * k; i++) A[i] = gen(0xA11CE, i);
  for (int64_t i = 0; i < k * n; i++) B[i] = gen(0xB0B, i);

  mx_ctx* ctx = nullptr;
  int rc = mx_init(&ctx, -1);
  if (rc != MX_OK) {
    fprintf(stderr, "mx_init: %s\n", mx_strerror(rc));
    return 1;
  }

  if (!strcmp(mode, "epilogue")) {
    // fp32 fused (A*B)^T + D against a naive fp64 CPU recompute
    std::vector<float> Af((size_t)(m * k)), Bf((size_t)(k * n)),
        Cf((size_t)(n * m)), Df((size_t)(n * m));
    for (int64_t i = 0; i < m * k; i++) Af[i] = (float)gen(0xA11CE, i);
    for (int64_t i = 0; i < k * n; i++) Bf[i] = (float)gen(0xB0B, i);
    for (int64_t i = 0; i < n * m; i++) Df[i] = (float)gen(0xADD, i);
    rc = mx_sgemm_epilogue(ctx, m, k, n, Af.data(), Bf.data(), Cf.data(),
                           /*transpose_c=*/1, Df.data());
    if (rc != MX_OK) {
      fprintf(stderr, "mx_sgemm_epilogue: %s\n", mx_strerror(rc));
      return 1;
    }
    double maxrel = 0;
    for (int64_t i = 0; i < m; i++)
      for (int64_t j = 0; j < n; j++) {
        double acc = 0;
        for (int64_t l = 0; l < k; l++)
          acc += (double)Af[l * m + i] * (double)Bf[j * k + l];
        acc += (double)Df[i * n + j];            // D is n x m col-major
        double d = (double)Cf[i * n + j] - acc;  // C_out is n x m
        double rel = (d < 0 ? -d : d) / (acc < 0 ? -acc : acc);
        if (rel > maxrel) maxrel = rel;
      }
    printf("epilogue %lldx%lldx%lld max_rel=%.3e %s\n", (long long)m,
           (long long)k, (long long)n, maxrel,
           maxrel < 1e-4 ? "OK" : "FAIL");
    mx_shutdown(ctx);
    return maxrel < 1e-4 ? 0 : 1;
  }

  if (!strcmp(mode, "verify")) {
    rc = mx_dgemm(ctx, m, k, n, A.data(), B.data(), C.data());
    if (rc != MX_OK) {
      fprintf(stderr, "mx_dgemm: %s\n", mx_strerror(rc));
      return 1;
    }
    double maxrel = 0;
    for (int64_t j = 0; j < n; j++)
      for (int64_t i = 0; i < m; i++) {
        double acc = 0;
        for (int64_t l = 0; l < k; l++)
          acc += A[l * m + i] * B[j * k + l];
        double d = C[j * m + i] - acc;
        double rel = (d < 0 ? -d : d) / (acc < 0 ? -acc : acc);
        if (rel > maxrel) maxrel = rel;
      }
    printf("verify %lldx%lldx%lld max_rel=%.3e %s\n", (long long)m,
           (long long)k, (long long)n, maxrel,
           maxrel < 1e-10 ? "OK" : "FAIL");
    mx_shutdown(ctx);
    return maxrel < 1e-10 ? 0 : 1;
  }

  for (int s = 0; s < warmup; s++)
    (void)mx_dgemm(ctx, m, k, n, A.data(), B.data(), C.data());
  double t0 = now_ms();
  double gemm_ms = 0;
  for (int s = 0; s < steps; s++) {
    rc = mx_dgemm(ctx, m, k, n, A.data(), B.data(), C.data());
    if (rc != MX_OK) {
      fprintf(stderr, "mx_dgemm: %s\n", mx_strerror(rc));
      return 1;
    }
    mx_stats_t st;
    mx_stats(ctx, &st);
    gemm_ms += st.gemm_ms;
  }
  double wall = now_ms() - t0;
  double tf = 2.0 * m * k * n * steps / (wall / 1e3) / 1e12;
  double tf_kernel = 2.0 * m * k * n * steps / (gemm_ms / 1e3) / 1e12;
  printf("{\"metric\": \"dense C=AxB TFLOP/s (fp64)\", \"value\": %.3f, "
         "\"kernel_tflops\": %.3f, \"ms_per_step\": %.2f, "
         "\"note\": \"host-buffer path incl. PCIe\"}\n",
         tf, tf_kernel, wall / steps);
  mx_shutdown(ctx);
  return 0;
}
