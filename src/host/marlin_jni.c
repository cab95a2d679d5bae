/* marlin_jni.c — JNI veneer over the C ABI (include/marlin_gpu.h).
 *
 * This is the binding a Marlin maintainer adds so the reference's Scala
 * host calls the MI355X engine at its native hand-off point — Breeze `*`
 * -> netlib-java JNI dgemm (SubMatrix.scala:87-105) and the whole
 * multiply (BlockMatrix.scala:149-220). The Scala object it implements
 * is edu.nju.pasalab.marlin.gpu.MarlinGpu.
 *
 * Compiled by __graft_entry__.build() iff a JDK is present on the box
 * (this dev container has none — probed in SURVEY.md):
 *   gcc -O2 -fPIC -shared marlin_jni.c -Iinclude \
 *       -I$JAVA_HOME/include -I$JAVA_HOME/include/linux \
 *       libmarlin_gpu.so -o libmarlin_jni.so
 */
#include <jni.h>

#include "marlin_gpu.h"

/* one ctx per executor JVM — one GPU per Spark executor, matching the
 * reference's one-BLAS-context-per-executor model */
static mx_ctx* g_ctx;

JNIEXPORT jint JNICALL
Java_edu_nju_pasalab_marlin_gpu_MarlinGpu_init(JNIEnv* env, jclass cls,
                                               jint device) {
  (void)env; (void)cls;
  return mx_init(&g_ctx, (int)device);
}

JNIEXPORT jint JNICALL
Java_edu_nju_pasalab_marlin_gpu_MarlinGpu_shutdown(JNIEnv* env, jclass cls) {
  (void)env; (void)cls;
  int rc = mx_shutdown(g_ctx);
  g_ctx = 0;
  return rc;
}

JNIEXPORT jstring JNICALL
Java_edu_nju_pasalab_marlin_gpu_MarlinGpu_strerror(JNIEnv* env, jclass cls,
                                                   jint code) {
  (void)cls;
  return (*env)->NewStringUTF(env, mx_strerror((int)code));
}

/* SubMatrix.multiply replacement (SubMatrix.scala:87-105): column-major
 * double[] tiles, C = A*B, or C += A*B when betaOne != 0 (the
 * reduceByKey add combiner, SubMatrix.scala:41-50, folded on-device). */
JNIEXPORT jint JNICALL
Java_edu_nju_pasalab_marlin_gpu_MarlinGpu_tileDgemmAcc(
    JNIEnv* env, jclass cls, jlong m, jlong k, jlong n, jdoubleArray a,
    jdoubleArray b, jdoubleArray c, jint betaOne) {
  (void)cls;
  jdouble* pa = (*env)->GetPrimitiveArrayCritical(env, a, 0);
  jdouble* pb = (*env)->GetPrimitiveArrayCritical(env, b, 0);
  jdouble* pc = (*env)->GetPrimitiveArrayCritical(env, c, 0);
  int rc = (pa && pb && pc)
               ? mx_tile_dgemm_acc(g_ctx, (int64_t)m, (int64_t)k, (int64_t)n,
                                   pa, pb, pc, (int)betaOne)
               : MX_EINVAL;
  if (pc) (*env)->ReleasePrimitiveArrayCritical(env, c, pc, 0);
  if (pb) (*env)->ReleasePrimitiveArrayCritical(env, b, pb, JNI_ABORT);
  if (pa) (*env)->ReleasePrimitiveArrayCritical(env, a, pa, JNI_ABORT);
  return rc;
}

/* Whole-multiply replacement (BlockMatrix.multiply,
 * BlockMatrix.scala:149-220) for the single-node multi-GPU deployment. */
JNIEXPORT jint JNICALL
Java_edu_nju_pasalab_marlin_gpu_MarlinGpu_dgemm(
    JNIEnv* env, jclass cls, jlong m, jlong k, jlong n, jdoubleArray a,
    jdoubleArray b, jdoubleArray c) {
  (void)cls;
  jdouble* pa = (*env)->GetPrimitiveArrayCritical(env, a, 0);
  jdouble* pb = (*env)->GetPrimitiveArrayCritical(env, b, 0);
  jdouble* pc = (*env)->GetPrimitiveArrayCritical(env, c, 0);
  int rc = (pa && pb && pc)
               ? mx_dgemm(g_ctx, (int64_t)m, (int64_t)k, (int64_t)n, pa, pb,
                          pc)
               : MX_EINVAL;
  if (pc) (*env)->ReleasePrimitiveArrayCritical(env, c, pc, 0);
  if (pb) (*env)->ReleasePrimitiveArrayCritical(env, b, pb, JNI_ABORT);
  if (pa) (*env)->ReleasePrimitiveArrayCritical(env, a, pa, JNI_ABORT);
  return rc;
}

/* fp32 multiply with fused transpose/add epilogue (config 5;
 * BlockMatrix.scala:514-523 + :344-452 composed). */
JNIEXPORT jint JNICALL
Java_edu_nju_pasalab_marlin_gpu_MarlinGpu_sgemmEpilogue(
    JNIEnv* env, jclass cls, jlong m, jlong k, jlong n, jfloatArray a,
    jfloatArray b, jfloatArray c, jint transposeC, jfloatArray addC) {
  (void)cls;
  jfloat* pa = (*env)->GetPrimitiveArrayCritical(env, a, 0);
  jfloat* pb = (*env)->GetPrimitiveArrayCritical(env, b, 0);
  jfloat* pc = (*env)->GetPrimitiveArrayCritical(env, c, 0);
  jfloat* pd = addC ? (*env)->GetPrimitiveArrayCritical(env, addC, 0) : 0;
  int rc = (pa && pb && pc)
               ? mx_sgemm_epilogue(g_ctx, (int64_t)m, (int64_t)k, (int64_t)n,
                                   pa, pb, pc, (int)transposeC, pd)
               : MX_EINVAL;
  if (pd) (*env)->ReleasePrimitiveArrayCritical(env, addC, pd, JNI_ABORT);
  if (pc) (*env)->ReleasePrimitiveArrayCritical(env, c, pc, 0);
  if (pb) (*env)->ReleasePrimitiveArrayCritical(env, b, pb, JNI_ABORT);
  if (pa) (*env)->ReleasePrimitiveArrayCritical(env, a, pa, JNI_ABORT);
  return rc;
}
