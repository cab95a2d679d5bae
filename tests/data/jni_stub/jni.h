/* Minimal jni.h STUB — test fixture only (tests/test_abi.py uses it to
 * syntax-check src/host/marlin_jni.c in a container without a JDK).
 * Covers exactly the JNI surface the veneer touches; never shipped,
 * never used by the product build (__graft_entry__.build() compiles the
 * veneer only against a real JDK's jni.h). */
#ifndef MARLIN_TEST_JNI_STUB_H
#define MARLIN_TEST_JNI_STUB_H

#define JNIEXPORT __attribute__((visibility("default")))
#define JNICALL
#define JNI_ABORT 2

typedef int jint;
typedef long long jlong;
typedef double jdouble;
typedef float jfloat;
typedef void* jclass;
typedef void* jstring;
typedef void* jdoubleArray;
typedef void* jfloatArray;

struct JNINativeInterface_;
typedef const struct JNINativeInterface_* JNIEnv;

struct JNINativeInterface_ {
  void* (*GetPrimitiveArrayCritical)(JNIEnv* env, void* array, int* isCopy);
  void (*ReleasePrimitiveArrayCritical)(JNIEnv* env, void* array, void* carray,
                                        jint mode);
  jstring (*NewStringUTF)(JNIEnv* env, const char* utf);
};

#endif
