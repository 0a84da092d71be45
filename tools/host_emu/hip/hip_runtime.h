// HIP runtime STUB for the CPU emulation build of the product pipeline
// (tools/host_emu): hipMalloc/Memcpy/etc become plain host memory ops and
// the kernel launchers are replaced by serial loops over the SAME per-row
// device code (scan_rowops.h).  TEST INFRASTRUCTURE ONLY — the real
// libvlogsql.so never uses this; the emu library is built separately and
// loaded via VQL_LIB by tests/test_emu_pipeline.py.
#pragma once

#include <chrono>
#include <cstdlib>
#include <cstring>

typedef int hipError_t;
constexpr hipError_t hipSuccess = 0;
typedef void* hipStream_t;
struct hipEventRec {
  std::chrono::steady_clock::time_point t;
};
typedef hipEventRec* hipEvent_t;

enum hipMemcpyKind {
  hipMemcpyHostToDevice,
  hipMemcpyDeviceToHost,
  hipMemcpyDeviceToDevice,
  hipMemcpyDefault,
};

static inline hipError_t hipSetDevice(int) { return hipSuccess; }
static inline const char* hipGetErrorString(hipError_t) { return "emu"; }
static inline hipError_t hipGetLastError() { return hipSuccess; }

template <typename T>
static inline hipError_t hipMalloc(T** p, size_t n) {
  *p = (T*)malloc(n ? n : 1);
  return *p ? hipSuccess : 1;
}
static inline hipError_t hipFree(void* p) {
  free(p);
  return hipSuccess;
}
static inline hipError_t hipMemcpy(void* d, const void* s, size_t n,
                                   hipMemcpyKind = hipMemcpyDefault) {
  memcpy(d, s, n);
  return hipSuccess;
}
static inline hipError_t hipMemcpyAsync(void* d, const void* s, size_t n,
                                        hipMemcpyKind, hipStream_t) {
  memcpy(d, s, n);
  return hipSuccess;
}
static inline hipError_t hipMemset(void* d, int v, size_t n) {
  memset(d, v, n);
  return hipSuccess;
}
static inline hipError_t hipMemsetAsync(void* d, int v, size_t n, hipStream_t) {
  memset(d, v, n);
  return hipSuccess;
}
static inline hipError_t hipStreamCreate(hipStream_t* s) {
  *s = nullptr;
  return hipSuccess;
}
static inline hipError_t hipStreamDestroy(hipStream_t) { return hipSuccess; }
static inline hipError_t hipStreamSynchronize(hipStream_t) { return hipSuccess; }
static inline hipError_t hipEventCreate(hipEvent_t* e) {
  *e = new hipEventRec();
  return hipSuccess;
}
static inline hipError_t hipEventDestroy(hipEvent_t e) {
  delete e;
  return hipSuccess;
}
static inline hipError_t hipEventRecord(hipEvent_t e, hipStream_t) {
  e->t = std::chrono::steady_clock::now();
  return hipSuccess;
}
static inline hipError_t hipEventElapsedTime(float* ms, hipEvent_t a,
                                             hipEvent_t b) {
  *ms = std::chrono::duration<float, std::milli>(b->t - a->t).count();
  return hipSuccess;
}
