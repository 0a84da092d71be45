#include "zstd_wrap.h"

#include <dlfcn.h>

namespace vl {

namespace {

// Hand-declared prototypes for the ~6 libzstd entry points we need (no zstd.h
// on this image; see SURVEY.md container facts).
typedef size_t (*fn_compressBound)(size_t);
typedef size_t (*fn_compress)(void*, size_t, const void*, size_t, int);
typedef unsigned (*fn_isError)(size_t);
typedef unsigned long long (*fn_getFrameContentSize)(const void*, size_t);
typedef size_t (*fn_decompress)(void*, size_t, const void*, size_t);

struct Zstd {
  fn_compressBound compressBound;
  fn_compress compress;
  fn_isError isError;
  fn_getFrameContentSize getFrameContentSize;
  fn_decompress decompress;

  Zstd() {
    void* h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("libzstd.so", RTLD_NOW | RTLD_GLOBAL);
    if (!h) fail("cannot dlopen libzstd.so.1");
    compressBound = (fn_compressBound)dlsym(h, "ZSTD_compressBound");
    compress = (fn_compress)dlsym(h, "ZSTD_compress");
    isError = (fn_isError)dlsym(h, "ZSTD_isError");
    getFrameContentSize = (fn_getFrameContentSize)dlsym(h, "ZSTD_getFrameContentSize");
    decompress = (fn_decompress)dlsym(h, "ZSTD_decompress");
    if (!compressBound || !compress || !isError || !getFrameContentSize || !decompress) {
      fail("missing ZSTD_* symbols in libzstd");
    }
  }
};

Zstd& z() {
  static Zstd instance;
  return instance;
}

constexpr unsigned long long kContentSizeUnknown = 0ULL - 1;  // ZSTD_CONTENTSIZE_UNKNOWN
constexpr unsigned long long kContentSizeError = 0ULL - 2;    // ZSTD_CONTENTSIZE_ERROR

}  // namespace

void zstd_compress(bytes& dst, const uint8_t* src, size_t n, int level) {
  size_t bound = z().compressBound(n);
  size_t old = dst.size();
  dst.resize(old + bound);
  size_t r = z().compress(dst.data() + old, bound, src, n, level);
  if (z().isError(r)) fail("ZSTD_compress failed");
  dst.resize(old + r);
}

void zstd_decompress(bytes& dst, const uint8_t* src, size_t n) {
  unsigned long long sz = z().getFrameContentSize(src, n);
  if (sz == kContentSizeError) fail("zstd: invalid frame");
  size_t old = dst.size();
  if (sz != kContentSizeUnknown) {
    dst.resize(old + size_t(sz));
    size_t r = z().decompress(dst.data() + old, size_t(sz), src, n);
    if (z().isError(r)) fail("ZSTD_decompress failed");
    dst.resize(old + r);
    return;
  }
  // Unknown content size: grow-and-retry (rare; gozstd writes sized frames).
  size_t cap = n * 4 + 64;
  for (int i = 0; i < 10; i++) {
    dst.resize(old + cap);
    size_t r = z().decompress(dst.data() + old, cap, src, n);
    if (!z().isError(r)) {
      dst.resize(old + r);
      return;
    }
    cap *= 2;
  }
  fail("zstd: cannot decompress frame with unknown content size");
}

}  // namespace vl
