// Thin zstd wrapper over the system libzstd.so.1 (no dev headers on this
// image — prototypes are hand-declared and the library is dlopen'ed).
//
// The reference compresses via valyala/gozstd (cgo->libzstd) or
// klauspost/compress selected by build tag
// (vendor/.../lib/encoding/zstd/zstd_{cgo,pure}.go); frames are standard zstd.
// Parity is on decompressed content, not on compressed bytes.
#pragma once

#include <cstddef>

#include "vl_base.h"

namespace vl {

// CompressZSTDLevel (vendor/.../lib/encoding/compress.go)
void zstd_compress(bytes& dst, const uint8_t* src, size_t n, int level);
// DecompressZSTD; appends to dst; throws on malformed frames.
void zstd_decompress(bytes& dst, const uint8_t* src, size_t n);

}  // namespace vl
