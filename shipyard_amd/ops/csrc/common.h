// Common helpers for shipyard_amd HIP kernels (gfx950 / MI355X only).
//
// These kernels implement the framework's data-plane hot paths — the
// MI355X-native equivalent of the reference's delegated native work
// (container layer decompress in dockerd, MD5/SHA256 integrity in
// convoy/util.py:461-508, blobxfer chunk hashing).  See SURVEY.md §2.5.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define SY_EXPORT extern "C" __attribute__((visibility("default")))

// Error contract: every entry point returns a hipError_t-compatible int
// (0 == success).  Kernel-level data errors (corrupt LZ4 stream, bounds)
// are reported through per-item status words, not the return code.
static inline int sy_check(hipError_t e) { return static_cast<int>(e); }

// wave64 is the CDNA scheduling quantum; hard-code 64 per the CDNA4 guide.
#ifndef SY_WAVE
#define SY_WAVE 64
#endif
