// Native file->HBM staging pipeline (host-side HIP C++).
//
// The data-plane transport under the shard stager: pread (optionally
// O_DIRECT, bypassing the page cache for true-NVMe numbers) into a
// ring of pinned (hipHostMalloc) buffers, each window pushed to the
// device with hipMemcpyAsync on the caller's stream; the next window's
// disk read overlaps the previous window's H2D.  This is the C++
// equivalent of shipyard_amd/data/stager.py's double-buffer loop with
// page-cache control and no interpreter on the per-window path
// (reference analogue: the blobxfer transport under convoy/data.py).

#include <errno.h>
#include <fcntl.h>
#include <stdint.h>
#include <string.h>
#include <sys/stat.h>
#include <unistd.h>

#include <chrono>

#include "common.h"

namespace {

constexpr int kRing = 2;

struct PinnedRing {
  void* buf[kRing] = {nullptr, nullptr};
  uint64_t bytes = 0;

  int ensure(uint64_t want) {
    if (bytes >= want && buf[0]) return 0;
    release();
    for (int i = 0; i < kRing; ++i) {
      hipError_t e = hipHostMalloc(&buf[i], want, hipHostMallocDefault);
      if (e != hipSuccess) {
        release();
        return (int)e;
      }
    }
    bytes = want;
    return 0;
  }

  void release() {
    for (int i = 0; i < kRing; ++i) {
      if (buf[i]) (void)hipHostFree(buf[i]);
      buf[i] = nullptr;
    }
    bytes = 0;
  }
};

PinnedRing g_ring;  // cached across calls (pinned alloc is ~ms)

}  // namespace

// Stage [file_off, file_off+n_bytes) of `path` into device memory at
// d_dst.  staging_bytes sets the window size (0 -> 64 MiB).  use_direct
// tries O_DIRECT first (falls back to buffered if open/read fails).
// Returns 0 on success, -errno for IO errors, hipError_t (>0) for HIP
// errors.  out_seconds (optional) receives wall time of the pipeline.
SY_EXPORT int sy_stage_file(const char* path, void* d_dst,
                            uint64_t file_off, uint64_t n_bytes,
                            uint64_t staging_bytes, int use_direct,
                            hipStream_t stream, double* out_seconds) {
  if (staging_bytes == 0) staging_bytes = 64ull << 20;
  // O_DIRECT needs 4 KiB-aligned offsets/sizes/buffers; pinned buffers
  // are page-aligned, so only check offset/length.
  int flags = O_RDONLY;
  bool direct = use_direct && (file_off % 4096 == 0);
  int fd = -1;
  if (direct) {
    fd = open(path, flags | O_DIRECT);
    if (fd < 0) direct = false;
  }
  if (fd < 0) fd = open(path, flags);
  if (fd < 0) return -errno;

  int rc = g_ring.ensure(staging_bytes);
  if (rc != 0) {
    close(fd);
    return rc;
  }

  hipEvent_t ev[kRing];
  for (int i = 0; i < kRing; ++i) {
    hipError_t e = hipEventCreateWithFlags(&ev[i], hipEventDisableTiming);
    if (e != hipSuccess) {
      close(fd);
      return (int)e;
    }
  }

  auto t0 = std::chrono::steady_clock::now();
  uint64_t off = 0;
  int idx = 0;
  bool used[kRing] = {false, false};
  int err = 0;
  while (off < n_bytes && err == 0) {
    uint64_t want = n_bytes - off;
    if (want > staging_bytes) want = staging_bytes;
    // O_DIRECT reads must be 4 KiB multiples except at EOF; round up
    // within the buffer and clamp after.
    if (used[idx]) {
      hipError_t e = hipEventSynchronize(ev[idx]);
      if (e != hipSuccess) { err = (int)e; break; }
    }
    uint64_t got = 0;
    while (got < want) {
      uint64_t ask = want - got;
      if (direct) ask = (ask + 4095) & ~4095ull;
      ssize_t r = pread(fd, (char*)g_ring.buf[idx] + got,
                        ask, (off_t)(file_off + off + got));
      if (r < 0) {
        if (errno == EINVAL && direct) {
          // fs refused O_DIRECT mid-stream: reopen buffered
          close(fd);
          fd = open(path, O_RDONLY);
          direct = false;
          if (fd < 0) { err = -errno; break; }
          continue;
        }
        err = -errno;
        break;
      }
      if (r == 0) { err = -EIO; break; }  // short file
      got += (uint64_t)r;
      if (got > want) got = want;  // O_DIRECT tail over-read
    }
    if (err != 0) break;
    hipError_t e = hipMemcpyAsync((char*)d_dst + off, g_ring.buf[idx],
                                  want, hipMemcpyHostToDevice, stream);
    if (e != hipSuccess) { err = (int)e; break; }
    e = hipEventRecord(ev[idx], stream);
    if (e != hipSuccess) { err = (int)e; break; }
    used[idx] = true;
    off += want;
    idx ^= 1;
  }
  if (err == 0) {
    hipError_t e = hipStreamSynchronize(stream);
    if (e != hipSuccess) err = (int)e;
  }
  if (out_seconds) {
    *out_seconds = std::chrono::duration<double>(
        std::chrono::steady_clock::now() - t0).count();
  }
  for (int i = 0; i < kRing; ++i) (void)hipEventDestroy(ev[i]);
  close(fd);
  return err;
}
