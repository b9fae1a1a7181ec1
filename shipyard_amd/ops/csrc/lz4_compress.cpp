// CPU LZ4 block compressor (host-side, multi-threaded over blocks).
//
// The SYSHARD authoring path: pack() compresses independent <=64 KiB
// blocks (decoded on the GPU by lz4_decode.hip).  The pure-python
// compressor does ~12 MB/s/core; this greedy hash-table matcher does
// the standard LZ4 block format at memory-ish speed, one thread per
// core across blocks.  (Reference analogue: dockerd's gzip — the
// reference never compresses itself; this framework authors its own
// layer format, so the writer must be fast too.)
//
// Format notes (must stay decodable by lz4_decode.hip + lz4py):
//   sequence = token (lit<<4 | (mlen-4)), LSIC extensions at 15,
//   literals, 2 B little-endian offset (1..65535), match extension.
//   Final sequence is literals-only.  Standard safety margins: last
//   match starts >= 12 B before end, match extension stops 5 B short.

#include <atomic>
#include <cstring>
#include <thread>
#include <vector>

#include "common.h"

namespace {

static inline uint32_t rd32(const uint8_t* p) {
  uint32_t v;
  memcpy(&v, p, 4);
  return v;
}

static inline uint32_t hash4(uint32_t v) {
  return (v * 2654435761u) >> 20;  // 12-bit table
}

// Compress one block; returns compressed size, or 0 when the result
// would not be smaller than n (caller stores the block raw).
static uint32_t compress_block(const uint8_t* src, uint32_t n,
                               uint8_t* dst, uint32_t cap) {
  uint32_t opos = 0;

  auto emit_len = [&](uint32_t len) -> bool {  // LSIC extension
    while (len >= 255) {
      if (opos >= cap) return false;
      dst[opos++] = 255;
      len -= 255;
    }
    if (opos >= cap) return false;
    dst[opos++] = (uint8_t)len;
    return true;
  };

  auto emit_seq = [&](uint32_t anchor, uint32_t lit_end, uint32_t offset,
                      uint32_t mlen) -> bool {
    const uint32_t litlen = lit_end - anchor;
    const uint32_t ml = mlen ? mlen - 4 : 0;
    if (opos >= cap) return false;
    dst[opos++] = (uint8_t)(((litlen < 15 ? litlen : 15) << 4) |
                            (ml < 15 ? ml : 15));
    if (litlen >= 15 && !emit_len(litlen - 15)) return false;
    if (opos + litlen > cap) return false;
    memcpy(dst + opos, src + anchor, litlen);
    opos += litlen;
    if (mlen == 0) return true;  // final literal-only sequence
    if (opos + 2 > cap) return false;
    dst[opos++] = (uint8_t)(offset & 0xFF);
    dst[opos++] = (uint8_t)(offset >> 8);
    if (ml >= 15 && !emit_len(ml - 15)) return false;
    return true;
  };

  if (n >= 13) {
    uint16_t ht[4096];
    memset(ht, 0, sizeof(ht));
    const uint32_t mflimit = n - 12;
    const uint32_t match_limit = n - 5;
    uint32_t anchor = 0, pos = 0;
    while (pos <= mflimit) {
      const uint32_t h = hash4(rd32(src + pos));
      const uint32_t ref = (uint32_t)ht[h];  // pos+1 encoding
      ht[h] = (uint16_t)(pos + 1);
      if (ref != 0 && ref - 1 < pos && pos - (ref - 1) <= 65535 &&
          rd32(src + (ref - 1)) == rd32(src + pos)) {
        const uint32_t r = ref - 1;
        uint32_t mlen = 4;
        while (pos + mlen < match_limit && src[r + mlen] == src[pos + mlen])
          ++mlen;
        if (!emit_seq(anchor, pos, pos - r, mlen)) return 0;
        pos += mlen;
        anchor = pos;
        // re-prime the table inside the skipped span (cheap, helps
        // repetitive data find the NEXT match quickly)
        if (pos <= mflimit) ht[hash4(rd32(src + pos - 2))] =
            (uint16_t)(pos - 1);
      } else {
        ++pos;
      }
    }
    if (!emit_seq(anchor, n, 0, 0)) return 0;
  } else {
    if (!emit_seq(0, n, 0, 0)) return 0;
  }
  return opos < n ? opos : 0;
}

}  // namespace

// dst is n_blocks x stride; out_lens[b] = compressed size or 0 (store
// raw).  threads <= 0 -> hardware_concurrency.
SY_EXPORT int sy_lz4_compress_blocks(const uint8_t* src, uint64_t total,
                                     uint32_t block_raw, uint8_t* dst,
                                     uint64_t stride, uint32_t* out_lens,
                                     uint32_t n_blocks, int threads) {
  if (n_blocks == 0) return 0;
  if (threads <= 0) {
    unsigned hc = std::thread::hardware_concurrency();
    threads = hc ? (int)hc : 4;
  }
  if ((uint32_t)threads > n_blocks) threads = (int)n_blocks;
  std::atomic<uint32_t> next{0};
  auto worker = [&]() {
    for (;;) {
      const uint32_t b = next.fetch_add(1);
      if (b >= n_blocks) return;
      const uint64_t off = (uint64_t)b * block_raw;
      const uint32_t len = (uint32_t)((total - off < block_raw)
                                          ? (total - off) : block_raw);
      out_lens[b] = compress_block(src + off, len, dst + b * stride,
                                   (uint32_t)stride);
    }
  };
  if (threads == 1) {
    worker();
  } else {
    std::vector<std::thread> pool;
    pool.reserve(threads);
    for (int t = 0; t < threads; ++t) pool.emplace_back(worker);
    for (auto& t : pool) t.join();
  }
  return 0;
}
