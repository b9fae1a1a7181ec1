// LZ4 block decompression for MI355X (gfx950).
//
// Role in the framework: the cascade-analogue image replicator and the
// shard stager store container layers / data shards as sequences of
// independently-compressed LZ4 blocks (raw block format, <= 64 KiB raw
// per block).  The reference delegates layer decompress to dockerd
// (reference cascade/cascade.py:500-571); here decode runs on the GPU so
// a layer goes NVMe -> pinned host -> HBM (hipMemcpyAsync) -> decoded in
// HBM without a CPU inflate pass.
//
// Design (CDNA4-first):
//  * One wave (64 lanes) per block.  Lane 0 parses the sequence stream
//    (token / LSIC lengths / offset) — parsing is inherently serial —
//    and broadcasts via __shfl; all 64 lanes then copy literals and
//    matches cooperatively.
//  * Output is staged in LDS (64 KiB per wave), not global memory:
//    match copies read bytes written by *other lanes* in previous
//    rounds, and LDS ordering via s_waitcnt lgkmcnt(0) is cheap and
//    wave-local, where global-memory ordering would need vmcnt drains
//    through L2.  The decoded block then streams LDS -> HBM with
//    coalesced 16 B stores — scattered byte writes never touch HBM.
//  * Overlapping matches (offset < length) use the doubling schedule:
//    round r may copy min(remaining, offset + done) bytes in parallel,
//    so rounds grow geometrically instead of byte-serial.
//  * LDS budget 64 KiB -> 2 concurrent blocks per CU (160 KiB LDS/CU);
//    grid = n_blocks waves; 512 blocks in flight across 256 CUs.  This
//    kernel is latency/parse bound per block, so throughput comes from
//    block-level parallelism, which container layers have in abundance.
//
// Workgroup = 1 wave (64 threads) to keep the LDS tile per-wave private
// (no __syncthreads needed across waves).

#include "common.h"

namespace {

constexpr int kBlockRaw = 64 * 1024;  // max raw bytes per LZ4 block

// status codes per block
enum : uint32_t {
  SY_LZ4_OK = 0,
  SY_LZ4_ERR_OFFSET = 1,    // match offset reaches before block start
  SY_LZ4_ERR_OVERFLOW = 2,  // output exceeded declared raw size
  SY_LZ4_ERR_TRUNC = 3,     // input ran out mid-sequence
  SY_LZ4_ERR_MISMATCH = 4,  // decoded size != declared raw size
};

__global__ __launch_bounds__(SY_WAVE) void lz4_decode_kernel(
    const uint8_t* __restrict__ comp, const uint64_t* __restrict__ in_off,
    const uint32_t* __restrict__ in_len, uint8_t* __restrict__ out,
    const uint64_t* __restrict__ out_off, const uint32_t* __restrict__ out_len,
    uint32_t* __restrict__ status, uint32_t n_blocks) {
  __shared__ uint8_t dst[kBlockRaw];

  const int lane = threadIdx.x;

  for (uint32_t blk = blockIdx.x; blk < n_blocks; blk += gridDim.x) {
    const uint8_t* src = comp + in_off[blk];
    const uint32_t slen = in_len[blk];
    const uint32_t rawlen = out_len[blk];
    uint32_t st = SY_LZ4_OK;

    // Lane 0 parses; sequence vars are broadcast each iteration.
    uint32_t pos = 0;   // input cursor
    uint32_t dpos = 0;  // output cursor
    for (;;) {
      uint32_t litlen = 0, mlen = 0, offset = 0, stop = 0;
      if (lane == 0) {
        if (pos >= slen) {
          stop = 1;
          if (dpos != rawlen) st = SY_LZ4_ERR_MISMATCH;
        } else {
          const uint32_t token = src[pos++];
          litlen = token >> 4;
          if (litlen == 15) {
            uint8_t b;
            do {
              if (pos >= slen) { st = SY_LZ4_ERR_TRUNC; break; }
              b = src[pos++];
              litlen += b;
            } while (b == 255);
          }
          if (st == SY_LZ4_OK) {
            if (dpos + litlen > rawlen || pos + litlen > slen) {
              st = SY_LZ4_ERR_OVERFLOW;
            } else {
              mlen = (token & 0xfu) + 4;
              // a block's last sequence is literals-only
              if (pos + litlen == slen) {
                stop = 2;  // copy literals then stop
              }
            }
          }
        }
      }
      st = __shfl(st, 0);
      if (st != SY_LZ4_OK) break;
      stop = __shfl(stop, 0);
      if (stop == 1) break;
      litlen = __shfl(litlen, 0);
      pos = __shfl(pos, 0);
      dpos = __shfl(dpos, 0);

      // ---- cooperative literal copy: global src -> LDS dst ----
      for (uint32_t i = lane; i < litlen; i += SY_WAVE) {
        dst[dpos + i] = src[pos + i];
      }
      pos += litlen;
      dpos += litlen;
      if (stop == 2) {
        if (lane == 0 && dpos != rawlen) st = SY_LZ4_ERR_MISMATCH;
        st = __shfl(st, 0);
        break;
      }

      if (lane == 0) {
        if (pos + 2 > slen) {
          st = SY_LZ4_ERR_TRUNC;
        } else {
          offset = (uint32_t)src[pos] | ((uint32_t)src[pos + 1] << 8);
          pos += 2;
          uint32_t ml = mlen;  // token nibble + 4, merged above
          if ((ml - 4) == 15) {
            uint8_t b;
            do {
              if (pos >= slen) { st = SY_LZ4_ERR_TRUNC; break; }
              b = src[pos++];
              ml += b;
            } while (b == 255);
          }
          mlen = ml;
          if (st == SY_LZ4_OK) {
            if (offset == 0 || offset > dpos) st = SY_LZ4_ERR_OFFSET;
            else if (dpos + mlen > rawlen) st = SY_LZ4_ERR_OVERFLOW;
          }
        }
      }
      st = __shfl(st, 0);
      if (st != SY_LZ4_OK) break;
      offset = __shfl(offset, 0);
      mlen = __shfl(mlen, 0);
      pos = __shfl(pos, 0);

      // ---- cooperative match copy with doubling over overlap ----
      // The decoded match is periodic with period `offset`.  Round r
      // may copy up to dist = done + offset bytes by reading at stride
      // -dist: dist is a multiple of offset by induction (offset, 2o,
      // 4o, ...), so dst[j - dist] == dst[j - offset], and every read
      // lands strictly below `done` (a completed round) — no
      // intra-round hazard.  Round sizes double: O(log(mlen/offset))
      // rounds instead of byte-serial.
      uint32_t done = 0;
      while (done < mlen) {
        const uint32_t dist = done + offset;
        const uint32_t n = min(mlen - done, dist);
        for (uint32_t i = lane; i < n; i += SY_WAVE) {
          dst[dpos + done + i] = dst[dpos + done + i - dist];
        }
        done += n;
        // order LDS writes before next round's cross-lane reads
        __builtin_amdgcn_s_waitcnt(0 /* vmcnt=0 lgkmcnt=0 ... */);
      }
      dpos += mlen;
    }

    if (lane == 0) status[blk] = st;
    if (st == SY_LZ4_OK) {
      // ---- stream LDS -> HBM, coalesced 16 B per lane ----
      uint8_t* g = out + out_off[blk];
      uint32_t n16 = rawlen >> 4;
      const uint4* s4 = reinterpret_cast<const uint4*>(dst);
      uint4* g4 = reinterpret_cast<uint4*>(g);  // out_off 16B-aligned (host)
      for (uint32_t i = lane; i < n16; i += SY_WAVE) g4[i] = s4[i];
      for (uint32_t i = (n16 << 4) + lane; i < rawlen; i += SY_WAVE)
        g[i] = dst[i];
    }
    // next loop iteration reuses dst; waves are independent, but lanes
    // must finish reading before any rewrite — wave lockstep plus the
    // waitcnt above suffices (single-wave workgroup).
    __builtin_amdgcn_s_waitcnt(0);
  }
}

}  // namespace

SY_EXPORT int sy_lz4_decode_blocks(const void* d_comp, const uint64_t* d_in_off,
                                   const uint32_t* d_in_len, void* d_out,
                                   const uint64_t* d_out_off,
                                   const uint32_t* d_out_len,
                                   uint32_t* d_status, uint32_t n_blocks,
                                   hipStream_t stream) {
  if (n_blocks == 0) return 0;
  uint32_t grid = n_blocks < 4096u ? n_blocks : 4096u;
  hipLaunchKernelGGL(lz4_decode_kernel, dim3(grid), dim3(SY_WAVE), 0, stream,
                     static_cast<const uint8_t*>(d_comp), d_in_off, d_in_len,
                     static_cast<uint8_t*>(d_out), d_out_off, d_out_len,
                     d_status, n_blocks);
  return sy_check(hipGetLastError());
}
