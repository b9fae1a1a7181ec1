// Batched gather-copy for STORED shard blocks (gfx950).
//
// SYSHARD stores incompressible blocks raw; decoding a mixed shard
// previously moved each stored block with a separate python-side
// device slice copy — at 4-32 KiB blocks the per-block host overhead
// dominated the whole stage pipeline (measured: 512 MB mixed corpus
// staged at 1.2-4 GB/s, bounded by O(n_blocks) python, not the bus).
// One kernel moves every stored block: one workgroup per block,
// uint4-vectorized body + byte tail.  comp offsets are 16 B aligned by
// the format; dst offsets are block_raw-aligned (block_raw is a 4 KiB
// multiple) except the final short block, whose tail loop handles any
// length.  (Reference analogue: none — dockerd inflates layers on the
// CPU, reference cascade/cascade.py:500-571.)

#include "common.h"

namespace {

__global__ __launch_bounds__(256) void gather_copy_kernel(
    const uint8_t* __restrict__ src, const uint64_t* __restrict__ src_off,
    uint8_t* __restrict__ dst, const uint64_t* __restrict__ dst_off,
    const uint32_t* __restrict__ len, uint32_t n_blocks) {
  for (uint32_t b = blockIdx.x; b < n_blocks; b += gridDim.x) {
    const uint8_t* s = src + src_off[b];
    uint8_t* d = dst + dst_off[b];
    const uint32_t n = len[b];
    const uint32_t n16 = n >> 4;
    const uint4* s4 = reinterpret_cast<const uint4*>(s);
    uint4* d4 = reinterpret_cast<uint4*>(d);
    for (uint32_t i = threadIdx.x; i < n16; i += blockDim.x) d4[i] = s4[i];
    for (uint32_t i = (n16 << 4) + threadIdx.x; i < n; i += blockDim.x)
      d[i] = s[i];
  }
}

}  // namespace

SY_EXPORT int sy_gather_copy(const void* d_src, const uint64_t* d_src_off,
                             void* d_dst, const uint64_t* d_dst_off,
                             const uint32_t* d_len, uint32_t n_blocks,
                             hipStream_t stream) {
  if (n_blocks == 0) return 0;
  uint32_t grid = n_blocks < (1u << 20) ? n_blocks : (1u << 20);
  hipLaunchKernelGGL(gather_copy_kernel, dim3(grid), dim3(256), 0, stream,
                     static_cast<const uint8_t*>(d_src), d_src_off,
                     static_cast<uint8_t*>(d_dst), d_dst_off, d_len,
                     n_blocks);
  return sy_check(hipGetLastError());
}
