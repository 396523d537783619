// Paged-KV-cache maintenance kernels for MI355X (gfx950).
//
// Cache layout (chosen for decode-attention coalescing on 64-lane waves):
//   k_cache / v_cache: [num_blocks, num_kv_heads, block_size, head_dim] bf16
// so one token-row of one head is head_dim*2 bytes contiguous (256 B at
// head_dim=128 = one full LDS/L2 line set; a wave reads it as 16 B/lane x 16
// lanes or 4 B/lane x 64 lanes).
//
// Parity: the reference engine's KV-block gather/scatter;
// P->D migration itself is hipMemcpyPeerAsync host-side (bindings.cpp).
#include "common.h"

namespace xllm {

// Scatter new k/v token vectors into the paged cache.
// k: [T, n_kv_heads, head_dim], slot_mapping[t] = block * block_size + off (-1 = skip)
__global__ void reshape_and_cache_kernel(
    unsigned short* __restrict__ k_cache,
    unsigned short* __restrict__ v_cache,
    const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v,
    const long* __restrict__ slot_mapping,
    const int n_kv_heads, const int head_dim, const int block_size,
    const long src_stride) {  // token stride of k/v (strided qkv views)
  const int token = blockIdx.x;
  const long slot = slot_mapping[token];
  if (slot < 0) return;
  const long blk = slot / block_size;
  const int off = (int)(slot % block_size);
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int nwaves = blockDim.x >> 6;

  for (int h = wid; h < n_kv_heads; h += nwaves) {
    const long src = (long)token * src_stride + (long)h * head_dim;
    const long dst = (((blk * n_kv_heads + h) * block_size) + off) * head_dim;
    // head_dim multiple of 8: lane moves 8 bf16 = 16 B
    for (int i = lane * 8; i < head_dim; i += 64 * 8) {
      *reinterpret_cast<ushort8_t*>(k_cache + dst + i) =
          *reinterpret_cast<const ushort8_t*>(k + src + i);
      *reinterpret_cast<ushort8_t*>(v_cache + dst + i) =
          *reinterpret_cast<const ushort8_t*>(v + src + i);
    }
  }
}

void launch_reshape_and_cache(unsigned short* k_cache, unsigned short* v_cache,
                              const unsigned short* k, const unsigned short* v,
                              const long* slot_mapping, int T, int n_kv_heads,
                              int head_dim, int block_size, long src_stride,
                              hipStream_t stream) {
  dim3 grid(T), block(256);
  hipLaunchKernelGGL(reshape_and_cache_kernel, grid, block, 0, stream, k_cache,
                     v_cache, k, v, slot_mapping, n_kv_heads, head_dim,
                     block_size, src_stride);
}

// Copy whole KV blocks within one device (cache defrag, swap, COW fork).
// pairs: [N, 2] (src_block, dst_block), applied to both k and v caches.
__global__ void copy_blocks_kernel(
    unsigned short* __restrict__ k_cache,
    unsigned short* __restrict__ v_cache,
    const long* __restrict__ pairs,
    const long block_numel) {  // n_kv_heads * block_size * head_dim
  const long src = pairs[blockIdx.x * 2];
  const long dst = pairs[blockIdx.x * 2 + 1];
  unsigned short* ks = k_cache + src * block_numel;
  unsigned short* kd = k_cache + dst * block_numel;
  unsigned short* vs = v_cache + src * block_numel;
  unsigned short* vd = v_cache + dst * block_numel;
  for (long i = (long)threadIdx.x * 8; i < block_numel; i += (long)blockDim.x * 8) {
    *reinterpret_cast<ushort8_t*>(kd + i) = *reinterpret_cast<ushort8_t*>(ks + i);
    *reinterpret_cast<ushort8_t*>(vd + i) = *reinterpret_cast<ushort8_t*>(vs + i);
  }
}

void launch_copy_blocks(unsigned short* k_cache, unsigned short* v_cache,
                        const long* pairs, int n_pairs, long block_numel,
                        hipStream_t stream) {
  if (n_pairs == 0) return;
  dim3 grid(n_pairs), block(256);
  hipLaunchKernelGGL(copy_blocks_kernel, grid, block, 0, stream, k_cache,
                     v_cache, pairs, block_numel);
}

// Gather scattered cache blocks into a contiguous staging buffer (and the
// inverse scatter) — the two halves of cross-GPU KV migration when the block
// lists are not contiguous: gather -> hipMemcpyPeerAsync -> scatter.
template <bool GATHER>
__global__ void gather_scatter_blocks_kernel(
    unsigned short* __restrict__ staging,  // [N, block_numel]
    unsigned short* __restrict__ cache,    // [num_blocks, block_numel]
    const long* __restrict__ block_ids,    // [N]
    const long block_numel) {
  const long cache_off = block_ids[blockIdx.x] * block_numel;
  const long stage_off = (long)blockIdx.x * block_numel;
  for (long i = (long)threadIdx.x * 8; i < block_numel; i += (long)blockDim.x * 8) {
    if constexpr (GATHER)
      *reinterpret_cast<ushort8_t*>(staging + stage_off + i) =
          *reinterpret_cast<ushort8_t*>(cache + cache_off + i);
    else
      *reinterpret_cast<ushort8_t*>(cache + cache_off + i) =
          *reinterpret_cast<ushort8_t*>(staging + stage_off + i);
  }
}

void launch_gather_blocks(unsigned short* staging, unsigned short* cache,
                          const long* block_ids, int n, long block_numel,
                          bool gather, hipStream_t stream) {
  if (n == 0) return;
  dim3 grid(n), block(256);
  if (gather)
    hipLaunchKernelGGL((gather_scatter_blocks_kernel<true>), grid, block, 0,
                       stream, staging, cache, block_ids, block_numel);
  else
    hipLaunchKernelGGL((gather_scatter_blocks_kernel<false>), grid, block, 0,
                       stream, staging, cache, block_ids, block_numel);
}

}  // namespace xllm
