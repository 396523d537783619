// Paged-attention DECODE kernel for MI355X (gfx950 / CDNA4).
//
// Memory-bound regime: one new query token per sequence reads the whole KV
// cache. The design streams K/V at wide coalesced granularity and keeps the
// whole online-softmax state in registers:
//
//   * one workgroup per (sequence, kv_head); 256 threads = 4 waves
//   * GQA: all G = n_qheads / n_kv_heads query heads of the kv_head are
//     processed together, so K/V are read ONCE for the whole group
//   * a wave processes one 16-token KV block per iteration: the 64 lanes are
//     split into 4 x 16-lane groups, each group owning one token; a lane
//     loads 8 bf16 (16 B) of the token's 128-dim row -> full 256-B coalesced
//     row per group, 1 KiB per wave instruction
//   * scores reduce within the 16-lane group (4 shfl_xor hops)
//   * each 16-lane group keeps independent online-softmax state (m, l,
//     acc[G][8]); the 16 partials (4 waves x 4 groups) merge through LDS at
//     the end (flash-decoding style merge)
//
// Parity: reference engine's paged-attention decode,
// re-designed for 64-lane waves + 8 TB/s HBM3E rather than ported.
#include "common.h"

namespace xllm {

#define PA_BLOCK_SIZE 16   // tokens per KV block (engine-wide constant)
#define PA_MAX_D 128       // max head_dim supported by this kernel

template <int G>
__global__ __launch_bounds__(256) void paged_attn_decode_kernel(
    unsigned short* __restrict__ out,        // [num_seqs, n_qheads, D] bf16
    const unsigned short* __restrict__ q,    // [num_seqs, n_qheads, D] bf16
    const unsigned short* __restrict__ k_cache,  // [blocks, n_kv, BS, D]
    const unsigned short* __restrict__ v_cache,
    const int* __restrict__ block_tables,    // [num_seqs, max_blocks]
    const int* __restrict__ seq_lens,        // [num_seqs]
    const float scale,
    const int n_kv_heads, const int D, const int max_blocks_per_seq,
    const long q_stride, const long out_stride) {
  const int seq = blockIdx.x / n_kv_heads;
  const int kvh = blockIdx.x % n_kv_heads;
  const int n_qheads = n_kv_heads * G;
  const int seq_len = seq_lens[seq];
  const int n_blocks = (seq_len + PA_BLOCK_SIZE - 1) / PA_BLOCK_SIZE;

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int tg = lane >> 4;    // 16-lane group (token slot)
  const int d8 = lane & 15;    // dim octet: this lane covers dims [8*d8, 8*d8+8)
  const int dvalid = (8 * d8) < D;  // lanes beyond head_dim idle (D<128)

  // ---- load q fragments (shared across all KV) --------------------------------
  float qf[G][8];
#pragma unroll
  for (int g = 0; g < G; g++) {
    const long qoff = (long)seq * q_stride + (long)(kvh * G + g) * D + 8 * d8;
    if (dvalid) {
      ushort8_t v = *reinterpret_cast<const ushort8_t*>(q + qoff);
#pragma unroll
      for (int j = 0; j < 8; j++) qf[g][j] = bf16_to_f32(v.x[j]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; j++) qf[g][j] = 0.0f;
    }
  }

  // ---- stream KV blocks, online softmax ---------------------------------------
  float m[G], l[G], acc[G][8];
#pragma unroll
  for (int g = 0; g < G; g++) {
    m[g] = -INFINITY;
    l[g] = 0.0f;
#pragma unroll
    for (int j = 0; j < 8; j++) acc[g][j] = 0.0f;
  }

  const int* btab = block_tables + (long)seq * max_blocks_per_seq;
  const long kv_head_stride = (long)PA_BLOCK_SIZE * D;

  for (int bi = wid; bi < n_blocks; bi += 4) {
    const long phys = btab[bi];
    const long base = (phys * n_kv_heads + kvh) * kv_head_stride;
#pragma unroll
    for (int j = 0; j < 4; j++) {
      const int t = j * 4 + tg;  // token within block, one per 16-lane group
      const int tok = bi * PA_BLOCK_SIZE + t;
      if (tok >= seq_len) continue;  // group-uniform branch
      const long roff = base + (long)t * D + 8 * d8;
      float kf[8];
      if (dvalid) {
        ushort8_t kv8 = *reinterpret_cast<const ushort8_t*>(k_cache + roff);
#pragma unroll
        for (int u = 0; u < 8; u++) kf[u] = bf16_to_f32(kv8.x[u]);
      } else {
#pragma unroll
        for (int u = 0; u < 8; u++) kf[u] = 0.0f;
      }
      float vf[8];
      if (dvalid) {
        ushort8_t vv8 = *reinterpret_cast<const ushort8_t*>(v_cache + roff);
#pragma unroll
        for (int u = 0; u < 8; u++) vf[u] = bf16_to_f32(vv8.x[u]);
      } else {
#pragma unroll
        for (int u = 0; u < 8; u++) vf[u] = 0.0f;
      }
#pragma unroll
      for (int g = 0; g < G; g++) {
        float s = 0.0f;
#pragma unroll
        for (int u = 0; u < 8; u++) s += qf[g][u] * kf[u];
        s = group_reduce_sum<16>(s) * scale;  // uniform across the 16 lanes
        const float m_new = fmaxf(m[g], s);
        const float alpha = __expf(m[g] - m_new);  // exp(-inf)=0 on first hit
        const float p = __expf(s - m_new);
        m[g] = m_new;
        l[g] = l[g] * alpha + p;
#pragma unroll
        for (int u = 0; u < 8; u++) acc[g][u] = acc[g][u] * alpha + p * vf[u];
      }
    }
  }

  // ---- merge the 16 partials through LDS --------------------------------------
  // layout: ml[16][G][2], accs[16][G][16 lanes][8]
  __shared__ float ml[16][G][2];
  __shared__ float accs[16][G][16][8];
  const int part = wid * 4 + tg;
  if (d8 == 0) {
#pragma unroll
    for (int g = 0; g < G; g++) {
      ml[part][g][0] = m[g];
      ml[part][g][1] = l[g];
    }
  }
#pragma unroll
  for (int g = 0; g < G; g++) {
#pragma unroll
    for (int u = 0; u < 8; u++) accs[part][g][d8][u] = acc[g][u];
  }
  __syncthreads();

  // wave `wid` merges heads g = wid, wid+4, ... ; lanes cover the 128 dims
  for (int g = wid; g < G; g += 4) {
    float m_star = -INFINITY;
#pragma unroll
    for (int p = 0; p < 16; p++) m_star = fmaxf(m_star, ml[p][g][0]);
    float l_tot = 0.0f;
    float o0 = 0.0f, o1 = 0.0f;  // lane covers dims (2*lane, 2*lane+1)
    const int pd8 = lane >> 2;         // octet holding dim 2*lane (=2*lane/8)
    const int pu = (2 * lane) & 7;     // position within octet
#pragma unroll
    for (int p = 0; p < 16; p++) {
      const float w = __expf(ml[p][g][0] - m_star);  // 0 if m=-inf
      l_tot += w * ml[p][g][1];
      o0 += w * accs[p][g][pd8][pu];
      o1 += w * accs[p][g][pd8][pu + 1];
    }
    const float inv = (l_tot > 0.0f) ? 1.0f / l_tot : 0.0f;
    if (2 * lane < D) {
      const long ooff = (long)seq * out_stride + (long)(kvh * G + g) * D + 2 * lane;
      out[ooff] = f32_to_bf16(o0 * inv);
      out[ooff + 1] = f32_to_bf16(o1 * inv);
    }
  }
}

#define PA_DISPATCH_G(GV)                                                     \
  hipLaunchKernelGGL((paged_attn_decode_kernel<GV>), grid, block, 0, stream,  \
                     out, q, k_cache, v_cache, block_tables, seq_lens, scale, \
                     n_kv_heads, D, max_blocks_per_seq, q_stride, out_stride)

void launch_paged_attn_decode(unsigned short* out, const unsigned short* q,
                              const unsigned short* k_cache,
                              const unsigned short* v_cache,
                              const int* block_tables, const int* seq_lens,
                              float scale, int num_seqs, int n_qheads,
                              int n_kv_heads, int D, int max_blocks_per_seq,
                              long q_stride, long out_stride,
                              hipStream_t stream) {
  dim3 grid(num_seqs * n_kv_heads), block(256);
  const int G = n_qheads / n_kv_heads;
  switch (G) {
    case 1: PA_DISPATCH_G(1); break;
    case 2: PA_DISPATCH_G(2); break;
    case 3: PA_DISPATCH_G(3); break;
    case 4: PA_DISPATCH_G(4); break;
    case 5: PA_DISPATCH_G(5); break;
    case 6: PA_DISPATCH_G(6); break;
    case 7: PA_DISPATCH_G(7); break;
    case 8: PA_DISPATCH_G(8); break;
    default: break;  // validated host-side
  }
}

}  // namespace xllm
