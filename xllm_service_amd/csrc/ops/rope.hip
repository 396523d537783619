// Fused rotary position embedding (RoPE) for MI355X (gfx950).
//
// NeoX / Llama "rotate-half" layout, applied in-place to q and k, which may
// be strided row views into a fused qkv projection output (no host-side
// .contiguous() copies). One 64-lane wave per (token, head) pair; the grid
// is sized by total wave-work so decode batches fill the chip.
//
// cos/sin precomputed host-side as f32 [max_pos, rot_dim] (first half cos):
// on-device trig would turn this memory-bound op VALU-bound (guide App. B).
//
// Parity: the reference engine's fused RoPE.
#include "common.h"

namespace xllm {

__global__ void rope_kernel(
    unsigned short* __restrict__ q,      // [T, n_qheads, D] rows stride q_stride
    unsigned short* __restrict__ k,      // [T, n_kheads, D] rows stride k_stride
    const long* __restrict__ positions,  // [T]
    const float* __restrict__ cos_sin,   // [max_pos, rot_dim]
    const int n_qheads, const int n_kheads,
    const int head_dim, const int rot_dim,
    const long q_stride, const long k_stride,
    const int T) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int waves_per_block = blockDim.x >> 6;
  const int total_heads = n_qheads + n_kheads;
  const long total_work = (long)T * total_heads;
  const int half = rot_dim / 2;

  for (long wk = blockIdx.x * (long)waves_per_block + wid; wk < total_work;
       wk += (long)gridDim.x * waves_per_block) {
    const int token = (int)(wk / total_heads);
    const int h = (int)(wk % total_heads);
    unsigned short* base =
        (h < n_qheads) ? q + token * q_stride + (long)h * head_dim
                       : k + token * k_stride + (long)(h - n_qheads) * head_dim;
    const float* cs = cos_sin + positions[token] * rot_dim;
    for (int i = lane; i < half; i += 64) {
      const float c = cs[i];
      const float s = cs[half + i];
      const float x1 = bf16_to_f32(base[i]);
      const float x2 = bf16_to_f32(base[i + half]);
      base[i] = f32_to_bf16(x1 * c - x2 * s);
      base[i + half] = f32_to_bf16(x2 * c + x1 * s);
    }
  }
}

void launch_rope(unsigned short* q, unsigned short* k, const long* positions,
                 const float* cos_sin, int T, int n_qheads, int n_kheads,
                 int head_dim, int rot_dim, long q_stride, long k_stride,
                 hipStream_t stream) {
  const long waves = (long)T * (n_qheads + n_kheads);
  long g = (waves + 3) / 4;  // 4 waves per 256-thread block
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  hipLaunchKernelGGL(rope_kernel, dim3((unsigned)g), dim3(256), 0, stream, q,
                     k, positions, cos_sin, n_qheads, n_kheads, head_dim,
                     rot_dim, q_stride, k_stride, T);
}

// Fused RoPE + KV-cache scatter: one pass ropes q in place, ropes k and
// writes it straight into the paged cache together with v (replaces the
// separate rope + reshape_and_cache launches and k's LDS->HBM round trip).
__global__ void fused_rope_cache_kernel(
    unsigned short* __restrict__ q,            // [T, n_q, D] strided rows
    const unsigned short* __restrict__ k,      // [T, n_kv, D] strided rows
    const unsigned short* __restrict__ v,      // [T, n_kv, D] strided rows
    unsigned short* __restrict__ k_cache,      // [blocks, n_kv, bs, D]
    unsigned short* __restrict__ v_cache,
    const long* __restrict__ positions,  // [T], or [3, T] when ms0 > 0
    const long* __restrict__ slot_mapping,     // [T] (-1 = skip cache write)
    const float* __restrict__ cos_sin,         // [max_pos, rot]
    const int n_q, const int n_kv, const int D, const int rot,
    const long q_stride, const long kv_stride, const int block_size,
    const int T,
    // M-RoPE (Qwen2-VL): cumulative frequency-section bounds; freq i takes
    // its position from row 0 (i < ms0: temporal), 1 (i < ms1: height) or
    // 2 (width) of the [3, T] positions. ms0 == 0 disables (1-D rope).
    const int ms0, const int ms1) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = blockDim.x >> 6;
  const int total_heads = n_q + 2 * n_kv;  // q-rope | k-rope+store | v-store
  const long total_work = (long)T * total_heads;
  const int half = rot / 2;

  for (long wk = blockIdx.x * (long)wpb + wid; wk < total_work;
       wk += (long)gridDim.x * wpb) {
    const int token = (int)(wk / total_heads);
    const int h = (int)(wk % total_heads);
    const long slot = slot_mapping[token];
    const float* cs = cos_sin + positions[token] * rot;
    auto cs_at = [&](int i) -> const float* {
      if (ms0 == 0) return cs;
      const int sec = i < ms0 ? 0 : (i < ms1 ? 1 : 2);
      return cos_sin + positions[(long)sec * T + token] * rot;
    };
    if (h < n_q) {                       // rope q in place
      unsigned short* base = q + token * q_stride + (long)h * D;
      for (int i = lane; i < half; i += 64) {
        const float* cf = cs_at(i);
        const float c = cf[i], s = cf[half + i];
        const float x1 = bf16_to_f32(base[i]);
        const float x2 = bf16_to_f32(base[i + half]);
        base[i] = f32_to_bf16(x1 * c - x2 * s);
        base[i + half] = f32_to_bf16(x2 * c + x1 * s);
      }
    } else if (h < n_q + n_kv) {         // rope k -> cache
      if (slot < 0) continue;
      const int kh = h - n_q;
      const unsigned short* src = k + token * kv_stride + (long)kh * D;
      unsigned short* dst =
          k_cache + (((slot / block_size) * n_kv + kh) * block_size +
                     slot % block_size) * (long)D;
      for (int i = lane; i < half; i += 64) {
        const float* cf = cs_at(i);
        const float c = cf[i], s = cf[half + i];
        const float x1 = bf16_to_f32(src[i]);
        const float x2 = bf16_to_f32(src[i + half]);
        dst[i] = f32_to_bf16(x1 * c - x2 * s);
        dst[i + half] = f32_to_bf16(x2 * c + x1 * s);
      }
      for (int i = rot + lane; i < D; i += 64) dst[i] = src[i];
    } else {                             // copy v -> cache (16 B vectors)
      if (slot < 0) continue;
      const int vh = h - n_q - n_kv;
      const unsigned short* src = v + token * kv_stride + (long)vh * D;
      unsigned short* dst =
          v_cache + (((slot / block_size) * n_kv + vh) * block_size +
                     slot % block_size) * (long)D;
      for (int i = lane * 8; i < D; i += 64 * 8)
        *reinterpret_cast<ushort8_t*>(dst + i) =
            *reinterpret_cast<const ushort8_t*>(src + i);
    }
  }
}

void launch_fused_rope_cache(unsigned short* q, const unsigned short* k,
                             const unsigned short* v, unsigned short* k_cache,
                             unsigned short* v_cache, const long* positions,
                             const long* slot_mapping, const float* cos_sin,
                             int T, int n_q, int n_kv, int D, int rot,
                             long q_stride, long kv_stride, int block_size,
                             int ms0, int ms1, hipStream_t stream) {
  const long waves = (long)T * (n_q + 2 * n_kv);
  long g = (waves + 3) / 4;
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  hipLaunchKernelGGL(fused_rope_cache_kernel, dim3((unsigned)g), dim3(256), 0,
                     stream, q, k, v, k_cache, v_cache, positions,
                     slot_mapping, cos_sin, n_q, n_kv, D, rot, q_stride,
                     kv_stride, block_size, T, ms0, ms1);
}

}  // namespace xllm
