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

}  // namespace xllm
