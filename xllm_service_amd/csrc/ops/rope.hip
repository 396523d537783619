// Fused rotary position embedding (RoPE) for MI355X (gfx950).
//
// NeoX / Llama "rotate-half" layout, applied in-place to q and k.
// cos/sin are precomputed host-side as an f32 table [max_pos, rot_dim]
// (first half cos(theta_i), second half sin(theta_i)): per the CDNA guide,
// on-device sinf/cosf turns this memory-bound op VALU-bound.
//
// Parity target: the reference engine's fused RoPE.
#include "common.h"

namespace xllm {

// q: [T, n_qheads * head_dim], k: [T, n_kheads * head_dim]; rot_dim <= head_dim.
// One wave handles one (token, head): lane i covers the rotation pair
// (i, i + rot/2) for i < rot/2 (rot/2 <= 64 for head_dim up to 128).
__global__ void rope_kernel(
    unsigned short* __restrict__ q,
    unsigned short* __restrict__ k,
    const long* __restrict__ positions,  // [T]
    const float* __restrict__ cos_sin,   // [max_pos, rot_dim]
    const int n_qheads, const int n_kheads,
    const int head_dim, const int rot_dim) {
  const int token = blockIdx.x;
  const int wid = threadIdx.x >> 6;       // wave in block (4 waves)
  const int lane = threadIdx.x & 63;
  const int nwaves = blockDim.x >> 6;
  const int half = rot_dim / 2;
  const long pos = positions[token];
  const float* cs = cos_sin + pos * rot_dim;

  const int total_heads = n_qheads + n_kheads;
  for (int h = wid; h < total_heads; h += nwaves) {
    unsigned short* base =
        (h < n_qheads) ? q + ((long)token * n_qheads + h) * head_dim
                       : k + ((long)token * n_kheads + (h - n_qheads)) * head_dim;
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
                 int head_dim, int rot_dim, hipStream_t stream) {
  dim3 grid(T), block(256);
  hipLaunchKernelGGL(rope_kernel, grid, block, 0, stream, q, k, positions,
                     cos_sin, n_qheads, n_kheads, head_dim, rot_dim);
}

}  // namespace xllm
