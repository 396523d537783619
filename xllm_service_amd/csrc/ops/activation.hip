// Fused activation kernels (SwiGLU / GEGLU) for MI355X (gfx950).
//
// out[t, i] = act(x[t, i]) * x[t, I + i] — one fused pass, bf16x8
// vectorized, flattened grid-stride over (t, i) so small decode batches
// still fill the chip (grid is sized by total work, not by rows).
#include "common.h"

namespace xllm {

template <bool GELU>
__global__ void act_and_mul_kernel(
    unsigned short* __restrict__ out,      // [T, I]
    const unsigned short* __restrict__ x,  // [T, 2*I]
    const int I, const long total_vec) {   // total_vec = T * I/8
  const int iv = I / 8;
  for (long v = blockIdx.x * (long)blockDim.x + threadIdx.x; v < total_vec;
       v += (long)gridDim.x * blockDim.x) {
    const long row = v / iv;
    const int col = (int)(v % iv) * 8;
    ushort8_t g = *reinterpret_cast<const ushort8_t*>(x + row * 2 * I + col);
    ushort8_t u = *reinterpret_cast<const ushort8_t*>(x + row * 2 * I + I + col);
    ushort8_t r;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float gf = bf16_to_f32(g.x[j]);
      float uf = bf16_to_f32(u.x[j]);
      float a;
      if constexpr (GELU) {
        float t = tanhf(0.7978845608028654f * (gf + 0.044715f * gf * gf * gf));
        a = 0.5f * gf * (1.0f + t);
      } else {
        a = gf / (1.0f + __expf(-gf));
      }
      r.x[j] = f32_to_bf16(a * uf);
    }
    *reinterpret_cast<ushort8_t*>(out + row * I + col) = r;
  }
}

static inline dim3 act_grid(long total_vec) {
  long g = (total_vec + 255) / 256;
  if (g > 2048) g = 2048;  // grid-stride the rest (Guideline 11)
  if (g < 1) g = 1;
  return dim3((unsigned)g);
}

void launch_silu_and_mul(unsigned short* out, const unsigned short* x, int T,
                         int I, hipStream_t stream) {
  const long total = (long)T * (I / 8);
  hipLaunchKernelGGL((act_and_mul_kernel<false>), act_grid(total), dim3(256),
                     0, stream, out, x, I, total);
}

void launch_gelu_and_mul(unsigned short* out, const unsigned short* x, int T,
                         int I, hipStream_t stream) {
  const long total = (long)T * (I / 8);
  hipLaunchKernelGGL((act_and_mul_kernel<true>), act_grid(total), dim3(256),
                     0, stream, out, x, I, total);
}

}  // namespace xllm
