// Fused activation kernels (SwiGLU) for MI355X (gfx950).
//
// out[t, i] = silu(x[t, i]) * x[t, I + i]  — one fused pass, bf16x8 vectorized.
#include "common.h"

namespace xllm {

__global__ void silu_and_mul_kernel(
    unsigned short* __restrict__ out,      // [T, I]
    const unsigned short* __restrict__ x,  // [T, 2*I]
    const int I) {
  const int row = blockIdx.x;
  const unsigned short* gate = x + (long)row * 2 * I;
  const unsigned short* up = gate + I;
  unsigned short* o = out + (long)row * I;
  for (int i = threadIdx.x * 8; i < I; i += blockDim.x * 8) {
    ushort8_t g = *reinterpret_cast<const ushort8_t*>(gate + i);
    ushort8_t u = *reinterpret_cast<const ushort8_t*>(up + i);
    ushort8_t r;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float gf = bf16_to_f32(g.x[j]);
      float uf = bf16_to_f32(u.x[j]);
      float silu = gf / (1.0f + __expf(-gf));
      r.x[j] = f32_to_bf16(silu * uf);
    }
    *reinterpret_cast<ushort8_t*>(o + i) = r;
  }
}

void launch_silu_and_mul(unsigned short* out, const unsigned short* x, int T,
                         int I, hipStream_t stream) {
  dim3 grid(T), block(256);
  hipLaunchKernelGGL(silu_and_mul_kernel, grid, block, 0, stream, out, x, I);
}

// GELU (tanh approx) * mul — used by some model families (e.g. vision MLPs).
__global__ void gelu_and_mul_kernel(
    unsigned short* __restrict__ out,
    const unsigned short* __restrict__ x,
    const int I) {
  const int row = blockIdx.x;
  const unsigned short* gate = x + (long)row * 2 * I;
  const unsigned short* up = gate + I;
  unsigned short* o = out + (long)row * I;
  for (int i = threadIdx.x * 8; i < I; i += blockDim.x * 8) {
    ushort8_t g = *reinterpret_cast<const ushort8_t*>(gate + i);
    ushort8_t u = *reinterpret_cast<const ushort8_t*>(up + i);
    ushort8_t r;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float gf = bf16_to_f32(g.x[j]);
      float uf = bf16_to_f32(u.x[j]);
      float t = tanhf(0.7978845608028654f * (gf + 0.044715f * gf * gf * gf));
      r.x[j] = f32_to_bf16(0.5f * gf * (1.0f + t) * uf);
    }
    *reinterpret_cast<ushort8_t*>(o + i) = r;
  }
}

void launch_gelu_and_mul(unsigned short* out, const unsigned short* x, int T,
                         int I, hipStream_t stream) {
  dim3 grid(T), block(256);
  hipLaunchKernelGGL(gelu_and_mul_kernel, grid, block, 0, stream, out, x, I);
}

}  // namespace xllm
