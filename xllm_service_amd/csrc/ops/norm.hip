// Fused RMSNorm kernels for MI355X (gfx950).
//
// Memory-bound: target HBM BW. bf16 I/O vectorized 16 B/lane (ushort8),
// fp32 accumulation, one workgroup per row (hidden sizes 1k-16k).
//
// Capability parity: the reference engine's fused RMSNorm (xLLM engine,
// absent submodule) — re-designed CDNA4-native.
#include "common.h"

namespace xllm {

// out[t, :] = x[t, :] / rms(x[t, :]) * w
template <bool FUSED_ADD>
__global__ void rmsnorm_kernel(
    unsigned short* __restrict__ out,       // [T, H] bf16 (= input if FUSED_ADD)
    unsigned short* __restrict__ residual,  // [T, H] bf16 io (FUSED_ADD only)
    const unsigned short* __restrict__ x,   // [T, H] bf16
    const unsigned short* __restrict__ w,   // [H] bf16
    const float eps,
    const int H) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int nthread = blockDim.x;
  __shared__ float red_tmp[8];

  const unsigned short* xr = x + (long)row * H;
  unsigned short* rr = FUSED_ADD ? residual + (long)row * H : nullptr;
  unsigned short* orow = out + (long)row * H;

  float sumsq = 0.0f;
  // pass 1: (optional residual add) + sum of squares; vectorized 8 bf16 = 16 B
  for (int i = tid * 8; i < H; i += nthread * 8) {
    ushort8_t v = *reinterpret_cast<const ushort8_t*>(xr + i);
    if constexpr (FUSED_ADD) {
      ushort8_t r = *reinterpret_cast<ushort8_t*>(rr + i);
#pragma unroll
      for (int j = 0; j < 8; j++) {
        float f = bf16_to_f32(v.x[j]) + bf16_to_f32(r.x[j]);
        v.x[j] = f32_to_bf16(f);
      }
      // write the new residual back (residual stream carries the sum)
      *reinterpret_cast<ushort8_t*>(rr + i) = v;
    }
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float f = bf16_to_f32(v.x[j]);
      sumsq += f * f;
    }
    if constexpr (FUSED_ADD) {
      // stash the summed row in out so pass 2 reads it from there (L2-hot)
      *reinterpret_cast<ushort8_t*>(orow + i) = v;
    }
  }
  sumsq = block_reduce_sum(sumsq, red_tmp);
  const float rrms = rsqrtf(sumsq / (float)H + eps);

  // pass 2: scale (rows are L2-resident after pass 1)
  for (int i = tid * 8; i < H; i += nthread * 8) {
    ushort8_t v = *reinterpret_cast<const ushort8_t*>((FUSED_ADD ? orow : xr) + i);
    ushort8_t wv = *reinterpret_cast<const ushort8_t*>(w + i);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      v.x[j] = f32_to_bf16(bf16_to_f32(v.x[j]) * rrms * bf16_to_f32(wv.x[j]));
    }
    *reinterpret_cast<ushort8_t*>(orow + i) = v;
  }
}

void launch_rmsnorm(unsigned short* out, const unsigned short* x,
                    const unsigned short* w, float eps, int T, int H,
                    hipStream_t stream) {
  dim3 grid(T), block(256);
  hipLaunchKernelGGL((rmsnorm_kernel<false>), grid, block, 0, stream, out,
                     nullptr, x, w, eps, H);
}

void launch_fused_add_rmsnorm(unsigned short* x, unsigned short* residual,
                              const unsigned short* w, float eps, int T, int H,
                              hipStream_t stream) {
  // in-place: residual += x ; x = rmsnorm(residual) * w
  dim3 grid(T), block(256);
  hipLaunchKernelGGL((rmsnorm_kernel<true>), grid, block, 0, stream, x,
                     residual, x, w, eps, H);
}

}  // namespace xllm
