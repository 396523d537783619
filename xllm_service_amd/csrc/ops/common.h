// Common device helpers for the MI355X (gfx950 / CDNA4) kernels.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  * wavefront = 64 lanes; all block sizes are multiples of 64
//  * bf16 global loads are vectorized as ushort4/ushort8 (8-16 B per lane)
//  * fp32 accumulation everywhere; bf16 storage
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define XLLM_WAVE 64
#define XLLM_DEV __device__ __forceinline__

namespace xllm {

using bf16 = __hip_bfloat16;

// ---- vector types for wide loads -------------------------------------------------
struct alignas(8) ushort4_t { unsigned short x[4]; };
struct alignas(16) ushort8_t { unsigned short x[8]; };
struct alignas(16) float4_t { float x[4]; };

XLLM_DEV float bf16_to_f32(unsigned short u) {
  union { float f; unsigned int i; } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}

XLLM_DEV unsigned short f32_to_bf16(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  // round-to-nearest-even
  unsigned int lsb = (c.i >> 16) & 1;
  c.i += 0x7fff + lsb;
  return (unsigned short)(c.i >> 16);
}

// ---- wave reductions -------------------------------------------------------------
XLLM_DEV float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

XLLM_DEV float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Reduce within a contiguous group of `W` lanes (W a power of two <= 64).
template <int W>
XLLM_DEV float group_reduce_sum(float v) {
#pragma unroll
  for (int off = W / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// ---- block reduction through LDS (for 256-thread blocks = 4 waves) ---------------
// `tmp` must have >= blockDim.x / 64 floats.
XLLM_DEV float block_reduce_sum(float v, float* tmp) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) tmp[wid] = v;
  __syncthreads();
  const int nw = blockDim.x >> 6;
  v = (threadIdx.x < nw) ? tmp[threadIdx.x] : 0.0f;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  }
  if (threadIdx.x == 0) tmp[0] = v;
  __syncthreads();
  return tmp[0];
}

}  // namespace xllm
