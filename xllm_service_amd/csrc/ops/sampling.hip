// Sampling kernels for MI355X (gfx950).
//
// Greedy argmax over the vocab dimension (bf16 logits), one workgroup per
// row, vectorized 16 B/lane loads. Stochastic top-k/top-p sampling is
// composed host-side from torch ops on GPU (it is not on the hot path at
// the same rate as argmax for the bench configs).
#include "common.h"

namespace xllm {

__global__ void greedy_sample_kernel(
    long* __restrict__ out,                     // [B]
    const unsigned short* __restrict__ logits,  // [B, V] bf16
    const int V) {
  const int row = blockIdx.x;
  const unsigned short* lg = logits + (long)row * V;
  float best = -INFINITY;
  int best_idx = 0;
  const int V8 = V & ~7;
  for (int i = threadIdx.x * 8; i < V8; i += blockDim.x * 8) {
    ushort8_t v = *reinterpret_cast<const ushort8_t*>(lg + i);
#pragma unroll
    for (int j = 0; j < 8; j++) {
      float f = bf16_to_f32(v.x[j]);
      if (f > best) { best = f; best_idx = i + j; }
    }
  }
  // vocab tail (V not a multiple of 8)
  for (int i = V8 + threadIdx.x; i < V; i += blockDim.x) {
    float f = bf16_to_f32(lg[i]);
    if (f > best) { best = f; best_idx = i; }
  }

  // reduce (max, argmin-index-on-tie) across the block
  __shared__ float sv[256];
  __shared__ int si[256];
  sv[threadIdx.x] = best;
  si[threadIdx.x] = best_idx;
  __syncthreads();
  for (int stride = blockDim.x / 2; stride > 0; stride >>= 1) {
    if (threadIdx.x < stride) {
      const float o = sv[threadIdx.x + stride];
      const int oi = si[threadIdx.x + stride];
      if (o > sv[threadIdx.x] ||
          (o == sv[threadIdx.x] && oi < si[threadIdx.x])) {
        sv[threadIdx.x] = o;
        si[threadIdx.x] = oi;
      }
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) out[row] = si[0];
}

void launch_greedy_sample(long* out, const unsigned short* logits, int B,
                          int V, hipStream_t stream) {
  dim3 grid(B), block(256);
  hipLaunchKernelGGL(greedy_sample_kernel, grid, block, 0, stream, out, logits,
                     V);
}

}  // namespace xllm
