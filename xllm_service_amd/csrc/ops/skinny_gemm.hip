// Skinny (decode-batch) GEMM for MI355X (gfx950): C[M,N] = A[M,K] @ W[N,K]^T.
//
// Library GEMMs pick tilings with too few workgroups for M <= 128 decode
// batches: the weight stream bottlenecks on a fraction of the chip's CUs
// (measured 0.9-2.3 TB/s for the qkv/o projections vs a 6.3 TB/s chip
// stream rate — scripts/bench_kernels.py). This kernel split-Ks the weight
// so (N/64) x ceil(K/KCHUNK) workgroups stream W exactly once, A stays
// L2-resident, and fp32 partials reduce in a second tiny kernel (both
// kernels live inside the decode hipGraph).
//
// No LDS at all: A and B fragments load straight from global as 16-B
// per-lane reads (both operands K-contiguous), MFMA 16x16x32 accumulates.
#include "common.h"

namespace xllm {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

#define SG_BN 64
#define SG_KCHUNK 1024

template <int MT>  // number of 16-row m-tiles (M <= MT*16)
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    float* __restrict__ ws,                // [SK, MT*16, N] fp32 partials
    const unsigned short* __restrict__ a,  // [M, K] bf16
    const unsigned short* __restrict__ w,  // [N, K] bf16
    const int M, const int N, const int K, const int SK) {
  const int ntile = blockIdx.x / SK;
  const int sk = blockIdx.x % SK;
  const int n0 = ntile * SG_BN;
  const int k_begin = sk * SG_KCHUNK;
  const int k_end = min(k_begin + SG_KCHUNK, K);

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int frow = lane & 15;
  const int fcol8 = (lane >> 4) * 8;
  const int crow4 = (lane >> 4) * 4;
  const int col = n0 + wid * 16 + frow;   // this wave's B column (= W row)
  const bool col_ok = col < N;

  f32x4 acc[MT];
#pragma unroll
  for (int m = 0; m < MT; m++) acc[m] = f32x4{0, 0, 0, 0};

  // Guards are hoisted into CLAMPED base pointers: an in-loop conditional
  // around each load makes hipcc branch + drain vmcnt per element (guide
  // §5 ".s-level traps" (c)). Clamped rows produce garbage partials only
  // for output rows/cols the epilogue never stores.
  const unsigned short* wbase =
      w + (long)(col_ok ? col : N - 1) * K + fcol8;
  const unsigned short* abase[MT];
#pragma unroll
  for (int m = 0; m < MT; m++) {
    const int row = m * 16 + frow;
    abase[m] = a + (long)(row < M ? row : M - 1) * K + fcol8;
  }
  // unroll by 2 K-steps so 2 B-loads + 2*MT A-loads stay in flight
  for (int k0 = k_begin; k0 < k_end; k0 += 64) {
    bf16x8 bk0 = *reinterpret_cast<const bf16x8*>(wbase + k0);
    bf16x8 bk1 = *reinterpret_cast<const bf16x8*>(wbase + k0 + 32);
    bf16x8 av0[MT], av1[MT];
#pragma unroll
    for (int m = 0; m < MT; m++) {
      av0[m] = *reinterpret_cast<const bf16x8*>(abase[m] + k0);
      av1[m] = *reinterpret_cast<const bf16x8*>(abase[m] + k0 + 32);
    }
#pragma unroll
    for (int m = 0; m < MT; m++)
      acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av0[m], bk0, acc[m],
                                                       0, 0, 0);
#pragma unroll
    for (int m = 0; m < MT; m++)
      acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av1[m], bk1, acc[m],
                                                       0, 0, 0);
  }

  // partials in C layout: ws[sk][m*16 + crow4 + r][col]
  float* wsk = ws + (long)sk * (MT * 16) * N;
#pragma unroll
  for (int m = 0; m < MT; m++) {
#pragma unroll
    for (int r = 0; r < 4; r++) {
      if (col_ok)
        wsk[(long)(m * 16 + crow4 + r) * N + col] = acc[m][r];
    }
  }
}

__global__ void skinny_reduce_kernel(
    unsigned short* __restrict__ c,        // [M, N] bf16
    const float* __restrict__ ws,          // [SK, MPAD, N]
    const unsigned short* __restrict__ bias,  // [N] or nullptr
    const int M, const int N, const int MPAD, const int SK) {
  const long total4 = (long)M * N / 4;    // N is a multiple of 4
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total4;
       i += (long)gridDim.x * blockDim.x) {
    const long row = (i * 4) / N;
    const int colb = (int)((i * 4) % N);
    float4_t s = {0, 0, 0, 0};
    for (int sk = 0; sk < SK; sk++) {
      const float* p = ws + ((long)sk * MPAD + row) * N + colb;
#pragma unroll
      for (int j = 0; j < 4; j++) s.x[j] += p[j];
    }
    if (bias != nullptr) {
#pragma unroll
      for (int j = 0; j < 4; j++) s.x[j] += bf16_to_f32(bias[colb + j]);
    }
    ushort4_t o;
#pragma unroll
    for (int j = 0; j < 4; j++) o.x[j] = f32_to_bf16(s.x[j]);
    *reinterpret_cast<ushort4_t*>(c + row * N + colb) = o;
  }
}

void launch_skinny_gemm(unsigned short* c, const unsigned short* a,
                        const unsigned short* w, const unsigned short* bias,
                        float* ws, int M, int N, int K, hipStream_t stream) {
  const int SK = (K + SG_KCHUNK - 1) / SG_KCHUNK;
  const int ntiles = (N + SG_BN - 1) / SG_BN;
  const int MT = (M + 15) / 16;
  dim3 grid(ntiles * SK), block(256);
  switch (MT) {
    case 1: hipLaunchKernelGGL((skinny_gemm_kernel<1>), grid, block, 0,
                               stream, ws, a, w, M, N, K, SK); break;
    case 2: hipLaunchKernelGGL((skinny_gemm_kernel<2>), grid, block, 0,
                               stream, ws, a, w, M, N, K, SK); break;
    case 3: hipLaunchKernelGGL((skinny_gemm_kernel<3>), grid, block, 0,
                               stream, ws, a, w, M, N, K, SK); break;
    case 4: hipLaunchKernelGGL((skinny_gemm_kernel<4>), grid, block, 0,
                               stream, ws, a, w, M, N, K, SK); break;
    case 5: hipLaunchKernelGGL((skinny_gemm_kernel<5>), grid, block, 0,
                               stream, ws, a, w, M, N, K, SK); break;
    case 7: hipLaunchKernelGGL((skinny_gemm_kernel<7>), grid, block, 0,
                               stream, ws, a, w, M, N, K, SK); break;
    case 6: hipLaunchKernelGGL((skinny_gemm_kernel<6>), grid, block, 0,
                               stream, ws, a, w, M, N, K, SK); break;
    case 8: hipLaunchKernelGGL((skinny_gemm_kernel<8>), grid, block, 0,
                               stream, ws, a, w, M, N, K, SK); break;
    default: break;  // host guards M <= 128
  }
  const int MPAD = MT * 16;
  long total4 = (long)M * N / 4;
  long rg = (total4 + 255) / 256;
  if (rg > 1024) rg = 1024;
  hipLaunchKernelGGL(skinny_reduce_kernel, dim3((unsigned)rg), block, 0,
                     stream, c, ws, bias, M, N, MPAD, SK);
}

}  // namespace xllm
