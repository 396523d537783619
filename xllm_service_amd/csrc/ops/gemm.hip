// Hand-written MFMA GEMM for the vision-encoder projections (gfx950).
//
// C[M,N] = A[M,K] @ B[K,N] + bias, bf16 in / bf16 out, fp32 accumulate.
// Used for the Qwen2-VL stage-E projections (patch embed: K=1176 padded to
// a multiple of 32 host-side; patch merger). Canonical CDNA GEMM anatomy
// (guide §5): 128x128 block tile, 4 waves each owning a 64x64 sub-tile of
// 4x4 16x16 fragments, K-tiles of 32 staged through padded LDS, MFMA
// accumulation in AGPRs. B is stored [N, K] (torch linear weight layout) so
// both A and B tiles load K-contiguously.
#include "common.h"

namespace xllm {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

#define GM_BM 128
#define GM_BN 128
#define GM_BK 32
#define GM_PAD 8

__global__ __launch_bounds__(256) void mfma_gemm_kernel(
    unsigned short* __restrict__ c,        // [M, N] bf16
    const unsigned short* __restrict__ a,  // [M, K] bf16
    const unsigned short* __restrict__ b,  // [N, K] bf16 (weight layout)
    const unsigned short* __restrict__ bias,  // [N] or nullptr
    const int M, const int N, const int K) {
  const int tiles_n = (N + GM_BN - 1) / GM_BN;
  const int bm = (blockIdx.x / tiles_n) * GM_BM;
  const int bn = (blockIdx.x % tiles_n) * GM_BN;

  const int wid = threadIdx.x >> 6;   // wave -> 2x2 sub-tile grid
  const int lane = threadIdx.x & 63;
  const int frow = lane & 15;
  const int fcol8 = (lane >> 4) * 8;
  const int crow4 = (lane >> 4) * 4;
  const int wm = (wid >> 1) * 64;     // wave row offset in tile
  const int wn = (wid & 1) * 64;

  __shared__ unsigned short a_lds[GM_BM][GM_BK + GM_PAD];
  __shared__ unsigned short b_lds[GM_BN][GM_BK + GM_PAD];

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; i++)
#pragma unroll
    for (int j = 0; j < 4; j++) acc[i][j] = f32x4{0, 0, 0, 0};

  for (int k0 = 0; k0 < K; k0 += GM_BK) {
    // stage A and B tiles: 128 rows x 32 cols each; thread copies one
    // 8-elem vector of each (256 threads x 8 x 2 = rows*32)
    {
      const int r = threadIdx.x >> 1;          // 0..127
      const int cc = (threadIdx.x & 1) * 16;   // 0 or 16
#pragma unroll
      for (int half = 0; half < 2; half++) {
        const int col = cc + half * 8;
        ushort8_t va, vb;
        const int ar = bm + r, kcol = k0 + col;
        if (ar < M && kcol < K) {
          va = *reinterpret_cast<const ushort8_t*>(a + (long)ar * K + kcol);
        } else {
#pragma unroll
          for (int j = 0; j < 8; j++) va.x[j] = 0;
        }
        const int br = bn + r;
        if (br < N && kcol < K) {
          vb = *reinterpret_cast<const ushort8_t*>(b + (long)br * K + kcol);
        } else {
#pragma unroll
          for (int j = 0; j < 8; j++) vb.x[j] = 0;
        }
        *reinterpret_cast<ushort8_t*>(&a_lds[r][col]) = va;
        *reinterpret_cast<ushort8_t*>(&b_lds[r][col]) = vb;
      }
    }
    __syncthreads();
#pragma unroll
    for (int ks = 0; ks < GM_BK / 32; ks++) {
      // A fragments for the wave's 4 row-tiles, B for 4 col-tiles
      bf16x8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; i++)
        af[i] = *reinterpret_cast<const bf16x8*>(
            &a_lds[wm + i * 16 + frow][ks * 32 + fcol8]);
#pragma unroll
      for (int j = 0; j < 4; j++)
        bf[j] = *reinterpret_cast<const bf16x8*>(
            &b_lds[wn + j * 16 + frow][ks * 32 + fcol8]);
#pragma unroll
      for (int i = 0; i < 4; i++)
#pragma unroll
        for (int j = 0; j < 4; j++)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: C[row = bm+wm+i*16+crow4+r][col = bn+wn+j*16+frow]
#pragma unroll
  for (int i = 0; i < 4; i++) {
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int row = bm + wm + i * 16 + crow4 + r;
      if (row >= M) continue;
#pragma unroll
      for (int j = 0; j < 4; j++) {
        const int col = bn + wn + j * 16 + frow;
        if (col >= N) continue;
        float v = acc[i][j][r];
        if (bias != nullptr) v += bf16_to_f32(bias[col]);
        c[(long)row * N + col] = f32_to_bf16(v);
      }
    }
  }
}

void launch_mfma_gemm(unsigned short* c, const unsigned short* a,
                      const unsigned short* b, const unsigned short* bias,
                      int M, int N, int K, hipStream_t stream) {
  const int tiles = ((M + GM_BM - 1) / GM_BM) * ((N + GM_BN - 1) / GM_BN);
  hipLaunchKernelGGL(mfma_gemm_kernel, dim3(tiles), dim3(256), 0, stream, c,
                     a, b, bias, M, N, K);
}

}  // namespace xllm
