// Packed-weight streaming GEMM for decode batches on MI355X (gfx950):
// C[M,N] = A[M,K] @ W[N,K]^T with W PRE-PACKED into per-wave stream order.
//
// Why: decode-layer GEMMs are pure weight streaming (W bytes dominate A/C
// by 50-400x), and the chip's DRAM streams fastest when each wave reads
// LONG CONTIGUOUS spans. The round-1 skinny kernel (skinny_gemm.hip) loads
// MFMA b-fragments straight from the row-major weight: 16 parallel row
// streams each advancing 64 B per 32-k step — measured 1.7-3.2 TB/s cold.
// The decode-attention kernel, whose waves read ~KB-contiguous K-cache
// tiles, streams 3.4-4.8 TB/s cold with the same fragment shapes
// (docs/TODO_ROUND2.md round-1 PMC analysis). Weights are static at
// inference, so we simply pre-pack W into the exact fragment order a wave
// consumes:
//
//   P[strip n/16][k/32][kgroup 4][row 16][8 k]   (bf16)
//
// i.e. per 16-column strip, the K/32 one-KB fragment blocks lie back to
// back — a wave assigned (strip, k-range) reads ONE contiguous span of
// 16*K_range bytes with lane l taking bytes [16l, 16l+16) of each block
// (identical to the b-fragment the MFMA wants, so zero shuffling).
//
//   * one wave per (strip, split-k chunk); a workgroup = 4 adjacent strips
//   * A[M,K] row-major fragments re-read per strip — A is tiny and L2-hot
//   * 64-k steps double-buffered exactly like skinny_gemm (that scheme is
//     measured to give full load/MFMA overlap; see its header notes)
//   * split-K partials reduce with skinny_reduce_kernel; SK==1 writes
//     bf16 directly
//
// Parity: reference engine decode-path linears.
#include "common.h"

namespace xllm {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8_p;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4_p;

static inline int packed_gemm_mt(int M) {
  if (M <= 16) return 1;
  if (M <= 32) return 2;
  if (M <= 64) return 4;
  return 8;   // M <= 128
}

int packed_gemm_pick_s(int N) {
  // strips per wave: amortizes the packed-A loads over S weight slabs
  // (A instructions halve per W byte at S=2); larger N affords larger S
  // while keeping >= 2 WGs per CU before split-K
  if (N >= 16384 && N % 256 == 0) return 4;
  if (N >= 8192 && N % 128 == 0) return 2;
  return 1;
}

int packed_gemm_splitk(int M, int N, int K, int S) {
  // k-chunks are 256-aligned so the branchless pipeline has an exact
  // trip count (binding requires K % 256 == 0)
  const int wgs = N / (64 * S);            // groups of 4 waves x S strips
  int sk = (768 + wgs - 1) / wgs;          // ~3 WGs per CU
  const int max_sk = K / 256 > 0 ? K / 256 : 1;
  if (sk > max_sk) sk = max_sk;
  if (sk > 32) sk = 32;
  if (sk < 1) sk = 1;
  return sk;
}

// A[M,K] row-major -> fragment-order Ap[K/32][MT][4][16][8]: the MFMA
// a-operand becomes ONE contiguous full-line load per (m-tile, 32k) instead
// of a 16-row gather. The gathers were the wall: the no-A probe streams W
// at 2.3-5.2 TB/s while the gather version sits at 1.3-1.7 (per-CU
// load-path bound, guide's "fragment-shaped loads" trap).
template <int MT>
__global__ __launch_bounds__(256) void pack_a_kernel(
    unsigned short* __restrict__ ap, const unsigned short* __restrict__ a,
    const int M, const int K) {
  const int k32 = blockIdx.x;
  const int lane = threadIdx.x & 63;
  const int m = threadIdx.x >> 6;          // 4 m-tiles per block (MT <= 8)
  const int frow = lane & 15;
  const int fcol8 = (lane >> 4) * 8;
  for (int mt = m; mt < MT; mt += 4) {
    const int row = mt * 16 + frow;
    const int src = (row < M ? row : M - 1) * K + k32 * 32 + fcol8;
    *reinterpret_cast<ushort8_t*>(ap + ((long)(k32 * MT + mt) << 9) +
                                  lane * 8) =
        *reinterpret_cast<const ushort8_t*>(a + src);
  }
}

template <int MT, int S, bool DIRECT, bool NOA = false>
__global__ __launch_bounds__(256) void packed_gemm_kernel(
    float* __restrict__ ws,                // [SK, MT*16, N] fp32 partials
    unsigned short* __restrict__ c,        // [M, N] bf16 (DIRECT only)
    const unsigned short* __restrict__ a,  // packed Ap[K/32][MT][4][16][8]
    const unsigned short* __restrict__ w,  // packed P[N/16][K/32][4][16][8]
    const unsigned short* __restrict__ bias,  // [N] or nullptr (DIRECT)
    const int M, const int N, const int K, const int SK) {
  const int group = blockIdx.x / SK;       // 64*S-column group
  const int sk = blockIdx.x % SK;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int strip0 = (group * 4 + wid) * S;  // this wave's S strips
  // ceil-divided k-range, 256-aligned: with K % 256 == 0 every chunk is
  // an exact multiple of the PF*64 superstep — the pipeline loop carries
  // NO branches (an in-loop uniform branch splits the stages into basic
  // blocks and the scheduler drains vmcnt at every boundary; measured
  // 1.0-1.6 TB/s vs branchless streaming)
  const int kper = ((K / 256 + SK - 1) / SK) * 256;
  const int k_begin = sk * kper;
  const int k_end = min(k_begin + kper, K);

  const int frow = lane & 15;
  const int fcol8 = (lane >> 4) * 8;
  const int crow4 = (lane >> 4) * 4;

  f32x4_p acc[S][MT];
#pragma unroll
  for (int j = 0; j < S; j++)
#pragma unroll
    for (int m = 0; m < MT; m++) acc[j][m] = f32x4_p{0, 0, 0, 0};

  // wave-contiguous W stream: strip slab + k offset + this lane's 16 B
  // (one 512-element block per 32 k). SGPR base pointer + one 32-bit
  // per-lane element offset so loads take the saddr+voffset form — a
  // computed per-lane pointer makes the allocator drain vmcnt between
  // loads (skinny_gemm.hip header, measured 1.8 TB/s). Max W is the
  // lm_head (128256 x 4096 bf16): element offsets fit 31 bits.
  int wofs[S];
#pragma unroll
  for (int j = 0; j < S; j++)
    wofs[j] = (strip0 + j) * 16 * K + k_begin * 16 + lane * 8;
  const int aofs = lane * 8;               // within a packed A block

  auto load_w = [&](bf16x8_p (&bb)[S][2], int krel) {
    // krel relative to k_begin; two 32-k blocks = 1 KB contiguous each
#pragma unroll
    for (int j = 0; j < S; j++) {
      bb[j][0] = *reinterpret_cast<const bf16x8_p*>(w + wofs[j] + krel * 16);
      bb[j][1] =
          *reinterpret_cast<const bf16x8_p*>(w + wofs[j] + (krel + 32) * 16);
    }
  };
  auto load_a = [&](bf16x8_p (&aa)[2][MT], int k0) {
    if (NOA) return;  // probe variant: W stream only (garbage output)
    const int b0 = (k0 >> 5) * MT;         // packed block index of k0
#pragma unroll
    for (int m = 0; m < MT; m++) {
      aa[0][m] = *reinterpret_cast<const bf16x8_p*>(
          a + ((b0 + m) << 9) + aofs);
      aa[1][m] = *reinterpret_cast<const bf16x8_p*>(
          a + ((b0 + MT + m) << 9) + aofs);
    }
  };
  auto mfmas = [&](bf16x8_p (&aa)[2][MT], bf16x8_p (&bb)[S][2]) {
#pragma unroll
    for (int h = 0; h < 2; h++)
#pragma unroll
      for (int j = 0; j < S; j++)
#pragma unroll
        for (int m = 0; m < MT; m++)
          acc[j][m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              aa[h][m], bb[j][h], acc[j][m], 0, 0, 0);
  };

  // 64-k steps with a PF-deep W pipeline (4 KB in flight per wave: the
  // MFMA tail of one step is ~50 ns while HBM latency is ~1 us, so a
  // 2-deep buffer leaves waves parked at s_waitcnt — measured 1.1-2.0
  // TB/s; depth 4 puts enough independent loads in flight). A is L2-hot
  // (every strip re-reads it) and keeps a 2-deep buffer. All buffer
  // indices are compile-time (unrolled stage loop): phase variables make
  // the allocator insert vmcnt drains (see skinny_gemm.hip notes).
  constexpr int PF = S >= 4 ? 2 : S == 2 ? 3 : 4;   // ~PF*S*2KB in flight
  const int k_last = k_end - 64;
  auto clampk = [&](int k) { return k < k_last ? k : k_last; };
  bf16x8_p wq[PF][S][2], aq[2][2][MT];
  if (k_begin < k_end) {
#pragma unroll
    for (int s = 0; s < PF; s++)
      load_w(wq[s], clampk(k_begin + s * 64) - k_begin);
    load_a(aq[0], k_begin);
    load_a(aq[1], clampk(k_begin + 64));
  }
  for (int k0 = k_begin; k0 < k_end; k0 += PF * 64) {
#pragma unroll
    for (int s = 0; s < PF; s++) {
      const int kcur = k0 + s * 64;
      if (s & 1) {
        mfmas(aq[1], wq[s]);
        load_w(wq[s], clampk(kcur + PF * 64) - k_begin);
        load_a(aq[1], clampk(kcur + 128));
      } else {
        mfmas(aq[0], wq[s]);
        load_w(wq[s], clampk(kcur + PF * 64) - k_begin);
        load_a(aq[0], clampk(kcur + 128));
      }
    }
  }

  // epilogue: C row = m*16 + crow4 + r, col = (strip0+j)*16 + frow
#pragma unroll
  for (int j = 0; j < S; j++) {
    const int col = (strip0 + j) * 16 + frow;
    if (col >= N) continue;
    if (DIRECT) {
      const float bv = (bias != nullptr) ? bf16_to_f32(bias[col]) : 0.0f;
#pragma unroll
      for (int m = 0; m < MT; m++)
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const int row = m * 16 + crow4 + r;
          if (row < M)
            c[(long)row * N + col] = f32_to_bf16(acc[j][m][r] + bv);
        }
    } else {
      // split-K partials to ws[sk][MT*16][N] + separate reduce kernel.
      // (fp32 ATOMIC accumulation into [MT*16, N] was measured SLOWER:
      // +6.5 us on the o-projection — global fp32 atomic throughput is
      // far below plain-store rate at 64 atomics/wave)
      float* wsk = ws + (long)sk * (MT * 16) * N;
#pragma unroll
      for (int m = 0; m < MT; m++)
#pragma unroll
        for (int r = 0; r < 4; r++)
          wsk[(long)(m * 16 + crow4 + r) * N + col] = acc[j][m][r];
    }
  }
}

void launch_skinny_reduce(unsigned short* c, const float* ws,
                          const unsigned short* bias, int M, int N, int MPAD,
                          int SK, hipStream_t stream);

void launch_packed_gemm_probe(unsigned short* c, const unsigned short* a,
                              const unsigned short* w, float* ws, int M,
                              int N, int K, hipStream_t stream) {
  // timing probe: identical W stream, A loads compiled out (S = 1)
  const int SK = packed_gemm_splitk(M, N, K, 1);
  dim3 grid((N / 64) * SK), block(256);
  if (SK == 1) {
    hipLaunchKernelGGL((packed_gemm_kernel<4, 1, true, true>), grid, block, 0,
                       stream, ws, c, a, w, nullptr, M, N, K, SK);
    return;
  }
  hipLaunchKernelGGL((packed_gemm_kernel<4, 1, false, true>), grid, block, 0,
                     stream, ws, c, a, w, nullptr, M, N, K, SK);
  launch_skinny_reduce(c, ws, nullptr, M, N, 64, SK, stream);
}

void launch_packed_gemm(unsigned short* c, const unsigned short* a,
                        unsigned short* ap, const unsigned short* w,
                        const unsigned short* bias, float* ws, int M, int N,
                        int K, int S, hipStream_t stream) {
  if (S <= 0) S = packed_gemm_pick_s(N);
  const int SK = packed_gemm_splitk(M, N, K, S);
  const int MT = packed_gemm_mt(M);
  switch (MT) {
    case 1: hipLaunchKernelGGL((pack_a_kernel<1>), dim3(K / 32), dim3(256), 0,
                               stream, ap, a, M, K); break;
    case 2: hipLaunchKernelGGL((pack_a_kernel<2>), dim3(K / 32), dim3(256), 0,
                               stream, ap, a, M, K); break;
    case 4: hipLaunchKernelGGL((pack_a_kernel<4>), dim3(K / 32), dim3(256), 0,
                               stream, ap, a, M, K); break;
    default: hipLaunchKernelGGL((pack_a_kernel<8>), dim3(K / 32), dim3(256), 0,
                                stream, ap, a, M, K); break;
  }
  const unsigned short* apc = ap;
  dim3 grid((N / (64 * S)) * SK), block(256);
#define PG_LAUNCH1(MTV, SV)                                                    \
  do {                                                                         \
    if (SK == 1) {                                                             \
      hipLaunchKernelGGL((packed_gemm_kernel<MTV, SV, true>), grid, block, 0,  \
                         stream, ws, c, apc, w, bias, M, N, K, SK);            \
      return;                                                                  \
    }                                                                          \
    hipLaunchKernelGGL((packed_gemm_kernel<MTV, SV, false>), grid, block, 0,   \
                       stream, ws, c, apc, w, bias, M, N, K, SK);              \
  } while (0)
#define PG_LAUNCH(MTV)                                                         \
  do {                                                                         \
    switch (S) {                                                               \
      case 1: PG_LAUNCH1(MTV, 1); break;                                       \
      case 2: PG_LAUNCH1(MTV, 2); break;                                       \
      default: PG_LAUNCH1(MTV, 4); break;                                      \
    }                                                                          \
  } while (0)
  switch (MT) {
    case 1: PG_LAUNCH(1); break;
    case 2: PG_LAUNCH(2); break;
    case 4: PG_LAUNCH(4); break;
    default: PG_LAUNCH(8); break;
  }
#undef PG_LAUNCH
#undef PG_LAUNCH1
  launch_skinny_reduce(c, ws, bias, M, N, MT * 16, SK, stream);
}

}  // namespace xllm
