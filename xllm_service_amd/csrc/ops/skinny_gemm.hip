// Skinny (decode-batch) GEMM for MI355X (gfx950): C[M,N] = A[M,K] @ W[N,K]^T.
//
// Library GEMMs pick macro-tiles shaped for square problems: at decode
// batches (M <= 64) their small N-tiles re-read A per tile and leave the
// weight stream at 1.5-2.8 TB/s (scripts/bench_kernels.py vs the ~6.3 TB/s
// chip stream rate). Decode-layer GEMMs are pure weight streaming — W bytes
// dominate 50-400x over A/C — so this kernel is built like the decode
// attention K-stream instead of like a GEMM:
//
//   * each wave owns 64 output columns (4 MFMA n-strips), so one set of
//     A-fragments (M <= 64 rows, L2-resident, ~0.1-1.8 MB total) feeds 16
//     MFMAs per 32-k step: A traffic is 1/4 of W and always L2-hot
//   * W rows stream as direct B-fragments (8 contiguous k per lane = one
//     16-B load), double-buffered a FULL k-step ahead with clamped-address
//     loads (no in-loop branches — guide trap 4(c))
//   * accumulators live in AGPRs; the kernel is ~3 waves/SIMD
//   * split-K across ceil-divided k-ranges fills all 256 CUs even for the
//     N=4096 projections; fp32 partials reduce in a tiny second kernel.
//     SK == 1 writes bf16 directly (no workspace round-trip)
//
// Parity: the reference engine's decode-path linear layers.
#include "common.h"

namespace xllm {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

// MT = number of 16-row m-tiles (M <= MT*16); S = 16/MT n-strips per wave
// keeps the accumulator footprint at a constant 64 AGPRs for any M <= 256.
static inline int skinny_gemm_mt(int M) {
  if (M <= 16) return 1;
  if (M <= 32) return 2;
  if (M <= 64) return 4;
  if (M <= 128) return 8;
  return 16;
}

static inline int skinny_gemm_strips(int MT) {  // n-strips per wave
  const int s = 16 / MT;
  return s > 8 ? 8 : s;      // cap: 16 strips of B double-buffer spills
}

static inline int skinny_gemm_wgn(int M) {   // output cols per workgroup
  return 4 * 16 * skinny_gemm_strips(skinny_gemm_mt(M));
}

int skinny_gemm_splitk(int M, int N, int K) {
  // enough workgroups to fill 256 CUs ~2x, bounded by k-granularity
  const int wgn = skinny_gemm_wgn(M);
  const int strips = (N + wgn - 1) / wgn;
  int sk = (1024 + strips - 1) / strips;   // ~4 WGs per CU: parallel
  const int max_sk = K / 256 > 0 ? K / 256 : 1;  // latency chains
  if (sk > max_sk) sk = max_sk;
  if (sk > 32) sk = 32;
  if (sk < 1) sk = 1;
  return sk;
}

template <int MT, bool DIRECT>  // DIRECT: write bf16 C, add bias inline
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    float* __restrict__ ws,                // [SK, 64, N] fp32 partials
    unsigned short* __restrict__ c,        // [M, N] bf16 (DIRECT only)
    const unsigned short* __restrict__ a,  // [M, K] bf16
    const unsigned short* __restrict__ w,  // [N, K] bf16
    const unsigned short* __restrict__ bias,  // [N] or nullptr (DIRECT)
    const int M, const int N, const int K, const int SK) {
  constexpr int S = (16 / MT) > 8 ? 8 : (16 / MT);  // n-strips per wave
  const int ntile = blockIdx.x / SK;
  const int sk = blockIdx.x % SK;
  // ceil-divided k-range, 64-aligned so the unrolled step never straddles
  const int kper = ((K / 64 + SK - 1) / SK) * 64;
  const int k_begin = sk * kper;
  const int k_end = min(k_begin + kper, K);

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int frow = lane & 15;
  const int fcol8 = (lane >> 4) * 8;
  const int crow4 = (lane >> 4) * 4;
  // this wave's S*16 columns
  const int n_wave = (ntile * 4 + wid) * S * 16;

  f32x4 acc[S][MT];
#pragma unroll
  for (int s = 0; s < S; s++)
#pragma unroll
    for (int m = 0; m < MT; m++) acc[s][m] = f32x4{0, 0, 0, 0};

  // Addressing: UNIFORM base pointers (SGPR pairs, advanced by the loop)
  // plus clamped per-lane 32-bit element offsets (one static VGPR each) so
  // loads take the saddr+voffset form. With per-lane 64-bit pointers the
  // allocator aliased load destinations onto the address pairs and drained
  // vmcnt between every load (measured: zero overlap, 1.8 TB/s).
  // Clamping (not branching) keeps the stream unconditional — trap 4(c).
  // Max W is the lm_head (128256 x 4096 bf16 = 1.05 GB): byte offsets fit
  // 31 bits.
  int wofs[S];
#pragma unroll
  for (int s = 0; s < S; s++) {
    const int col = n_wave + s * 16 + frow;
    wofs[s] = (col < N ? col : N - 1) * K + fcol8;
  }
  int aofs[MT];
#pragma unroll
  for (int m = 0; m < MT; m++) {
    const int row = m * 16 + frow;
    aofs[m] = (row < M ? row : M - 1) * K + fcol8;
  }

  auto load_b = [&](bf16x8 (&bb)[S], int k0) {
#pragma unroll
    for (int s = 0; s < S; s++)
      bb[s] = *reinterpret_cast<const bf16x8*>(w + k0 + wofs[s]);
  };
  auto load_a = [&](bf16x8 (&aa)[MT], int k0) {
#pragma unroll
    for (int m = 0; m < MT; m++)
      aa[m] = *reinterpret_cast<const bf16x8*>(a + k0 + aofs[m]);
  };

  // k-loop, 64 k per iteration as two explicitly-named half-steps: each
  // half's loads are issued a full half-step before their MFMAs consume
  // them. No phase variable — a `phase ? b_b : b_a` ternary made the
  // allocator insert register copies and a vmcnt drain ladder between
  // every load (measured: zero overlap). Out-of-range prefetches clamp to
  // the last in-range chunk (harmless re-read, no branches).
  auto mfmas = [&](bf16x8 (&aa)[MT], bf16x8 (&bb)[S]) {
#pragma unroll
    for (int s = 0; s < S; s++)
#pragma unroll
      for (int m = 0; m < MT; m++)
        acc[s][m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aa[m], bb[s],
                                                            acc[s][m], 0, 0, 0);
  };
  const int k_last = k_end - 32;            // (k_end - k_begin) % 64 == 0
  bf16x8 b_a[S], b_b[S], a_a[MT], a_b[MT];
  if (k_begin < k_end) { load_b(b_a, k_begin); load_a(a_a, k_begin); }
  for (int k0 = k_begin; k0 < k_end; k0 += 64) {
    const int kh2 = k0 + 32;                 // always < k_end (64-aligned)
    load_b(b_b, kh2); load_a(a_b, kh2);
    mfmas(a_a, b_a);
    const int kn = min(k0 + 64, k_last);     // clamped next-iter prefetch
    load_b(b_a, kn); load_a(a_a, kn);
    mfmas(a_b, b_b);
  }

  // epilogue: C-frag row m = mt*16 + crow4 + r, col = n_wave + s*16 + frow
#pragma unroll
  for (int s = 0; s < S; s++) {
    const int col = n_wave + s * 16 + frow;
    if (col >= N) continue;
    if (DIRECT) {
      const float bv = (bias != nullptr) ? bf16_to_f32(bias[col]) : 0.0f;
#pragma unroll
      for (int m = 0; m < MT; m++)
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const int row = m * 16 + crow4 + r;
          if (row < M)
            c[(long)row * N + col] = f32_to_bf16(acc[s][m][r] + bv);
        }
    } else {
      float* wsk = ws + (long)sk * (MT * 16) * N;
#pragma unroll
      for (int m = 0; m < MT; m++)
#pragma unroll
        for (int r = 0; r < 4; r++)
          wsk[(long)(m * 16 + crow4 + r) * N + col] = acc[s][m][r];
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

void launch_skinny_reduce(unsigned short* c, const float* ws,
                          const unsigned short* bias, int M, int N, int MPAD,
                          int SK, hipStream_t stream) {
  long total4 = (long)M * N / 4;
  long rg = (total4 + 255) / 256;
  if (rg > 1024) rg = 1024;
  hipLaunchKernelGGL(skinny_reduce_kernel, dim3((unsigned)rg), dim3(256), 0,
                     stream, c, ws, bias, M, N, MPAD, SK);
}

void launch_skinny_gemm(unsigned short* c, const unsigned short* a,
                        const unsigned short* w, const unsigned short* bias,
                        float* ws, int M, int N, int K, hipStream_t stream) {
  const int SK = skinny_gemm_splitk(M, N, K);
  const int MT = skinny_gemm_mt(M);
  const int wgn = skinny_gemm_wgn(M);
  const int ntiles = (N + wgn - 1) / wgn;
  dim3 grid(ntiles * SK), block(256);
#define SG_LAUNCH(MTV)                                                         \
  do {                                                                         \
    if (SK == 1) {                                                             \
      hipLaunchKernelGGL((skinny_gemm_kernel<MTV, true>), grid, block, 0,      \
                         stream, ws, c, a, w, bias, M, N, K, SK);              \
      return;                                                                  \
    }                                                                          \
    hipLaunchKernelGGL((skinny_gemm_kernel<MTV, false>), grid, block, 0,       \
                       stream, ws, c, a, w, bias, M, N, K, SK);                \
  } while (0)
  switch (MT) {
    case 1: SG_LAUNCH(1); break;
    case 2: SG_LAUNCH(2); break;
    case 4: SG_LAUNCH(4); break;
    case 8: SG_LAUNCH(8); break;
    default: SG_LAUNCH(16); break;
  }
#undef SG_LAUNCH
  long total4 = (long)M * N / 4;
  long rg = (total4 + 255) / 256;
  if (rg > 1024) rg = 1024;
  hipLaunchKernelGGL(skinny_reduce_kernel, dim3((unsigned)rg), block, 0,
                     stream, c, ws, bias, M, N, MT * 16, SK);
}

}  // namespace xllm
