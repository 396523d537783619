// Fused RMSNorm kernels for MI355X (gfx950).
//
// Single-pass, register-resident: each row is handled by one workgroup of
// up to 512 threads; a thread keeps its <=32 bf16 elements in registers
// between the sum-of-squares reduction and the scale, so the row is read
// once and written once (the fused-add variant also reads/writes the
// residual stream once). Sized for decode batches (small T) as well as
// prefill (large T): block width scales with H, not with T.
//
// Capability parity: the reference engine's fused RMSNorm.
#include "common.h"

namespace xllm {

// ELEMS = number of ushort8 octets each thread owns (H <= block*8*ELEMS)
template <bool FUSED_ADD, int ELEMS>
__global__ void rmsnorm_kernel(
    unsigned short* __restrict__ out,       // [T, H] bf16 (= x if FUSED_ADD)
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

  float v[ELEMS][8];
  float sumsq = 0.0f;
#pragma unroll
  for (int e = 0; e < ELEMS; e++) {
    const int i = (tid + e * nthread) * 8;
    if (i < H) {
      ushort8_t u = *reinterpret_cast<const ushort8_t*>(xr + i);
#pragma unroll
      for (int j = 0; j < 8; j++) v[e][j] = bf16_to_f32(u.x[j]);
      if constexpr (FUSED_ADD) {
        ushort8_t r = *reinterpret_cast<const ushort8_t*>(rr + i);
        ushort8_t rw;
#pragma unroll
        for (int j = 0; j < 8; j++) {
          v[e][j] += bf16_to_f32(r.x[j]);
          rw.x[j] = f32_to_bf16(v[e][j]);
        }
        *reinterpret_cast<ushort8_t*>(rr + i) = rw;
      }
#pragma unroll
      for (int j = 0; j < 8; j++) sumsq += v[e][j] * v[e][j];
    }
  }
  sumsq = block_reduce_sum(sumsq, red_tmp);
  const float rrms = rsqrtf(sumsq / (float)H + eps);

#pragma unroll
  for (int e = 0; e < ELEMS; e++) {
    const int i = (tid + e * nthread) * 8;
    if (i < H) {
      ushort8_t wv = *reinterpret_cast<const ushort8_t*>(w + i);
      ushort8_t o;
#pragma unroll
      for (int j = 0; j < 8; j++)
        o.x[j] = f32_to_bf16(v[e][j] * rrms * bf16_to_f32(wv.x[j]));
      *reinterpret_cast<ushort8_t*>(orow + i) = o;
    }
  }
}

static inline int norm_block(int H) {
  int octets = H / 8;
  int b = octets < 512 ? octets : 512;
  return ((b + 63) / 64) * 64;  // multiple of the 64-lane wave
}

#define RMS_LAUNCH(FUSED, E)                                              \
  hipLaunchKernelGGL((rmsnorm_kernel<FUSED, E>), dim3(T), dim3(block), 0, \
                     stream, out, residual, x, w, eps, H)

template <bool FUSED>
static void rmsnorm_dispatch(unsigned short* out, unsigned short* residual,
                             const unsigned short* x, const unsigned short* w,
                             float eps, int T, int H, hipStream_t stream) {
  const int block = norm_block(H);
  const int elems = (H + block * 8 - 1) / (block * 8);
  switch (elems) {
    case 1: RMS_LAUNCH(FUSED, 1); break;
    case 2: RMS_LAUNCH(FUSED, 2); break;
    case 3: RMS_LAUNCH(FUSED, 3); break;
    case 4: RMS_LAUNCH(FUSED, 4); break;
    default: RMS_LAUNCH(FUSED, 8); break;  // up to H = 32768
  }
}

void launch_rmsnorm(unsigned short* out, const unsigned short* x,
                    const unsigned short* w, float eps, int T, int H,
                    hipStream_t stream) {
  rmsnorm_dispatch<false>(out, nullptr, x, w, eps, T, H, stream);
}

void launch_fused_add_rmsnorm(unsigned short* x, unsigned short* residual,
                              const unsigned short* w, float eps, int T, int H,
                              hipStream_t stream) {
  // in-place: residual += x ; x = rmsnorm(residual) * w
  rmsnorm_dispatch<true>(x, residual, x, w, eps, T, H, stream);
}

}  // namespace xllm
