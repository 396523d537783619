// Paged-attention PREFILL kernel for MI355X (gfx950 / CDNA4), MFMA-based.
//
// Flash-style online-softmax attention where K/V come from the paged KV
// cache (chunked prefill and prefix-cache hits share this one code path:
// new tokens' K/V are scattered into the cache first, then this runs).
//
// v3 structure (v2 at 4 waves x 32 q rows allocated 247 VGPR + 76 AGPR =
// 1 wave/SIMD — zero latency hiding, 111 TF; v1 was the guide's "2-phase
// stall" shape at ~76 TF):
//   * workgroup = 512 threads = 8 waves; one workgroup per
//     (128-row q tile, q_head); each wave owns 16 q rows (1 MFMA m-tile),
//     halving per-wave register state -> 2 waves/SIMD occupancy with the
//     same K/V staging amortization (128 q rows per staged tile)
//   * KV tile = 64 keys (4 cache blocks) staged cooperatively:
//     K row-major [64][128+8] (padded against the 16-way ds_read_b128
//     conflict), V into two tr16 images so P·V B-fragments come from
//     ds_read_b64_tr_b16 (see paged_attn_decode.hip for the image math)
//   * per wave and KV tile: 16 QK^T MFMAs -> online softmax (row stats
//     reduced over the 16 lanes holding a row's columns) -> P via LDS ->
//     16 P·V MFMAs
//
// MFMA fragment layout assumptions (verified on hardware by
// tests/test_gpu_ops.py::test_mfma_fragment_layout):
//   A: lane l holds A[l&15][(l>>4)*8 + j], j=0..7  (8 contiguous bf16)
//   B: lane l holds B[(l>>4)*8 + j][l&15]
//   C: lane l holds C[(l>>4)*4 + r][l&15], r=0..3
//
// Parity: reference engine's paged-attention prefill.
#include "common.h"

namespace xllm {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;
typedef __attribute__((__vector_size__(2 * sizeof(unsigned)))) unsigned u32x2;

#define PF_QW 16           // q rows per wave (1 m-tile)
#define PF_WGQ 128         // q rows per workgroup (8 waves)
#define PF_KBLK 64         // keys per KV tile (4 cache blocks)
#define PF_D 128           // head_dim (required)
#define PF_KPAD 8          // pad elements per K row
#define PF_PPAD 8          // pad elements per P row
#define PF_BS 16           // cache block size

__global__ __launch_bounds__(512) __attribute__((amdgpu_waves_per_eu(4)))
void paged_attn_prefill_kernel(
    unsigned short* __restrict__ out,        // [total_q, n_qheads, D] bf16
    const unsigned short* __restrict__ q,    // [total_q, n_qheads, D] bf16
    const unsigned short* __restrict__ k_cache,  // [blocks, n_kv, 16, D]
    const unsigned short* __restrict__ v_cache,
    const int* __restrict__ block_tables,    // [num_seqs, max_blocks]
    const int* __restrict__ cu_q,            // [num_seqs+1] query offsets
    const int* __restrict__ seq_lens,        // [num_seqs] total key len
    const int* __restrict__ tile_seq,        // [total_tiles] tile -> seq
    const int* __restrict__ tile_q0,         // [total_tiles] tile -> local q0
    const float scale,
    const int n_qheads, const int n_kv_heads, const int max_blocks_per_seq,
    const long q_stride, const long out_stride) {
  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (n_qheads / n_kv_heads);
  const int seq = tile_seq[tile];
  const int q0 = tile_q0[tile];
  const int q_start = cu_q[seq];
  const int q_len = cu_q[seq + 1] - q_start;
  const int seq_len = seq_lens[seq];
  const int ctx = seq_len - q_len;

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int frow = lane & 15;
  const int fcol8 = (lane >> 4) * 8;
  const int crow4 = (lane >> 4) * 4;

  const int wq0 = q0 + wid * PF_QW;
  const int wq_rows = min(PF_QW, q_len - wq0);
  const bool wave_active = wq_rows > 0;
  const int wave_kmax = wave_active ? (ctx + wq0 + wq_rows) : 0;
  const int wg_q_end = min(q0 + PF_WGQ, q_len);
  const int wg_kmax = ctx + wg_q_end;

  // v4: SINGLE-buffered K/V staging (52 KB total) + registers capped at
  // 128 so TWO 8-wave workgroups co-reside per CU. That costs a second
  // barrier per tile, but with 2 WGs the other workgroup's MFMAs cover
  // every barrier/stage stall — the round-1 single-WG profile parked 45%
  // of wave-cycles at SQ_WAIT with nothing to switch to.
  __shared__ unsigned short k_lds[PF_KBLK][PF_D + PF_KPAD];
  __shared__ unsigned short v_img[2][4096];     // tr16 image per 32-key half
  __shared__ unsigned short p_lds[8][PF_QW][PF_KBLK + PF_PPAD];

  // ---- Q fragment: one m-tile, lane row frow -------------------------------
  bf16x8 qf[4];
  {
    const int qrow = wq0 + frow;
    if (wave_active && qrow < q_len) {
      const long qoff = (long)(q_start + qrow) * q_stride + (long)qh * PF_D;
#pragma unroll
      for (int ks = 0; ks < 4; ks++)
        qf[ks] = *reinterpret_cast<const bf16x8*>(q + qoff + ks * 32 + fcol8);
    } else {
#pragma unroll
      for (int ks = 0; ks < 4; ks++) qf[ks] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  float m_r[4], l_r[4];
  f32x4 o_acc[8];
#pragma unroll
  for (int r = 0; r < 4; r++) { m_r[r] = -INFINITY; l_r[r] = 0.0f; }
#pragma unroll
  for (int n = 0; n < 8; n++) o_acc[n] = f32x4{0, 0, 0, 0};

  const int* btab = block_tables + (long)seq * max_blocks_per_seq;
  const long head_stride = (long)PF_BS * PF_D;

  // ---- T14 async-stage split (guide Guideline 15): the next tile's global
  // loads issue BEFORE this tile's compute; the LDS write lands after the
  // read barrier. Each of the 512 threads owns 2 fixed 8-element strips
  // of the 64x128 tile (threadIdx*8 + it*4096).
  ushort8_t stg_k[2], stg_v[2];
  const int sidx0 = threadIdx.x * 8;

  auto stage_load = [&](int kv0) {
#pragma unroll
    for (int it = 0; it < 2; it++) {
      const int idx = sidx0 + it * 4096;
      const int kt = idx / PF_D;
      const int d = idx % PF_D;
      const int tok = kv0 + kt;
      // clamp (not branch) the address: an in-loop conditional around each
      // load makes hipcc branch + drain per element (guide traps 4(c))
      const int tok_c = tok < seq_len ? tok : seq_len - 1;
      const long base =
          ((long)btab[tok_c / PF_BS] * n_kv_heads + kvh) * head_stride +
          (long)(tok_c % PF_BS) * PF_D + d;
      stg_k[it] = *reinterpret_cast<const ushort8_t*>(k_cache + base);
      stg_v[it] = *reinterpret_cast<const ushort8_t*>(v_cache + base);
    }
  };

  auto stage_write = [&](int kv0) {
#pragma unroll
    for (int it = 0; it < 2; it++) {
      const int idx = sidx0 + it * 4096;
      const int kt = idx / PF_D;
      const int d = idx % PF_D;
      ushort8_t kvv = stg_k[it], vvv = stg_v[it];
      if (kv0 + kt >= seq_len) {
#pragma unroll
        for (int j = 0; j < 8; j++) { kvv.x[j] = 0; vvv.x[j] = 0; }
      }
      *reinterpret_cast<ushort8_t*>(&k_lds[kt][d]) = kvv;
      const int k32 = kt & 31;
      const int koff =
          ((k32 >> 2) & 1) * 256 + (k32 >> 3) * 64 + (k32 & 3) * 16;
      *reinterpret_cast<ushort8_t*>(
          &v_img[kt >> 5][(d >> 4) * 512 + koff + (d & 15)]) = vvv;
    }
  };

  stage_load(0);
  stage_write(0);
  __syncthreads();

  for (int kv0 = 0; kv0 < wg_kmax; kv0 += PF_KBLK) {
    const bool has_next = kv0 + PF_KBLK < wg_kmax;
    if (has_next) stage_load(kv0 + PF_KBLK);  // overlaps this tile's compute

    if (wave_active && kv0 < wave_kmax) {
      // ---- S = Q K^T: 4 col-tiles x 4 k-steps -----------------------------
      f32x4 s[4];
#pragma unroll
      for (int n = 0; n < 4; n++) s[n] = f32x4{0, 0, 0, 0};
#pragma unroll
      for (int n = 0; n < 4; n++) {
#pragma unroll
        for (int ks = 0; ks < 4; ks++) {
          bf16x8 bk = *reinterpret_cast<const bf16x8*>(
              &k_lds[n * 16 + frow][ks * 32 + fcol8]);
          s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[ks], bk, s[n],
                                                         0, 0, 0);
        }
      }
      // ---- mask + online softmax ------------------------------------------
      float alpha[4];
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const int qrow = wq0 + crow4 + r;
        const int qpos = ctx + qrow;
        float smax = -INFINITY;
        float pv[4];
#pragma unroll
        for (int n = 0; n < 4; n++) {
          const int key = kv0 + n * 16 + frow;
          float sv = s[n][r] * scale;
          if (key > qpos || key >= seq_len || qrow >= q_len) sv = -INFINITY;
          pv[n] = sv;
          smax = fmaxf(smax, sv);
        }
#pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          smax = fmaxf(smax, __shfl_xor(smax, off, 64));
        const float m_new = fmaxf(m_r[r], smax);
        alpha[r] = (m_new == -INFINITY) ? 1.0f : __expf(m_r[r] - m_new);
        float psum = 0.0f;
#pragma unroll
        for (int n = 0; n < 4; n++) {
          const float p = (pv[n] == -INFINITY) ? 0.0f : __expf(pv[n] - m_new);
          psum += p;
          p_lds[wid][crow4 + r][n * 16 + frow] = f32_to_bf16(p);
        }
#pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          psum += __shfl_xor(psum, off, 64);
        m_r[r] = m_new;
        l_r[r] = l_r[r] * alpha[r] + psum;
      }
#pragma unroll
      for (int n = 0; n < 8; n++) {
#pragma unroll
        for (int r = 0; r < 4; r++) o_acc[n][r] *= alpha[r];
      }
      // ---- O += P V over the two 32-key halves ----------------------------
      // (B-fragments from the tr16 image; per-lane tr address = image base
      // + (l>>4)*128B + (l&15)*8B; each 32-key half needs TWO 8-read
      // batches — one per 4 d-tiles — reusing one 16-VGPR tr[] set)
      const unsigned vbase =
          (unsigned)(unsigned long long)(&v_img[0][0]) +
          ((lane >> 4) * 128u + (lane & 15) * 8u);
      u32x2 tr[8];
#define PF_TR8(OFF)                                                        \
      asm volatile(                                                        \
          "ds_read_b64_tr_b16 %[t0], %[a] offset:" #OFF "+0\n\t"           \
          "ds_read_b64_tr_b16 %[t1], %[a] offset:" #OFF "+512\n\t"         \
          "ds_read_b64_tr_b16 %[t2], %[a] offset:" #OFF "+1024\n\t"        \
          "ds_read_b64_tr_b16 %[t3], %[a] offset:" #OFF "+1536\n\t"        \
          "ds_read_b64_tr_b16 %[t4], %[a] offset:" #OFF "+2048\n\t"        \
          "ds_read_b64_tr_b16 %[t5], %[a] offset:" #OFF "+2560\n\t"        \
          "ds_read_b64_tr_b16 %[t6], %[a] offset:" #OFF "+3072\n\t"        \
          "ds_read_b64_tr_b16 %[t7], %[a] offset:" #OFF "+3584\n\t"        \
          "s_waitcnt lgkmcnt(0)"                                           \
          : [t0] "=&v"(tr[0]), [t1] "=&v"(tr[1]), [t2] "=&v"(tr[2]),       \
            [t3] "=&v"(tr[3]), [t4] "=&v"(tr[4]), [t5] "=&v"(tr[5]),       \
            [t6] "=&v"(tr[6]), [t7] "=&v"(tr[7])                           \
          : [a] "v"(vbase)                                                 \
          : "memory")
#define PF_PV4(HALF)                                                       \
      do {                                                                 \
        __builtin_amdgcn_sched_barrier(0);                                 \
        _Pragma("unroll")                                                  \
        for (int nn = 0; nn < 4; nn++) {                                   \
          const int n = (HALF) * 4 + nn;                                   \
          bf16x8 bv;                                                       \
          unsigned* bw = reinterpret_cast<unsigned*>(&bv);                 \
          bw[0] = tr[2 * nn][0];                                           \
          bw[1] = tr[2 * nn][1];                                           \
          bw[2] = tr[2 * nn + 1][0];                                       \
          bw[3] = tr[2 * nn + 1][1];                                       \
          o_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(               \
              pa, bv, o_acc[n], 0, 0, 0);                                  \
        }                                                                  \
      } while (0)
      {
        bf16x8 pa = *reinterpret_cast<const bf16x8*>(
            &p_lds[wid][frow][fcol8]);
        PF_TR8(0);     PF_PV4(0);
        PF_TR8(4096);  PF_PV4(1);
      }
      {
        bf16x8 pa = *reinterpret_cast<const bf16x8*>(
            &p_lds[wid][frow][32 + fcol8]);
        PF_TR8(8192);  PF_PV4(0);
        PF_TR8(12288); PF_PV4(1);
      }
#undef PF_TR8
#undef PF_PV4
    }
    // single buffer: all waves must be done READING this tile before the
    // next tile's write lands (the co-resident second workgroup hides
    // both barriers)
    __syncthreads();
    if (has_next) {
      stage_write(kv0 + PF_KBLK);
      __syncthreads();
    }
  }

  // ---- epilogue ------------------------------------------------------------
  if (wave_active) {
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int qrow = wq0 + crow4 + r;
      if (qrow >= q_len) continue;
      const float inv = (l_r[r] > 0.0f) ? 1.0f / l_r[r] : 0.0f;
      const long obase =
          (long)(q_start + qrow) * out_stride + (long)qh * PF_D;
#pragma unroll
      for (int n = 0; n < 8; n++)
        out[obase + n * 16 + frow] = f32_to_bf16(o_acc[n][r] * inv);
    }
  }
}

void launch_paged_attn_prefill(unsigned short* out, const unsigned short* q,
                               const unsigned short* k_cache,
                               const unsigned short* v_cache,
                               const int* block_tables, const int* cu_q,
                               const int* seq_lens, const int* tile_seq,
                               const int* tile_q0, int total_tiles, float scale,
                               int n_qheads, int n_kv_heads,
                               int max_blocks_per_seq, long q_stride,
                               long out_stride, hipStream_t stream) {
  dim3 grid(total_tiles, n_qheads), block(512);
  hipLaunchKernelGGL(paged_attn_prefill_kernel, grid, block, 0, stream, out, q,
                     k_cache, v_cache, block_tables, cu_q, seq_lens, tile_seq,
                     tile_q0, scale, n_qheads, n_kv_heads, max_blocks_per_seq,
                     q_stride, out_stride);
}

// ---------------------------------------------------------------------------------
// MFMA layout probe: C[16][16] = A[16][32] x B[32][16] with the fragment
// layout assumptions documented above. tests/test_gpu_ops.py checks this
// against torch.matmul on random data — if the lane mappings are wrong, that
// test fails (and so would the prefill kernel).
__global__ void mfma_probe_kernel(float* __restrict__ c,
                                  const unsigned short* __restrict__ a,
                                  const unsigned short* __restrict__ b) {
  const int lane = threadIdx.x & 63;
  const int frow = lane & 15;
  const int fcol8 = (lane >> 4) * 8;
  bf16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; j++) {
    af[j] = (short)a[frow * 32 + fcol8 + j];   // A[frow][fcol8+j]
    bf[j] = (short)b[(fcol8 + j) * 16 + frow]; // B[fcol8+j][frow]
  }
  f32x4 acc = {0, 0, 0, 0};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; r++) c[((lane >> 4) * 4 + r) * 16 + frow] = acc[r];
}

void launch_mfma_probe(float* c, const unsigned short* a,
                       const unsigned short* b, hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream, c, a, b);
}

}  // namespace xllm
