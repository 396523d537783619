// Paged-attention PREFILL kernel for MI355X (gfx950 / CDNA4), MFMA-based.
//
// Flash-style online-softmax attention where K/V come from the paged KV
// cache (so chunked prefill and prefix-cache hits share one code path: new
// tokens' K/V are scattered into the cache first, then this kernel runs).
//
// Structure (v1 — correctness-first MFMA, single-buffered LDS):
//   * workgroup = 256 threads = 4 waves; one workgroup per
//     (sequence q-tile of 64 rows, q_head); each wave owns 16 q rows
//   * per KV tile of 32 keys (2 cache blocks):
//       - all threads cooperatively stage K [32][128] row-major (+8 pad) and
//         V^T [128][32] (+8 pad) into LDS (padding kills the 16-way
//         ds_read_b128 bank conflict of 256-B rows; Guideline 4)
//       - each wave: S = Q·K^T via 8x mfma_f32_16x16x32_bf16, causal mask,
//         online softmax (row stats reduced over the 16 lanes holding a row's
//         columns), P round-trips through a per-wave LDS scratch to reshape
//         C-layout -> A-fragment layout, then O += P·V via 8 MFMAs
//   * epilogue: O /= l, bf16 store
//
// MFMA fragment layout assumptions (verified by tests/test_gpu_mfma.py on
// hardware via the mfma_probe_16x16x32 op):
//   A: lane l holds A[l&15][(l>>4)*8 + j], j=0..7  (8 contiguous bf16)
//   B: lane l holds B[(l>>4)*8 + j][l&15]
//   C: lane l holds C[(l>>4)*4 + r][l&15], r=0..3
//
// Parity: reference engine's paged-attention prefill.
#include "common.h"

namespace xllm {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

#define PF_QBLK 16         // q rows per wave
#define PF_KBLK 32         // keys per KV tile (= 2 cache blocks)
#define PF_D 128           // head_dim (required)
#define PF_KPAD 8          // pad elements per K row
#define PF_VPAD 8          // pad elements per V^T row
#define PF_BS 16           // cache block size

__global__ __launch_bounds__(256) void paged_attn_prefill_kernel(
    unsigned short* __restrict__ out,        // [total_q, n_qheads, D] bf16
    const unsigned short* __restrict__ q,    // [total_q, n_qheads, D] bf16
    const unsigned short* __restrict__ k_cache,  // [blocks, n_kv, 16, D]
    const unsigned short* __restrict__ v_cache,
    const int* __restrict__ block_tables,    // [num_seqs, max_blocks]
    const int* __restrict__ cu_q,            // [num_seqs+1] query offsets
    const int* __restrict__ seq_lens,        // [num_seqs] total key len
    const int* __restrict__ tile_seq,        // [total_tiles] tile -> seq
    const int* __restrict__ tile_q0,         // [total_tiles] tile -> local q row
    const float scale,
    const int n_qheads, const int n_kv_heads, const int max_blocks_per_seq,
    const long q_stride, const long out_stride) {
  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (n_qheads / n_kv_heads);
  const int seq = tile_seq[tile];
  const int q0 = tile_q0[tile];                  // first local q row of this WG
  const int q_start = cu_q[seq];
  const int q_len = cu_q[seq + 1] - q_start;
  const int seq_len = seq_lens[seq];
  const int ctx = seq_len - q_len;               // keys before the new chunk

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int frow = lane & 15;                    // fragment row (A) / col (B,C)
  const int fcol8 = (lane >> 4) * 8;             // fragment 8-col base (A,B)
  const int crow4 = (lane >> 4) * 4;             // fragment 4-row base (C)

  // wave's q rows: [q0 + wid*16, +16) ∩ [0, q_len)
  const int wq0 = q0 + wid * PF_QBLK;
  const int wq_rows = min(PF_QBLK, q_len - wq0);     // may be <= 0
  const bool wave_active = wq_rows > 0;
  // last key this wave may see (causal): global pos of its last q row
  const int wave_kmax = wave_active ? (ctx + wq0 + wq_rows) : 0;
  // workgroup loop bound: keys needed by the deepest wave in this WG
  const int wg_q_end = min(q0 + 64, q_len);
  const int wg_kmax = ctx + wg_q_end;            // == max over waves

  __shared__ unsigned short k_lds[PF_KBLK][PF_D + PF_KPAD];
  // tr16-compatible V image (see paged_attn_decode.hip): element (k, d) at
  // (d>>4)*512 + ((k>>2)&1)*256 + (k>>3)*64 + (k&3)*16 + (d&15)
  __shared__ unsigned short vt_lds[4096];
  __shared__ unsigned short p_lds[4][PF_QBLK][PF_KBLK + PF_VPAD];

  // ---- load Q fragments (lane holds row frow, cols ks*32+fcol8..+8) ----------
  bf16x8 qf[4];
  if (wave_active && frow < wq_rows) {
    const long qrow = (long)(q_start + wq0 + frow) * q_stride + (long)qh * PF_D;
#pragma unroll
    for (int ks = 0; ks < 4; ks++)
      qf[ks] = *reinterpret_cast<const bf16x8*>(q + qrow + ks * 32 + fcol8);
  } else {
#pragma unroll
    for (int ks = 0; ks < 4; ks++) qf[ks] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
  }

  // ---- online state ----------------------------------------------------------
  float m_r[4], l_r[4];      // per C fragment row r (q row crow4 + r)
  f32x4 o_acc[8];            // 8 d-tiles of O in C layout
#pragma unroll
  for (int r = 0; r < 4; r++) { m_r[r] = -INFINITY; l_r[r] = 0.0f; }
#pragma unroll
  for (int n = 0; n < 8; n++) o_acc[n] = f32x4{0, 0, 0, 0};

  const int* btab = block_tables + (long)seq * max_blocks_per_seq;
  const long kv_head_stride = (long)PF_BS * PF_D;

  for (int kv0 = 0; kv0 < wg_kmax; kv0 += PF_KBLK) {
    const int kleft = min(PF_KBLK, seq_len - kv0);  // valid keys this tile
    // ---- stage K and V^T (cooperative, 256 threads) --------------------------
    // K: rows are (key) tokens; thread t covers elements of flattened [32][128]
    {
      // each thread copies 16 bf16 of K (32*128/256) as 2x ushort8
      for (int idx = threadIdx.x * 8; idx < PF_KBLK * PF_D; idx += 256 * 8) {
        const int kt = idx / PF_D;         // key within tile
        const int d = idx % PF_D;
        ushort8_t val;
        if (kt < kleft) {
          const int tok = kv0 + kt;
          const long phys = btab[tok / PF_BS];
          const long off = (phys * n_kv_heads + kvh) * kv_head_stride +
                           (long)(tok % PF_BS) * PF_D + d;
          val = *reinterpret_cast<const ushort8_t*>(k_cache + off);
        } else {
#pragma unroll
          for (int j = 0; j < 8; j++) val.x[j] = 0;
        }
        *reinterpret_cast<ushort8_t*>(&k_lds[kt][d]) = val;
      }
      // V into the tr16 image: vector loads AND vector writes
      for (int idx = threadIdx.x * 8; idx < PF_KBLK * PF_D; idx += 256 * 8) {
        const int kt = idx / PF_D;
        const int d = idx % PF_D;
        ushort8_t val;
        if (kt < kleft) {
          const int tok = kv0 + kt;
          const long phys = btab[tok / PF_BS];
          const long off = (phys * n_kv_heads + kvh) * kv_head_stride +
                           (long)(tok % PF_BS) * PF_D + d;
          val = *reinterpret_cast<const ushort8_t*>(v_cache + off);
        } else {
#pragma unroll
          for (int j = 0; j < 8; j++) val.x[j] = 0;
        }
        const int koff =
            ((kt >> 2) & 1) * 256 + (kt >> 3) * 64 + (kt & 3) * 16;
        *reinterpret_cast<ushort8_t*>(
            &vt_lds[(d >> 4) * 512 + koff + (d & 15)]) = val;
      }
    }
    __syncthreads();

    // waves whose q rows can't see this tile skip compute (still barrier)
    if (wave_active && kv0 < wave_kmax) {
      // ---- S = Q K^T : 2 col-tiles x 4 k-steps -------------------------------
      f32x4 s[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
#pragma unroll
      for (int n = 0; n < 2; n++) {
#pragma unroll
        for (int ks = 0; ks < 4; ks++) {
          // B frag: K^T[d = ks*32+fcol8+j][key = n*16+frow] = K[key][d...]
          bf16x8 bk = *reinterpret_cast<const bf16x8*>(
              &k_lds[n * 16 + frow][ks * 32 + fcol8]);
          s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[ks], bk, s[n], 0, 0, 0);
        }
      }
      // ---- mask + online softmax --------------------------------------------
      // element (r, n): q row wq0+crow4+r, key kv0 + n*16 + frow
      float p_val[2][4];
      float alpha[4];
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const int qrow = wq0 + crow4 + r;
        const int qpos = ctx + qrow;  // last visible key index
        float smax = -INFINITY;
#pragma unroll
        for (int n = 0; n < 2; n++) {
          const int key = kv0 + n * 16 + frow;
          float sv = s[n][r] * scale;
          if (key > qpos || key >= seq_len || qrow >= q_len) sv = -INFINITY;
          p_val[n][r] = sv;
          smax = fmaxf(smax, sv);
        }
        // reduce max over the 16 lanes holding this row's 32 cols
#pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          smax = fmaxf(smax, __shfl_xor(smax, off, 64));
        const float m_new = fmaxf(m_r[r], smax);
        // m_new can be -inf for fully-masked rows; keep alpha finite
        alpha[r] = (m_new == -INFINITY) ? 1.0f : __expf(m_r[r] - m_new);
        float psum = 0.0f;
#pragma unroll
        for (int n = 0; n < 2; n++) {
          const float p = (p_val[n][r] == -INFINITY)
                              ? 0.0f
                              : __expf(p_val[n][r] - m_new);
          p_val[n][r] = p;
          psum += p;
        }
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) psum += __shfl_xor(psum, off, 64);
        m_r[r] = m_new;
        l_r[r] = l_r[r] * alpha[r] + psum;
      }
      // ---- P (C layout) -> LDS -> A fragments -------------------------------
#pragma unroll
      for (int n = 0; n < 2; n++) {
#pragma unroll
        for (int r = 0; r < 4; r++)
          p_lds[wid][crow4 + r][n * 16 + frow] = f32_to_bf16(p_val[n][r]);
      }
      // rescale O by alpha (row r of each C frag)
#pragma unroll
      for (int n = 0; n < 8; n++) {
#pragma unroll
        for (int r = 0; r < 4; r++) o_acc[n][r] *= alpha[r];
      }
      // A frag of P: lane holds P[frow][fcol8 + j]
      bf16x8 pa = *reinterpret_cast<const bf16x8*>(&p_lds[wid][frow][fcol8]);
      // ---- O += P V (B-fragments via hardware transpose reads) --------------
      {
        typedef __attribute__((__vector_size__(2 * sizeof(unsigned)))) unsigned u32x2;
        const unsigned vaddr =
            (unsigned)(unsigned long long)(&vt_lds[0]) +
            ((lane >> 4) * 128u + (lane & 15) * 8u);
        u32x2 tr[8];
        u32x2 tr2[8];
        asm volatile(
            "ds_read_b64_tr_b16 %[t0], %[a] offset:0\n\t"
            "ds_read_b64_tr_b16 %[t1], %[a] offset:512\n\t"
            "ds_read_b64_tr_b16 %[t2], %[a] offset:1024\n\t"
            "ds_read_b64_tr_b16 %[t3], %[a] offset:1536\n\t"
            "ds_read_b64_tr_b16 %[t4], %[a] offset:2048\n\t"
            "ds_read_b64_tr_b16 %[t5], %[a] offset:2560\n\t"
            "ds_read_b64_tr_b16 %[t6], %[a] offset:3072\n\t"
            "ds_read_b64_tr_b16 %[t7], %[a] offset:3584\n\t"
            "ds_read_b64_tr_b16 %[u0], %[a] offset:4096\n\t"
            "ds_read_b64_tr_b16 %[u1], %[a] offset:4608\n\t"
            "ds_read_b64_tr_b16 %[u2], %[a] offset:5120\n\t"
            "ds_read_b64_tr_b16 %[u3], %[a] offset:5632\n\t"
            "ds_read_b64_tr_b16 %[u4], %[a] offset:6144\n\t"
            "ds_read_b64_tr_b16 %[u5], %[a] offset:6656\n\t"
            "ds_read_b64_tr_b16 %[u6], %[a] offset:7168\n\t"
            "ds_read_b64_tr_b16 %[u7], %[a] offset:7680\n\t"
            "s_waitcnt lgkmcnt(0)"
            : [t0] "=&v"(tr[0]), [t1] "=&v"(tr[1]), [t2] "=&v"(tr[2]),
              [t3] "=&v"(tr[3]), [t4] "=&v"(tr[4]), [t5] "=&v"(tr[5]),
              [t6] "=&v"(tr[6]), [t7] "=&v"(tr[7]),
              [u0] "=&v"(tr2[0]), [u1] "=&v"(tr2[1]), [u2] "=&v"(tr2[2]),
              [u3] "=&v"(tr2[3]), [u4] "=&v"(tr2[4]), [u5] "=&v"(tr2[5]),
              [u6] "=&v"(tr2[6]), [u7] "=&v"(tr2[7])
            : [a] "v"(vaddr)
            : "memory");
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int n = 0; n < 8; n++) {
          bf16x8 bv;
          unsigned* bw = reinterpret_cast<unsigned*>(&bv);
          u32x2* src = (n < 4) ? tr : tr2;
          const int nn = n & 3;
          bw[0] = src[2 * nn][0];
          bw[1] = src[2 * nn][1];
          bw[2] = src[2 * nn + 1][0];
          bw[3] = src[2 * nn + 1][1];
          o_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, bv, o_acc[n],
                                                             0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: O /= l, store ----------------------------------------------
  if (wave_active) {
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int qrow = wq0 + crow4 + r;
      if (qrow >= q_len) continue;
      const float inv = (l_r[r] > 0.0f) ? 1.0f / l_r[r] : 0.0f;
      const long obase = (long)(q_start + qrow) * out_stride + (long)qh * PF_D;
#pragma unroll
      for (int n = 0; n < 8; n++) {
        out[obase + n * 16 + frow] = f32_to_bf16(o_acc[n][r] * inv);
      }
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
  dim3 grid(total_tiles, n_qheads), block(256);
  hipLaunchKernelGGL(paged_attn_prefill_kernel, grid, block, 0, stream, out, q,
                     k_cache, v_cache, block_tables, cu_q, seq_lens, tile_seq,
                     tile_q0, scale, n_qheads, n_kv_heads, max_blocks_per_seq,
                     q_stride, out_stride);
}

// ---------------------------------------------------------------------------------
// MFMA layout probe: C[16][16] = A[16][32] x B[32][16] with the fragment
// layout assumptions documented above. tests/test_gpu_mfma.py checks this
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
