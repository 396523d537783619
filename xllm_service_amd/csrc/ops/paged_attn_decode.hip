// Paged-attention DECODE kernel for MI355X (gfx950 / CDNA4) — MFMA-based.
//
// v2 design: the GQA head group IS the MFMA M dimension. For one
// (sequence, kv_head), the G query heads (G <= 16, zero-padded to 16) form
// the A operand rows, so QK^T and P·V are mfma_f32_16x16x32_bf16 chains and
// the VALU does only softmax bookkeeping — v1's shuffle-reduce design was
// VALU-bound at ~2.1 TB/s of KV stream; here MFMA issue is ~free and the
// kernel streams K/V at memory speed.
//
//   * one workgroup (4 waves) per (seq, kv_head); wave w owns KV tiles
//     w, w+4, w+8... of 32 keys (2 cache blocks) — sequence-partitioned
//     flash-decoding, merged through LDS at the end
//   * K feeds the QK^T B-fragment STRAIGHT from the paged cache (the
//     [tok][d] cache row layout is exactly the B-fragment gather: 16-B
//     vector loads per lane, no staging)
//   * V bounces through a per-wave LDS transpose to build P·V B-fragments
//   * per-row (=per-head) online softmax identical to the prefill kernel
//
// Fragment layouts as verified by tests/test_gpu_ops.py::test_mfma_fragment_layout.
// Parity: reference engine's paged-attention decode.
#include "common.h"

namespace xllm {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

#define PA_BS 16           // cache block size (tokens)
#define PA_KBLK 32         // keys per tile (2 cache blocks)
#define PA_D 128           // head_dim handled by the MFMA path
#define PA_PAD 8           // LDS row padding (elements)

// NOTE round-2: forcing 3 waves/SIMD via __launch_bounds__(256, 3) caps the
// unified VGPR+AGPR file at 166 and the allocator spills the K prefetch to
// SCRATCH — measured 3.1 TB/s vs 5.3 at the natural 232V+32A allocation
// (S=64 L=2048). The occupancy squeeze in docs/TODO_ROUND2.md is a dead
// end on gfx950's unified file; the kernel already streams 5.1-5.6 TB/s at
// serving shapes (B>=64, L>=1100).
template <int G>
__global__ __launch_bounds__(256) void paged_attn_decode_kernel(
    float* __restrict__ ws_ml,               // [S*n_kv*W][G][2] partial m,l
    float* __restrict__ ws_o,                // [S*n_kv*W][G][D] partial O
    const unsigned short* __restrict__ q,    // [num_seqs, n_qheads, D] bf16
    const unsigned short* __restrict__ k_cache,  // [blocks, n_kv, BS, D]
    const unsigned short* __restrict__ v_cache,
    const int* __restrict__ block_tables,    // [num_seqs, max_blocks]
    const int* __restrict__ seq_lens,        // [num_seqs]
    const float scale,
    const int n_kv_heads, const int max_blocks_per_seq,
    const long q_stride, const int P,       // P partitions per (seq, kvh)
    unsigned short* __restrict__ out,       // [S, n_qheads, D] (P == 1)
    const long out_stride) {
  const int sk = blockIdx.x / P;            // (seq, kvh) unit
  const int wgp = blockIdx.x % P;           // partition within the unit
  const int seq = sk / n_kv_heads;
  const int kvh = sk % n_kv_heads;
  const int n_qheads = n_kv_heads * G;
  const int seq_len = seq_lens[seq];
  const int n_tiles = (seq_len + PA_KBLK - 1) / PA_KBLK;
  const int nwaves = P * 4;                 // total partials per unit

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int frow = lane & 15;
  const int fcol8 = (lane >> 4) * 8;
  const int crow4 = (lane >> 4) * 4;

  // per-wave LDS scratch + cross-wave merge buffers.
  // vt_lds holds the tr16-compatible V image: per n-tile (16 d) and k-half
  // t, four [4 k][16 d] row-major blocks (one per 16-lane group), so P·V
  // B-fragments come from ds_read_b64_tr_b16 (guide T10) instead of a
  // scalar transpose. Element (k, d) lives at
  //   (d>>4)*512 + ((k>>2)&1)*256 + (k>>3)*64 + (k&3)*16 + (d&15).
  __shared__ unsigned short p_lds[4][16][PA_KBLK + PA_PAD];
  __shared__ unsigned short vt_lds[4][4096];

  // ---- Q as A-fragment: row = head-in-group (zero-padded to 16) -----------
  bf16x8 qf[4];
  if (frow < G) {
    const long qoff = (long)seq * q_stride + (long)(kvh * G + frow) * PA_D;
#pragma unroll
    for (int ks = 0; ks < 4; ks++)
      qf[ks] = *reinterpret_cast<const bf16x8*>(q + qoff + ks * 32 + fcol8);
  } else {
#pragma unroll
    for (int ks = 0; ks < 4; ks++) qf[ks] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
  }

  float m_r[4], l_r[4];
  f32x4 o_acc[8];
#pragma unroll
  for (int r = 0; r < 4; r++) { m_r[r] = -INFINITY; l_r[r] = 0.0f; }
#pragma unroll
  for (int n = 0; n < 8; n++) o_acc[n] = f32x4{0, 0, 0, 0};

  const int* btab = block_tables + (long)seq * max_blocks_per_seq;
  const long head_stride = (long)PA_BS * PA_D;

  // K/V loads of a tile: clamped-address loads (no per-load branch — guide
  // trap 4(c)); lanes beyond seq_len re-read the last token. K garbage is
  // masked to -inf in the softmax; V garbage is harmless because its P
  // weight is exactly 0 (finite * 0 = 0 — clamping, unlike zero-fill via a
  // branch, keeps the load stream unconditional).
  // Cache-block index per 16-token half: min(t0/16+n, last_blk) is exact
  // for clamped lanes too (the boundary tile's in-range lanes live in the
  // same block the clamp resolves to) and — crucially — UNIFORM across the
  // wave, so btab[...] compiles to a scalar load instead of a per-lane
  // VMEM load sitting in the K/V address chain every tile.
  const int last_blk = (seq_len - 1) / PA_BS;
  auto load_blks = [&](int tile, int& b0, int& b1) {
    // uniform per tile: min(2*tile+n, last_blk) is exact for clamped lanes
    b0 = btab[min(tile * (PA_KBLK / PA_BS), last_blk)];
    b1 = btab[min(tile * (PA_KBLK / PA_BS) + 1, last_blk)];
  };
  auto base_of = [&](int blk) -> long {
    return ((long)blk * n_kv_heads + kvh) * head_stride;
  };
  auto load_k = [&](bf16x8 (&kb)[2][4], int tile, int b0, int b1) {
    const int t0 = tile * PA_KBLK;
#pragma unroll
    for (int n = 0; n < 2; n++) {
      const int tok = t0 + n * 16 + frow;
      const int tok_c = tok < seq_len ? tok : seq_len - 1;
      const long rbase = base_of(n ? b1 : b0) +
                         (long)(tok_c % PA_BS) * PA_D;
#pragma unroll
      for (int ks = 0; ks < 4; ks++)
        kb[n][ks] = *reinterpret_cast<const bf16x8*>(
            k_cache + rbase + ks * 32 + fcol8);
    }
  };
  // lane covers token tv = lane>>1, d-half dv = (lane&1)*64
  const int tv = lane >> 1;
  const int dv = (lane & 1) * 64;
  const int koff = ((tv >> 2) & 1) * 256 + (tv >> 3) * 64 + (tv & 3) * 16;
  auto stage_v = [&](int tile, int b0, int b1) {
    // V is loaded and written to the tr16 LDS image in-tile with transient
    // registers (8-16 live): K owns the persistent prefetch buffers, and
    // holding V too pushes past 256 regs into scratch spills (measured).
    const int tok = tile * PA_KBLK + tv;
    const int tok_c = tok < seq_len ? tok : seq_len - 1;
    const long vbase = base_of(tv >> 4 ? b1 : b0) +
                       (long)(tok_c % PA_BS) * PA_D + dv;
#pragma unroll
    for (int mseg = 0; mseg < 8; mseg++) {
      const ushort8_t vv = *reinterpret_cast<const ushort8_t*>(
          v_cache + vbase + mseg * 8);
      const int d0 = dv + mseg * 8;
      *reinterpret_cast<ushort8_t*>(
          &vt_lds[wid][(d0 >> 4) * 512 + koff + (d0 & 15)]) = vv;
    }
  };

  // Pipeline: K is double-buffered (each tile's K loads issue a FULL tile
  // ahead), V stages in-tile through transient regs, and the uniform block
  // ids both need are scalar-loaded TWO iterations ahead (cb/fb/gb
  // rotation) so the btab lookup never sits in any load's address chain.
  bf16x8 kb_a[2][4], kb_b[2][4];
  const int tile0 = wgp * 4 + wid;
  int cb0 = 0, cb1 = 0;      // ids for the tile being computed
  int fb0 = 0, fb1 = 0;      // ids for the next tile (K prefetch target)
  if (tile0 < n_tiles) {
    load_blks(tile0, cb0, cb1);
    load_k(kb_a, tile0, cb0, cb1);
    if (tile0 + nwaves < n_tiles) load_blks(tile0 + nwaves, fb0, fb1);
  }

  for (int tile = tile0, phase = 0; tile < n_tiles;
       tile += nwaves, phase ^= 1) {
    const int t0 = tile * PA_KBLK;
    bf16x8 (&kb)[2][4] = phase ? kb_b : kb_a;
    bf16x8 (&kb_next)[2][4] = phase ? kb_a : kb_b;
    // ids for tile+2*nwaves: a full iteration before their first use
    int gb0 = 0, gb1 = 0;
    if (tile + 2 * nwaves < n_tiles) load_blks(tile + 2 * nwaves, gb0, gb1);
    // K loads for the next tile: full-tile latency distance
    if (tile + nwaves < n_tiles) load_k(kb_next, tile + nwaves, fb0, fb1);

    // ---- S = Q K^T from the prefetched fragments --------------------------
    f32x4 s[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
#pragma unroll
    for (int n = 0; n < 2; n++) {
#pragma unroll
      for (int ks = 0; ks < 4; ks++)
        s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[ks], kb[n][ks],
                                                       s[n], 0, 0, 0);
    }
    // ---- stage V into the tr16 image (in-tile, transient regs) ------------
    stage_v(tile, cb0, cb1);
    cb0 = fb0; cb1 = fb1; fb0 = gb0; fb1 = gb1;

    // ---- mask + online softmax (rows are heads) ---------------------------
    float p_val[2][4];
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; r++) {
      float smax = -INFINITY;
#pragma unroll
      for (int n = 0; n < 2; n++) {
        float sv = s[n][r] * scale;
        if (t0 + n * 16 + frow >= seq_len) sv = -INFINITY;
        p_val[n][r] = sv;
        smax = fmaxf(smax, sv);
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        smax = fmaxf(smax, __shfl_xor(smax, off, 64));
      const float m_new = fmaxf(m_r[r], smax);
      alpha[r] = (m_new == -INFINITY) ? 1.0f : __expf(m_r[r] - m_new);
      float psum = 0.0f;
#pragma unroll
      for (int n = 0; n < 2; n++) {
        const float p = (p_val[n][r] == -INFINITY)
                            ? 0.0f : __expf(p_val[n][r] - m_new);
        p_val[n][r] = p;
        psum += p;
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) psum += __shfl_xor(psum, off, 64);
      m_r[r] = m_new;
      l_r[r] = l_r[r] * alpha[r] + psum;
    }
    // P -> LDS -> A-fragment
#pragma unroll
    for (int n = 0; n < 2; n++) {
#pragma unroll
      for (int r = 0; r < 4; r++)
        p_lds[wid][crow4 + r][n * 16 + frow] = f32_to_bf16(p_val[n][r]);
    }
#pragma unroll
    for (int n = 0; n < 8; n++) {
#pragma unroll
      for (int r = 0; r < 4; r++) o_acc[n][r] *= alpha[r];
    }
    bf16x8 pa = *reinterpret_cast<const bf16x8*>(&p_lds[wid][frow][fcol8]);
    // ---- O += P V (B-fragments via hardware transpose reads) --------------
    // per tr read a 16-lane group gathers one [4 k][16 d] block; lane
    // address = image + (l>>4)*128B + (l&15)*8B, sub-tiles via offset:
    {
      typedef __attribute__((__vector_size__(2 * sizeof(unsigned)))) unsigned u32x2;
      // low 32 bits of a generic LDS pointer ARE the LDS byte offset
      // (the shared aperture is 2^32-aligned on amdgcn)
      const unsigned vaddr =
          (unsigned)(unsigned long long)(&vt_lds[wid][0]) +
          ((lane >> 4) * 128u + (lane & 15) * 8u);
      // one half-batch of 8 tr reads live at a time (16 result VGPRs,
      // reused across halves: the kernel is register-occupancy-bound)
#define PA_TR8(OFF0)                                                       \
      asm volatile(                                                        \
          "ds_read_b64_tr_b16 %[t0], %[a] offset:" #OFF0 "+0\n\t"          \
          "ds_read_b64_tr_b16 %[t1], %[a] offset:" #OFF0 "+512\n\t"        \
          "ds_read_b64_tr_b16 %[t2], %[a] offset:" #OFF0 "+1024\n\t"       \
          "ds_read_b64_tr_b16 %[t3], %[a] offset:" #OFF0 "+1536\n\t"       \
          "ds_read_b64_tr_b16 %[t4], %[a] offset:" #OFF0 "+2048\n\t"       \
          "ds_read_b64_tr_b16 %[t5], %[a] offset:" #OFF0 "+2560\n\t"       \
          "ds_read_b64_tr_b16 %[t6], %[a] offset:" #OFF0 "+3072\n\t"       \
          "ds_read_b64_tr_b16 %[t7], %[a] offset:" #OFF0 "+3584\n\t"       \
          "s_waitcnt lgkmcnt(0)"                                           \
          : [t0] "=&v"(tr[0]), [t1] "=&v"(tr[1]), [t2] "=&v"(tr[2]),       \
            [t3] "=&v"(tr[3]), [t4] "=&v"(tr[4]), [t5] "=&v"(tr[5]),       \
            [t6] "=&v"(tr[6]), [t7] "=&v"(tr[7])                           \
          : [a] "v"(vaddr)                                                 \
          : "memory")
      u32x2 tr[8];
#pragma unroll
      for (int half = 0; half < 2; half++) {
        if (half == 0) { PA_TR8(0); } else { PA_TR8(4096); }
        __builtin_amdgcn_sched_barrier(0);  // MFMAs stay below the wait
#pragma unroll
        for (int nn = 0; nn < 4; nn++) {
          const int n = half * 4 + nn;
          bf16x8 bv;
          unsigned* bw = reinterpret_cast<unsigned*>(&bv);
          bw[0] = tr[2 * nn][0];
          bw[1] = tr[2 * nn][1];
          bw[2] = tr[2 * nn + 1][0];
          bw[3] = tr[2 * nn + 1][1];
          o_acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, bv, o_acc[n],
                                                             0, 0, 0);
        }
      }
#undef PA_TR8
    }
  }

  if (P == 1) {
    // ---- fused merge: combine the 4 per-wave partials through LDS and
    // write the output directly — saves the merge-kernel launch plus the
    // global ws_o/ws_ml round-trip (~8 MB at batch 64). vt_lds (32 KB) and
    // p_lds are dead by now and are reused as the merge buffers.
    float* mo = reinterpret_cast<float*>(&vt_lds[0][0]);    // [4][G][128]
    float* mml = reinterpret_cast<float*>(&p_lds[0][0][0]); // [4][G][2]
    __syncthreads();
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int row = crow4 + r;
      if (row < G) {
        if (frow == 0) {
          mml[(wid * G + row) * 2 + 0] = m_r[r];
          mml[(wid * G + row) * 2 + 1] = l_r[r];
        }
#pragma unroll
        for (int n = 0; n < 8; n++)
          mo[((wid * G + row) * PA_D) + n * 16 + frow] = o_acc[n][r];
      }
    }
    __syncthreads();
    if (wid == 0) {
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const int row = crow4 + r;
        if (row >= G) continue;
        float m_star = -INFINITY;
#pragma unroll
        for (int w = 0; w < 4; w++)
          m_star = fmaxf(m_star, mml[(w * G + row) * 2]);
        float wgt[4], l_tot = 0.0f;
#pragma unroll
        for (int w = 0; w < 4; w++) {
          const float mw = mml[(w * G + row) * 2];
          wgt[w] = (mw == -INFINITY) ? 0.0f : __expf(mw - m_star);
          l_tot += wgt[w] * mml[(w * G + row) * 2 + 1];
        }
        const float inv = (l_tot > 0.0f) ? 1.0f / l_tot : 0.0f;
#pragma unroll
        for (int n = 0; n < 8; n++) {
          float o = 0.0f;
#pragma unroll
          for (int w = 0; w < 4; w++)
            o += wgt[w] * mo[((w * G + row) * PA_D) + n * 16 + frow];
          out[(long)seq * out_stride + (long)(kvh * G + row) * PA_D +
              n * 16 + frow] = f32_to_bf16(o * inv);
        }
      }
    }
    return;
  }

  // ---- write this wave's partial (no cross-wave sync: the tiny merge
  // kernel below combines the P*4 partials per (seq, kvh) unit) ------------
  const long part = ((long)sk * nwaves + wgp * 4 + wid);
  if (frow == 0) {
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int row = crow4 + r;
      if (row < G) {
        ws_ml[(part * G + row) * 2 + 0] = m_r[r];
        ws_ml[(part * G + row) * 2 + 1] = l_r[r];
      }
    }
  }
#pragma unroll
  for (int n = 0; n < 8; n++) {
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int row = crow4 + r;
      if (row < G)
        ws_o[(part * G + row) * PA_D + n * 16 + frow] = o_acc[n][r];
    }
  }
}

// Combine the W = P*4 per-wave partials of each (seq, kvh, g) row.
__global__ void paged_attn_decode_merge_kernel(
    unsigned short* __restrict__ out,        // [S, n_qheads, D]
    const float* __restrict__ ws_ml,         // [S*n_kv*W][G][2]
    const float* __restrict__ ws_o,          // [S*n_kv*W][G][D]
    const int G, const int n_kv_heads, const int W, const long out_stride,
    const long total) {                      // total = S * n_kv * G * D
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int d = (int)(idx % PA_D);
    long t = idx / PA_D;
    const int g = (int)(t % G);
    t /= G;
    const int kvh = (int)(t % n_kv_heads);
    const long seq = t / n_kv_heads;
    const long unit = (seq * n_kv_heads + kvh);
    float m_star = -INFINITY;
    for (int w = 0; w < W; w++)
      m_star = fmaxf(m_star, ws_ml[((unit * W + w) * G + g) * 2]);
    float l_tot = 0.0f, o = 0.0f;
    for (int w = 0; w < W; w++) {
      const float mw = ws_ml[((unit * W + w) * G + g) * 2];
      const float wgt = (mw == -INFINITY) ? 0.0f : __expf(mw - m_star);
      l_tot += wgt * ws_ml[((unit * W + w) * G + g) * 2 + 1];
      o += wgt * ws_o[((unit * W + w) * G + g) * PA_D + d];
    }
    const float inv = (l_tot > 0.0f) ? 1.0f / l_tot : 0.0f;
    out[seq * out_stride + (long)(kvh * G + g) * PA_D + d] =
        f32_to_bf16(o * inv);
  }
}

// ---------------- fallback (head_dim != 128): v1 shuffle-reduce design ------
template <int G>
__global__ __launch_bounds__(256) void paged_attn_decode_small_kernel(
    unsigned short* __restrict__ out, const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k_cache,
    const unsigned short* __restrict__ v_cache,
    const int* __restrict__ block_tables, const int* __restrict__ seq_lens,
    const float scale, const int n_kv_heads, const int D,
    const int max_blocks_per_seq, const long q_stride, const long out_stride) {
  const int seq = blockIdx.x / n_kv_heads;
  const int kvh = blockIdx.x % n_kv_heads;
  const int n_qheads = n_kv_heads * G;
  const int seq_len = seq_lens[seq];
  const int n_blocks = (seq_len + PA_BS - 1) / PA_BS;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int tg = lane >> 4;
  const int d8 = lane & 15;
  const int dvalid = (8 * d8) < D;

  float qf[G][8];
#pragma unroll
  for (int g = 0; g < G; g++) {
    const long qoff = (long)seq * q_stride + (long)(kvh * G + g) * D + 8 * d8;
    if (dvalid) {
      ushort8_t v = *reinterpret_cast<const ushort8_t*>(q + qoff);
#pragma unroll
      for (int j = 0; j < 8; j++) qf[g][j] = bf16_to_f32(v.x[j]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; j++) qf[g][j] = 0.0f;
    }
  }
  float m[G], l[G], acc[G][8];
#pragma unroll
  for (int g = 0; g < G; g++) {
    m[g] = -INFINITY; l[g] = 0.0f;
#pragma unroll
    for (int j = 0; j < 8; j++) acc[g][j] = 0.0f;
  }
  const int* btab = block_tables + (long)seq * max_blocks_per_seq;
  const long head_stride = (long)PA_BS * D;
  for (int bi = wid; bi < n_blocks; bi += 4) {
    const long base = ((long)btab[bi] * n_kv_heads + kvh) * head_stride;
#pragma unroll
    for (int j = 0; j < 4; j++) {
      const int t = j * 4 + tg;
      const int tok = bi * PA_BS + t;
      if (tok >= seq_len) continue;
      const long roff = base + (long)t * D + 8 * d8;
      float kf[8], vf[8];
      if (dvalid) {
        ushort8_t k8 = *reinterpret_cast<const ushort8_t*>(k_cache + roff);
        ushort8_t v8 = *reinterpret_cast<const ushort8_t*>(v_cache + roff);
#pragma unroll
        for (int u = 0; u < 8; u++) { kf[u] = bf16_to_f32(k8.x[u]); vf[u] = bf16_to_f32(v8.x[u]); }
      } else {
#pragma unroll
        for (int u = 0; u < 8; u++) { kf[u] = 0.0f; vf[u] = 0.0f; }
      }
#pragma unroll
      for (int g = 0; g < G; g++) {
        float s = 0.0f;
#pragma unroll
        for (int u = 0; u < 8; u++) s += qf[g][u] * kf[u];
        s = group_reduce_sum<16>(s) * scale;
        const float m_new = fmaxf(m[g], s);
        const float a = __expf(m[g] - m_new);
        const float p = __expf(s - m_new);
        m[g] = m_new;
        l[g] = l[g] * a + p;
#pragma unroll
        for (int u = 0; u < 8; u++) acc[g][u] = acc[g][u] * a + p * vf[u];
      }
    }
  }
  __shared__ float ml[16][G][2];
  __shared__ float accs[16][G][16][8];
  const int part = wid * 4 + tg;
  if (d8 == 0) {
#pragma unroll
    for (int g = 0; g < G; g++) { ml[part][g][0] = m[g]; ml[part][g][1] = l[g]; }
  }
#pragma unroll
  for (int g = 0; g < G; g++) {
#pragma unroll
    for (int u = 0; u < 8; u++) accs[part][g][d8][u] = acc[g][u];
  }
  __syncthreads();
  for (int g = wid; g < G; g += 4) {
    float m_star = -INFINITY;
#pragma unroll
    for (int p = 0; p < 16; p++) m_star = fmaxf(m_star, ml[p][g][0]);
    float l_tot = 0.0f, o0 = 0.0f, o1 = 0.0f;
    const int pd8 = lane >> 2;
    const int pu = (2 * lane) & 7;
#pragma unroll
    for (int p = 0; p < 16; p++) {
      const float mw = ml[p][g][0];
      const float w = (mw == -INFINITY) ? 0.0f : __expf(mw - m_star);
      l_tot += w * ml[p][g][1];
      o0 += w * accs[p][g][pd8][pu];
      o1 += w * accs[p][g][pd8][pu + 1];
    }
    const float inv = (l_tot > 0.0f) ? 1.0f / l_tot : 0.0f;
    if (2 * lane < D) {
      const long ooff = (long)seq * out_stride + (long)(kvh * G + g) * D + 2 * lane;
      out[ooff] = f32_to_bf16(o0 * inv);
      out[ooff + 1] = f32_to_bf16(o1 * inv);
    }
  }
}

#define PA_DISPATCH_G(GV)                                                      \
  do {                                                                         \
    if (D == PA_D) {                                                           \
      hipLaunchKernelGGL((paged_attn_decode_kernel<GV>), grid, block, 0,       \
                         stream, ws_ml, ws_o, q, k_cache, v_cache,             \
                         block_tables, seq_lens, scale, n_kv_heads,            \
                         max_blocks_per_seq, q_stride, P, out, out_stride);    \
      if (P > 1)                                                               \
        hipLaunchKernelGGL(paged_attn_decode_merge_kernel, mgrid, block, 0,    \
                           stream, out, ws_ml, ws_o, GV, n_kv_heads, P * 4,    \
                           out_stride, total);                                 \
    } else {                                                                   \
      hipLaunchKernelGGL((paged_attn_decode_small_kernel<GV>), grid, block, 0, \
                         stream, out, q, k_cache, v_cache, block_tables,       \
                         seq_lens, scale, n_kv_heads, D, max_blocks_per_seq,   \
                         q_stride, out_stride);                                \
    }                                                                          \
  } while (0)

int paged_attn_decode_partitions(int num_seqs, int n_kv_heads) {
  // partition only when there are too few (seq, kv_head) units to fill the
  // chip: extra partitions shrink per-wave work and cost merge overhead
  const int units = num_seqs * n_kv_heads;
  int P = (512 + units - 1) / units;
  if (P < 1) P = 1;
  if (P > 8) P = 8;
  return P;
}

void launch_paged_attn_decode(unsigned short* out, const unsigned short* q,
                              const unsigned short* k_cache,
                              const unsigned short* v_cache,
                              const int* block_tables, const int* seq_lens,
                              float* ws_ml, float* ws_o,
                              float scale, int num_seqs, int n_qheads,
                              int n_kv_heads, int D, int max_blocks_per_seq,
                              long q_stride, long out_stride,
                              hipStream_t stream) {
  const int P = paged_attn_decode_partitions(num_seqs, n_kv_heads);
  dim3 grid(num_seqs * n_kv_heads * (D == PA_D ? P : 1)), block(256);
  const int G = n_qheads / n_kv_heads;
  const long total = (long)num_seqs * n_kv_heads * G * PA_D;
  long mg = (total + 255) / 256;
  if (mg > 2048) mg = 2048;
  dim3 mgrid((unsigned)mg);
  switch (G) {
    case 1: PA_DISPATCH_G(1); break;
    case 2: PA_DISPATCH_G(2); break;
    case 3: PA_DISPATCH_G(3); break;
    case 4: PA_DISPATCH_G(4); break;
    case 5: PA_DISPATCH_G(5); break;
    case 6: PA_DISPATCH_G(6); break;
    case 7: PA_DISPATCH_G(7); break;
    case 8: PA_DISPATCH_G(8); break;
    default: break;  // validated host-side
  }
}

}  // namespace xllm
