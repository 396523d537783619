// PyTorch bindings for the MI355X (gfx950) HIP kernels.
//
// All tensor-shape/dtype validation lives here so the kernels stay lean.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <vector>
#include <cstring>

namespace xllm {
void launch_rmsnorm(unsigned short*, const unsigned short*,
                    const unsigned short*, float, int, int, hipStream_t);
void launch_fused_add_rmsnorm(unsigned short*, unsigned short*,
                              const unsigned short*, float, int, int,
                              hipStream_t);
void launch_rope(unsigned short*, unsigned short*, const long*, const float*,
                 int, int, int, int, int, long, long, hipStream_t);
void launch_fused_rope_cache(unsigned short*, const unsigned short*,
                             const unsigned short*, unsigned short*,
                             unsigned short*, const long*, const long*,
                             const float*, int, int, int, int, int, long,
                             long, int, int, int, hipStream_t);
void launch_silu_and_mul(unsigned short*, const unsigned short*, int, int,
                         hipStream_t);
void launch_gelu_and_mul(unsigned short*, const unsigned short*, int, int,
                         hipStream_t);
void launch_reshape_and_cache(unsigned short*, unsigned short*,
                              const unsigned short*, const unsigned short*,
                              const long*, int, int, int, int, long,
                              hipStream_t);
void launch_copy_blocks(unsigned short*, unsigned short*, const long*, int,
                        long, hipStream_t);
void launch_gather_blocks(unsigned short*, unsigned short*, const long*, int,
                          long, bool, hipStream_t);
void launch_paged_attn_decode(unsigned short*, const unsigned short*,
                              const unsigned short*, const unsigned short*,
                              const int*, const int*, float*, float*, float,
                              int, int, int, int, int, long, long,
                              hipStream_t);
int paged_attn_decode_partitions(int, int);
void launch_paged_attn_prefill(unsigned short*, const unsigned short*,
                               const unsigned short*, const unsigned short*,
                               const int*, const int*, const int*, const int*,
                               const int*, int, float, int, int, int, long,
                               long, hipStream_t);
void launch_greedy_sample(long*, const unsigned short*, int, int, hipStream_t);
void launch_mfma_gemm(unsigned short*, const unsigned short*,
                      const unsigned short*, const unsigned short*, int, int,
                      int, hipStream_t);
int skinny_gemm_splitk(int M, int N, int K);
void launch_skinny_gemm(unsigned short*, const unsigned short*,
                        const unsigned short*, const unsigned short*, float*,
                        int, int, int, hipStream_t);
int packed_gemm_splitk(int M, int N, int K, int S);
void launch_packed_gemm(unsigned short*, const unsigned short*,
                        unsigned short*, const unsigned short*,
                        const unsigned short*, float*, int, int, int, int,
                        hipStream_t);
int packed_gemm_pick_s(int N);
void launch_packed_gemm_probe(unsigned short*, const unsigned short*,
                              const unsigned short*, float*, int, int, int,
                              hipStream_t);
void launch_mfma_probe(float*, const unsigned short*, const unsigned short*,
                       hipStream_t);
}  // namespace xllm

namespace {

#define CHECK_BF16_CUDA(t)                                      \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");             \
  TORCH_CHECK((t).dtype() == torch::kBFloat16, #t " must be bf16"); \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

// bf16 [T, H, D] (or [T, H*D]) view: innermost contiguous, heads packed;
// rows may be strided (views into a fused qkv output)
#define CHECK_BF16_ROWVIEW(t)                                              \
  TORCH_CHECK((t).is_cuda() && (t).dtype() == torch::kBFloat16,            \
              #t " must be bf16 on GPU");                                  \
  TORCH_CHECK((t).stride(-1) == 1, #t " innermost dim must be contiguous");\
  TORCH_CHECK((t).dim() < 3 || (t).stride(1) == (t).size(2),               \
              #t " heads must be packed")

inline unsigned short* u16(torch::Tensor& t) {
  return reinterpret_cast<unsigned short*>(t.data_ptr());
}
inline const unsigned short* u16c(const torch::Tensor& t) {
  return reinterpret_cast<const unsigned short*>(t.data_ptr());
}
inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor w, double eps) {
  CHECK_BF16_CUDA(out); CHECK_BF16_CUDA(x); CHECK_BF16_CUDA(w);
  const int H = x.size(-1);
  const int T = x.numel() / H;
  TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
  xllm::launch_rmsnorm(u16(out), u16c(x), u16c(w), (float)eps, T, H,
                       cur_stream());
}

void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor w, double eps) {
  CHECK_BF16_CUDA(x); CHECK_BF16_CUDA(residual); CHECK_BF16_CUDA(w);
  const int H = x.size(-1);
  const int T = x.numel() / H;
  TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
  xllm::launch_fused_add_rmsnorm(u16(x), u16(residual), u16c(w), (float)eps, T,
                                 H, cur_stream());
}

void rope(torch::Tensor positions, torch::Tensor q, torch::Tensor k,
          torch::Tensor cos_sin, long head_dim, long rot_dim) {
  CHECK_BF16_ROWVIEW(q); CHECK_BF16_ROWVIEW(k);
  TORCH_CHECK(positions.dtype() == torch::kLong && positions.is_cuda());
  TORCH_CHECK(cos_sin.dtype() == torch::kFloat && cos_sin.is_cuda());
  const int T = positions.size(0);
  const int n_q = q.numel() / T / head_dim;
  const int n_k = k.numel() / T / head_dim;
  xllm::launch_rope(u16(q), u16(k), positions.data_ptr<long>(),
                    cos_sin.data_ptr<float>(), T, n_q, n_k, (int)head_dim,
                    (int)rot_dim, q.stride(0), k.stride(0), cur_stream());
}

void fused_rope_cache(torch::Tensor positions, torch::Tensor q,
                      torch::Tensor k, torch::Tensor v,
                      torch::Tensor k_cache, torch::Tensor v_cache,
                      torch::Tensor slot_mapping, torch::Tensor cos_sin,
                      long rot_dim, long ms0, long ms1) {
  // ms0/ms1: cumulative M-RoPE frequency-section bounds; 0 = 1-D rope.
  // With ms0 > 0, positions is [3, T] (temporal / height / width rows).
  CHECK_BF16_ROWVIEW(q); CHECK_BF16_ROWVIEW(k); CHECK_BF16_ROWVIEW(v);
  CHECK_BF16_CUDA(k_cache); CHECK_BF16_CUDA(v_cache);
  TORCH_CHECK(positions.dtype() == torch::kLong && positions.is_cuda() &&
              positions.is_contiguous());
  TORCH_CHECK(ms0 == 0 ? positions.dim() == 1
                       : (positions.dim() == 2 && positions.size(0) == 3));
  TORCH_CHECK(slot_mapping.dtype() == torch::kLong && slot_mapping.is_cuda());
  TORCH_CHECK(cos_sin.dtype() == torch::kFloat && cos_sin.is_cuda());
  TORCH_CHECK(k.stride(0) == v.stride(0));
  const int T = positions.size(positions.dim() - 1);
  const int n_kv = k_cache.size(1);
  const int bs = k_cache.size(2);
  const int D = k_cache.size(3);
  const int n_q = q.numel() / T / D;
  xllm::launch_fused_rope_cache(
      u16(q), u16c(k), u16c(v), u16(k_cache), u16(v_cache),
      positions.data_ptr<long>(), slot_mapping.data_ptr<long>(),
      cos_sin.data_ptr<float>(), T, n_q, n_kv, D, (int)rot_dim, q.stride(0),
      k.stride(0), bs, (int)ms0, (int)ms1, cur_stream());
}

void silu_and_mul(torch::Tensor out, torch::Tensor x) {
  CHECK_BF16_CUDA(out); CHECK_BF16_CUDA(x);
  const int I = out.size(-1);
  const int T = out.numel() / I;
  TORCH_CHECK(x.size(-1) == 2 * I && I % 8 == 0);
  xllm::launch_silu_and_mul(u16(out), u16c(x), T, I, cur_stream());
}

void gelu_and_mul(torch::Tensor out, torch::Tensor x) {
  CHECK_BF16_CUDA(out); CHECK_BF16_CUDA(x);
  const int I = out.size(-1);
  const int T = out.numel() / I;
  TORCH_CHECK(x.size(-1) == 2 * I && I % 8 == 0);
  xllm::launch_gelu_and_mul(u16(out), u16c(x), T, I, cur_stream());
}

void reshape_and_cache(torch::Tensor k, torch::Tensor v,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor slot_mapping) {
  CHECK_BF16_ROWVIEW(k); CHECK_BF16_ROWVIEW(v);
  CHECK_BF16_CUDA(k_cache); CHECK_BF16_CUDA(v_cache);
  TORCH_CHECK(slot_mapping.dtype() == torch::kLong && slot_mapping.is_cuda());
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v must share row stride");
  const int T = k.size(0);
  const int n_kv = k_cache.size(1);
  const int bs = k_cache.size(2);
  const int D = k_cache.size(3);
  TORCH_CHECK(D % 8 == 0);
  xllm::launch_reshape_and_cache(u16(k_cache), u16(v_cache), u16c(k), u16c(v),
                                 slot_mapping.data_ptr<long>(), T, n_kv, D, bs,
                                 k.stride(0), cur_stream());
}

void copy_blocks(torch::Tensor k_cache, torch::Tensor v_cache,
                 torch::Tensor pairs) {
  CHECK_BF16_CUDA(k_cache); CHECK_BF16_CUDA(v_cache);
  TORCH_CHECK(pairs.dtype() == torch::kLong && pairs.is_cuda() &&
              pairs.is_contiguous());
  const long numel = (long)k_cache.size(1) * k_cache.size(2) * k_cache.size(3);
  xllm::launch_copy_blocks(u16(k_cache), u16(v_cache), pairs.data_ptr<long>(),
                           pairs.size(0), numel, cur_stream());
}

void gather_blocks(torch::Tensor staging, torch::Tensor cache,
                   torch::Tensor block_ids, bool gather) {
  CHECK_BF16_CUDA(staging); CHECK_BF16_CUDA(cache);
  TORCH_CHECK(block_ids.dtype() == torch::kLong && block_ids.is_cuda());
  const long numel = (long)cache.size(1) * cache.size(2) * cache.size(3);
  xllm::launch_gather_blocks(u16(staging), u16(cache),
                             block_ids.data_ptr<long>(), block_ids.size(0),
                             numel, gather, cur_stream());
}

void paged_attn_decode(torch::Tensor out, torch::Tensor q,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor block_tables, torch::Tensor seq_lens,
                       double scale) {
  CHECK_BF16_ROWVIEW(out); CHECK_BF16_ROWVIEW(q);
  CHECK_BF16_CUDA(k_cache); CHECK_BF16_CUDA(v_cache);
  TORCH_CHECK(block_tables.dtype() == torch::kInt && block_tables.is_cuda());
  TORCH_CHECK(seq_lens.dtype() == torch::kInt && seq_lens.is_cuda());
  const int num_seqs = q.size(0);
  const int n_qheads = q.size(1);
  const int D = q.size(2);
  const int n_kv = k_cache.size(1);
  TORCH_CHECK(k_cache.size(2) == 16, "block_size must be 16");
  TORCH_CHECK(D <= 128 && D % 8 == 0, "head_dim must be <=128, mult of 8");
  TORCH_CHECK(n_qheads % n_kv == 0 && n_qheads / n_kv <= 8,
              "GQA group must be <= 8");
  const int G = n_qheads / n_kv;
  const int W = xllm::paged_attn_decode_partitions(num_seqs, n_kv) * 4;
  auto fopt = q.options().dtype(torch::kFloat);
  torch::Tensor ws_ml, ws_o;
  if (D == 128 && W > 4) {  // P == 1 merges in-kernel: no workspace
    ws_ml = torch::empty({(long)num_seqs * n_kv * W * G * 2}, fopt);
    ws_o = torch::empty({(long)num_seqs * n_kv * W * G * 128}, fopt);
  } else {  // small-head fallback kernel merges in-workgroup
    ws_ml = torch::empty({1}, fopt);
    ws_o = torch::empty({1}, fopt);
  }
  xllm::launch_paged_attn_decode(
      u16(out), u16c(q), u16c(k_cache), u16c(v_cache),
      block_tables.data_ptr<int>(), seq_lens.data_ptr<int>(),
      ws_ml.data_ptr<float>(), ws_o.data_ptr<float>(), (float)scale,
      num_seqs, n_qheads, n_kv, D, block_tables.size(1), q.stride(0),
      out.stride(0), cur_stream());
}

void paged_attn_prefill(torch::Tensor out, torch::Tensor q,
                        torch::Tensor k_cache, torch::Tensor v_cache,
                        torch::Tensor block_tables, torch::Tensor cu_q,
                        torch::Tensor seq_lens, torch::Tensor tile_seq,
                        torch::Tensor tile_q0, double scale) {
  CHECK_BF16_ROWVIEW(out); CHECK_BF16_ROWVIEW(q);
  CHECK_BF16_CUDA(k_cache); CHECK_BF16_CUDA(v_cache);
  for (auto* t : {&block_tables, &cu_q, &seq_lens, &tile_seq, &tile_q0}) {
    TORCH_CHECK(t->dtype() == torch::kInt && t->is_cuda() && t->is_contiguous());
  }
  const int n_qheads = q.size(1);
  const int D = q.size(2);
  const int n_kv = k_cache.size(1);
  TORCH_CHECK(D == 128, "prefill kernel requires head_dim == 128");
  TORCH_CHECK(k_cache.size(2) == 16, "block_size must be 16");
  xllm::launch_paged_attn_prefill(
      u16(out), u16c(q), u16c(k_cache), u16c(v_cache),
      block_tables.data_ptr<int>(), cu_q.data_ptr<int>(),
      seq_lens.data_ptr<int>(), tile_seq.data_ptr<int>(),
      tile_q0.data_ptr<int>(), tile_seq.size(0), (float)scale, n_qheads, n_kv,
      block_tables.size(1), q.stride(0), out.stride(0), cur_stream());
}

torch::Tensor greedy_sample(torch::Tensor logits) {
  CHECK_BF16_CUDA(logits);
  const int B = logits.size(0);
  const int V = logits.size(1);
  auto out = torch::empty({B}, logits.options().dtype(torch::kLong));
  xllm::launch_greedy_sample(out.data_ptr<long>(), u16c(logits), B, V,
                             cur_stream());
  return out;
}

torch::Tensor mfma_gemm(torch::Tensor a, torch::Tensor b,
                        c10::optional<torch::Tensor> bias) {
  // a: [M, K]; b: [N, K] (torch weight layout); returns [M, N]
  CHECK_BF16_CUDA(a); CHECK_BF16_CUDA(b);
  TORCH_CHECK(a.size(1) == b.size(1), "K mismatch");
  TORCH_CHECK(a.size(1) % 8 == 0, "K must be a multiple of 8");
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  auto c = torch::empty({M, N}, a.options());
  const unsigned short* bp = nullptr;
  if (bias.has_value()) {
    CHECK_BF16_CUDA(bias.value());
    bp = u16c(bias.value());
  }
  xllm::launch_mfma_gemm(u16(c), u16c(a), u16c(b), bp, M, N, K,
                         cur_stream());
  return c;
}

torch::Tensor skinny_gemm(torch::Tensor a, torch::Tensor b,
                          c10::optional<torch::Tensor> bias) {
  // a: [M, K]; b: [N, K]; returns [M, N]
  CHECK_BF16_CUDA(a); CHECK_BF16_CUDA(b);
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(a.size(1) == b.size(1) && M <= 256);
  TORCH_CHECK(K % 64 == 0 && N % 4 == 0);
  auto c = torch::empty({M, N}, a.options());
  const int SK = xllm::skinny_gemm_splitk(M, N, K);
  const long mpad = ((M + 15) / 16) * 16;
  torch::Tensor ws;
  if (SK > 1)
    ws = torch::empty({(long)SK * mpad * N}, a.options().dtype(torch::kFloat));
  else
    ws = torch::empty({1}, a.options().dtype(torch::kFloat));
  const unsigned short* bp = nullptr;
  if (bias.has_value()) bp = u16c(bias.value());
  xllm::launch_skinny_gemm(u16(c), u16c(a), u16c(b), bp,
                           ws.data_ptr<float>(), M, N, K, cur_stream());
  return c;
}

torch::Tensor packed_gemm(torch::Tensor a, torch::Tensor wp, int64_t N,
                          c10::optional<torch::Tensor> bias,
                          int64_t s_override) {
  // a: [M, K]; wp: W[N,K] pre-packed by ops.pack_gemm_weight; returns [M, N]
  CHECK_BF16_CUDA(a); CHECK_BF16_CUDA(wp);
  const int M = a.size(0), K = a.size(1);
  TORCH_CHECK(wp.numel() == (long)N * K, "packed weight size mismatch");
  TORCH_CHECK(K % 256 == 0 && N % 64 == 0 && M <= 128);
  auto c = torch::empty({M, (long)N}, a.options());
  const int S = s_override > 0 ? (int)s_override
                               : xllm::packed_gemm_pick_s((int)N);
  TORCH_CHECK(N % (64 * S) == 0, "N not divisible for S");
  const int SK = xllm::packed_gemm_splitk(M, (int)N, K, S);
  const long mpad = M <= 16 ? 16 : M <= 32 ? 32 : M <= 64 ? 64 : 128;
  torch::Tensor ws;
  if (SK > 1)
    ws = torch::empty({(long)SK * mpad * N}, a.options().dtype(torch::kFloat));
  else
    ws = torch::empty({1}, a.options().dtype(torch::kFloat));
  const unsigned short* bp = nullptr;
  if (bias.has_value()) bp = u16c(bias.value());
  const long mt = mpad / 16;
  auto ap = torch::empty({(long)(K / 32) * mt * 512}, a.options());
  xllm::launch_packed_gemm(u16(c), u16c(a), u16(ap), u16c(wp), bp,
                           ws.data_ptr<float>(), M, (int)N, K, S,
                           cur_stream());
  return c;
}

torch::Tensor packed_gemm_probe(torch::Tensor a, torch::Tensor wp,
                                int64_t N) {
  // W-stream timing probe: output is garbage (A loads compiled out)
  CHECK_BF16_CUDA(a); CHECK_BF16_CUDA(wp);
  const int M = a.size(0), K = a.size(1);
  auto c = torch::empty({M, (long)N}, a.options());
  const int SK = xllm::packed_gemm_splitk(M, (int)N, K, 1);
  torch::Tensor ws;
  if (SK > 1)
    ws = torch::empty({(long)SK * 64 * N}, a.options().dtype(torch::kFloat));
  else
    ws = torch::empty({1}, a.options().dtype(torch::kFloat));
  xllm::launch_packed_gemm_probe(u16(c), u16c(a), u16c(wp),
                                 ws.data_ptr<float>(), M, (int)N, K,
                                 cur_stream());
  return c;
}

torch::Tensor mfma_probe_16x16x32(torch::Tensor a, torch::Tensor b) {
  CHECK_BF16_CUDA(a); CHECK_BF16_CUDA(b);
  TORCH_CHECK(a.size(0) == 16 && a.size(1) == 32);
  TORCH_CHECK(b.size(0) == 32 && b.size(1) == 16);
  auto c = torch::empty({16, 16}, a.options().dtype(torch::kFloat));
  xllm::launch_mfma_probe(c.data_ptr<float>(), u16c(a), u16c(b), cur_stream());
  return c;
}

// ------ cross-GPU KV-block migration over xGMI ----------------------------------
// Direct peer copy of whole blocks: both caches must have identical geometry.
// Contiguous runs of (src,dst) pairs collapse into single hipMemcpyPeerAsync
// calls on the CURRENT stream (callers use a dedicated side stream to overlap
// with decode; see engine/kv_migration.py).
void migrate_blocks_peer(torch::Tensor dst_cache, long dst_device,
                         torch::Tensor src_cache, long src_device,
                         std::vector<long> src_blocks,
                         std::vector<long> dst_blocks) {
  TORCH_CHECK(src_blocks.size() == dst_blocks.size());
  const long numel =
      (long)src_cache.size(1) * src_cache.size(2) * src_cache.size(3);
  const long bytes = numel * 2;  // bf16
  char* dst = reinterpret_cast<char*>(dst_cache.data_ptr());
  const char* src = reinterpret_cast<const char*>(src_cache.data_ptr());
  hipStream_t stream = cur_stream();
  size_t i = 0;
  while (i < src_blocks.size()) {
    // collapse contiguous runs
    size_t j = i + 1;
    while (j < src_blocks.size() && src_blocks[j] == src_blocks[j - 1] + 1 &&
           dst_blocks[j] == dst_blocks[j - 1] + 1)
      j++;
    const long n = (long)(j - i);
    hipError_t err = hipMemcpyPeerAsync(
        dst + dst_blocks[i] * bytes, (int)dst_device,
        src + src_blocks[i] * bytes, (int)src_device, (size_t)(n * bytes),
        stream);
    TORCH_CHECK(err == hipSuccess, "hipMemcpyPeerAsync failed: ",
                hipGetErrorString(err));
    i = j;
  }
}

// ---- HIP IPC: cross-process cache sharing for same-node KV migration ----
// A decode worker opens the prefill worker's cache once at LinkInstance
// time, then pulls blocks with device-to-device copies over xGMI.
std::vector<char> ipc_get_handle(torch::Tensor t) {
  TORCH_CHECK(t.is_cuda());
  hipIpcMemHandle_t h;
  hipError_t err = hipIpcGetMemHandle(&h, t.data_ptr());
  TORCH_CHECK(err == hipSuccess, "hipIpcGetMemHandle: ",
              hipGetErrorString(err));
  return std::vector<char>(reinterpret_cast<char*>(&h),
                           reinterpret_cast<char*>(&h) + sizeof(h));
}

int64_t ipc_open_handle(std::vector<char> handle_bytes, long device) {
  TORCH_CHECK(handle_bytes.size() == sizeof(hipIpcMemHandle_t));
  hipIpcMemHandle_t h;
  memcpy(&h, handle_bytes.data(), sizeof(h));
  hipSetDevice((int)device);
  void* ptr = nullptr;
  hipError_t err =
      hipIpcOpenMemHandle(&ptr, h, hipIpcMemLazyEnablePeerAccess);
  TORCH_CHECK(err == hipSuccess, "hipIpcOpenMemHandle: ",
              hipGetErrorString(err));
  return reinterpret_cast<int64_t>(ptr);
}

void ipc_close_handle(int64_t ptr) {
  hipIpcCloseMemHandle(reinterpret_cast<void*>(ptr));
}

// Pull blocks from a peer cache (opened via IPC) into the local cache.
// Both caches share geometry; contiguous runs collapse into single copies.
void migrate_blocks_from_ptr(torch::Tensor dst_cache, int64_t src_ptr,
                             long src_device, long dst_device,
                             std::vector<long> src_blocks,
                             std::vector<long> dst_blocks) {
  TORCH_CHECK(src_blocks.size() == dst_blocks.size());
  const long numel =
      (long)dst_cache.size(1) * dst_cache.size(2) * dst_cache.size(3);
  const long bytes = numel * dst_cache.element_size();
  char* dst = reinterpret_cast<char*>(dst_cache.data_ptr());
  const char* src = reinterpret_cast<const char*>(src_ptr);
  hipStream_t stream = cur_stream();
  size_t i = 0;
  while (i < src_blocks.size()) {
    size_t j = i + 1;
    while (j < src_blocks.size() && src_blocks[j] == src_blocks[j - 1] + 1 &&
           dst_blocks[j] == dst_blocks[j - 1] + 1)
      j++;
    const long n = (long)(j - i);
    hipError_t err;
    if (src_device == dst_device) {
      err = hipMemcpyAsync(dst + dst_blocks[i] * bytes,
                           src + src_blocks[i] * bytes, (size_t)(n * bytes),
                           hipMemcpyDeviceToDevice, stream);
    } else {
      err = hipMemcpyPeerAsync(dst + dst_blocks[i] * bytes, (int)dst_device,
                               src + src_blocks[i] * bytes, (int)src_device,
                               (size_t)(n * bytes), stream);
    }
    TORCH_CHECK(err == hipSuccess, "block migration copy failed: ",
                hipGetErrorString(err));
    i = j;
  }
}

// ---- KV block swap GPU<->CPU (the "dram" cache tier) ---------------------
// Contiguous (src,dst) runs collapse into single async copies on the
// current stream; callers use a side stream + synchronize.
void swap_blocks(torch::Tensor dst_cache, torch::Tensor src_cache,
                 std::vector<long> src_blocks, std::vector<long> dst_blocks) {
  TORCH_CHECK(src_blocks.size() == dst_blocks.size());
  TORCH_CHECK(src_cache.element_size() == dst_cache.element_size());
  const long numel =
      (long)src_cache.size(1) * src_cache.size(2) * src_cache.size(3);
  const long bytes = numel * src_cache.element_size();
  char* dst = reinterpret_cast<char*>(dst_cache.data_ptr());
  const char* src = reinterpret_cast<const char*>(src_cache.data_ptr());
  hipMemcpyKind kind =
      src_cache.is_cuda()
          ? (dst_cache.is_cuda() ? hipMemcpyDeviceToDevice
                                 : hipMemcpyDeviceToHost)
          : hipMemcpyHostToDevice;
  hipStream_t stream = cur_stream();
  size_t i = 0;
  while (i < src_blocks.size()) {
    size_t j = i + 1;
    while (j < src_blocks.size() && src_blocks[j] == src_blocks[j - 1] + 1 &&
           dst_blocks[j] == dst_blocks[j - 1] + 1)
      j++;
    const long n = (long)(j - i);
    hipError_t err = hipMemcpyAsync(dst + dst_blocks[i] * bytes,
                                    src + src_blocks[i] * bytes,
                                    (size_t)(n * bytes), kind, stream);
    TORCH_CHECK(err == hipSuccess, "swap_blocks copy failed: ",
                hipGetErrorString(err));
    i = j;
  }
}

void enable_peer_access(long device, long peer) {
  int can = 0;
  hipError_t err = hipDeviceCanAccessPeer(&can, (int)device, (int)peer);
  TORCH_CHECK(err == hipSuccess && can, "no P2P path between GPUs ", device,
              " and ", peer);
  hipSetDevice((int)device);
  err = hipDeviceEnablePeerAccess((int)peer, 0);
  TORCH_CHECK(err == hipSuccess || err == hipErrorPeerAccessAlreadyEnabled,
              "hipDeviceEnablePeerAccess failed: ", hipGetErrorString(err));
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm);
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm);
  m.def("rope", &rope);
  m.def("fused_rope_cache", &fused_rope_cache);
  m.def("silu_and_mul", &silu_and_mul);
  m.def("gelu_and_mul", &gelu_and_mul);
  m.def("reshape_and_cache", &reshape_and_cache);
  m.def("copy_blocks", &copy_blocks);
  m.def("gather_blocks", &gather_blocks);
  m.def("paged_attn_decode", &paged_attn_decode);
  m.def("paged_attn_prefill", &paged_attn_prefill);
  m.def("greedy_sample", &greedy_sample);
  m.def("mfma_probe_16x16x32", &mfma_probe_16x16x32);
  m.def("mfma_gemm", &mfma_gemm);
  m.def("skinny_gemm", &skinny_gemm);
  m.def("packed_gemm", &packed_gemm);
  m.def("packed_gemm_probe", &packed_gemm_probe);
  m.def("migrate_blocks_peer", &migrate_blocks_peer);
  m.def("ipc_get_handle", &ipc_get_handle);
  m.def("ipc_open_handle", &ipc_open_handle);
  m.def("ipc_close_handle", &ipc_close_handle);
  m.def("migrate_blocks_from_ptr", &migrate_blocks_from_ptr);
  m.def("swap_blocks", &swap_blocks);
  m.def("enable_peer_access", &enable_peer_access);
}
