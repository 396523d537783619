"""xllm_service_amd — an MI355X-native prefill/decode-disaggregated LLM
serving framework with the capabilities of jd-opensource/xllm-service.

Layers (top to bottom):
  service/      the master: OpenAI HTTP front end, RPC plane, global scheduler,
                instance manager, load-balance policies (parity with the
                reference's xllm_service/ tree, re-designed — SURVEY.md §2)
  registry/     embedded etcd-style metadata store (leases, watches, txns)
  engine/       per-GPU worker: continuous batching, paged KV cache,
                PD-disaggregation, KV migration over xGMI
  models/       model definitions (Llama family, Qwen2-VL, OPT)
  ops/          CDNA4 HIP kernels (gfx950) + CPU reference implementations
  distributed/  tensor parallelism over RCCL/xGMI
  tokenizer/, chat_template/   host-side text processing
"""

__version__ = "0.1.0"
