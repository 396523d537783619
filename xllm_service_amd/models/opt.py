"""OPT decoder (OPT-125m on the CPU control-plane
path). Standard LayerNorm + learned positions + GELU; shares the paged-KV
attention path with the rest of the engine."""
from __future__ import annotations

import math
from typing import List, Tuple

import torch
import torch.nn as nn

from xllm_service_amd import ops
from xllm_service_amd.engine.metadata import AttnMetadata
from xllm_service_amd.models.config import ModelConfig


class OPTAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype):
        super().__init__()
        self.n_heads = cfg.num_heads
        self.head_dim = cfg.head_dim
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        h = cfg.hidden_size
        self.qkv_proj = nn.Linear(h, 3 * h, bias=True, dtype=dtype)
        self.out_proj = nn.Linear(h, h, bias=True, dtype=dtype)

    def forward(self, x, kv_cache, meta: AttnMetadata):
        T = x.shape[0]
        q, k, v = self.qkv_proj(x).chunk(3, dim=-1)
        qh = q.contiguous().view(T, self.n_heads, self.head_dim)
        kh = k.contiguous().view(T, self.n_heads, self.head_dim)
        vh = v.contiguous().view(T, self.n_heads, self.head_dim)
        k_cache, v_cache = kv_cache
        ops.reshape_and_cache(kh, vh, k_cache, v_cache, meta.slot_mapping)
        np_, nd = meta.num_prefill_tokens, meta.num_decode_tokens
        out = torch.empty_like(qh)
        if np_:
            out[:np_] = ops.paged_attn_prefill(
                qh[:np_], k_cache, v_cache, meta.prefill_block_tables,
                meta.cu_q, meta.prefill_seq_lens, self.scale)
        if nd:
            out[np_:] = ops.paged_attn_decode(
                qh[np_:], k_cache, v_cache, meta.decode_block_tables,
                meta.decode_seq_lens, self.scale)
        return self.out_proj(out.view(T, -1))


class OPTLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype):
        super().__init__()
        h = cfg.hidden_size
        self.attn = OPTAttention(cfg, dtype)
        self.ln1 = nn.LayerNorm(h, dtype=dtype)
        self.ln2 = nn.LayerNorm(h, dtype=dtype)
        self.fc1 = nn.Linear(h, cfg.intermediate_size, bias=True, dtype=dtype)
        self.fc2 = nn.Linear(cfg.intermediate_size, h, bias=True, dtype=dtype)

    def forward(self, x, kv_cache, meta):
        x = x + self.attn(self.ln1(x), kv_cache, meta)
        x = x + self.fc2(torch.nn.functional.gelu(self.fc1(self.ln2(x))))
        return x


class OPTForCausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype=torch.float32):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size, dtype=dtype)
        # OPT's learned positional table has a +2 offset (HF convention)
        self.embed_pos = nn.Embedding(cfg.max_position + 2, cfg.hidden_size,
                                      dtype=dtype)
        self.layers = nn.ModuleList(
            [OPTLayer(cfg, dtype) for _ in range(cfg.num_layers)])
        self.final_ln = nn.LayerNorm(cfg.hidden_size, dtype=dtype)

    @property
    def local_kv_heads(self) -> int:
        return self.cfg.num_kv_heads

    def random_init(self, seed: int = 0):
        gen = torch.Generator().manual_seed(seed)
        for name, p in self.named_parameters():
            with torch.no_grad():
                if "ln" in name and name.endswith("weight"):
                    p.fill_(1.0)
                elif "ln" in name or "bias" in name:
                    p.zero_()
                else:
                    p.copy_((torch.randn(p.shape, generator=gen) * 0.02
                             ).to(p.dtype))

    def forward(self, input_ids, positions, kv_caches: List[Tuple],
                meta: AttnMetadata, inputs_embeds=None):
        x = inputs_embeds if inputs_embeds is not None else self.embed(input_ids)
        x = x + self.embed_pos(positions + 2)
        for i, layer in enumerate(self.layers):
            x = layer(x, kv_caches[i], meta)
        return self.final_ln(x)

    def compute_logits(self, hidden):
        return hidden @ self.embed.weight.t()
