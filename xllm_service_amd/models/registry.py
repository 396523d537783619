"""Model factory."""
from __future__ import annotations

import torch

from .config import ModelConfig


def create_model(cfg: ModelConfig, dtype=torch.bfloat16):
    if cfg.architecture == "llama":
        from .llama import LlamaForCausalLM
        return LlamaForCausalLM(cfg, dtype=dtype)
    if cfg.architecture == "opt":
        from .opt import OPTForCausalLM
        return OPTForCausalLM(cfg, dtype=dtype)
    if cfg.architecture in ("qwen2", "qwen2_vl"):
        # plain Qwen2 text models share the VL language tower exactly
        # (llama stack + qkv bias); the vision tower is a separate class
        from .qwen2_vl import Qwen2VLForCausalLM
        return Qwen2VLForCausalLM(cfg, dtype=dtype)
    raise ValueError(f"unknown architecture {cfg.architecture}")
