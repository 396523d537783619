"""Model configuration + named presets (random-init; no network access)."""
from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class ModelConfig:
    name: str = "llama"
    architecture: str = "llama"          # llama | opt | qwen2 | qwen2_vl
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    rope_theta: float = 500000.0
    max_position: int = 8192
    rms_eps: float = 1e-5
    tie_word_embeddings: bool = False
    hidden_act: str = "silu"
    # vision tower (qwen2_vl only)
    vision: dict = field(default_factory=dict)
    image_pad_token_id: int = -1
    # M-RoPE frequency split (temporal, height, width) in frequency units,
    # summing to rot_dim // 2. Empty = plain 1-D RoPE.
    # (reference: Qwen2-VL 3-D rotary position embedding)
    mrope_section: tuple = ()

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim


PRESETS = {
    # flagship bench model
    "llama-3-8b": ModelConfig(
        name="llama-3-8b", vocab_size=128256, hidden_size=4096,
        intermediate_size=14336, num_layers=32, num_heads=32, num_kv_heads=8,
        head_dim=128, rope_theta=500000.0, max_position=8192),
    # small CPU-path / test configs
    "llama-tiny": ModelConfig(
        name="llama-tiny", vocab_size=1024, hidden_size=256,
        intermediate_size=512, num_layers=2, num_heads=4, num_kv_heads=2,
        head_dim=64, rope_theta=10000.0, max_position=2048),
    "llama-debug-128": ModelConfig(
        name="llama-debug-128", vocab_size=512, hidden_size=512,
        intermediate_size=1024, num_layers=2, num_heads=4, num_kv_heads=2,
        head_dim=128, rope_theta=10000.0, max_position=2048),
    # OPT-125m (CPU control-plane check)
    "opt-125m": ModelConfig(
        name="opt-125m", architecture="opt", vocab_size=50272,
        hidden_size=768, intermediate_size=3072, num_layers=12, num_heads=12,
        num_kv_heads=12, head_dim=64, max_position=2048, hidden_act="gelu"),
    # Qwen2-7B (text-only: the VL language tower without a vision stage)
    "qwen2-7b": ModelConfig(
        name="qwen2-7b", architecture="qwen2", vocab_size=152064,
        hidden_size=3584, intermediate_size=18944, num_layers=28,
        num_heads=28, num_kv_heads=4, head_dim=128, rope_theta=1000000.0,
        max_position=8192),
    "qwen2-tiny": ModelConfig(
        name="qwen2-tiny", architecture="qwen2", vocab_size=1024,
        hidden_size=256, intermediate_size=512, num_layers=2, num_heads=4,
        num_kv_heads=2, head_dim=64, rope_theta=10000.0, max_position=2048),
    # Qwen2-VL-7B — language tower dims
    "qwen2-vl-7b": ModelConfig(
        name="qwen2-vl-7b", architecture="qwen2_vl", vocab_size=152064,
        hidden_size=3584, intermediate_size=18944, num_layers=28,
        num_heads=28, num_kv_heads=4, head_dim=128, rope_theta=1000000.0,
        max_position=8192, image_pad_token_id=151655,
        mrope_section=(16, 24, 24),
        vision=dict(depth=32, embed_dim=1280, num_heads=16, patch_size=14,
                    spatial_merge_size=2, out_hidden_size=3584)),
    # small multimodal config with head_dim 128 for GPU tests
    "qwen2-vl-debug": ModelConfig(
        name="qwen2-vl-debug", architecture="qwen2_vl", vocab_size=1024,
        hidden_size=512, intermediate_size=1024, num_layers=2, num_heads=4,
        num_kv_heads=2, head_dim=128, rope_theta=10000.0, max_position=2048,
        image_pad_token_id=9, mrope_section=(16, 24, 24),
        vision=dict(depth=2, embed_dim=256, num_heads=4, patch_size=14,
                    spatial_merge_size=2, out_hidden_size=512)),
    # tiny multimodal config with the MFMA head_dim (GPU EPD checks)
    "qwen2-vl-tiny128": ModelConfig(
        name="qwen2-vl-tiny128", architecture="qwen2_vl", vocab_size=1024,
        hidden_size=512, intermediate_size=1024, num_layers=2, num_heads=4,
        num_kv_heads=2, head_dim=128, rope_theta=10000.0, max_position=2048,
        image_pad_token_id=9, mrope_section=(16, 24, 24),
        vision=dict(depth=2, embed_dim=64, num_heads=4, patch_size=14,
                    spatial_merge_size=2, out_hidden_size=512)),
    # tiny multimodal config for CPU EPD tests
    "qwen2-vl-tiny": ModelConfig(
        name="qwen2-vl-tiny", architecture="qwen2_vl", vocab_size=1024,
        hidden_size=256, intermediate_size=512, num_layers=2, num_heads=4,
        num_kv_heads=2, head_dim=64, rope_theta=10000.0, max_position=2048,
        image_pad_token_id=9, mrope_section=(8, 12, 12),
        vision=dict(depth=2, embed_dim=64, num_heads=4, patch_size=14,
                    spatial_merge_size=2, out_hidden_size=256)),
}


def get_config(name: str) -> ModelConfig:
    if name not in PRESETS:
        raise KeyError(f"unknown model preset {name!r}; have {list(PRESETS)}")
    return PRESETS[name]
