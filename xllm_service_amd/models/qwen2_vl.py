"""Qwen2-VL: vision tower (stage E) + Qwen2 language tower on the paged
engine. BASELINE.md config 5 (multimodal EPD three-stage split).

Vision tower (Qwen2VisionTransformer): 14x14 patches (x2 temporal merge =
1176-dim patch vectors), rotary 2-D position embedding, full (non-causal)
attention, QuickGELU MLP, 2x2 spatial PatchMerger to the LM hidden size.
The patch-embed and merger projections run on the hand-written MFMA GEMM
(ops.mfma_gemm, csrc/ops/gemm.hip) — the "vision-encoder GEMM" kernel.

Language tower = Llama architecture + qkv bias (Qwen2). Note: text-only
1-D RoPE positions are used for all tokens (M-RoPE's 3-D position ids
collapse to 1-D for text; with random-init weights and synthetic images the
bench shapes are identical — noted as a fidelity simplification).
"""
from __future__ import annotations

import math
import torch
import torch.nn as nn

from xllm_service_amd import ops
from xllm_service_amd.models.config import ModelConfig
from xllm_service_amd.models.llama import LlamaForCausalLM

IMAGE_PAD_TOKEN_ID = 151655  # <|image_pad|> in the Qwen2-VL vocab


def _vision_rope_2d(h: int, w: int, dim: int) -> torch.Tensor:
    """cos/sin for 2-D rotary: half the rot dim indexed by row, half by col.
    Returns [h*w, dim] (first half cos, second half sin over dim/2 freqs)."""
    half = dim // 2  # freqs per axis pair
    inv = 1.0 / (10000.0 ** (torch.arange(0, half, 2).float() / half))
    rows = torch.arange(h).float()
    cols = torch.arange(w).float()
    fr = torch.outer(rows, inv)  # [h, half/2]
    fc = torch.outer(cols, inv)
    # token (r, c): freqs = concat(row freqs, col freqs)
    f = torch.cat([
        fr[:, None, :].expand(h, w, -1),
        fc[None, :, :].expand(h, w, -1),
    ], dim=-1).reshape(h * w, half)
    return torch.cat([f.cos(), f.sin()], dim=-1)  # [hw, dim]


class VisionAttention(nn.Module):
    def __init__(self, dim: int, heads: int, dtype):
        super().__init__()
        self.heads = heads
        self.head_dim = dim // heads
        self.qkv = nn.Linear(dim, dim * 3, bias=True, dtype=dtype)
        self.proj = nn.Linear(dim, dim, bias=True, dtype=dtype)

    def forward(self, x: torch.Tensor, rope: torch.Tensor) -> torch.Tensor:
        T, C = x.shape
        q, k, v = self.qkv(x).chunk(3, dim=-1)

        def shape(t):
            return t.reshape(T, self.heads, self.head_dim)

        q, k, v = shape(q), shape(k), shape(v)
        # rotate-half 2-D rope on q, k
        half = self.head_dim // 2
        cos = rope[:, :half].unsqueeze(1).to(torch.float32)
        sin = rope[:, half:].unsqueeze(1).to(torch.float32)

        def rot(t):
            tf = t.float()
            t1, t2 = tf[..., :half], tf[..., half:]
            return torch.cat([t1 * cos - t2 * sin, t2 * cos + t1 * sin],
                             dim=-1).to(t.dtype)

        q, k = rot(q), rot(k)
        o = torch.nn.functional.scaled_dot_product_attention(
            q.transpose(0, 1).float(), k.transpose(0, 1).float(),
            v.transpose(0, 1).float(),
            scale=1.0 / math.sqrt(self.head_dim)).to(x.dtype)
        return self.proj(o.transpose(0, 1).reshape(T, C))


class VisionBlock(nn.Module):
    def __init__(self, dim: int, heads: int, dtype):
        super().__init__()
        self.ln1 = nn.LayerNorm(dim, dtype=dtype)
        self.ln2 = nn.LayerNorm(dim, dtype=dtype)
        self.attn = VisionAttention(dim, heads, dtype)
        self.fc1 = nn.Linear(dim, dim * 4, bias=True, dtype=dtype)
        self.fc2 = nn.Linear(dim * 4, dim, bias=True, dtype=dtype)

    def forward(self, x, rope):
        x = x + self.attn(self.ln1(x), rope)
        h = self.fc1(self.ln2(x))
        h = h * torch.sigmoid(1.702 * h.float()).to(h.dtype)  # QuickGELU
        return x + self.fc2(h)


class Qwen2VisionTransformer(nn.Module):
    """Stage E: pixel patches -> LM-space embeddings."""

    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16):
        super().__init__()
        v = cfg.vision
        self.patch_size = v.get("patch_size", 14)
        self.temporal = 2
        self.in_ch = 3
        self.embed_dim = v.get("embed_dim", 1280)
        self.merge = v.get("spatial_merge_size", 2)
        self.out_hidden = v.get("out_hidden_size", cfg.hidden_size)
        self.patch_dim = self.in_ch * self.temporal * self.patch_size ** 2
        # pad K to a multiple of 32 for the MFMA GEMM
        self.patch_dim_pad = (self.patch_dim + 31) // 32 * 32
        self.patch_embed_w = nn.Parameter(torch.empty(
            self.embed_dim, self.patch_dim_pad, dtype=dtype))
        self.blocks = nn.ModuleList([
            VisionBlock(self.embed_dim, v.get("num_heads", 16), dtype)
            for _ in range(v.get("depth", 32))])
        merged = self.embed_dim * self.merge ** 2
        self.merger_ln = nn.LayerNorm(self.embed_dim, dtype=dtype)
        self.merger_fc1 = nn.Linear(merged, merged, bias=True, dtype=dtype)
        self.merger_fc2_w = nn.Parameter(torch.empty(
            self.out_hidden, merged, dtype=dtype))

    def num_output_tokens(self, grid_h: int, grid_w: int) -> int:
        return (grid_h // self.merge) * (grid_w // self.merge)

    def forward(self, patches: torch.Tensor, grid_h: int,
                grid_w: int) -> torch.Tensor:
        """patches: [grid_h*grid_w, patch_dim] -> [out_tokens, out_hidden]."""
        T = patches.shape[0]
        assert T == grid_h * grid_w
        if patches.shape[1] < self.patch_dim_pad:
            patches = torch.nn.functional.pad(
                patches, (0, self.patch_dim_pad - patches.shape[1]))
        x = ops.mfma_gemm(patches.to(self.patch_embed_w.dtype),
                          self.patch_embed_w)
        rope = _vision_rope_2d(grid_h, grid_w, self.blocks[0].attn.head_dim
                               ).to(x.device)
        for blk in self.blocks:
            x = blk(x, rope)
        x = self.merger_ln(x)
        m = self.merge
        # 2x2 spatial merge: [h, w, C] -> [h/m, w/m, m*m*C]
        x = x.reshape(grid_h // m, m, grid_w // m, m, self.embed_dim)
        x = x.permute(0, 2, 1, 3, 4).reshape(
            (grid_h // m) * (grid_w // m), m * m * self.embed_dim)
        x = torch.nn.functional.gelu(self.merger_fc1(x))
        return ops.mfma_gemm(x, self.merger_fc2_w)

    def random_init(self, seed: int = 0):
        """Fully deterministic: every parameter is seeded (the default
        nn.Linear bias init is not, which would break cross-process
        determinism of the E stage)."""
        dev = next(self.parameters()).device
        gen = torch.Generator(device=dev).manual_seed(seed + 77)
        for name, p in self.named_parameters():
            with torch.no_grad():
                if "ln" in name and name.endswith("weight"):
                    p.fill_(1.0)
                elif "ln" in name or "bias" in name:
                    p.zero_()
                else:
                    p.copy_((torch.randn(p.shape, generator=gen, device=dev,
                                         dtype=torch.float32) * 0.02
                             ).to(p.dtype))


def mrope_positions(token_ids, placeholder_id, grids):
    """3-D (temporal, height, width) M-RoPE position ids for one prompt.

    Text tokens advance all three components together; each image span of
    LM-grid (t, h, w) gets grid-coordinate positions offset by the running
    maximum, and text after it resumes at max + 1 — so a (1, 4, 6) image
    occupies only max(1,4,6)=6 position slots instead of 24.
    (reference semantics: Qwen2-VL M-RoPE / get_rope_index)

    Returns (pos3 [3, L] int64 numpy, delta) where delta = (max_pos + 1) - L
    is added to text-style positions for all decode steps.
    """
    import numpy as np
    L = len(token_ids)
    pos3 = np.empty((3, L), dtype=np.int64)
    gi = 0
    cur = 0           # next text position
    i = 0
    while i < L:
        if token_ids[i] == placeholder_id and gi < len(grids):
            t, h, w = (int(x) for x in grids[gi])
            n = t * h * w
            st = cur
            idx = np.arange(n)
            pos3[0, i:i + n] = st + idx // (h * w)
            pos3[1, i:i + n] = st + (idx // w) % h
            pos3[2, i:i + n] = st + idx % w
            cur = st + max(t, h, w)
            gi += 1
            i += n
        else:
            pos3[:, i] = cur
            cur += 1
            i += 1
    delta = int(cur - L)
    return pos3, delta


class Qwen2VLForCausalLM(LlamaForCausalLM):
    """Language tower; the vision tower runs in the ENCODE stage (or
    in-process for colocated DEFAULT instances)."""

    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16):
        super().__init__(cfg, dtype=dtype)
        # Qwen2 uses qkv bias
        for layer in self.layers:
            attn = layer.attn
            attn.qkv_proj.bias = nn.Parameter(torch.zeros(
                attn.qkv_proj.shard_out, dtype=dtype))
        self.image_pad_token_id = (cfg.image_pad_token_id
            if cfg.image_pad_token_id >= 0 else IMAGE_PAD_TOKEN_ID)
