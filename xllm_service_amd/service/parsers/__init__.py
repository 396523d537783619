"""Chat-output parsing: reasoning extraction + tool-call extraction, with
streaming (incremental) variants.

Parity with the reference's parser bridge (xllm_chat_parse_bridge.cpp,
SURVEY.md 2.12): model type inferred from the model id; "auto" silently
disables parsing for unknown model families.
"""
from __future__ import annotations

from .reasoning import ReasoningParser, StreamingReasoningParser
from .tool_call import (StreamingToolCallParser, ToolCall, ToolCallParser,
                        parse_tool_calls)

# model family -> (reasoning tags, tool-call tags)
_FAMILIES = {
    "qwen3": {"reasoning": ("<think>", "</think>"),
              "tool": ("<tool_call>", "</tool_call>")},
    "qwen2": {"reasoning": None,
              "tool": ("<tool_call>", "</tool_call>")},
    "deepseek_v3": {"reasoning": ("<think>", "</think>"),
                    "tool": ("<｜tool▁call▁begin｜>", "<｜tool▁call▁end｜>")},
    "kimi_k2": {"reasoning": ("<think>", "</think>"),
                "tool": ("<|tool_call_begin|>", "<|tool_call_end|>")},
    "glm4_moe": {"reasoning": ("<think>", "</think>"),
                 "tool": ("<tool_call>", "</tool_call>")},
    "step3": {"reasoning": ("<think>", "</think>"),
              "tool": ("<tool_call>", "</tool_call>")},
    "llama": {"reasoning": None, "tool": ("<|python_tag|>", "<|eom_id|>")},
}


def infer_model_family(model_id: str) -> str | None:
    m = (model_id or "").lower()
    for probe, fam in (("qwen3", "qwen3"), ("qwen-3", "qwen3"),
                       ("qwen2", "qwen2"), ("qwen-2", "qwen2"),
                       ("deepseek-v3", "deepseek_v3"),
                       ("deepseek_v3", "deepseek_v3"),
                       ("deepseek-v32", "deepseek_v3"),
                       ("kimi-k2", "kimi_k2"), ("kimi_k2", "kimi_k2"),
                       ("glm-4", "glm4_moe"), ("glm4", "glm4_moe"),
                       ("step3", "step3"), ("step-3", "step3"),
                       ("llama", "llama")):
        if probe in m:
            return fam
    return None


def make_parsers(model_id: str, mode: str = "auto"):
    """Returns (ReasoningParser|None, ToolCallParser|None)."""
    fam = infer_model_family(model_id) if mode == "auto" else mode
    spec = _FAMILIES.get(fam or "")
    if spec is None:
        return None, None  # unknown family: parsing disabled (reference behaviour)
    rp = ReasoningParser(*spec["reasoning"]) if spec["reasoning"] else None
    tp = ToolCallParser(*spec["tool"]) if spec["tool"] else None
    return rp, tp


def make_stream_parsers(model_id: str, mode: str = "auto"):
    fam = infer_model_family(model_id) if mode == "auto" else mode
    spec = _FAMILIES.get(fam or "")
    if spec is None:
        return None, None
    rp = (StreamingReasoningParser(*spec["reasoning"])
          if spec["reasoning"] else None)
    tp = (StreamingToolCallParser(*spec["tool"]) if spec["tool"] else None)
    return rp, tp
