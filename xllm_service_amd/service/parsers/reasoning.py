"""Reasoning-content extraction (<think>...</think> and variants).
(reference role: xllm/parser/reasoning_parser.h used by the bridge)"""
from __future__ import annotations

from typing import Optional, Tuple


class ReasoningParser:
    def __init__(self, start_tag: str, end_tag: str):
        self.start = start_tag
        self.end = end_tag

    def extract(self, text: str) -> Tuple[Optional[str], str]:
        """-> (reasoning_content | None, content). Handles the common case
        where the model omits the opening tag but emits the closing one."""
        if self.start in text:
            pre, rest = text.split(self.start, 1)
            if self.end in rest:
                reasoning, content = rest.split(self.end, 1)
                return reasoning.strip("\n"), (pre + content).lstrip("\n")
            return rest.strip("\n"), ""   # unterminated: all reasoning
        if self.end in text:
            reasoning, content = text.split(self.end, 1)
            return reasoning.strip("\n"), content.lstrip("\n")
        return None, text


class StreamingReasoningParser:
    """Incremental variant: feed deltas, get (reasoning_delta, content_delta).

    States: maybe-start -> reasoning -> content."""

    def __init__(self, start_tag: str, end_tag: str):
        self.start = start_tag
        self.end = end_tag
        self.buf = ""
        self.state = "init"   # init | reasoning | content

    def feed(self, delta: str) -> Tuple[str, str]:
        self.buf += delta
        r_out, c_out = "", ""
        while True:
            if self.state == "init":
                if self.buf.startswith(self.start):
                    self.buf = self.buf[len(self.start):]
                    self.state = "reasoning"
                    continue
                if len(self.buf) < len(self.start) and self.start.startswith(self.buf):
                    break  # could still be the opening tag
                # no opening tag: models that skip <think> go straight to
                # reasoning if an end tag may come later, else content.
                self.state = "content"
                continue
            if self.state == "reasoning":
                idx = self.buf.find(self.end)
                if idx >= 0:
                    r_out += self.buf[:idx]
                    self.buf = self.buf[idx + len(self.end):].lstrip("\n")
                    self.state = "content"
                    continue
                keep = self._tail_overlap(self.buf, self.end)
                emit = self.buf[:len(self.buf) - keep]
                r_out += emit
                self.buf = self.buf[len(emit):]
                break
            if self.state == "content":
                c_out += self.buf
                self.buf = ""
                break
        return r_out, c_out

    @staticmethod
    def _tail_overlap(text: str, tag: str) -> int:
        for n in range(min(len(tag) - 1, len(text)), 0, -1):
            if tag.startswith(text[-n:]):
                return n
        return 0
