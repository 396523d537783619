"""Tool-call extraction: `<tool_call>{json}</tool_call>`-style blocks (and
family-specific tag variants), non-stream and incremental.
(reference role: xllm/function_call/function_call_parser.h)"""
from __future__ import annotations

import json
import uuid
from dataclasses import dataclass, field
from typing import List, Optional, Tuple


@dataclass
class ToolCall:
    name: str
    arguments: str           # JSON string (OpenAI wire format)
    id: str = field(default_factory=lambda: f"call_{uuid.uuid4().hex[:24]}")


def _parse_block(block: str) -> Optional[ToolCall]:
    block = block.strip()
    try:
        obj = json.loads(block)
    except json.JSONDecodeError:
        # deepseek-style: "name\n```json\n{...}\n```"
        if "```json" in block:
            head, rest = block.split("```json", 1)
            body = rest.split("```", 1)[0]
            try:
                return ToolCall(name=head.strip().strip("\n"),
                                arguments=json.dumps(json.loads(body)))
            except json.JSONDecodeError:
                return None
        return None
    if isinstance(obj, dict):
        name = obj.get("name")
        args = obj.get("arguments", obj.get("parameters", {}))
        if name:
            return ToolCall(name=name, arguments=json.dumps(args))
    return None


class ToolCallParser:
    def __init__(self, start_tag: str, end_tag: str):
        self.start = start_tag
        self.end = end_tag

    def extract(self, text: str) -> Tuple[str, List[ToolCall]]:
        """-> (content without tool blocks, tool calls)."""
        calls: List[ToolCall] = []
        out = []
        rest = text
        while self.start in rest:
            pre, after = rest.split(self.start, 1)
            out.append(pre)
            if self.end in after:
                block, rest = after.split(self.end, 1)
            else:
                block, rest = after, ""
            tc = _parse_block(block)
            if tc:
                calls.append(tc)
        out.append(rest)
        return "".join(out).strip(), calls


class StreamingToolCallParser:
    """Incremental: feed text deltas; emits (content_delta, [completed tool
    calls]); unstreamed buffered args flush on finish (reference:
    response_handler.cpp:292-308)."""

    def __init__(self, start_tag: str, end_tag: str):
        self.start = start_tag
        self.end = end_tag
        self.buf = ""
        self.in_block = False

    def feed(self, delta: str) -> Tuple[str, List[ToolCall]]:
        self.buf += delta
        content = ""
        calls: List[ToolCall] = []
        while True:
            if not self.in_block:
                idx = self.buf.find(self.start)
                if idx >= 0:
                    content += self.buf[:idx]
                    self.buf = self.buf[idx + len(self.start):]
                    self.in_block = True
                    continue
                keep = self._tail_overlap(self.buf, self.start)
                emit = self.buf[:len(self.buf) - keep]
                content += emit
                self.buf = self.buf[len(emit):]
                break
            idx = self.buf.find(self.end)
            if idx >= 0:
                tc = _parse_block(self.buf[:idx])
                if tc:
                    calls.append(tc)
                self.buf = self.buf[idx + len(self.end):]
                self.in_block = False
                continue
            break  # wait for more of the block
        return content, calls

    def flush(self) -> List[ToolCall]:
        """At stream end: parse any unterminated block."""
        if self.in_block and self.buf.strip():
            tc = _parse_block(self.buf)
            self.buf = ""
            self.in_block = False
            return [tc] if tc else []
        return []

    @staticmethod
    def _tail_overlap(text: str, tag: str) -> int:
        for n in range(min(len(tag) - 1, len(text)), 0, -1):
            if tag.startswith(text[-n:]):
                return n
        return 0


def parse_tool_calls(text: str, start: str = "<tool_call>",
                     end: str = "</tool_call>"):
    return ToolCallParser(start, end).extract(text)
