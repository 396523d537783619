"""ResponseHandler: OpenAI wire serialization for the four delivery paths
(chat/completion x stream/non-stream), with reasoning + tool-call parsing.
(reference: scheduler/response_handler.{h,cpp})
"""
from __future__ import annotations

import asyncio
import json
from typing import AsyncIterator, Optional

from xllm_service_amd.tokenizer import IncrementalDecoder, Tokenizer

from .parsers import make_parsers, make_stream_parsers
from .request import GenerationDelta, ServiceRequest

STREAM_TIMEOUT_S = 600.0


def _sse(obj) -> str:
    return f"data: {json.dumps(obj, ensure_ascii=False)}\n\n"


async def _next_delta(queue, hold: list) -> GenerationDelta:
    """Await one delta, then greedily coalesce any backlog of plain token
    deltas into it (one SSE write then carries the merged text). Deltas
    carrying errors, text overrides or logprobs are never merged across —
    they are parked in `hold` (a per-stream 1-slot list) and delivered on
    the next call."""
    if hold:
        return hold.pop()
    gen: GenerationDelta = await asyncio.wait_for(queue.get(),
                                                  STREAM_TIMEOUT_S)
    if gen.error or gen.finished or gen.text is not None or gen.logprobs:
        return gen
    while True:
        try:
            nxt: GenerationDelta = queue.get_nowait()
        except asyncio.QueueEmpty:
            break
        if nxt.error or nxt.text is not None or nxt.logprobs:
            hold.append(nxt)    # deliver right after the merged delta
            break
        gen.token_ids = list(gen.token_ids) + list(nxt.token_ids)
        if nxt.finished:
            gen.finished = True
            gen.finish_reason = nxt.finish_reason
            gen.usage_prompt_tokens = nxt.usage_prompt_tokens
            gen.usage_completion_tokens = nxt.usage_completion_tokens
            break
    return gen


class ResponseHandler:
    def __init__(self, tokenizer: Tokenizer, parser_mode: str = "auto"):
        self.tokenizer = tokenizer
        self.parser_mode = parser_mode

    # ------------------------------------------------------------ chat stream
    async def stream_chat(self, req: ServiceRequest,
                          on_cancel) -> AsyncIterator[str]:
        rid = req.service_request_id
        created = int(req.created)
        model = req.model
        include_usage = bool(req.params.get("include_usage"))
        dec = IncrementalDecoder(self.tokenizer)
        rp, tp = make_stream_parsers(model, self.parser_mode)
        sent_role = False
        tool_idx = 0
        emitted_tool = False
        usage = None

        def chunk(delta: dict, finish: Optional[str] = None, lps=None):
            choice = {"index": 0, "delta": delta, "finish_reason": finish}
            if lps:
                choice["logprobs"] = {"content": lps}
            return _sse({
                "id": rid, "object": "chat.completion.chunk",
                "created": created, "model": model,
                "choices": [choice]})

        def chat_lps(gen):
            if not gen.logprobs:
                return None
            return [{"token": self.tokenizer.decode([t]),
                     "logprob": lp["token_logprob"] if lp else None,
                     "top_logprobs": [
                         {"token": self.tokenizer.decode([tt]), "logprob": vv}
                         for tt, vv in (lp.get("top") or {}).items()]}
                    for t, lp in zip(gen.token_ids, gen.logprobs)]

        held: list = []
        try:
            while True:
                gen = await _next_delta(req.output_queue, held)
                if gen.error:
                    yield _sse({"error": {"message": gen.error,
                                          "type": "server_error"}})
                    break
                text = (gen.text if gen.text is not None
                        else dec.push(gen.token_ids) if gen.token_ids else "")
                if not sent_role and (text or gen.finished):
                    yield chunk({"role": "assistant", "content": ""})
                    sent_role = True
                reasoning_delta, content = (None, text)
                if rp is not None and text:
                    reasoning_delta, content = rp.feed(text)
                    if reasoning_delta:
                        yield chunk({"reasoning_content": reasoning_delta})
                if tp is not None and content:
                    content, calls = tp.feed(content)
                    for tc in calls:
                        yield chunk({"tool_calls": [{
                            "index": tool_idx, "id": tc.id, "type": "function",
                            "function": {"name": tc.name,
                                         "arguments": tc.arguments}}]})
                        tool_idx += 1
                        emitted_tool = True
                if content:
                    yield chunk({"content": content}, lps=chat_lps(gen))
                elif gen.logprobs:
                    # tokens arrived but the incremental decoder is holding
                    # partial UTF-8: ship the logprobs with an empty delta
                    yield chunk({}, lps=chat_lps(gen))
                if gen.finished:
                    if tp is not None:
                        for tc in tp.flush():
                            yield chunk({"tool_calls": [{
                                "index": tool_idx, "id": tc.id,
                                "type": "function",
                                "function": {"name": tc.name,
                                             "arguments": tc.arguments}}]})
                            tool_idx += 1
                            emitted_tool = True
                    finish = gen.finish_reason or "stop"
                    if emitted_tool and finish == "stop":
                        finish = "tool_calls"
                    yield chunk({}, finish=finish)
                    usage = {
                        "prompt_tokens": gen.usage_prompt_tokens,
                        "completion_tokens": gen.usage_completion_tokens,
                        "total_tokens": gen.usage_prompt_tokens +
                        gen.usage_completion_tokens}
                    break
            if include_usage and usage:
                yield _sse({"id": rid, "object": "chat.completion.chunk",
                            "created": created, "model": model,
                            "choices": [], "usage": usage})
            yield "data: [DONE]\n\n"
        except (asyncio.CancelledError, GeneratorExit):
            await on_cancel(req)
            raise
        except asyncio.TimeoutError:
            yield _sse({"error": {"message": "stream timeout",
                                  "type": "server_error"}})
            await on_cancel(req)

    # -------------------------------------------------------- chat non-stream
    async def collect_chat(self, req: ServiceRequest) -> dict:
        token_ids, text, usage, finish, err = await self._collect(req)
        if err:
            return {"error": {"message": err, "type": "server_error"}}
        rp, tp = make_parsers(req.model, self.parser_mode)
        reasoning = None
        tool_calls = []
        if rp is not None:
            reasoning, text = rp.extract(text)
        if tp is not None:
            text, tool_calls = tp.extract(text)
        if tool_calls and finish == "stop":
            finish = "tool_calls"
        msg = {"role": "assistant", "content": text}
        if reasoning:
            msg["reasoning_content"] = reasoning
        if tool_calls:
            msg["tool_calls"] = [{
                "id": tc.id, "type": "function",
                "function": {"name": tc.name, "arguments": tc.arguments}}
                for tc in tool_calls]
        choice = {"index": 0, "message": msg, "finish_reason": finish}
        raw_lps = getattr(req, "_collected_logprobs", [])
        if raw_lps:
            choice["logprobs"] = {"content": [
                {"token": self.tokenizer.decode([t]),
                 "logprob": lp["token_logprob"] if lp else None,
                 "top_logprobs": [
                     {"token": self.tokenizer.decode([tt]), "logprob": vv}
                     for tt, vv in (lp.get("top") or {}).items()]}
                for t, lp in zip(token_ids, raw_lps)]}
        return {
            "id": req.service_request_id, "object": "chat.completion",
            "created": int(req.created), "model": req.model,
            "choices": [choice],
            "usage": usage}

    # ------------------------------------------------------ completion paths
    async def stream_completion(self, req: ServiceRequest, on_cancel,
                                echo_text: Optional[str] = None
                                ) -> AsyncIterator[str]:
        rid = req.service_request_id
        created = int(req.created)
        dec = IncrementalDecoder(self.tokenizer)
        if echo_text:   # OpenAI completions echo=true: prompt leads the text
            yield _sse({"id": rid, "object": "text_completion",
                        "created": created, "model": req.model,
                        "choices": [{"index": 0, "text": echo_text,
                                     "finish_reason": None}]})
        held: list = []
        try:
            while True:
                gen = await _next_delta(req.output_queue, held)
                if gen.error:
                    yield _sse({"error": {"message": gen.error,
                                          "type": "server_error"}})
                    break
                text = (gen.text if gen.text is not None
                        else dec.push(gen.token_ids) if gen.token_ids else "")
                if text or gen.logprobs:
                    choice = {"index": 0, "text": text, "finish_reason": None}
                    lp = self._completion_logprobs(gen.token_ids,
                                                   gen.logprobs or [])
                    if lp is not None:
                        choice["logprobs"] = lp
                    yield _sse({"id": rid, "object": "text_completion",
                                "created": created, "model": req.model,
                                "choices": [choice]})
                if gen.finished:
                    yield _sse({"id": rid, "object": "text_completion",
                                "created": created, "model": req.model,
                                "choices": [{"index": 0, "text": "",
                                             "finish_reason":
                                             gen.finish_reason or "stop"}],
                                "usage": {
                                    "prompt_tokens": gen.usage_prompt_tokens,
                                    "completion_tokens":
                                    gen.usage_completion_tokens,
                                    "total_tokens": gen.usage_prompt_tokens +
                                    gen.usage_completion_tokens}})
                    break
            yield "data: [DONE]\n\n"
        except (asyncio.CancelledError, GeneratorExit):
            await on_cancel(req)
            raise
        except asyncio.TimeoutError:
            yield _sse({"error": {"message": "stream timeout",
                                  "type": "server_error"}})
            await on_cancel(req)

    async def collect_completion(self, req: ServiceRequest) -> dict:
        token_ids, text, usage, finish, err = await self._collect(req)
        if err:
            return {"error": {"message": err, "type": "server_error"}}
        choice = {"index": 0, "text": text, "finish_reason": finish}
        lp = self._completion_logprobs(
            token_ids, getattr(req, "_collected_logprobs", []))
        if lp is not None:
            choice["logprobs"] = lp
        return {
            "id": req.service_request_id, "object": "text_completion",
            "created": int(req.created), "model": req.model,
            "choices": [choice],
            "usage": usage}

    # ---------------------------------------------------------------- common
    async def _collect(self, req: ServiceRequest):
        """Drain the request to completion. Returns
        (token_ids, text, usage, finish_reason, error); text honours
        per-delta text overrides (text-level stop trimming)."""
        token_ids = []
        text_parts = []
        dec = IncrementalDecoder(self.tokenizer)
        logprobs = []
        usage = {"prompt_tokens": 0, "completion_tokens": 0, "total_tokens": 0}
        finish = "stop"
        while True:
            gen: GenerationDelta = await asyncio.wait_for(
                req.output_queue.get(), STREAM_TIMEOUT_S)
            if gen.error:
                return token_ids, "".join(text_parts), usage, finish, gen.error
            token_ids.extend(gen.token_ids)
            text_parts.append(gen.text if gen.text is not None
                              else dec.push(gen.token_ids))
            if gen.logprobs:
                logprobs.extend(gen.logprobs)
            if gen.finished:
                finish = gen.finish_reason or "stop"
                usage = {"prompt_tokens": gen.usage_prompt_tokens,
                         "completion_tokens": gen.usage_completion_tokens,
                         "total_tokens": gen.usage_prompt_tokens +
                         gen.usage_completion_tokens}
                req._collected_logprobs = logprobs
                return token_ids, "".join(text_parts), usage, finish, None

    def _completion_logprobs(self, token_ids, logprobs):
        """OpenAI completion-style logprobs block."""
        if not logprobs:
            return None
        return {
            "tokens": [self.tokenizer.decode([t]) for t in token_ids],
            "token_logprobs": [lp["token_logprob"] if lp else None
                               for lp in logprobs],
            "top_logprobs": [
                {self.tokenizer.decode([t]): v
                 for t, v in (lp.get("top") or {}).items()} if lp else None
                for lp in logprobs],
        }
