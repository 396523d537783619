"""OpenAI-compatible HTTP front end (reference: http_service/).

Routes: /hello, /v1/completions, /v1/chat/completions, /v1/models,
/v1/embeddings (501, like the reference), /metrics. Readiness gating: when
no viable instance group exists the API answers 503 (the reference stops
its HTTP listener; answering 503 is the same contract to a load balancer).
"""
from __future__ import annotations

import logging
from typing import Any, Dict, List, Optional, Union

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, Response, StreamingResponse
from pydantic import BaseModel, Field

from . import metrics
from .request import ServiceRequest, make_request_id
from .scheduler import SchedulerError, ServiceScheduler

log = logging.getLogger("xllm.http")


class CompletionBody(BaseModel):
    model: str = ""
    prompt: Union[str, List[int], List[str], None] = ""
    max_tokens: int = 16
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = -1
    n: int = 1
    stream: bool = False
    stream_options: Optional[Dict[str, Any]] = None
    stop: Union[str, List[str], None] = None
    seed: Optional[int] = None
    logprobs: Optional[int] = None
    echo: bool = False
    ignore_eos: bool = False
    offline: bool = False
    min_tokens: int = 0


class ChatBody(BaseModel):
    model: str = ""
    messages: List[Dict[str, Any]] = Field(default_factory=list)
    tools: Optional[List[Dict[str, Any]]] = None
    tool_choice: Union[str, Dict[str, Any], None] = None
    chat_template_kwargs: Optional[Dict[str, Any]] = None
    max_tokens: Optional[int] = None
    max_completion_tokens: Optional[int] = None
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = -1
    n: int = 1
    stream: bool = False
    stream_options: Optional[Dict[str, Any]] = None
    stop: Union[str, List[str], None] = None
    seed: Optional[int] = None
    logprobs: bool = False          # OpenAI chat: bool + top_logprobs int
    top_logprobs: Optional[int] = None
    ignore_eos: bool = False
    offline: bool = False
    min_tokens: int = 0


def build_app(master) -> FastAPI:
    """master: service.master.Master (owns scheduler + response handler)."""
    app = FastAPI(title="xllm-service-amd", docs_url=None, redoc_url=None)

    def scheduler() -> ServiceScheduler:
        return master.scheduler

    @app.get("/hello")
    @app.post("/hello")
    async def hello(request: Request):
        try:
            body = await request.json()
        except Exception:
            body = None
        return {"message": "hello", "echo": body}

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/metrics")
    async def metrics_route():
        return Response(content=metrics.render(),
                        media_type="text/plain; version=0.0.4")

    @app.get("/v1/models")
    async def models():
        return {"object": "list",
                "data": [{"id": m, "object": "model",
                          "owned_by": "xllm-service-amd"}
                         for m in master.served_models()]}

    @app.post("/v1/messages")
    async def anthropic_messages(request: Request):
        """Anthropic-style messages API (the reference compiles an
        `anthropic` proto alongside the OpenAI ones — SURVEY.md 2.9).
        Translated onto the chat pipeline; supports system, max_tokens,
        temperature, stop_sequences and SSE streaming."""
        body = await request.json()
        na = _not_ready()
        if na is not None:
            return na
        messages = []
        if body.get("system"):
            messages.append({"role": "system", "content": body["system"]})
        messages.extend(body.get("messages", []))
        try:
            prompt_text, token_ids = scheduler().tokenize_chat(messages)
        except Exception as e:
            return JSONResponse({"type": "error", "error": {
                "type": "invalid_request_error", "message": str(e)}},
                status_code=400)
        params = dict(temperature=body.get("temperature", 1.0),
                      top_p=body.get("top_p", 1.0),
                      top_k=body.get("top_k", -1) or -1,
                      max_tokens=body.get("max_tokens", 256),
                      ignore_eos=False, stop_token_ids=[])
        req = ServiceRequest(
            service_request_id=make_request_id("msg"),
            kind="chat", model=body.get("model") or master.model_id,
            stream=bool(body.get("stream")), token_ids=token_ids,
            prompt_text=prompt_text, params=params,
            stop_texts=[s for s in (body.get("stop_sequences") or []) if s])
        sch = scheduler()
        try:
            sch.schedule(req)
            await sch.dispatch(req)
        except SchedulerError as e:
            return JSONResponse({"type": "error", "error": {
                "type": "overloaded_error", "message": str(e)}},
                status_code=e.status_code)

        async def on_cancel(r):
            await sch.cancel_request(r, reason="client disconnected")

        if req.stream:
            async def gen():
                import json as _json
                yield ("event: message_start\ndata: " + _json.dumps({
                    "type": "message_start", "message": {
                        "id": req.service_request_id, "type": "message",
                        "role": "assistant", "content": [],
                        "model": req.model}}) + "\n\n")
                yield ("event: content_block_start\ndata: " + _json.dumps({
                    "type": "content_block_start", "index": 0,
                    "content_block": {"type": "text", "text": ""}}) + "\n\n")
                from xllm_service_amd.tokenizer import IncrementalDecoder
                dec = IncrementalDecoder(master.tokenizer)
                import asyncio as _aio
                try:
                    while True:
                        d = await _aio.wait_for(req.output_queue.get(), 600.0)
                        if d.error:
                            break
                        text = (d.text if d.text is not None
                                else dec.push(d.token_ids)
                                if d.token_ids else "")
                        if text:
                            yield ("event: content_block_delta\ndata: " +
                                   _json.dumps({
                                       "type": "content_block_delta",
                                       "index": 0,
                                       "delta": {"type": "text_delta",
                                                 "text": text}}) + "\n\n")
                        if d.finished:
                            stop = ("max_tokens"
                                    if d.finish_reason == "length"
                                    else "end_turn")
                            yield ("event: message_delta\ndata: " +
                                   _json.dumps({
                                       "type": "message_delta",
                                       "delta": {"stop_reason": stop},
                                       "usage": {"output_tokens":
                                                 d.usage_completion_tokens}})
                                   + "\n\n")
                            break
                    yield ("event: message_stop\ndata: " + _json.dumps(
                        {"type": "message_stop"}) + "\n\n")
                except (GeneratorExit, Exception):
                    await on_cancel(req)
                    raise
            return StreamingResponse(gen(), media_type="text/event-stream")

        token_ids_out, text_out, usage, finish, err = \
            await master.response_handler._collect(req)
        if err:
            return JSONResponse({"type": "error", "error": {
                "type": "api_error", "message": err}}, status_code=500)
        return {
            "id": req.service_request_id, "type": "message",
            "role": "assistant", "model": req.model,
            "content": [{"type": "text", "text": text_out}],
            "stop_reason": ("max_tokens" if finish == "length"
                            else "end_turn"),
            "usage": {"input_tokens": usage["prompt_tokens"],
                      "output_tokens": usage["completion_tokens"]},
        }

    @app.post("/admin/reload_flags")
    async def reload_flags(request: Request):
        """Runtime-reloadable knobs (reference: brpc-reloadable target_ttft /
        target_tpot gflags)."""
        body = await request.json()
        changed = {}
        from .policies import SloAwarePolicy
        pol = scheduler().policy
        if "target_ttft_ms" in body:
            master.opts.target_ttft_ms = float(body["target_ttft_ms"])
            if isinstance(pol, SloAwarePolicy):
                pol.target_ttft_ms = master.opts.target_ttft_ms
            changed["target_ttft_ms"] = master.opts.target_ttft_ms
        if "target_tpot_ms" in body:
            master.opts.target_tpot_ms = float(body["target_tpot_ms"])
            if isinstance(pol, SloAwarePolicy):
                pol.target_tpot_ms = master.opts.target_tpot_ms
            changed["target_tpot_ms"] = master.opts.target_tpot_ms
        if "enable_request_trace" in body:
            master.tracer.enabled = bool(body["enable_request_trace"])
            changed["enable_request_trace"] = master.tracer.enabled
        return {"reloaded": changed}

    @app.post("/v1/embeddings")
    async def embeddings():
        return JSONResponse({"error": {"message": "not support embeddings",
                                       "type": "invalid_request_error"}},
                            status_code=501)

    def _not_ready() -> Optional[JSONResponse]:
        if not scheduler().has_available_instances():
            metrics.REQUEST_ERROR_TOTAL.labels(
                method="any", reason="no_instances").inc()
            return JSONResponse(
                {"error": {"message": "no available instances",
                           "type": "server_error"}}, status_code=503)
        return None

    def _bad_params(body) -> Optional[JSONResponse]:
        mt = getattr(body, "max_tokens", None)
        if mt is None:
            mt = getattr(body, "max_completion_tokens", None)
        if mt is not None and mt < 1:
            # the old params plumbing turned 0 into the default (128):
            # reject like OpenAI instead of silently generating
            return JSONResponse(
                {"error": {"message": "max_tokens must be at least 1",
                           "type": "invalid_request_error"}},
                status_code=400)
        if getattr(body, "n", 1) != 1:
            # better a clear 400 than silently returning one choice
            return JSONResponse(
                {"error": {"message": "n != 1 is not supported",
                           "type": "invalid_request_error"}},
                status_code=400)
        return None

    @app.post("/v1/completions")
    async def completions(body: CompletionBody, request: Request):
        metrics.REQUEST_IN_TOTAL.labels(method="completion").inc()
        na = _not_ready() or _bad_params(body)
        if na is not None:
            return na
        if isinstance(body.prompt, list) and body.prompt \
                and isinstance(body.prompt[0], int):
            token_ids = list(body.prompt)
            prompt_text = ""
        elif isinstance(body.prompt, list) and len(body.prompt) > 1 \
                and isinstance(body.prompt[0], str):
            # OpenAI batches string lists into multiple choices; a silent
            # concatenation would serve the wrong completion
            return JSONResponse(
                {"error": {"message": "batched prompt lists are not "
                                       "supported; send one request per "
                                       "prompt",
                           "type": "invalid_request_error"}},
                status_code=400)
        else:
            prompt_text = body.prompt if isinstance(body.prompt, str) \
                else "".join(body.prompt or [])
            token_ids = scheduler().tokenizer.encode(prompt_text)
        params = _sampling_dict(body)
        stop_texts = params.pop("stop_texts", [])
        req = ServiceRequest(
            service_request_id=make_request_id("cmpl"),
            kind="completion", model=body.model or master.model_id,
            stream=body.stream, token_ids=token_ids, prompt_text=prompt_text,
            params=params, offline=body.offline, stop_texts=stop_texts)
        return await _run(req, request, body.stream, chat=False)

    @app.post("/v1/chat/completions")
    async def chat_completions(body: ChatBody, request: Request):
        metrics.REQUEST_IN_TOTAL.labels(method="chat").inc()
        na = _not_ready() or _bad_params(body)
        if na is not None:
            return na
        try:
            prompt_text, token_ids = scheduler().tokenize_chat(
                body.messages, tools=body.tools, tool_choice=body.tool_choice,
                chat_template_kwargs=body.chat_template_kwargs)
        except Exception as e:
            return JSONResponse(
                {"error": {"message": f"chat template error: {e}",
                           "type": "invalid_request_error"}}, status_code=400)
        # multimodal: collect image parts; offline environment accepts the
        # synthetic form {"type": "image", "grid": [gh, gw], "seed": n}
        images = []
        for msg in body.messages:
            content = msg.get("content")
            if not isinstance(content, list):
                continue
            for part in content:
                if part.get("type") == "image":
                    gh, gw = part.get("grid", [28, 28])
                    images.append(dict(grid_h=int(gh), grid_w=int(gw),
                                       seed=int(part.get("seed", 0))))
                elif part.get("type") == "image_url":
                    return JSONResponse(
                        {"error": {"message":
                                   "image_url fetch unavailable offline; "
                                   "use {'type':'image','grid':[h,w]}",
                                   "type": "invalid_request_error"}},
                        status_code=400)
        multimodal = None
        if images:
            try:
                from xllm_service_amd.models.config import get_config
                mcfg = get_config(body.model or master.model_id)
                merge = mcfg.vision.get("spatial_merge_size", 2)
                pad_id = mcfg.image_pad_token_id
                assert pad_id >= 0
            except Exception:
                return JSONResponse(
                    {"error": {"message": "model is not multimodal",
                               "type": "invalid_request_error"}},
                    status_code=400)
            n_img_tokens = sum((im["grid_h"] // merge) * (im["grid_w"] // merge)
                               for im in images)
            # vision tokens precede the text tokens
            token_ids = [pad_id] * n_img_tokens + token_ids
            multimodal = {"images": images}
        params = _sampling_dict(body)
        stop_texts = params.pop("stop_texts", [])
        if body.max_completion_tokens is not None:
            params["max_tokens"] = body.max_completion_tokens
        elif body.max_tokens is None:
            params["max_tokens"] = 512
        req = ServiceRequest(
            service_request_id=make_request_id("chatcmpl"),
            kind="chat", model=body.model or master.model_id,
            stream=body.stream, token_ids=token_ids, prompt_text=prompt_text,
            params=params, offline=body.offline, multimodal=multimodal,
            stop_texts=stop_texts)
        return await _run(req, request, body.stream, chat=True)

    def _sampling_dict(body) -> Dict[str, Any]:
        lp = getattr(body, "logprobs", None)
        if isinstance(lp, bool):    # chat API: logprobs=true + top_logprobs=N
            lp = (getattr(body, "top_logprobs", None) or 0) if lp else None
        d = dict(temperature=body.temperature, top_p=body.top_p,
                 top_k=body.top_k, max_tokens=body.max_tokens or 16,
                 min_tokens=body.min_tokens, seed=body.seed,
                 ignore_eos=body.ignore_eos, logprobs=lp)
        if body.stream_options:
            d["include_usage"] = bool(
                body.stream_options.get("include_usage"))
        stops = body.stop if isinstance(body.stop, list) else (
            [body.stop] if body.stop else [])
        stop_ids = []
        stop_seqs = []
        for s in stops:
            ids = scheduler().tokenizer.encode(s)
            if len(ids) == 1:
                stop_ids.append(ids[0])
            elif ids:
                stop_seqs.append(ids)
        d["stop_token_ids"] = stop_ids
        d["stop_sequences"] = stop_seqs
        # raw stop strings for the service-layer text scanner (popped off
        # before the params dict travels to the worker)
        d["stop_texts"] = [s for s in stops if s]
        if getattr(body, "echo", False):
            d["echo"] = True
        return d

    async def _run(req: ServiceRequest, http_request: Request, stream: bool,
                   chat: bool):
        sch = scheduler()
        # non-stream disconnects are detected by polling this (streams
        # cancel via GeneratorExit) — see ServiceScheduler.handle_generation
        req.http_request = http_request
        master.tracer.trace(req.service_request_id, "request_in",
                            {"kind": req.kind, "tokens": len(req.token_ids)})
        try:
            sch.schedule(req)
            await sch.dispatch(req)
        except SchedulerError as e:
            metrics.REQUEST_ERROR_TOTAL.labels(
                method=req.kind, reason="schedule").inc()
            return JSONResponse({"error": {"message": str(e),
                                           "type": "server_error"}},
                                status_code=e.status_code)

        async def on_cancel(r):
            await sch.cancel_request(r, reason="client disconnected")

        rh = master.response_handler
        echo = bool(req.params.get("echo")) and not chat
        if stream:
            gen = (rh.stream_chat(req, on_cancel) if chat
                   else rh.stream_completion(req, on_cancel,
                                             echo_text=req.prompt_text
                                             if echo else None))
            return StreamingResponse(gen, media_type="text/event-stream")
        result = await (rh.collect_chat(req) if chat
                        else rh.collect_completion(req))
        if echo and "choices" in result:
            for c in result["choices"]:
                c["text"] = req.prompt_text + c["text"]
        status = 500 if "error" in result else 200
        master.tracer.trace(req.service_request_id, "response_out", result)
        return JSONResponse(result, status_code=status)

    return app
