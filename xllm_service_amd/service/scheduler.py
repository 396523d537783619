"""ServiceScheduler: the master's request orchestrator.

Responsibilities (reference: scheduler/scheduler.{h,cpp}):
  * schedule(): chat-template + tokenize host-side, pick a (prefill, decode)
    pair via the configured policy, pin the request to the instances'
    incarnations
  * dispatch the request to the prefill instance over RPC (fire-and-forget;
    tokens return via the Generations push path)
  * handle_generation(): per-request ordered delivery through an asyncio
    queue (replaces the reference's 128 single-thread output lanes), client
    disconnect -> cancel + abort RPC to the instance
  * clear_requests_on_failed_instance(): cancel in-flight requests bound to
    a dead (name, incarnation)
  * SLO metric ingestion (TTFT/TPOT observations feed the predictors)
"""
from __future__ import annotations

import logging
import time
from typing import Any, Dict, List, Optional

from xllm_service_amd.chat_template import JinjaChatTemplate
from xllm_service_amd.tokenizer import Tokenizer

from . import metrics
from .instance_mgr import InstanceMgr
from .kvcache_mgr import GlobalKVCacheMgr
from .policies import LoadBalancePolicy, SloAwarePolicy
from .request import GenerationDelta, ServiceRequest
from .tracer import RequestTracer

log = logging.getLogger("xllm.scheduler")


class SchedulerError(Exception):
    def __init__(self, message: str, status_code: int = 503):
        super().__init__(message)
        self.status_code = status_code


class ServiceScheduler:
    def __init__(self, instance_mgr: InstanceMgr, kv_mgr: GlobalKVCacheMgr,
                 policy: LoadBalancePolicy, tokenizer: Tokenizer,
                 chat_template: JinjaChatTemplate,
                 tracer: Optional[RequestTracer] = None):
        self.mgr = instance_mgr
        self.kv = kv_mgr
        self.policy = policy
        self.tokenizer = tokenizer
        self.chat_template = chat_template
        self.tracer = tracer or RequestTracer(False)
        self.requests: Dict[str, ServiceRequest] = {}

    # ---- scheduling ---------------------------------------------------------
    def has_available_instances(self) -> bool:
        return self.mgr.has_available_instances()

    def tokenize_chat(self, messages, tools=None, tool_choice=None,
                      chat_template_kwargs=None) -> tuple[str, List[int]]:
        prompt = self.chat_template.apply(
            messages, tools=tools, tool_choice=tool_choice,
            chat_template_kwargs=chat_template_kwargs)
        return prompt, self.tokenizer.encode(prompt)

    _rr_encode = 0

    def schedule(self, req: ServiceRequest) -> None:
        """Bind req to an instance pair (plus the E-stage instance for
        multimodal requests); raises SchedulerError if none."""
        pair = self.policy.select_instances_pair(req.token_ids)
        if not pair.ok:
            raise SchedulerError("no available instances")
        if req.multimodal and req.multimodal.get("images"):
            encodes = self.mgr.schedulable_encodes()
            if encodes:  # E/P/D three-stage split; else colocated vision
                ServiceScheduler._rr_encode += 1
                req.encode_name = encodes[
                    ServiceScheduler._rr_encode % len(encodes)].name
        if pair.prefill is not None:
            req.prefill_name = pair.prefill.name
            req.prefill_incarnation = pair.prefill.meta.incarnation_id
            pair.prefill.num_scheduled += 1
            pair.prefill.num_prefill_unfinished += 1
            pair.prefill.pending_prefill_tokens += len(req.token_ids)
        if req.stop_texts:
            from .stop_scanner import StopTextScanner
            req.stop_scanner = StopTextScanner(self.tokenizer, req.stop_texts)
        if pair.decode is not None:
            req.decode_name = pair.decode.name
            req.decode_incarnation = pair.decode.meta.incarnation_id
            pair.decode.num_decoding += 1
        elif pair.prefill is not None:
            # colocated (DEFAULT): the prefill instance also decodes, so the
            # request must be decode-bound to it for failure cancellation
            req.decode_name = pair.prefill.name
            req.decode_incarnation = pair.prefill.meta.incarnation_id
            pair.prefill.num_decoding += 1
        req.scheduled_at = time.monotonic()

    async def dispatch(self, req: ServiceRequest) -> None:
        """Forward the request to its first-stage instance (E for
        multimodal when an encode pool exists, else P)."""
        target_name = req.encode_name or req.prefill_name or req.decode_name
        inst = self.mgr.get(target_name)
        if inst is None or inst.conn is None:
            raise SchedulerError(f"instance {target_name} unavailable")
        self.requests[req.service_request_id] = req
        metrics.ACTIVE_REQUESTS.inc()
        payload = dict(
            service_request_id=req.service_request_id,
            token_ids=req.token_ids,
            params=req.params,
            offline=req.offline,
            routing=dict(prefill_name=req.prefill_name,
                         decode_name=req.decode_name,
                         encode_name=req.encode_name,
                         prefill_incarnation=req.prefill_incarnation,
                         decode_incarnation=req.decode_incarnation),
            multimodal=req.multimodal,
        )
        self.tracer.trace(req.service_request_id, "dispatch", payload)
        try:
            # fire-and-forget like the reference: tokens come back via the
            # generations push, not this call
            await inst.conn.notify("execute_request", **payload)
        except Exception as e:
            self._drop_request(req, f"dispatch failed: {e}")
            raise SchedulerError(f"dispatch to {target_name} failed: {e}")

    # ---- generation ingestion (from worker RPC push) ------------------------
    async def handle_generation(self, gen: Dict[str, Any]) -> bool:
        """Returns False if the request is gone/cancelled (the worker should
        abort it)."""
        rid = gen.get("service_request_id")
        req = self.requests.get(rid)
        if req is None:
            return False
        if req.is_disconnected():
            await self.cancel_request(req, reason="client disconnected")
            return False
        if req.http_request is not None and not req.stream:
            # non-stream clients have no generator to cancel; poll the HTTP
            # connection (rate-limited so the hot path stays cheap). Stream
            # paths cancel via GeneratorExit already.
            now0 = time.monotonic()
            if now0 - req._last_disc_check > 0.5:
                req._last_disc_check = now0
                try:
                    if await req.http_request.is_disconnected():
                        await self.cancel_request(
                            req, reason="client disconnected")
                        return False
                except Exception:
                    pass

        now = time.monotonic()
        toks = list(gen.get("token_ids") or [])
        if toks:
            if req.first_token_at is None:
                req.first_token_at = now
                if req.scheduled_at is not None:
                    ttft_ms = (now - req.scheduled_at) * 1000
                    metrics.TTFT_MS.observe(ttft_ms)
                    if isinstance(self.policy, SloAwarePolicy) and req.prefill_name:
                        self.policy.observe_ttft(req.prefill_name,
                                                 len(req.token_ids), ttft_ms)
            elif req.last_token_at is not None:
                itl_ms = (now - req.last_token_at) * 1000 / max(len(toks), 1)
                metrics.ITL_MS.observe(itl_ms)
                if isinstance(self.policy, SloAwarePolicy) and req.decode_name:
                    inst = self.mgr.get(req.decode_name)
                    batch = inst.num_decoding if inst else 1
                    self.policy.observe_tpot(req.decode_name, batch,
                                             len(req.token_ids), itl_ms)
            req.last_token_at = now
            req.num_generated += len(toks)
            metrics.GENERATED_TOKENS.inc(len(toks))

        if gen.get("finished_on_prefill"):
            req.prefill_finished = True
            inst = self.mgr.get(req.prefill_name or "")
            if inst:
                inst.num_prefill_unfinished = max(
                    0, inst.num_prefill_unfinished - 1)
                inst.pending_prefill_tokens = max(
                    0, inst.pending_prefill_tokens - len(req.token_ids))

        # text-level stop strings (OpenAI semantics): scan decoded text,
        # trim at the match; engine token-level matching is the fast path
        finished = bool(gen.get("finished"))
        finish_reason = gen.get("finish_reason")
        text_override = None
        scan = req.stop_scanner
        if scan is not None and not gen.get("error"):
            stop_hit = False
            if toks:
                toks, text_override, stop_hit = scan.feed(toks)
            if stop_hit:
                finished, finish_reason = True, "stop"
                # the engine may still be generating: stop it
                for name in {req.prefill_name, req.decode_name} - {None}:
                    inst = self.mgr.get(name)
                    if inst and inst.conn:
                        try:
                            await inst.conn.notify(
                                "abort_request", service_request_id=rid)
                        except Exception:
                            pass
            elif finished:
                toks = toks + scan.flush()

        delta = GenerationDelta(
            token_ids=toks,
            finished=finished,
            finish_reason=finish_reason,
            text=text_override,
            finished_on_prefill=bool(gen.get("finished_on_prefill")),
            usage_prompt_tokens=gen.get("prompt_tokens", len(req.token_ids)),
            usage_completion_tokens=gen.get("completion_tokens",
                                            req.num_generated),
            logprobs=gen.get("logprobs"),
            error=gen.get("error"),
        )
        await req.output_queue.put(delta)
        if delta.finished or delta.error:
            self.finish_request(req)
        return True

    def finish_request(self, req: ServiceRequest):
        if self.requests.pop(req.service_request_id, None) is not None:
            metrics.ACTIVE_REQUESTS.dec()
            inst = self.mgr.get(req.decode_name or "")
            if inst:
                inst.num_decoding = max(0, inst.num_decoding - 1)

    async def cancel_request(self, req: ServiceRequest, reason: str):
        metrics.REQUEST_CANCEL_TOTAL.labels(reason=reason).inc()
        self.finish_request(req)
        # unblock any pending collector (non-stream handler task)
        try:
            req.output_queue.put_nowait(GenerationDelta(
                token_ids=[], finished=True, finish_reason="abort",
                error=f"cancelled: {reason}"))
        except Exception:
            pass
        # tell the instance(s) to stop computing
        for name in {req.prefill_name, req.decode_name} - {None}:
            inst = self.mgr.get(name)
            if inst and inst.conn:
                try:
                    await inst.conn.notify("abort_request",
                                           service_request_id=req.service_request_id)
                except Exception:
                    pass

    def _drop_request(self, req: ServiceRequest, reason: str):
        log.warning("dropping request %s: %s", req.service_request_id, reason)
        self.finish_request(req)

    # ---- failure propagation ------------------------------------------------
    async def clear_requests_on_failed_instance(self, name: str,
                                                incarnation: int):
        """Cancel requests bound to a failed (name, incarnation):
        prefill-bound only while prefill is unfinished; decode-bound always.
        (reference: scheduler.cpp:443-482)"""
        self.kv.remove_instance(name)
        victims: List[ServiceRequest] = []
        for req in list(self.requests.values()):
            if (req.prefill_name == name
                    and req.prefill_incarnation == incarnation
                    and not req.prefill_finished):
                victims.append(req)
            elif (req.decode_name == name
                    and req.decode_incarnation == incarnation):
                victims.append(req)
        for req in victims:
            await req.output_queue.put(GenerationDelta(
                token_ids=[], finished=True, finish_reason="abort",
                error="Instance is failed and deleted"))
            self.finish_request(req)
            metrics.REQUEST_CANCEL_TOTAL.labels(reason="instance_failed").inc()
        if victims:
            log.warning("cancelled %d requests on failed instance %s",
                        len(victims), name)
