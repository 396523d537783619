"""Master-side request state (reference: request/request.h + common/xllm/output.h)."""
from __future__ import annotations

import asyncio
import time
import uuid
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional


def short_uuid() -> str:
    return uuid.uuid4().hex[:12]


def make_request_id(kind: str) -> str:
    return f"{kind}-{short_uuid()}"


@dataclass
class GenerationDelta:
    """One token batch pushed by an instance for one request."""
    token_ids: List[int]
    finished: bool = False
    finish_reason: Optional[str] = None
    finished_on_prefill: bool = False
    usage_prompt_tokens: int = 0
    usage_completion_tokens: int = 0
    logprobs: Optional[List[float]] = None
    error: Optional[str] = None
    # text-level stop trimming: when set, this is the delta's text and the
    # response handler must use it instead of decoding token_ids
    text: Optional[str] = None


@dataclass
class ServiceRequest:
    service_request_id: str
    kind: str                     # "completion" | "chat"
    model: str
    stream: bool
    token_ids: List[int] = field(default_factory=list)
    prompt_text: str = ""
    params: Dict[str, Any] = field(default_factory=dict)   # sampling etc.
    offline: bool = False         # offline batch job (preemptible)
    created: float = field(default_factory=time.time)
    # routing (bound at schedule time)
    prefill_name: Optional[str] = None
    prefill_incarnation: int = -1
    decode_name: Optional[str] = None
    decode_incarnation: int = -1
    encode_name: Optional[str] = None   # multimodal E-stage
    # multimodal payload (image grid etc.)
    multimodal: Optional[Dict[str, Any]] = None
    # lifecycle
    prefill_finished: bool = False
    first_token_at: Optional[float] = None
    last_token_at: Optional[float] = None
    num_generated: int = 0
    scheduled_at: Optional[float] = None
    # per-request ordered delivery lane
    output_queue: "asyncio.Queue[GenerationDelta]" = field(
        default_factory=asyncio.Queue)
    is_disconnected: Callable[[], bool] = lambda: False
    # non-stream disconnect detection: the live HTTP request object
    # (awaited .is_disconnected(), rate-limited in handle_generation)
    http_request: Optional[Any] = None
    _last_disc_check: float = 0.0
    # OpenAI text-level stop strings (engine token-match is the fast path;
    # the scanner is authoritative — see service/stop_scanner.py)
    stop_texts: List[str] = field(default_factory=list)
    stop_scanner: Optional[Any] = None
    trace_cb: Optional[Callable[[str, Any], None]] = None
