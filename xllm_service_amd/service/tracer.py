"""Request tracer: JSONL log of every request/response payload.
(reference: http_service/request_tracer.cpp — flag enable_request_trace)"""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Any


class RequestTracer:
    def __init__(self, enabled: bool = False,
                 path: str = "trace/trace.jsonl"):
        self.enabled = enabled
        self.path = path
        self._lock = threading.Lock()
        self._fh = None
        if enabled:
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
            self._fh = open(path, "a", buffering=1)

    def trace(self, service_request_id: str, direction: str, data: Any):
        if not self.enabled:
            return
        if self._fh is None:  # enabled at runtime via /admin/reload_flags
            os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
            self._fh = open(self.path, "a", buffering=1)
        rec = {"timestamp": time.time(),
               "service_request_id": service_request_id,
               "direction": direction, "data": data}
        line = json.dumps(rec, default=str)
        with self._lock:
            self._fh.write(line + "\n")

    def close(self):
        if self._fh:
            self._fh.close()
            self._fh = None
