"""Prometheus metrics (reference: common/metrics.{h,cpp} over bvar; here on
prometheus_client, served at /metrics — which the reference left a TODO)."""
from __future__ import annotations

from prometheus_client import (CollectorRegistry, Counter, Gauge, Histogram,
                               generate_latest)

REGISTRY = CollectorRegistry()

REQUEST_IN_TOTAL = Counter(
    "server_request_in_total", "Requests accepted", ["method"],
    registry=REGISTRY)
REQUEST_ERROR_TOTAL = Counter(
    "server_request_error_total", "Requests failed", ["method", "reason"],
    registry=REGISTRY)
REQUEST_CANCEL_TOTAL = Counter(
    "server_request_cancel_total", "Requests cancelled", ["reason"],
    registry=REGISTRY)
TTFT_MS = Histogram(
    "time_to_first_token_latency_milliseconds", "TTFT (ms)",
    buckets=(10, 25, 50, 100, 200, 400, 800, 1000, 1500, 2500, 5000, 10000),
    registry=REGISTRY)
ITL_MS = Histogram(
    "inter_token_latency_milliseconds", "Inter-token latency (ms)",
    buckets=(1, 2, 5, 10, 20, 30, 50, 75, 100, 200, 500, 1000),
    registry=REGISTRY)
ACTIVE_REQUESTS = Gauge(
    "server_active_requests", "In-flight requests", registry=REGISTRY)
AVAILABLE_INSTANCES = Gauge(
    "cluster_schedulable_instances", "Schedulable instances", ["side"],
    registry=REGISTRY)
GENERATED_TOKENS = Counter(
    "generated_tokens_total", "Generated tokens", registry=REGISTRY)


def render() -> bytes:
    return generate_latest(REGISTRY)
