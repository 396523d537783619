"""Control-plane types shared by master, workers and replicas.

Parity with the reference's common/types.h (instance metadata, status
machine, load/latency metrics, registry key namespaces) — re-expressed as
plain dataclasses serialised to JSON/msgpack.
"""
from __future__ import annotations

import enum
import time
from dataclasses import asdict, dataclass, field
from typing import Any, Dict, List


class InstanceType(str, enum.Enum):
    DEFAULT = "DEFAULT"    # colocated prefill+decode
    PREFILL = "PREFILL"
    DECODE = "DECODE"
    MIX = "MIX"            # schedulable as either side
    ENCODE = "ENCODE"      # multimodal vision-encoder stage (EPD 3-stage)


class InstanceStatus(str, enum.Enum):
    ACTIVE = "ACTIVE"
    LEASE_LOST = "LEASE_LOST"   # lease expired but health probe passed (grace)
    SUSPECT = "SUSPECT"         # unhealthy; excluded from scheduling


# registry key namespaces (reference: common/types.h:33-35 key scheme)
KEY_SERVICE = "XLLM:SERVICE:"
KEY_MASTER = "XLLM:SERVICE:MASTER"
KEY_INSTANCE = {t: f"XLLM:{t.value}:" for t in InstanceType}
KEY_CACHE = "XLLM:CACHE:"
KEY_LOADMETRICS = "XLLM:LOADMETRICS:"


@dataclass
class InstanceMetaInfo:
    name: str
    itype: str = InstanceType.DEFAULT.value
    rpc_host: str = "127.0.0.1"
    rpc_port: int = 0
    http_port: int = 0
    device_index: int = -1            # CUDA/HIP ordinal of the worker's GPU
    cluster_ids: List[int] = field(default_factory=list)
    device_ips: List[str] = field(default_factory=list)
    ports: List[int] = field(default_factory=list)
    dp_size: int = 1
    tp_size: int = 1
    k_cache_ids: List[int] = field(default_factory=list)
    v_cache_ids: List[int] = field(default_factory=list)
    num_kv_blocks: int = 0
    block_size: int = 16
    model: str = ""
    incarnation_id: int = 0
    register_ts_ms: int = field(default_factory=lambda: int(time.time() * 1000))
    # TTFT/TPOT profiling samples for the SLO predictor:
    # [(num_tokens, ttft_ms)], [(batch, tokens, tpot_ms)]
    ttft_profile: List[List[float]] = field(default_factory=list)
    tpot_profile: List[List[float]] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return asdict(self)

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "InstanceMetaInfo":
        return cls(**{k: v for k, v in d.items()
                      if k in cls.__dataclass_fields__})


@dataclass
class LoadMetrics:
    waiting_requests_num: int = 0
    running_requests_num: int = 0
    gpu_cache_usage_perc: float = 0.0

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in (d or {}).items()
                      if k in cls.__dataclass_fields__})


@dataclass
class LatencyMetrics:
    recent_max_ttft_ms: float = 0.0
    recent_max_tbt_ms: float = 0.0

    def to_dict(self):
        return asdict(self)

    @classmethod
    def from_dict(cls, d):
        return cls(**{k: v for k, v in (d or {}).items()
                      if k in cls.__dataclass_fields__})


@dataclass
class KvCacheEvent:
    """Per-heartbeat delta of an instance's prefix-cache contents."""
    stored: List[bytes] = field(default_factory=list)
    removed: List[bytes] = field(default_factory=list)
    offloaded: List[bytes] = field(default_factory=list)  # hbm -> dram tier
