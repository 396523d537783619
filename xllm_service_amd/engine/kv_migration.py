"""Cross-GPU KV-block migration over xGMI (the PD-disaggregation data plane).

Design: on one 8xMI355X node every GPU pair is one hop over
7x~153 GB/s xGMI links, so the reference's device-network KV-transfer
(LinkInstance with device_ips/ports) degenerates to:
  * at link time, the prefill worker exports hipIpc handles for its KV cache
    tensors (one per layer, k and v), sent to the decode worker over RPC
  * the decode worker opens them once (hipIpcOpenMemHandle with lazy peer
    access = hipDeviceEnablePeerAccess under the hood)
  * per migration, the decode worker PULLS the prompt's blocks with
    hipMemcpyPeerAsync per contiguous run, per layer, on a dedicated side
    stream so copies overlap with the decode compute stream

The serialized-bytes RPC transport (engine.export_block_bytes) remains the
fallback for cross-node peers and for the CPU test path.
"""
from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

log = logging.getLogger("xllm.kv_migration")


def available() -> bool:
    try:
        from xllm_service_amd import _ops
        return torch.cuda.is_available() and hasattr(_ops, "ipc_open_handle")
    except ImportError:
        return False


def export_cache_handles(engine) -> dict:
    """Prefill-side: IPC handles + geometry for every layer's k/v cache."""
    from xllm_service_amd import _ops
    handles = []
    for (kc, vc) in engine.runner.kv_caches:
        handles.append([bytes(_ops.ipc_get_handle(kc)),
                        bytes(_ops.ipc_get_handle(vc))])
    kc0 = engine.runner.kv_caches[0][0]
    return dict(
        handles=handles,
        device=engine.device.index or 0,
        num_blocks=int(kc0.shape[0]),
        shape=list(kc0.shape[1:]),
    )


@dataclass
class PeerCache:
    """Decode-side view of one peer's opened cache (per layer k/v ptrs)."""
    src_device: int
    ptrs: List[List[int]] = field(default_factory=list)  # [layer][k,v]

    def close(self):
        from xllm_service_amd import _ops
        for pk, pv in self.ptrs:
            try:
                _ops.ipc_close_handle(pk)
                _ops.ipc_close_handle(pv)
            except Exception:
                pass
        self.ptrs = []


class MigrationManager:
    """Decode-side registry of opened peer caches + the pull operation."""

    def __init__(self, engine):
        self.engine = engine
        self.peers: Dict[str, PeerCache] = {}
        self.stream = (torch.cuda.Stream(engine.device)
                       if engine.device.type == "cuda" else None)

    def open_peer(self, name: str, exported: dict):
        from xllm_service_amd import _ops
        self.close_peer(name)
        pc = PeerCache(src_device=int(exported["device"]))
        my_dev = self.engine.device.index or 0
        if pc.src_device != my_dev:
            try:
                _ops.enable_peer_access(my_dev, pc.src_device)
            except Exception as e:  # same-device IPC needs no peer access
                log.warning("peer access %d->%d: %s", my_dev, pc.src_device, e)
        for hk, hv in exported["handles"]:
            pk = _ops.ipc_open_handle(list(hk), my_dev)
            pv = _ops.ipc_open_handle(list(hv), my_dev)
            pc.ptrs.append([pk, pv])
        self.peers[name] = pc
        log.info("opened IPC cache of peer %s (device %d, %d layers)", name,
                 pc.src_device, len(pc.ptrs))

    def close_peer(self, name: str):
        pc = self.peers.pop(name, None)
        if pc:
            pc.close()

    def has_peer(self, name: str) -> bool:
        return name in self.peers

    def pull_blocks(self, src_name: str, src_blocks: List[int],
                    dst_blocks: List[int]):
        """Copy blocks from the peer's cache into ours (all layers), on the
        side stream; synchronizes before returning so the caller may
        activate the sequence immediately."""
        from xllm_service_amd import _ops
        pc = self.peers[src_name]
        my_dev = self.engine.device.index or 0
        with torch.cuda.stream(self.stream):
            for layer, (kc, vc) in enumerate(self.engine.runner.kv_caches):
                pk, pv = pc.ptrs[layer]
                _ops.migrate_blocks_from_ptr(kc, pk, pc.src_device, my_dev,
                                             list(src_blocks), list(dst_blocks))
                _ops.migrate_blocks_from_ptr(vc, pv, pc.src_device, my_dev,
                                             list(src_blocks), list(dst_blocks))
        self.stream.synchronize()


def migrate_in_xgmi(engine, peer_meta, src_blocks, dst_blocks):
    raise RuntimeError(
        "migrate_in_xgmi requires the worker's MigrationManager path")
