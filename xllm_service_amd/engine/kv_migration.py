"""Cross-GPU KV-block migration over xGMI (the PD-disaggregation data plane).

Design: on one 8xMI355X node every GPU pair is one hop over
7x~153 GB/s xGMI links, so the reference's device-network KV-transfer
(LinkInstance with device_ips/ports) degenerates to:
  * at link time the prefill worker exports its per-layer KV cache tensors
    with torch's CUDA-IPC storage sharing (dmabuf handles on this driver —
    HSA_ENABLE_IPC_MODE_LEGACY=0); sharing through the storage layer keeps
    allocator-pool offsets correct, which raw hipIpcGetMemHandle on a
    tensor data_ptr would lose
  * the decode worker materialises tensor views of the peer's caches once,
    then PULLS blocks per migration with hipMemcpyPeerAsync per contiguous
    run (ops.migrate_blocks_peer), on a dedicated side stream so copies
    overlap the decode compute stream

The serialized-bytes RPC transport (engine.export_block_bytes) remains the
fallback for cross-node peers and the CPU test path.
"""
from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Dict, List

import torch

from xllm_service_amd import ops

log = logging.getLogger("xllm.kv_migration")


def available() -> bool:
    return torch.cuda.is_available() and ops.HAS_EXT


def _share_tensor(t: torch.Tensor) -> dict:
    storage = t.untyped_storage()
    (device, handle, storage_size, storage_offset, ref_handle, ref_offset,
     event_handle, event_sync) = storage._share_cuda_()
    return dict(device=device, handle=bytes(handle),
                storage_size=storage_size, storage_offset=storage_offset,
                ref_handle=bytes(ref_handle) if ref_handle else b"",
                ref_offset=ref_offset,
                event_handle=bytes(event_handle) if event_handle else b"",
                event_sync=bool(event_sync),
                shape=list(t.shape), dtype=str(t.dtype).split(".")[-1],
                tensor_offset=t.storage_offset())


def _open_tensor(d: dict) -> torch.Tensor:
    storage = torch.UntypedStorage._new_shared_cuda(
        d["device"], d["handle"], d["storage_size"], d["storage_offset"],
        d["ref_handle"], d["ref_offset"], d["event_handle"], d["event_sync"])
    dtype = getattr(torch, d["dtype"])
    t = torch.empty(0, dtype=dtype, device=f"cuda:{d['device']}")
    t.set_(storage, d["tensor_offset"], d["shape"])
    return t


def export_cache_handles(engine) -> dict:
    """Prefill-side: shareable descriptors for every layer's k/v cache."""
    handles = []
    for (kc, vc) in engine.runner.kv_caches:
        handles.append([_share_tensor(kc), _share_tensor(vc)])
    return dict(handles=handles, device=engine.device.index or 0)


@dataclass
class PeerCache:
    src_device: int
    views: List[List[torch.Tensor]] = field(default_factory=list)

    def close(self):
        self.views = []


class MigrationManager:
    """Decode-side registry of opened peer caches + the pull operation."""

    def __init__(self, engine):
        self.engine = engine
        self.peers: Dict[str, PeerCache] = {}
        self.stream = (torch.cuda.Stream(engine.device)
                       if engine.device.type == "cuda" else None)

    def open_peer(self, name: str, exported: dict):
        self.close_peer(name)
        pc = PeerCache(src_device=int(exported["device"]))
        my_dev = self.engine.device.index or 0
        if pc.src_device != my_dev:
            try:
                ops.enable_peer_access(my_dev, pc.src_device)
            except Exception as e:
                log.warning("peer access %d->%d: %s", my_dev, pc.src_device, e)
        for hk, hv in exported["handles"]:
            pc.views.append([_open_tensor(hk), _open_tensor(hv)])
        self.peers[name] = pc
        log.info("opened shared cache of peer %s (device %d, %d layers)",
                 name, pc.src_device, len(pc.views))

    def close_peer(self, name: str):
        pc = self.peers.pop(name, None)
        if pc:
            pc.close()

    def has_peer(self, name: str) -> bool:
        return name in self.peers

    def pull_blocks_async(self, src_name: str, src_blocks: List[int],
                          dst_blocks: List[int]) -> "torch.cuda.Event":
        """Start the block pull on the dedicated side stream and return the
        completion event — the decode loop keeps stepping while xGMI copies
        fly; the sequence activates only when the event fires (engine
        pending-migration poll). Safe to call from any thread."""
        pc = self.peers[src_name]
        my_dev = self.engine.device.index or 0
        ev = torch.cuda.Event()
        with torch.cuda.stream(self.stream):
            for layer, (kc, vc) in enumerate(self.engine.runner.kv_caches):
                sk, sv = pc.views[layer]
                ops.migrate_blocks_peer(kc, my_dev, sk, pc.src_device,
                                        src_blocks, dst_blocks)
                ops.migrate_blocks_peer(vc, my_dev, sv, pc.src_device,
                                        src_blocks, dst_blocks)
            ev.record(self.stream)
        return ev

    def pull_blocks(self, src_name: str, src_blocks: List[int],
                    dst_blocks: List[int]):
        """Synchronous pull (tests / compatibility)."""
        self.pull_blocks_async(src_name, src_blocks, dst_blocks).synchronize()
