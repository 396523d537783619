"""Embedded etcd-style metadata store: leased keys, prefix watches, txns.

The reference externalises cluster state to an etcd cluster
(reference: xllm_service/scheduler/etcd_client/etcd_client.h:38-161). For a
single-node 8xMI355X deployment an embedded store hosted by the master (and
reachable over TCP by workers/replicas, registry/server.py) removes that
external dependency while keeping the same semantics the control plane
needs: PUT/GET/DELETE with namespaces, TTL leases with keepalive, prefix
scans, create-if-absent txn (master election), and prefix watches that
deliver PUT/DELETE events carrying previous values.

KVStore is the synchronous core; it is wrapped by the asyncio TCP server and
also used directly in-process by unit tests.
"""
from __future__ import annotations

import itertools
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple


@dataclass
class WatchEvent:
    type: str          # "put" | "delete"
    key: str
    value: Optional[bytes]       # new value (put) or None
    prev_value: Optional[bytes]  # previous value if any


@dataclass
class _Entry:
    value: bytes
    lease_id: int = 0
    version: int = 1


@dataclass
class _Lease:
    ttl: float
    expires_at: float
    keys: set = field(default_factory=set)


class KVStore:
    def __init__(self, clock: Callable[[], float] = time.monotonic):
        self._clock = clock
        self._lock = threading.RLock()
        self._data: Dict[str, _Entry] = {}
        self._leases: Dict[int, _Lease] = {}
        self._lease_ids = itertools.count(1)
        self._watch_ids = itertools.count(1)
        # watch_id -> (prefix, callback)
        self._watches: Dict[int, Tuple[str, Callable[[WatchEvent], None]]] = {}

    # ---- leases -------------------------------------------------------------
    def grant_lease(self, ttl: float) -> int:
        with self._lock:
            lid = next(self._lease_ids)
            self._leases[lid] = _Lease(ttl, self._clock() + ttl)
            return lid

    def keepalive(self, lease_id: int) -> bool:
        with self._lock:
            lease = self._leases.get(lease_id)
            if lease is None:
                return False
            lease.expires_at = self._clock() + lease.ttl
            return True

    def revoke_lease(self, lease_id: int):
        with self._lock:
            lease = self._leases.pop(lease_id, None)
            if lease is None:
                return
            for key in list(lease.keys):
                self._delete(key)

    def expire_leases(self) -> int:
        """Delete keys of expired leases; returns count of expired leases.
        Must be called periodically (the server runs a 100 ms ticker)."""
        now = self._clock()
        with self._lock:
            expired = [lid for lid, l in self._leases.items()
                       if l.expires_at <= now]
            for lid in expired:
                self.revoke_lease(lid)
            return len(expired)

    # ---- kv -----------------------------------------------------------------
    def put(self, key: str, value: bytes, lease_id: int = 0) -> None:
        with self._lock:
            if lease_id:
                lease = self._leases.get(lease_id)
                if lease is None:
                    raise KeyError(f"lease {lease_id} not found")
                lease.keys.add(key)
            prev = self._data.get(key)
            if prev is not None and prev.lease_id and prev.lease_id != lease_id:
                old_lease = self._leases.get(prev.lease_id)
                if old_lease:
                    old_lease.keys.discard(key)
            self._data[key] = _Entry(value, lease_id,
                                     (prev.version + 1) if prev else 1)
            self._notify(WatchEvent("put", key, value,
                                    prev.value if prev else None))

    def get(self, key: str) -> Optional[bytes]:
        with self._lock:
            e = self._data.get(key)
            return e.value if e else None

    def delete(self, key: str) -> bool:
        with self._lock:
            return self._delete(key)

    def _delete(self, key: str) -> bool:
        e = self._data.pop(key, None)
        if e is None:
            return False
        if e.lease_id:
            lease = self._leases.get(e.lease_id)
            if lease:
                lease.keys.discard(key)
        self._notify(WatchEvent("delete", key, None, e.value))
        return True

    def range(self, prefix: str) -> List[Tuple[str, bytes]]:
        with self._lock:
            return sorted((k, e.value) for k, e in self._data.items()
                          if k.startswith(prefix))

    def delete_prefix(self, prefix: str) -> int:
        with self._lock:
            keys = [k for k in self._data if k.startswith(prefix)]
            for k in keys:
                self._delete(k)
            return len(keys)

    # ---- txn ----------------------------------------------------------------
    def create_if_absent(self, key: str, value: bytes, lease_id: int = 0) -> bool:
        """Atomic compare(version==0)-then-create; the master-election
        primitive (reference: etcd_client.cpp:105-120)."""
        with self._lock:
            if key in self._data:
                return False
            self.put(key, value, lease_id)
            return True

    # ---- watches ------------------------------------------------------------
    def add_watch(self, prefix: str,
                  callback: Callable[[WatchEvent], None]) -> int:
        with self._lock:
            wid = next(self._watch_ids)
            self._watches[wid] = (prefix, callback)
            return wid

    def remove_watch(self, watch_id: int):
        with self._lock:
            self._watches.pop(watch_id, None)

    def _notify(self, ev: WatchEvent):
        for prefix, cb in list(self._watches.values()):
            if ev.key.startswith(prefix):
                try:
                    cb(ev)
                except Exception:  # watcher errors must not poison the store
                    pass
