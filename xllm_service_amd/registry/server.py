"""Registry service: the KVStore exposed over msgpack-RPC, plus the asyncio
client used by workers, masters and replicas.

The master process hosts one RegistryService; every other process connects
with RegistryClient. Watches are server-push notifications on the client's
connection. A 100 ms ticker expires leases server-side.
"""
from __future__ import annotations

import asyncio
import json
from typing import Any, Callable, List, Optional, Tuple

from xllm_service_amd.utils import msgrpc

from .store import KVStore, WatchEvent


class _RegistryHandler:
    def __init__(self, service: "RegistryService", conn: msgrpc.Connection):
        self.service = service
        self.conn = conn
        self.watch_ids: List[int] = []
        self.lease_ids: List[int] = []

    # ---- kv ----
    def rpc_put(self, conn, key: str, value: bytes, lease_id: int = 0):
        self.service.store.put(key, value, lease_id)
        return True

    def rpc_get(self, conn, key: str):
        return self.service.store.get(key)

    def rpc_delete(self, conn, key: str):
        return self.service.store.delete(key)

    def rpc_range(self, conn, prefix: str):
        return self.service.store.range(prefix)

    def rpc_delete_prefix(self, conn, prefix: str):
        return self.service.store.delete_prefix(prefix)

    def rpc_create_if_absent(self, conn, key: str, value: bytes,
                             lease_id: int = 0):
        return self.service.store.create_if_absent(key, value, lease_id)

    # ---- leases ----
    def rpc_grant_lease(self, conn, ttl: float):
        lid = self.service.store.grant_lease(ttl)
        self.lease_ids.append(lid)
        return lid

    def rpc_keepalive(self, conn, lease_id: int):
        return self.service.store.keepalive(lease_id)

    def rpc_revoke_lease(self, conn, lease_id: int):
        self.service.store.revoke_lease(lease_id)
        return True

    # ---- watches ----
    def rpc_watch(self, conn, prefix: str):
        loop = asyncio.get_running_loop()

        def cb(ev: WatchEvent):
            # hop back to the event loop; connection may be gone
            loop.call_soon_threadsafe(
                lambda: asyncio.ensure_future(self._push(ev)))

        wid = self.service.store.add_watch(prefix, cb)
        self.watch_ids.append(wid)
        return wid

    async def _push(self, ev: WatchEvent):
        try:
            await self.conn.notify("watch_event", type=ev.type, key=ev.key,
                                   value=ev.value, prev_value=ev.prev_value)
        except Exception:
            pass

    def rpc_cancel_watch(self, conn, watch_id: int):
        self.service.store.remove_watch(watch_id)
        return True


class RegistryService:
    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self.store = KVStore()
        self.server = msgrpc.Server(
            lambda conn: _RegistryHandler(self, conn), host, port)
        self._ticker: Optional[asyncio.Task] = None

    async def start(self) -> int:
        port = await self.server.start()
        self._ticker = asyncio.create_task(self._tick())
        return port

    async def _tick(self):
        while True:
            await asyncio.sleep(0.1)
            self.store.expire_leases()

    async def stop(self):
        if self._ticker:
            self._ticker.cancel()
        await self.server.stop()


class RegistryClient:
    """Asyncio client. Watch callbacks run on the client's event loop."""

    def __init__(self):
        self.conn: Optional[msgrpc.Connection] = None
        self._watch_cbs: List[Tuple[str, Callable[[WatchEvent], Any]]] = []

    async def connect(self, host: str, port: int):
        self.conn = await msgrpc.connect(host, port, handler=self)
        return self

    # server-push watch events
    async def on_watch_event(self, conn, type: str, key: str, value, prev_value):
        ev = WatchEvent(type, key, value, prev_value)
        for prefix, cb in list(self._watch_cbs):
            if key.startswith(prefix):
                r = cb(ev)
                if asyncio.iscoroutine(r):
                    await r

    # ---- kv ----
    async def put(self, key: str, value: bytes, lease_id: int = 0):
        return await self.conn.call("put", key=key, value=value,
                                    lease_id=lease_id)

    async def put_json(self, key: str, obj: Any, lease_id: int = 0):
        return await self.put(key, json.dumps(obj).encode(), lease_id)

    async def get(self, key: str) -> Optional[bytes]:
        return await self.conn.call("get", key=key)

    async def get_json(self, key: str) -> Optional[Any]:
        v = await self.get(key)
        return None if v is None else json.loads(v)

    async def delete(self, key: str) -> bool:
        return await self.conn.call("delete", key=key)

    async def range(self, prefix: str) -> List[Tuple[str, bytes]]:
        return [tuple(kv) for kv in await self.conn.call("range", prefix=prefix)]

    async def create_if_absent(self, key: str, value: bytes,
                               lease_id: int = 0) -> bool:
        return await self.conn.call("create_if_absent", key=key, value=value,
                                    lease_id=lease_id)

    # ---- leases ----
    async def grant_lease(self, ttl: float) -> int:
        return await self.conn.call("grant_lease", ttl=ttl)

    async def keepalive(self, lease_id: int) -> bool:
        return await self.conn.call("keepalive", lease_id=lease_id)

    async def revoke_lease(self, lease_id: int):
        return await self.conn.call("revoke_lease", lease_id=lease_id)

    # ---- watches ----
    async def watch(self, prefix: str,
                    callback: Callable[[WatchEvent], Any]) -> int:
        self._watch_cbs.append((prefix, callback))
        return await self.conn.call("watch", prefix=prefix)

    async def close(self):
        if self.conn:
            await self.conn.close()


def main():
    """Standalone registry daemon: python -m xllm_service_amd.registry.server
    (multi-replica deployments keep the registry outside any master)."""
    import argparse
    import logging
    ap = argparse.ArgumentParser(description="xllm-service-amd registry")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=12379)
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)

    async def run():
        svc = RegistryService(args.host, args.port)
        port = await svc.start()
        logging.info("registry listening on %s:%d", args.host, port)
        await asyncio.Event().wait()

    asyncio.run(run())


if __name__ == "__main__":
    main()
