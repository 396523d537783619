"""Registry tests: KVStore semantics + the TCP server/client + watches."""
import asyncio

import pytest

from xllm_service_amd.registry.server import RegistryClient, RegistryService
from xllm_service_amd.registry.store import KVStore


@pytest.fixture
def anyio_backend():
    return "asyncio"


def test_kvstore_basics():
    now = [0.0]
    kv = KVStore(clock=lambda: now[0])
    kv.put("a/1", b"x")
    assert kv.get("a/1") == b"x"
    kv.put("a/2", b"y")
    assert kv.range("a/") == [("a/1", b"x"), ("a/2", b"y")]
    assert kv.delete("a/1")
    assert kv.get("a/1") is None


def test_kvstore_lease_expiry():
    now = [0.0]
    kv = KVStore(clock=lambda: now[0])
    lid = kv.grant_lease(3.0)
    kv.put("svc/i1", b"meta", lease_id=lid)
    events = []
    kv.add_watch("svc/", events.append)
    now[0] = 2.0
    kv.keepalive(lid)
    now[0] = 4.0
    assert kv.expire_leases() == 0  # keepalive pushed expiry to 5.0
    now[0] = 5.5
    assert kv.expire_leases() == 1
    assert kv.get("svc/i1") is None
    assert events and events[0].type == "delete"
    assert events[0].prev_value == b"meta"


def test_kvstore_create_if_absent():
    kv = KVStore()
    assert kv.create_if_absent("master", b"a")
    assert not kv.create_if_absent("master", b"b")
    assert kv.get("master") == b"a"
    kv.delete("master")
    assert kv.create_if_absent("master", b"b")


@pytest.mark.anyio
async def test_registry_over_tcp():
    svc = RegistryService()
    port = await svc.start()
    c1 = await RegistryClient().connect("127.0.0.1", port)
    c2 = await RegistryClient().connect("127.0.0.1", port)

    seen = []

    async def on_event(ev):
        seen.append((ev.type, ev.key))

    await c2.watch("XLLM:PREFILL:", on_event)
    lid = await c1.grant_lease(0.5)
    await c1.put_json("XLLM:PREFILL:p0", {"name": "p0"}, lease_id=lid)
    assert (await c2.get_json("XLLM:PREFILL:p0"))["name"] == "p0"
    await asyncio.sleep(0.3)
    assert ("put", "XLLM:PREFILL:p0") in seen
    # let the lease expire (no keepalive)
    await asyncio.sleep(1.0)
    assert await c2.get("XLLM:PREFILL:p0") is None
    assert ("delete", "XLLM:PREFILL:p0") in seen
    # election txn
    assert await c1.create_if_absent("XLLM:SERVICE:MASTER", b"m1")
    assert not await c2.create_if_absent("XLLM:SERVICE:MASTER", b"m2")
    await c1.close()
    await c2.close()
    await svc.stop()
