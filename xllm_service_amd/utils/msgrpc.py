"""Minimal asyncio msgpack-RPC: the framework under the service's RPC plane.

Replaces the reference's brpc (protobuf-over-HTTP) with a compact
length-prefixed msgpack protocol (reference role: SURVEY.md 2.10 "RPC
framework"). One protocol serves the registry, the master<->worker RPC
plane, and tests.

Wire format: 4-byte big-endian length, then a msgpack array:
  [0, msgid, method, params]   request
  [1, msgid, error, result]    response (error: None | str)
  [2, method, params]          notification (no reply)

Handlers are objects exposing `rpc_<method>(**params)` coroutines or plain
functions; notifications dispatch to `on_<method>`.
"""
from __future__ import annotations

import asyncio
import itertools
import struct
from typing import Any, Callable, Dict, Optional

import msgpack

_LEN = struct.Struct(">I")
MAX_FRAME = 256 << 20


class RpcError(Exception):
    pass


class Connection:
    def __init__(self, reader: asyncio.StreamReader,
                 writer: asyncio.StreamWriter, handler: Any = None,
                 name: str = ""):
        self.reader = reader
        self.writer = writer
        self.handler = handler
        self.name = name
        self._msgids = itertools.count(1)
        self._pending: Dict[int, asyncio.Future] = {}
        self._task: Optional[asyncio.Task] = None
        self._send_lock = asyncio.Lock()
        self.closed = asyncio.Event()

    def start(self):
        self._task = asyncio.create_task(self._read_loop())
        return self._task

    async def _send(self, obj):
        data = msgpack.packb(obj, use_bin_type=True)
        async with self._send_lock:
            self.writer.write(_LEN.pack(len(data)) + data)
            await self.writer.drain()

    async def call(self, method: str, timeout: float = 30.0, **params) -> Any:
        msgid = next(self._msgids)
        fut = asyncio.get_running_loop().create_future()
        self._pending[msgid] = fut
        try:
            await self._send([0, msgid, method, params])
            return await asyncio.wait_for(fut, timeout)
        finally:
            self._pending.pop(msgid, None)

    async def notify(self, method: str, **params) -> None:
        await self._send([2, method, params])

    async def _read_loop(self):
        try:
            while True:
                hdr = await self.reader.readexactly(4)
                (ln,) = _LEN.unpack(hdr)
                if ln > MAX_FRAME:
                    raise RpcError(f"frame too large: {ln}")
                data = await self.reader.readexactly(ln)
                msg = msgpack.unpackb(data, raw=False, strict_map_key=False)
                kind = msg[0]
                if kind == 0:
                    asyncio.create_task(self._handle_request(msg[1], msg[2],
                                                             msg[3]))
                elif kind == 1:
                    fut = self._pending.get(msg[1])
                    if fut and not fut.done():
                        if msg[2] is not None:
                            fut.set_exception(RpcError(msg[2]))
                        else:
                            fut.set_result(msg[3])
                elif kind == 2:
                    asyncio.create_task(self._handle_notify(msg[1], msg[2]))
        except (asyncio.IncompleteReadError, ConnectionError,
                asyncio.CancelledError):
            pass
        finally:
            self.closed.set()
            for fut in self._pending.values():
                if not fut.done():
                    fut.set_exception(RpcError("connection closed"))
            try:
                self.writer.close()
            except Exception:
                pass

    async def _handle_request(self, msgid, method, params):
        err = result = None
        try:
            fn = getattr(self.handler, f"rpc_{method}", None)
            if fn is None:
                raise RpcError(f"no such method: {method}")
            result = fn(self, **(params or {}))
            if asyncio.iscoroutine(result):
                result = await result
        except Exception as e:  # noqa: BLE001 — error goes back on the wire
            err = f"{type(e).__name__}: {e}"
        try:
            await self._send([1, msgid, err, result])
        except Exception:
            pass

    async def _handle_notify(self, method, params):
        fn = getattr(self.handler, f"on_{method}", None)
        if fn is None:
            return
        try:
            r = fn(self, **(params or {}))
            if asyncio.iscoroutine(r):
                await r
        except Exception:
            pass

    async def close(self):
        try:
            self.writer.close()
            await self.writer.wait_closed()
        except Exception:
            pass


class Server:
    def __init__(self, handler_factory: Callable[[Connection], Any],
                 host: str = "127.0.0.1", port: int = 0):
        self.handler_factory = handler_factory
        self.host = host
        self.port = port
        self._server: Optional[asyncio.AbstractServer] = None
        self.connections: set[Connection] = set()

    async def start(self) -> int:
        self._server = await asyncio.start_server(self._on_conn, self.host,
                                                  self.port)
        self.port = self._server.sockets[0].getsockname()[1]
        return self.port

    async def _on_conn(self, reader, writer):
        conn = Connection(reader, writer)
        conn.handler = self.handler_factory(conn)
        self.connections.add(conn)
        task = conn.start()
        task.add_done_callback(lambda _t: self.connections.discard(conn))

    async def stop(self):
        if self._server:
            self._server.close()
            await self._server.wait_closed()
        for conn in list(self.connections):
            await conn.close()


async def connect(host: str, port: int, handler: Any = None) -> Connection:
    reader, writer = await asyncio.open_connection(host, port)
    conn = Connection(reader, writer, handler)
    conn.start()
    return conn
