"""Master: the cluster control plane binary (reference: master.{h,cpp} +
rpc_service/).

Wires: embedded registry (or an external one) -> InstanceMgr +
GlobalKVCacheMgr + policy -> ServiceScheduler -> RPC plane (worker-facing)
+ OpenAI HTTP front end. Master election via create-if-absent of
XLLM:SERVICE:MASTER with a TTL lease; non-masters watch that key and take
over on DELETE. A 3 s sync task uploads dirty KV-cache index + load metrics
for replica masters.
"""
from __future__ import annotations

import argparse
import asyncio
import logging
import socket
from dataclasses import dataclass
from typing import List, Optional

import uvicorn

from xllm_service_amd.chat_template import JinjaChatTemplate
from xllm_service_amd.registry.server import RegistryClient, RegistryService
from xllm_service_amd.tokenizer import Tokenizer, TokenizerFactory
from xllm_service_amd.utils import msgrpc

from .http_api import build_app
from .instance_mgr import InstanceMgr
from .kvcache_mgr import GlobalKVCacheMgr
from .policies import create_policy
from .response_handler import ResponseHandler
from .scheduler import ServiceScheduler
from .tracer import RequestTracer
from .types import KEY_MASTER, KEY_SERVICE

log = logging.getLogger("xllm.master")


@dataclass
class MasterOptions:
    """Service options (reference: common/options.h, 27 PROPERTYs +
    global_gflags.cpp, 30 flags — the ones that survive the re-design)."""
    http_host: str = "127.0.0.1"
    http_port: int = 8080
    rpc_host: str = "127.0.0.1"
    rpc_port: int = 0
    registry_host: str = "127.0.0.1"
    registry_port: int = 0            # 0 = ephemeral
    host_registry: bool = True        # embed the registry in this process
    model_id: str = "llama-3-8b"
    model_dir: Optional[str] = None   # tokenizer/chat-template source
    tokenizer_vocab: int = 512        # ByteTokenizer fallback vocab
    load_balance_policy: str = "CAR"  # RR | CAR | SLO_AWARE
    target_ttft_ms: float = 1000.0    # runtime-reloadable SLO knobs
    target_tpot_ms: float = 50.0
    block_size: int = 16
    enable_request_trace: bool = False
    trace_path: str = "trace/trace.jsonl"
    heartbeat_sync_s: float = 3.0
    service_lease_ttl_s: float = 3.0
    instance_probe_timeout_s: float = 1.0
    instance_probe_attempts: int = 2
    lease_lost_heartbeat_timeout_s: float = 3.0
    suspect_eviction_s: float = 15.0
    parser_mode: str = "auto"
    chat_template: Optional[str] = None


class _MasterRpcHandler:
    """Worker-facing RPC (reference proto XllmRpcService: Hello,
    Heartbeat, Generations, GetInstanceInfo, GetStatic*List)."""

    def __init__(self, master: "Master"):
        self.master = master

    def rpc_hello(self, conn, name: str = "") -> str:
        return "hello"

    def rpc_heartbeat(self, conn, name: str, incarnation: int,
                      load: dict = None, latency: dict = None,
                      kv_stored: List[bytes] = None,
                      kv_removed: List[bytes] = None) -> bool:
        m = self.master
        ok = m.instance_mgr.record_heartbeat(name, incarnation, load, latency)
        if ok and (kv_stored or kv_removed):
            m.kv_mgr.record_updated_kvcaches(name, kv_stored or [],
                                             kv_removed or [])
        return ok

    async def on_generations(self, conn, gens: List[dict]):
        for gen in gens:
            ok = await self.master.scheduler.handle_generation(gen)
            if not ok:
                # request gone/cancelled: tell the pushing instance to stop
                rid = gen.get("service_request_id")
                try:
                    await conn.notify("abort_request",
                                      service_request_id=rid)
                except Exception:
                    pass

    def rpc_get_instance_info(self, conn, name: str) -> Optional[dict]:
        inst = self.master.instance_mgr.get(name)
        return inst.meta.to_dict() if inst else None

    def rpc_get_static_prefill_list(self, conn) -> List[str]:
        return [i.name for i in self.master.instance_mgr.schedulable_prefills()]

    def rpc_get_static_decode_list(self, conn) -> List[str]:
        return [i.name for i in self.master.instance_mgr.schedulable_decodes()]


class Master:
    def __init__(self, opts: MasterOptions):
        self.opts = opts
        self.model_id = opts.model_id
        self.is_master = False
        self.registry_service: Optional[RegistryService] = None
        self.registry: Optional[RegistryClient] = None
        self.instance_mgr: Optional[InstanceMgr] = None
        self.kv_mgr: Optional[GlobalKVCacheMgr] = None
        self.scheduler: Optional[ServiceScheduler] = None
        self.response_handler: Optional[ResponseHandler] = None
        self.tokenizer: Optional[Tokenizer] = None
        self.tracer = RequestTracer(opts.enable_request_trace, opts.trace_path)
        self.rpc_server: Optional[msgrpc.Server] = None
        self._tasks: List[asyncio.Task] = []
        self._uvicorn: Optional[uvicorn.Server] = None
        self._lease_id: Optional[int] = None

    def served_models(self) -> List[str]:
        models = {self.model_id}
        for inst in self.instance_mgr.instances.values():
            if inst.meta.model:
                models.add(inst.meta.model)
        return sorted(models)

    # ---- lifecycle ----------------------------------------------------------
    async def start(self, serve_http: bool = True):
        opts = self.opts
        if opts.host_registry:
            self.registry_service = RegistryService(opts.registry_host,
                                                    opts.registry_port)
            opts.registry_port = await self.registry_service.start()
        self.registry = await RegistryClient().connect(opts.registry_host,
                                                       opts.registry_port)

        # worker-facing RPC plane
        handler = _MasterRpcHandler(self)
        self.rpc_server = msgrpc.Server(lambda conn: handler, opts.rpc_host,
                                        opts.rpc_port)
        opts.rpc_port = await self.rpc_server.start()

        # register this service + elect master
        self._lease_id = await self.registry.grant_lease(
            opts.service_lease_ttl_s)
        my_info = {"rpc_host": opts.rpc_host, "rpc_port": opts.rpc_port,
                   "http_port": opts.http_port}
        await self.registry.put_json(
            KEY_SERVICE + f"{opts.rpc_host}:{opts.rpc_port}", my_info,
            lease_id=self._lease_id)
        self.is_master = await self.registry.create_if_absent(
            KEY_MASTER, __import__("json").dumps(my_info).encode(),
            lease_id=self._lease_id)
        if not self.is_master:
            await self.registry.watch(KEY_MASTER, self._on_master_key_event)

        # text processing
        self.tokenizer = TokenizerFactory.create(opts.model_dir,
                                                 opts.tokenizer_vocab)
        chat_template = JinjaChatTemplate(opts.chat_template, opts.model_dir)

        # managers + policy + scheduler
        self.kv_mgr = GlobalKVCacheMgr(self.registry, opts.block_size,
                                       is_master=lambda: self.is_master)
        await self.kv_mgr.start()
        self.instance_mgr = InstanceMgr(
            self.registry,
            on_instance_failed=self._on_instance_failed,
            probe_timeout_s=opts.instance_probe_timeout_s,
            probe_attempts=opts.instance_probe_attempts,
            lease_lost_heartbeat_timeout_s=opts.lease_lost_heartbeat_timeout_s,
            suspect_eviction_s=opts.suspect_eviction_s,
            is_master=lambda: self.is_master)
        await self.instance_mgr.start()
        policy_kwargs = {}
        if opts.load_balance_policy.upper() == "SLO_AWARE":
            policy_kwargs = dict(target_ttft_ms=opts.target_ttft_ms,
                                 target_tpot_ms=opts.target_tpot_ms)
        policy = create_policy(opts.load_balance_policy, self.instance_mgr,
                               self.kv_mgr, **policy_kwargs)
        if hasattr(policy, "seed_from_meta"):
            self.instance_mgr.profile_seed_cb = policy.seed_from_meta
        self.scheduler = ServiceScheduler(self.instance_mgr, self.kv_mgr,
                                          policy, self.tokenizer,
                                          chat_template, self.tracer)
        self.response_handler = ResponseHandler(self.tokenizer,
                                                opts.parser_mode)

        self._tasks = [
            asyncio.create_task(self._keepalive_loop()),
            asyncio.create_task(self._sync_loop()),
        ]
        if serve_http:
            self._tasks.append(asyncio.create_task(self._serve_http()))
        log.info("master up: http=%s:%d rpc=%d registry=%d master=%s",
                 opts.http_host, opts.http_port, opts.rpc_port,
                 opts.registry_port, self.is_master)

    async def stop(self):
        for t in self._tasks:
            t.cancel()
        if self.registry and self._lease_id:
            try:  # release the service key + master election key promptly
                await self.registry.revoke_lease(self._lease_id)
            except Exception:
                pass
        if self._uvicorn:
            self._uvicorn.should_exit = True
        if self.instance_mgr:
            await self.instance_mgr.stop()
        if self.rpc_server:
            await self.rpc_server.stop()
        if self.registry:
            await self.registry.close()
        if self.registry_service:
            await self.registry_service.stop()
        self.tracer.close()

    async def _serve_http(self):
        app = build_app(self)
        config = uvicorn.Config(app, host=self.opts.http_host,
                                port=self.opts.http_port, log_level="warning",
                                access_log=False)
        self._uvicorn = uvicorn.Server(config)
        await self._uvicorn.serve()

    # ---- election -----------------------------------------------------------
    async def _on_master_key_event(self, ev):
        if ev.type == "delete" and not self.is_master:
            my_info = {"rpc_host": self.opts.rpc_host,
                       "rpc_port": self.opts.rpc_port,
                       "http_port": self.opts.http_port}
            won = await self.registry.create_if_absent(
                KEY_MASTER, __import__("json").dumps(my_info).encode(),
                lease_id=self._lease_id)
            if won:
                self.is_master = True
                log.warning("took over as master")

    # ---- background ---------------------------------------------------------
    async def _keepalive_loop(self):
        while True:
            await asyncio.sleep(self.opts.service_lease_ttl_s / 3)
            try:
                await self.registry.keepalive(self._lease_id)
            except Exception:
                pass

    async def _sync_loop(self):
        """Master-only 3 s upload of dirty cluster state for replicas."""
        while True:
            await asyncio.sleep(self.opts.heartbeat_sync_s)
            if not self.is_master:
                continue
            try:
                await self.kv_mgr.upload_kvcache()
                await self.instance_mgr.upload_load_metrics()
            except Exception:
                log.exception("state sync failed")

    async def _on_instance_failed(self, name: str, incarnation: int):
        await self.scheduler.clear_requests_on_failed_instance(name,
                                                               incarnation)


def main():
    ap = argparse.ArgumentParser(description="xllm-service-amd master")
    ap.add_argument("--http-host", default="0.0.0.0")
    ap.add_argument("--http-port", type=int, default=8080)
    ap.add_argument("--rpc-port", type=int, default=18080)
    ap.add_argument("--registry-port", type=int, default=12379)
    ap.add_argument("--model-id", default="llama-3-8b")
    ap.add_argument("--model-dir", default=None)
    ap.add_argument("--policy", default="CAR",
                    choices=["RR", "CAR", "SLO_AWARE"])
    ap.add_argument("--target-ttft-ms", type=float, default=1000.0)
    ap.add_argument("--target-tpot-ms", type=float, default=50.0)
    ap.add_argument("--enable-request-trace", action="store_true")
    args = ap.parse_args()

    logging.basicConfig(level=logging.INFO)
    opts = MasterOptions(
        http_host=args.http_host, http_port=args.http_port,
        rpc_host=socket.gethostbyname(socket.gethostname())
        if args.http_host == "0.0.0.0" else args.http_host,
        rpc_port=args.rpc_port, registry_port=args.registry_port,
        model_id=args.model_id, model_dir=args.model_dir,
        load_balance_policy=args.policy,
        target_ttft_ms=args.target_ttft_ms,
        target_tpot_ms=args.target_tpot_ms,
        enable_request_trace=args.enable_request_trace)

    async def run():
        master = Master(opts)
        await master.start()
        try:
            await asyncio.Event().wait()
        finally:
            await master.stop()

    asyncio.run(run())


if __name__ == "__main__":
    main()
