"""Worker process: one engine per GPU, speaking the instance-side contract.

Implements everything SURVEY.md §2.9 requires of an instance:
  1. serves execute_request (the XllmAPIService Completions equivalent)
  2. serves link_instance / unlink_instance with peer cluster info
  3. registers itself in the registry under XLLM:<TYPE>:<name> with a TTL
     lease + incarnation id
  4. heartbeats the master every ~3 s with LoadMetrics + KvCacheEvents +
     LatencyMetrics
  5. pushes generated tokens to the master ("generations", batched), with
     finished_on_prefill marking the TTFT token
  6. serves health probes
plus PD-disaggregation: prefill role computes the prompt + first token,
holds the KV blocks, migrates them to the decode peer (xGMI P2P on GPU via
kv_migration.py; serialized-bytes RPC transport elsewhere), then releases.

Threading model: asyncio loop for RPC/registry; the engine runs on its own
thread, fed through a command queue; outputs hop back via a thread-safe
queue drained by the push task.
"""
from __future__ import annotations

import asyncio
import concurrent.futures
import logging
import os
import queue
import threading
import time
from typing import Any, Dict, List, Optional

import torch

from xllm_service_amd.registry.server import RegistryClient
from xllm_service_amd.service.types import (KEY_INSTANCE, KEY_MASTER,
                                            InstanceMetaInfo, InstanceType)
from xllm_service_amd.utils import msgrpc

from .engine import LLMEngine
from .sampling import SamplingParams


def _image_grids(cfg, images):
    """LM-token grid (t, h, w) per image after the 2x2 spatial merge —
    the spans M-RoPE assigns 3-D position ids over."""
    m = int(cfg.vision.get("spatial_merge_size", 2)) if cfg.vision else 2
    return [[1, int(img["grid_h"]) // m, int(img["grid_w"]) // m]
            for img in images]


class VisionEncoder:
    """Stage-E vision tower runner (ENCODE instances; also used in-process
    by colocated DEFAULT instances serving multimodal models)."""

    def __init__(self, model_name: str, device: Optional[str], seed: int = 0):
        from xllm_service_amd.models.config import get_config
        from xllm_service_amd.models.qwen2_vl import Qwen2VisionTransformer
        self.cfg = get_config(model_name)
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self.tower = Qwen2VisionTransformer(self.cfg, dtype).to(self.device)
        self.tower.random_init(seed)
        self.tower.eval()

    @torch.inference_mode()
    def encode(self, images: List[Dict[str, Any]]) -> torch.Tensor:
        """images: [{grid_h, grid_w, seed}] with synthetic deterministic
        pixels (offline environment); returns [sum tokens, hidden]."""
        outs = []
        for img in images:
            gh, gw = int(img["grid_h"]), int(img["grid_w"])
            gen = torch.Generator().manual_seed(int(img.get("seed", 0)))
            patches = torch.randn(gh * gw, self.tower.patch_dim,
                                  generator=gen) * 0.5
            patches = patches.to(self.device, next(
                self.tower.parameters()).dtype)
            outs.append(self.tower(patches, gh, gw))
        return torch.cat(outs, dim=0)

log = logging.getLogger("xllm.worker")


def params_from_dict(d: Dict[str, Any]) -> SamplingParams:
    d = d or {}
    return SamplingParams(
        temperature=float(d.get("temperature", 0.0) or 0.0),
        top_p=float(d.get("top_p", 1.0) or 1.0),
        top_k=int(d.get("top_k", -1) or -1),
        max_tokens=int(d.get("max_tokens", 128) or 128),
        min_tokens=int(d.get("min_tokens", 0) or 0),
        stop_token_ids=list(d.get("stop_token_ids") or []),
        stop_sequences=[list(x) for x in (d.get("stop_sequences") or [])],
        ignore_eos=bool(d.get("ignore_eos", False)),
        seed=d.get("seed"),
        logprobs=d.get("logprobs"),
    )


class Worker:
    def __init__(self, name: str, itype: str = "DEFAULT",
                 model: str = "llama-tiny", device: Optional[str] = None,
                 registry_host: str = "127.0.0.1", registry_port: int = 0,
                 rpc_host: str = "127.0.0.1", rpc_port: int = 0,
                 heartbeat_s: float = 1.0, lease_ttl_s: float = 3.0,
                 eos_token_id: Optional[int] = None,
                 max_kv_blocks: Optional[int] = None,
                 relay_responses: bool = False,
                 push_interval_ms: float = 0.0,
                 engine_kwargs: Optional[Dict[str, Any]] = None):
        self.name = name
        self.itype = InstanceType(itype)
        self.model = model
        self.device = device
        self.registry_addr = (registry_host, registry_port)
        self.rpc_host = rpc_host
        self.rpc_port = rpc_port
        self.heartbeat_s = heartbeat_s
        self.lease_ttl_s = lease_ttl_s
        self.eos_token_id = eos_token_id
        self.max_kv_blocks = max_kv_blocks
        # decode→prefill→service response relay (reference's second
        # response topology): a DECODE instance routes its generations
        # through the prefill peer that migrated the request in
        self.relay_responses = relay_responses
        # push coalescing: hold non-urgent token pushes for up to this long
        # so the master sees one batched Generations per window instead of
        # one per engine step (the reference pushes batched
        # DisaggStreamGenerations too). First tokens (TTFT) and finishes
        # always flush immediately.
        self.push_interval_ms = push_interval_ms
        self.engine_kwargs = engine_kwargs or {}
        self.incarnation = int(time.time() * 1000)

        self.engine: Optional[LLMEngine] = None
        self.registry: Optional[RegistryClient] = None
        self.master_conn: Optional[msgrpc.Connection] = None
        self.rpc_server: Optional[msgrpc.Server] = None
        self.peers: Dict[str, InstanceMetaInfo] = {}
        self.peer_conns: Dict[str, msgrpc.Connection] = {}
        self.mig = None  # decode-side xGMI MigrationManager (lazy)
        self.encoder: Optional[VisionEncoder] = None  # ENCODE stage / VL
        # requests this PREFILL instance must migrate after the first token:
        # rid -> dict(routing/params)
        self.pending_migration: Dict[str, Dict[str, Any]] = {}
        # remember request params for usage accounting
        self.req_meta: Dict[str, Dict[str, Any]] = {}
        # TTFT/TPOT profiling samples measured from real steps, shipped in
        # InstanceMetaInfo so the master's SLO predictors start seeded
        # (reference: profiling samples in common/types.h registration)
        self.prof_ttft: List[List[float]] = []   # [num_tokens, ms]
        self.prof_tpot: List[List[float]] = []   # [batch, tokens, ms]
        self._prof_shipped = False

        self._cmd_q: "queue.Queue" = queue.Queue()
        self._out_q: "queue.Queue" = queue.Queue()
        self._stop = threading.Event()
        self._engine_thread: Optional[threading.Thread] = None
        self._tasks: List[asyncio.Task] = []
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._lease_id: Optional[int] = None
        self._ttft_samples: List[float] = []
        self._tbt_samples: List[float] = []

    # ------------------------------------------------------------------ setup
    async def start(self):
        self._loop = asyncio.get_running_loop()
        if self.itype == InstanceType.ENCODE:
            # stage-E instance: vision tower only, no LM engine
            self.encoder = await self._loop.run_in_executor(
                None, lambda: VisionEncoder(self.model, self.device))
        else:
            # engine init is slow (weights); do it off-loop
            self.engine = await self._loop.run_in_executor(
                None, self._make_engine)
            self._engine_thread = threading.Thread(target=self._engine_loop,
                                                   daemon=True,
                                                   name=f"engine-{self.name}")
            self._engine_thread.start()

        self.rpc_server = msgrpc.Server(lambda conn: self, self.rpc_host,
                                        self.rpc_port)
        self.rpc_port = await self.rpc_server.start()

        self.registry = await RegistryClient().connect(*self.registry_addr)
        self._lease_id = await self.registry.grant_lease(self.lease_ttl_s)
        await self.registry.put_json(self._regkey(), self.meta().to_dict(),
                                     lease_id=self._lease_id)
        await self._connect_master()
        self._tasks = [
            asyncio.create_task(self._keepalive_loop()),
            asyncio.create_task(self._heartbeat_loop()),
            asyncio.create_task(self._push_loop()),
        ]
        log.info("worker %s (%s) up: rpc=%s:%d", self.name, self.itype.value,
                 self.rpc_host, self.rpc_port)

    def _make_engine(self) -> LLMEngine:
        eng = LLMEngine(self.model, device=self.device,
                        max_kv_blocks=self.max_kv_blocks,
                        **self.engine_kwargs)
        eng.eos_token_id = self.eos_token_id
        return eng

    def _regkey(self) -> str:
        return KEY_INSTANCE[self.itype] + self.name

    def meta(self) -> InstanceMetaInfo:
        dev = -1
        if self.engine is not None and self.engine.device.type == "cuda":
            dev = self.engine.device.index or 0
        return InstanceMetaInfo(
            name=self.name, itype=self.itype.value,
            rpc_host=self.rpc_host, rpc_port=self.rpc_port,
            device_index=dev,
            cluster_ids=[dev] if dev >= 0 else [],
            num_kv_blocks=self.engine.block_manager.num_blocks
            if self.engine else 0,
            dp_size=1,
            block_size=16, model=self.model,
            incarnation_id=self.incarnation,
            k_cache_ids=list(range(self.engine.cfg.num_layers))
            if self.engine else [],
            v_cache_ids=list(range(self.engine.cfg.num_layers))
            if self.engine else [],
            ttft_profile=list(self.prof_ttft),
            tpot_profile=list(self.prof_tpot),
        )

    async def _connect_master(self):
        async def on_master_change(ev):
            if ev.type == "put":
                await self._dial_master()

        await self.registry.watch(KEY_MASTER, on_master_change)
        await self._dial_master()

    async def _dial_master(self):
        info = await self.registry.get_json(KEY_MASTER)
        if not info:
            return
        host, port = info["rpc_host"], info["rpc_port"]
        if self.master_conn and not self.master_conn.closed.is_set():
            return
        try:
            self.master_conn = await msgrpc.connect(host, port, handler=self)
            await self.master_conn.call("hello", name=self.name)
        except OSError as e:
            log.warning("worker %s: master dial failed: %s", self.name, e)

    async def stop(self):
        self._stop.set()
        for t in self._tasks:
            t.cancel()
        if self._engine_thread:
            self._engine_thread.join(timeout=5)
        if self.rpc_server:
            await self.rpc_server.stop()
        for c in self.peer_conns.values():
            await c.close()
        if self.master_conn:
            await self.master_conn.close()
        if self.registry:
            if self._lease_id:
                try:
                    await self.registry.revoke_lease(self._lease_id)
                except Exception:
                    pass
            await self.registry.close()

    # ---------------------------------------------------------- engine thread
    def _engine_loop(self):
        while not self._stop.is_set():
            did = False
            try:
                while True:
                    fn, fut = self._cmd_q.get_nowait()
                    try:
                        res = fn()
                        if fut:
                            fut.set_result(res)
                    except Exception as e:  # noqa: BLE001
                        if fut:
                            fut.set_exception(e)
                        else:
                            log.exception("engine cmd failed")
                    did = True
            except queue.Empty:
                pass
            if self.engine.has_work():
                t_step = time.monotonic()
                try:
                    outs = self.engine.step()
                except Exception:
                    # a poisoned batch must not kill the worker: drop the
                    # running sequences with an abort so the master errors
                    # them out instead of hanging the clients
                    log.exception("engine step failed; aborting running seqs")
                    import traceback
                    err = traceback.format_exc().strip().splitlines()[-1]
                    from .engine import StepOutput
                    bad = [s.request_id
                           for s in list(self.engine.scheduler.running)
                           ] + [s.request_id
                                for s in list(self.engine.scheduler.waiting)]
                    outs = []
                    for rid in bad:
                        self.engine.abort_request(rid)
                        outs.append(StepOutput(
                            request_id=rid, new_token_ids=[], finished=True,
                            finish_reason="abort",
                            error=f"engine step failed: {err}"))
                else:
                    ms = (time.monotonic() - t_step) * 1000.0
                    st = self.engine.stats
                    if st.last_prefill_tokens > 0 and len(self.prof_ttft) < 64:
                        self.prof_ttft.append(
                            [float(st.last_prefill_tokens), ms])
                    elif (st.last_prefill_tokens == 0 and st.last_decodes > 0
                          and len(self.prof_tpot) < 64):
                        self.prof_tpot.append(
                            [float(st.last_decodes),
                             float(st.last_decodes), ms])
                if outs:
                    self._out_q.put(outs)
                did = True
            if not did:
                time.sleep(0.002)

    async def _run_on_engine(self, fn):
        fut = concurrent.futures.Future()
        self._cmd_q.put((fn, fut))
        return await asyncio.wrap_future(fut)

    def _post_to_engine(self, fn):
        self._cmd_q.put((fn, None))

    # --------------------------------------------------------------- RPC API
    def rpc_health(self, conn) -> bool:
        return True

    def rpc_get_info(self, conn) -> dict:
        return self.meta().to_dict()

    async def rpc_link_instance(self, conn, peer: dict) -> bool:
        meta = InstanceMetaInfo.from_dict(peer)
        self.peers[meta.name] = meta
        log.info("worker %s: linked peer %s (%s)", self.name, meta.name,
                 meta.itype)
        return True

    async def rpc_unlink_instance(self, conn, peer_name: str) -> bool:
        self.peers.pop(peer_name, None)
        c = self.peer_conns.pop(peer_name, None)
        if c:
            await c.close()
        if self.mig is not None:
            await self._run_on_engine(
                lambda: self.mig.close_peer(peer_name))
        return True

    async def rpc_export_cache(self, conn) -> dict:
        """Prefill-side: IPC handles of the KV cache for xGMI migration."""
        from .kv_migration import export_cache_handles
        return await self._run_on_engine(
            lambda: export_cache_handles(self.engine))

    async def _peer_conn(self, name: str) -> Optional[msgrpc.Connection]:
        c = self.peer_conns.get(name)
        if c and not c.closed.is_set():
            return c
        meta = self.peers.get(name)
        if meta is None:
            return None
        try:
            c = await msgrpc.connect(meta.rpc_host, meta.rpc_port)
            self.peer_conns[name] = c
            return c
        except OSError:
            return None

    # execute_request arrives as a notification (fire-and-forget dispatch)
    async def on_execute_request(self, conn, service_request_id: str,
                                 token_ids: List[int], params: dict,
                                 routing: dict, offline: bool = False,
                                 multimodal: Optional[dict] = None):
        routing = routing or {}
        # ---- stage E: run the vision tower, then hand off to prefill ------
        if self.itype == InstanceType.ENCODE:
            await self._encode_and_forward(service_request_id, token_ids,
                                           params, routing, offline,
                                           multimodal)
            return
        sp = params_from_dict(params)
        if self.engine is not None and (
                not token_ids
                or len(token_ids) >= self.engine.max_model_len):
            if self.master_conn:
                try:
                    await self.master_conn.notify("generations", gens=[dict(
                        service_request_id=service_request_id, token_ids=[],
                        finished=True, finish_reason="abort",
                        error=("empty prompt" if not token_ids else
                               f"prompt length {len(token_ids)} exceeds "
                               f"max_model_len {self.engine.max_model_len}"))])
                except Exception:
                    pass
            return
        self.req_meta[service_request_id] = dict(
            params=params, routing=routing, prompt_len=len(token_ids),
            offline=offline, multimodal=multimodal)
        mm_embeds = None
        mm_grids = None
        if multimodal:
            if multimodal.get("embeds_b") is not None:
                import numpy as np
                raw = np.frombuffer(multimodal["embeds_b"], dtype=np.uint8)
                mm_embeds = torch.from_numpy(raw.copy()).view(
                    torch.bfloat16 if multimodal.get("dtype") == "bfloat16"
                    else torch.float32).reshape(multimodal["embeds_shape"])
                mm_grids = multimodal.get("grids")
            elif multimodal.get("images"):
                # colocated multimodal: run the vision tower in-process
                if self.encoder is None:
                    self.encoder = await self._loop.run_in_executor(
                        None, lambda: VisionEncoder(self.model, self.device))
                mm_embeds = await self._loop.run_in_executor(
                    None, lambda: self.encoder.encode(multimodal["images"]))
                mm_grids = _image_grids(self.encoder.cfg, multimodal["images"])
        decode_name = routing.get("decode_name")
        do_migrate = (self.itype == InstanceType.PREFILL
                      and decode_name and decode_name != self.name)
        if do_migrate:
            # prefill role: produce exactly the first token, hold blocks
            first_sp = params_from_dict(params)
            first_sp.max_tokens = 1
            first_sp.ignore_eos = True
            self.pending_migration[service_request_id] = dict(
                routing=routing, params=params, token_ids=list(token_ids),
                offline=offline)
            self._post_to_engine(
                lambda: self.engine.add_request(
                    service_request_id, token_ids, first_sp,
                    priority=1 if offline else 0, hold_blocks=True,
                    mm_embeds=mm_embeds, mm_grids=mm_grids))
        else:
            self._post_to_engine(
                lambda: self.engine.add_request(
                    service_request_id, token_ids, sp,
                    priority=1 if offline else 0, mm_embeds=mm_embeds,
                    mm_grids=mm_grids))

    async def _encode_and_forward(self, rid, token_ids, params, routing,
                                  offline, multimodal):
        target = routing.get("prefill_name") or routing.get("decode_name")
        try:
            embeds = await self._loop.run_in_executor(
                None, lambda: self.encoder.encode(
                    (multimodal or {}).get("images", [])))
            conn = await self._peer_conn(target)
            if conn is None:
                raise RuntimeError(f"prefill peer {target} unreachable")
            payload_mm = dict(
                embeds_b=embeds.cpu().view(torch.uint8).numpy().tobytes()
                if embeds.dtype != torch.float32
                else embeds.cpu().numpy().tobytes(),
                dtype=str(embeds.dtype).split(".")[-1],
                embeds_shape=list(embeds.shape),
                grids=_image_grids(self.encoder.cfg,
                                   (multimodal or {}).get("images", [])))
            await conn.notify("execute_request",
                              service_request_id=rid, token_ids=token_ids,
                              params=params, routing=routing,
                              offline=offline, multimodal=payload_mm)
        except Exception as e:
            log.warning("encode stage failed for %s: %s", rid, e)
            if self.master_conn:
                try:
                    await self.master_conn.notify("generations", gens=[dict(
                        service_request_id=rid, token_ids=[], finished=True,
                        finish_reason="abort",
                        error=f"vision encode failed: {e}")])
                except Exception:
                    pass

    async def on_relay_generations(self, conn, gens):
        """decode→prefill→service relay hop: forward a decode peer's
        generations to the master over this instance's master link."""
        if self.master_conn and not self.master_conn.closed.is_set():
            try:
                await self.master_conn.notify("generations", gens=gens)
                return
            except Exception:
                pass
        log.warning("worker %s: relay forward to master failed", self.name)

    async def on_role_change(self, conn, new_type: str):
        """SLO-aware adaptive P:D flip (master -> worker): adopt the new
        role for FUTURE requests (in-flight old-role work drains
        naturally), re-partition the KV pool — a decode role reclaims the
        prefix-cache reserve for long-lived decode KV — and refresh the
        registration meta so GetInstanceInfo reflects the new type."""
        old = self.itype
        try:
            self.itype = InstanceType(new_type)
        except ValueError:
            return
        if self.engine is not None and self.itype == InstanceType.DECODE:
            released = await self._run_on_engine(
                lambda: self.engine.block_manager.reset_prefix_cache())
            log.info("worker %s: role %s -> %s, released %d cached blocks",
                     self.name, old.value, new_type, released)
        else:
            log.info("worker %s: role %s -> %s", self.name, old.value,
                     new_type)
        # same key + incarnation: the master treats this as a meta refresh
        if self.registry is not None:
            try:
                await self.registry.put_json(self._regkey(),
                                             self.meta().to_dict(),
                                             lease_id=self._lease_id)
            except Exception:
                pass

    def on_abort_request(self, conn, service_request_id: str):
        self.pending_migration.pop(service_request_id, None)
        self._post_to_engine(
            lambda: self.engine.abort_request(service_request_id))

    # decode side of a migration (called by the prefill worker)
    async def rpc_migrate_in(self, conn, service_request_id: str,
                             prompt_token_ids: List[int],
                             first_token_ids: List[int], params: dict,
                             n_blocks: int, transport: str,
                             data: Optional[bytes] = None,
                             src_name: Optional[str] = None,
                             src_blocks: Optional[List[int]] = None,
                             offline: bool = False,
                             mrope_delta: int = 0) -> bool:
        sp = params_from_dict(params)
        # account the token(s) the prefill already produced
        sp.max_tokens = max(sp.max_tokens, 1)
        meta = self.req_meta.setdefault(service_request_id, {})
        meta.update(prompt_len=len(prompt_token_ids), params=params,
                    relay_via=src_name)

        if transport == "xgmi":
            # open the peer's cache over IPC once, then pull over xGMI
            from .kv_migration import MigrationManager
            if self.mig is None:
                self.mig = MigrationManager(self.engine)
            if not self.mig.has_peer(src_name):
                pconn = await self._peer_conn(src_name)
                if pconn is None:
                    raise RuntimeError(f"peer {src_name} unreachable")
                exported = await pconn.call("export_cache", timeout=30.0)
                await self._run_on_engine(
                    lambda: self.mig.open_peer(src_name, exported))

        # allocate destination blocks on the engine thread (block manager
        # is engine-thread state), then pull OFF the engine thread so
        # decode steps keep running during the copy (SURVEY hard-part 2)
        blocks = await self._run_on_engine(
            lambda: self.engine.alloc_migration_blocks(n_blocks))
        event = None
        try:
            if transport == "bytes":
                await self._loop.run_in_executor(
                    None, lambda: self.engine.import_block_bytes(blocks,
                                                                 data))
            elif transport == "xgmi":
                # copies fly on the migration side stream; the engine
                # activates the sequence when the event fires
                event = self.mig.pull_blocks_async(src_name, src_blocks,
                                                   blocks)
            else:
                raise ValueError(f"unknown transport {transport}")
        except Exception:
            self._post_to_engine(lambda: self.engine.free_blocks(blocks))
            raise
        self._post_to_engine(
            lambda: self.engine.enqueue_migrated_request(
                service_request_id, prompt_token_ids, first_token_ids,
                blocks, sp, priority=1 if offline else 0,
                mrope_delta=mrope_delta, event=event))
        if event is not None:
            # don't answer the prefill peer until the copy has landed: it
            # releases its held source blocks when this RPC returns
            await self._loop.run_in_executor(None, event.synchronize)
        return True

    # ------------------------------------------------------------- push loop
    async def _push_loop(self):
        """Drain engine outputs and push batched Generations to the master.

        With push_interval_ms > 0, plain decode tokens coalesce per request
        for up to one interval (one notify carries many tokens per request),
        which keeps the master's per-token work O(tokens/interval) instead
        of O(engine steps x batch). First tokens (TTFT), finishes and
        migrations always flush immediately."""
        pending: Dict[str, dict] = {}
        urgent = False
        interval = self.push_interval_ms / 1000.0
        last_flush = time.monotonic()
        get_timeout = 0.1 if interval <= 0 else min(0.1, interval / 2)
        while not self._stop.is_set():
            outs = await self._loop.run_in_executor(
                None, self._out_q_get, get_timeout)
            migrations = []
            batches = [outs] if outs else []
            while True:
                try:
                    batches.append(self._out_q.get_nowait())
                except queue.Empty:
                    break
            for o in (o for b in batches for o in b):
                rid = o.request_id
                # heartbeat LatencyMetrics: per-request TTFT / inter-token
                # gaps measured at emission (recent_max_* in the reference)
                if o.ttft_ms is not None and len(self._ttft_samples) < 4096:
                    self._ttft_samples.append(o.ttft_ms)
                elif o.tbt_ms and len(self._tbt_samples) < 4096:
                    self._tbt_samples.append(o.tbt_ms)
                mig = self.pending_migration.get(rid)
                if mig is not None and o.finished:
                    # first token produced by prefill: announce + migrate
                    pending[rid] = dict(
                        service_request_id=rid,
                        token_ids=o.new_token_ids,
                        finished=False, finished_on_prefill=True,
                        prompt_tokens=o.num_prompt_tokens,
                        completion_tokens=o.num_output_tokens)
                    migrations.append((rid, self.pending_migration.pop(rid),
                                       o.new_token_ids))
                    urgent = True
                    continue
                g = pending.get(rid)
                if g is None:
                    pending[rid] = g = dict(
                        service_request_id=rid, token_ids=[],
                        finished=False, logprobs=None, finish_reason=None,
                        finished_on_prefill=False, prompt_tokens=0,
                        completion_tokens=0)
                    if self.relay_responses:
                        g["_via"] = (self.req_meta.get(rid) or
                                     {}).get("relay_via")
                g["token_ids"] = g["token_ids"] + o.new_token_ids
                if o.logprobs:
                    g["logprobs"] = (g["logprobs"] or []) + o.logprobs
                g["finished"] = o.finished
                g["finish_reason"] = o.finish_reason
                if getattr(o, "error", None):
                    g["error"] = o.error
                g["finished_on_prefill"] = g["finished_on_prefill"] or (
                    o.first_token and
                    self.itype in (InstanceType.DEFAULT, InstanceType.MIX,
                                   InstanceType.PREFILL))
                g["prompt_tokens"] = o.num_prompt_tokens
                g["completion_tokens"] = o.num_output_tokens
                if o.finished:
                    self.req_meta.pop(rid, None)
                    urgent = True
                if o.first_token:
                    urgent = True
            now = time.monotonic()
            if pending and (urgent or interval <= 0
                            or now - last_flush >= interval):
                gens = list(pending.values())
                pending.clear()
                urgent = False
                last_flush = now
                await self._push_gens(gens)
            for rid, mig, first_toks in migrations:
                asyncio.create_task(self._do_migration(rid, mig, first_toks))

    async def _push_gens(self, gens: List[dict]):
        relayed: Dict[str, list] = {}
        if self.relay_responses:
            direct = []
            for g in gens:
                via = g.pop("_via", None) or (
                    self.req_meta.get(g["service_request_id"]) or
                    {}).get("relay_via")
                if via:
                    relayed.setdefault(via, []).append(g)
                else:
                    direct.append(g)
            gens = direct
        for via, batch in relayed.items():
            conn = await self._peer_conn(via)
            try:
                if conn is None:
                    raise RuntimeError(f"relay peer {via} unreachable")
                await conn.notify("relay_generations", gens=batch)
            except Exception:
                log.warning("worker %s: relay via %s failed, pushing "
                            "direct", self.name, via)
                gens.extend(batch)
        if gens and self.master_conn and not self.master_conn.closed.is_set():
            try:
                await self.master_conn.notify("generations", gens=gens)
            except Exception:
                log.warning("worker %s: generations push failed", self.name)

    def _out_q_get(self, timeout: float = 0.1):
        try:
            return self._out_q.get(timeout=timeout)
        except queue.Empty:
            return None

    async def _do_migration(self, rid: str, mig: dict, first_toks: List[int]):
        decode_name = mig["routing"]["decode_name"]
        conn = await self._peer_conn(decode_name)
        try:
            if conn is None:
                raise RuntimeError(f"decode peer {decode_name} unreachable")
            blocks = await self._run_on_engine(
                lambda: self.engine.held_block_table(rid))
            peer = self.peers.get(decode_name)
            transport = self._pick_transport(peer)
            kwargs: Dict[str, Any] = dict(
                service_request_id=rid,
                prompt_token_ids=mig["token_ids"],
                first_token_ids=first_toks,
                params=mig["params"], n_blocks=len(blocks),
                transport=transport, src_name=self.name,
                offline=mig.get("offline", False),
                mrope_delta=await self._run_on_engine(
                    lambda: self.engine.held_mrope_delta(rid)))
            if transport == "bytes":
                kwargs["data"] = await self._run_on_engine(
                    lambda: self.engine.export_block_bytes(blocks))
            else:
                kwargs["src_blocks"] = blocks
            try:
                await conn.call("migrate_in", timeout=60.0, **kwargs)
            except Exception as e:
                if transport != "xgmi":
                    raise
                # xGMI/IPC path failed on this topology: fall back to the
                # serialized-bytes transport rather than killing the request
                log.warning("xgmi migration of %s failed (%s); retrying "
                            "with bytes transport", rid, e)
                kwargs["transport"] = "bytes"
                kwargs.pop("src_blocks", None)
                kwargs["data"] = await self._run_on_engine(
                    lambda: self.engine.export_block_bytes(blocks))
                await conn.call("migrate_in", timeout=60.0, **kwargs)
        except Exception as e:
            log.warning("migration of %s to %s failed: %s", rid, decode_name, e)
            # tell the master the request died (client will retry)
            if self.master_conn:
                try:
                    await self.master_conn.notify("generations", gens=[dict(
                        service_request_id=rid, token_ids=[], finished=True,
                        finish_reason="abort",
                        error=f"kv migration failed: {e}")])
                except Exception:
                    pass
        finally:
            self._post_to_engine(lambda: self.engine.release_held(rid))

    def _pick_transport(self, peer: Optional[InstanceMetaInfo]) -> str:
        if (peer is not None and peer.device_index >= 0
                and self.engine.device.type == "cuda"):
            try:
                from . import kv_migration
                if kv_migration.available():
                    return "xgmi"
            except Exception:
                pass
        return "bytes"

    # ---------------------------------------------------------- housekeeping
    async def _keepalive_loop(self):
        while not self._stop.is_set():
            await asyncio.sleep(self.lease_ttl_s / 3)
            try:
                ok = await self.registry.keepalive(self._lease_id)
                if not ok:  # lease expired (e.g. long GC pause): re-register
                    self._lease_id = await self.registry.grant_lease(
                        self.lease_ttl_s)
                    await self.registry.put_json(self._regkey(),
                                                 self.meta().to_dict(),
                                                 lease_id=self._lease_id)
            except Exception:
                pass

    async def _heartbeat_loop(self):
        while not self._stop.is_set():
            await asyncio.sleep(self.heartbeat_s)
            if self.master_conn is None or self.master_conn.closed.is_set():
                await self._dial_master()
                if self.master_conn is None or self.master_conn.closed.is_set():
                    continue
            try:
                if self.engine is None:  # ENCODE stage: trivial heartbeat
                    await self.master_conn.call(
                        "heartbeat", timeout=5.0, name=self.name,
                        incarnation=self.incarnation,
                        load=dict(waiting_requests_num=0,
                                  running_requests_num=0,
                                  gpu_cache_usage_perc=0.0),
                        latency={}, kv_stored=[], kv_removed=[])
                    continue
                ev = self.engine.block_manager.events.drain()
                st = self.engine.stats
                await self.master_conn.call(
                    "heartbeat", timeout=5.0,
                    name=self.name, incarnation=self.incarnation,
                    load=dict(
                        waiting_requests_num=st.num_waiting,
                        running_requests_num=st.num_running,
                        gpu_cache_usage_perc=st.kv_usage),
                    latency=dict(
                        recent_max_ttft_ms=max(self._ttft_samples or [0.0]),
                        recent_max_tbt_ms=max(self._tbt_samples or [0.0])),
                    kv_stored=[bytes(h) for h in ev.stored],
                    kv_removed=[bytes(h) for h in ev.removed])
                if log.isEnabledFor(logging.INFO):
                    now_hb = time.monotonic()
                    prev = getattr(self, "_hb_prev", None)
                    self._hb_prev = (now_hb, st.steps, st.generated_tokens)
                    ss = sorted(self._ttft_samples) or [0.0]
                    if prev is not None:
                        dt = now_hb - prev[0]
                        log.info(
                            "engine ttft ms p50=%.0f n=%d waiting=%d "
                            "running=%d kv=%.2f | %.1f steps/s %.0f tok/s",
                            ss[len(ss) // 2], len(self._ttft_samples),
                            st.num_waiting, st.num_running, st.kv_usage,
                            (st.steps - prev[1]) / dt,
                            (st.generated_tokens - prev[2]) / dt)
                self._ttft_samples.clear()
                self._tbt_samples.clear()
                # ship the first batch of profiling samples by re-PUTting
                # the registration meta (same incarnation: a refresh)
                if (not self._prof_shipped and self.registry is not None
                        and (len(self.prof_ttft) >= 8
                             or len(self.prof_tpot) >= 8)):
                    await self.registry.put_json(
                        self._regkey(), self.meta().to_dict(),
                        lease_id=self._lease_id)
                    self._prof_shipped = True
            except Exception:
                pass


def main():
    """CLI: python -m xllm_service_amd.engine.worker --name w0 ..."""
    import argparse
    ap = argparse.ArgumentParser(description="xllm-service-amd worker")
    ap.add_argument("--name", required=True)
    ap.add_argument("--type", default="DEFAULT",
                    choices=[t.value for t in InstanceType])
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--device", default=None)
    ap.add_argument("--registry-host", default="127.0.0.1")
    ap.add_argument("--registry-port", type=int, required=True)
    ap.add_argument("--rpc-host", default="127.0.0.1")
    ap.add_argument("--rpc-port", type=int, default=0)
    ap.add_argument("--max-kv-blocks", type=int, default=None)
    ap.add_argument("--max-num-seqs", type=int, default=256)
    ap.add_argument("--max-batched-tokens", type=int, default=8192)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--no-graphs", action="store_true")
    ap.add_argument("--ssd-swap-dir", default=None,
                    help="directory for the SSD KV swap tier (below DRAM)")
    ap.add_argument("--prefill-hold-ms", type=float, default=25.0,
                    help="batch scattered prompt arrivals into fewer eager "
                         "prefill steps (decodes keep the graph path); "
                         "bounded TTFT cost")
    ap.add_argument("--push-interval-ms", type=float, default=0.0,
                    help="coalesce plain token pushes to the master for up "
                         "to this long (first tokens/finishes always flush "
                         "immediately); 0 = push every engine step")
    ap.add_argument("--relay-responses", action="store_true",
                    help="DECODE instances route generations through their "
                         "prefill peer (decode->prefill->service topology)")
    ap.add_argument("--tp", type=int, default=1,
                    help="tensor-parallel group size; launch one process "
                         "per rank with RANK/WORLD_SIZE/MASTER_ADDR set "
                         "(torchrun-style); rank 0 serves RPC, ranks >0 "
                         "run the follower loop")
    ap.add_argument("--load-state", default=None,
                    help="full (tp=1) state-dict .pt to shard-load")
    args = ap.parse_args()
    logging.basicConfig(level=logging.INFO)

    if args.tp > 1 and int(os.environ.get("RANK", "0")) > 0:
        # TP follower rank: shard engine + replay loop, no service plumbing
        from .engine import LLMEngine
        eng = LLMEngine(args.model, device=args.device,
                        max_kv_blocks=args.max_kv_blocks,
                        tp_size=args.tp, load_state_path=args.load_state,
                        seed=args.seed,
                        enable_graphs=not args.no_graphs,
                        max_num_seqs=args.max_num_seqs,
                        max_batched_tokens=args.max_batched_tokens)
        logging.info("TP follower rank %s up", os.environ["RANK"])
        eng.follower_loop()
        return

    worker = Worker(
        args.name, args.type, model=args.model, device=args.device,
        registry_host=args.registry_host, registry_port=args.registry_port,
        rpc_host=args.rpc_host, rpc_port=args.rpc_port,
        max_kv_blocks=args.max_kv_blocks,
        relay_responses=args.relay_responses,
        push_interval_ms=args.push_interval_ms,
        engine_kwargs=dict(seed=args.seed, max_num_seqs=args.max_num_seqs,
                           max_batched_tokens=args.max_batched_tokens,
                           prefill_hold_ms=args.prefill_hold_ms,
                           enable_graphs=not args.no_graphs,
                           ssd_swap_dir=args.ssd_swap_dir,
                           tp_size=args.tp,
                           load_state_path=args.load_state))

    async def run():
        await worker.start()
        try:
            await asyncio.Event().wait()
        finally:
            await worker.stop()

    asyncio.run(run())


if __name__ == "__main__":
    main()
