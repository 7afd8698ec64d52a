"""Worker orchestrator: DHT + RPC handler + backend + announcer.

Parity target: reference Server / ModuleContainer / ModuleAnnouncerThread
(server/server.py:103-1007). One process per worker: an asyncio loop hosts
the RPC endpoint and DHT node; compute runs on the backend's worker thread.
"""
from __future__ import annotations

import asyncio
import threading
import time
import uuid
from typing import List, Optional, Sequence, Tuple

import torch

from bloombee_amd.data_structures import (ServerInfo, ServerState,
                                          declare_active_modules,
                                          get_remote_module_infos, module_uids)
from bloombee_amd.models.base import ModelConfig, resolve_config
from bloombee_amd.net.dht import Dht
from bloombee_amd.net.rpc import RpcServer
from bloombee_amd.server.backend import StackBackend
from bloombee_amd.server.block_selection import choose_best_blocks
from bloombee_amd.server.handler import ConnectionHandler
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


class Server:
    def __init__(
        self,
        model: str | ModelConfig,
        *,
        initial_peers: Optional[Sequence[Tuple[str, int]]] = None,
        host: str = "127.0.0.1",
        port: int = 0,
        dht_port: int = 0,
        block_indices: Optional[Tuple[int, int]] = None,
        num_blocks: Optional[int] = None,
        device: str = "cpu",
        seed: int = 0,
        kv_max_tokens: int = 1 << 18,
        update_period: float = 30.0,
        session_max_idle: float = 600.0,
        expiration: Optional[float] = None,
        checkpoint_dir: Optional[str] = None,
        throughput: float = 1.0,
        network_rps: Optional[float] = None,
        model_name: Optional[str] = None,
        offload_policy=None,
        adapters: Optional[dict] = None,
        identity_path: Optional[str] = None,
        max_batch_size: int = 2048,
        announce_host: Optional[str] = None,
        announce_port: Optional[int] = None,
    ):
        self.config = model if isinstance(model, ModelConfig) else resolve_config(model)
        self.model_name = model_name or (model if isinstance(model, str)
                                         else self.config.model_type)
        # serving a converted real checkpoint: the model path IS the weight dir
        if checkpoint_dir is None and isinstance(model, str):
            from bloombee_amd.server.from_pretrained import is_converted
            if is_converted(model):
                checkpoint_dir = model
        self.device = device
        # persistent identity (ref --identity_path): keeps routing history,
        # bans and DHT records stable across restarts
        if identity_path:
            import os as _os
            if _os.path.exists(identity_path):
                with open(identity_path) as f:
                    self.peer_id = f.read().strip()
            else:
                self.peer_id = uuid.uuid4().hex[:16]
                _os.makedirs(_os.path.dirname(identity_path) or ".",
                             exist_ok=True)
                with open(identity_path, "w") as f:
                    f.write(self.peer_id)
        else:
            self.peer_id = uuid.uuid4().hex[:16]
        self.max_batch_size = max_batch_size
        self.update_period = update_period
        self.session_max_idle = session_max_idle
        self.expiration = expiration or max(60.0, 2 * update_period)
        self.network_rps = network_rps
        self.throughput = (min(throughput, network_rps)
                           if network_rps is not None else throughput)
        self.host = host
        # NAT / multi-homed hosts (ref --announce_maddrs): advertise a
        # different address than the bind address
        self.announce_host = announce_host
        self.announce_port = announce_port
        self.dht = Dht(initial_peers=list(initial_peers or []), host=host,
                       port=dht_port)

        L = self.config.num_hidden_layers
        # auto-chosen ranges opt in to runtime rebalancing (ref server.py:
        # 479-542 rebuild loop); explicit --block-indices pins the range
        self._auto_blocks = block_indices is None
        self._num_blocks = num_blocks or L
        if block_indices is None:
            infos = get_remote_module_infos(
                self.dht, module_uids(self.model_name, L))
            blocks = choose_best_blocks(self._num_blocks, infos)
            block_indices = (blocks[0], blocks[-1] + 1)
        self.block_range = block_indices
        self.uids = [f"{self.model_name}.{i}"
                     for i in range(block_indices[0], block_indices[1])]

        self._backend_kw = dict(device=device, seed=seed,
                                kv_max_tokens=kv_max_tokens,
                                checkpoint_dir=checkpoint_dir,
                                offload_policy=offload_policy,
                                max_batch_size=max_batch_size)
        self.backend = StackBackend(self.config, block_indices[0],
                                    block_indices[1], **self._backend_kw)
        if adapters:
            # preload LoRA adapters onto this range's blocks (ref
            # run_server --adapters + utils/peft.py download-per-block;
            # offline layout: {dir}/block{i}/ with *.lora_{a,b}.npy)
            import os as _os

            from bloombee_amd.utils.peft import (add_adapter_to_block,
                                                 load_adapter)
            for name, root in adapters.items():
                n_loaded = 0
                for i, blk in enumerate(self.backend.stack.blocks):
                    bdir = _os.path.join(root, f"block{block_indices[0] + i}")
                    if _os.path.isdir(bdir):
                        add_adapter_to_block(blk, name, load_adapter(
                            bdir, dtype=self.config.dtype))
                        n_loaded += 1
                logger.info("adapter %r: %d/%d blocks", name, n_loaded,
                            len(self.backend.stack.blocks))
        self.rpc = RpcServer(host, port)
        self.handler = ConnectionHandler(self.backend, self.rpc)
        from bloombee_amd.server.reachability import attach_reachability
        attach_reachability(self.rpc)
        self._next_pings = {}
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._thread: Optional[threading.Thread] = None
        self._announcer: Optional[threading.Thread] = None
        self._stop = threading.Event()
        self.endpoint: Optional[Tuple[str, int]] = None

    # ------------------------------------------------------------------
    def _server_info(self) -> ServerInfo:
        from bloombee_amd.net.channels import channels
        return ServerInfo(
            state=ServerState.ONLINE,
            host=self.announce_host or self.endpoint[0],
            port=self.announce_port or self.endpoint[1],
            throughput=self.throughput,
            cache_tokens_left=self.backend.kv_pool.tokens_left,
            torch_dtype=self.config.torch_dtype,
            device=self.device,
            start_block=self.block_range[0], end_block=self.block_range[1],
            next_pings=dict(self._next_pings),
            network_rps=self.network_rps,
            dist_rank=channels.rank,
        )

    def _measure_next_pings(self) -> None:
        """RTTs to servers hosting the blocks after ours — gossiped as
        next_pings for the client's min-latency router (ref ModuleAnnouncer
        next-server pings, server/server.py:957-1007)."""
        try:
            from bloombee_amd.utils.ping import PingAggregator

            L = self.config.num_hidden_layers
            nxt = self.block_range[1] % L
            infos = get_remote_module_infos(
                self.dht, [f"{self.model_name}.{nxt}"])
            eps = [(srv.host, srv.port) for srv in infos[0].servers.values()
                   if (srv.host, srv.port) != tuple(self.endpoint or ())]
            if eps:
                agg = PingAggregator(timeout=2.0)
                agg.ping_many(eps)
                self._next_pings = agg.to_dict()
        except Exception as e:  # noqa: BLE001
            logger.debug("next-ping measurement failed: %s", e)

    def _announce_loop(self):
        while not self._stop.is_set():
            try:
                self._measure_next_pings()
                declare_active_modules(self.dht, self.uids, self.peer_id,
                                       self._server_info(),
                                       time.time() + self.expiration)
                self.backend.reap_idle_sessions(self.session_max_idle)
                self._maybe_rebalance()
            except Exception as e:  # noqa: BLE001
                logger.warning("announce failed: %s", e)
            self._stop.wait(self.update_period)

    # -- runtime rebalancing (ref server.py:479-542 rebuild loop) ---------
    def _maybe_rebalance(self) -> None:
        """If re-placing this server would raise the swarm's bottleneck
        throughput enough (block_selection.should_choose_other_blocks —
        dead code in round 1), gracefully re-host a better block range.
        Only auto-placed servers move; in-flight sessions on the old range
        fail over via the client's history replay."""
        if not self._auto_blocks or self._stop.is_set():
            return
        import random as _random

        from bloombee_amd.server.block_selection import \
            should_choose_other_blocks
        L = self.config.num_hidden_layers
        uids = module_uids(self.model_name, L)
        infos = get_remote_module_infos(self.dht, uids)
        if not should_choose_other_blocks(self.peer_id, infos):
            return
        # double-check after a random stagger so two under-used neighbors
        # don't both jump onto the same orphaned range
        self._stop.wait(_random.uniform(0.05, 0.5) * self.update_period)
        infos = get_remote_module_infos(self.dht, uids)
        if self._stop.is_set() or not should_choose_other_blocks(
                self.peer_id, infos):
            return
        stripped = [
            type(i)(uid=i.uid, servers={k: v for k, v in i.servers.items()
                                        if k != self.peer_id})
            for i in infos]
        blocks = choose_best_blocks(min(self._num_blocks, L), stripped)
        new_range = (blocks[0], blocks[-1] + 1)
        if new_range == tuple(self.block_range):
            return
        logger.info("rebalancing: moving from blocks [%d:%d) to [%d:%d)",
                    *self.block_range, *new_range)
        old_backend = self.backend
        old_uids = self.uids
        new_backend = StackBackend(self.config, new_range[0], new_range[1],
                                   **self._backend_kw)
        # atomic swap: the handler reads self.backend per call
        self.backend = new_backend
        self.handler.backend = new_backend
        self.block_range = new_range
        self.uids = [f"{self.model_name}.{i}"
                     for i in range(new_range[0], new_range[1])]
        # retire the old records immediately (state OFFLINE) and announce
        # the new range
        try:
            off_info = self._server_info()
            off_info.state = ServerState.OFFLINE
            declare_active_modules(self.dht, old_uids, self.peer_id,
                                   off_info, time.time() + self.expiration)
            declare_active_modules(self.dht, self.uids, self.peer_id,
                                   self._server_info(),
                                   time.time() + self.expiration)
        finally:
            old_backend.shutdown()

    # ------------------------------------------------------------------
    def run_in_background(self) -> Tuple[str, int]:
        """Start the RPC endpoint + announcer; returns the block endpoint."""
        started = threading.Event()

        def run():
            loop = asyncio.new_event_loop()
            asyncio.set_event_loop(loop)
            self._loop = loop
            self.endpoint = loop.run_until_complete(self.rpc.start())
            started.set()
            loop.run_forever()

        self._thread = threading.Thread(target=run, daemon=True,
                                        name=f"server-{self.peer_id[:6]}")
        self._thread.start()
        if not started.wait(timeout=30):
            raise RuntimeError("server failed to start")
        declare_active_modules(self.dht, self.uids, self.peer_id,
                               self._server_info(), time.time() + self.expiration)
        self._announcer = threading.Thread(target=self._announce_loop,
                                           daemon=True, name="announcer")
        self._announcer.start()
        logger.info("server %s serving %s.[%d:%d) at %s", self.peer_id,
                    self.model_name, *self.block_range, self.endpoint)
        return self.endpoint

    def run(self):
        """Foreground serve (CLI entry)."""
        self.run_in_background()
        try:
            while not self._stop.is_set():
                time.sleep(1.0)
        except KeyboardInterrupt:
            pass
        finally:
            self.shutdown()

    def shutdown(self):
        self._stop.set()
        # the announcer must stop FIRST: an in-flight iteration could
        # re-announce ONLINE after the OFFLINE retirement below
        if self._announcer is not None:
            self._announcer.join(timeout=10)
        # retire this server's records NOW (ref: servers declare OFFLINE on
        # graceful shutdown) so routing and rebalancing don't wait out the
        # DHT expiration to notice the blocks are orphaned
        try:
            if self.endpoint is not None:
                info = self._server_info()
                info.state = ServerState.OFFLINE
                declare_active_modules(self.dht, self.uids, self.peer_id,
                                       info, time.time() + self.expiration)
        except Exception:  # noqa: BLE001 — best effort on the way down
            pass
        if self._loop is not None:
            fut = asyncio.run_coroutine_threadsafe(self.rpc.stop(), self._loop)
            try:
                fut.result(timeout=5)
            except Exception:
                pass

            def _drain_and_stop(loop=self._loop):
                # cancel stragglers (e.g. s2s push clients' read loops) so
                # the loop is not GC'd with pending tasks
                for t in asyncio.all_tasks(loop):
                    t.cancel()
                loop.stop()

            self._loop.call_soon_threadsafe(_drain_and_stop)
        if self._thread is not None:
            self._thread.join(timeout=5)
        self.backend.shutdown()
        self.dht.shutdown()
