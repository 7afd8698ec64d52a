"""Server-side unit tests: block selection, throughput measurement,
telemetry, reachability (mirror reference test_server_stats/aux tiers)."""
import asyncio
import time

import pytest
import torch

from bloombee_amd.data_structures import (RemoteModuleInfo, ServerInfo,
                                          ServerState)
from bloombee_amd.server.block_selection import (block_throughputs,
                                                 choose_best_blocks,
                                                 should_choose_other_blocks)
from bloombee_amd.server.throughput import measure_compute_rps
from bloombee_amd.utils.telemetry import StageTimes


def _infos(cov):
    """cov: list of per-block total throughput."""
    out = []
    for i, t in enumerate(cov):
        servers = {}
        if t > 0:
            servers[f"p{i}"] = ServerInfo(throughput=t)
        out.append(RemoteModuleInfo(uid=f"m.{i}", servers=servers))
    return out


def test_choose_best_blocks_picks_weakest_window():
    infos = _infos([5, 5, 0, 0, 5, 5])
    assert choose_best_blocks(2, infos) == [2, 3]
    assert block_throughputs(infos) == [5, 5, 0, 0, 5, 5]


def test_should_choose_other_blocks():
    # our server p0 sits on well-covered blocks while block 3 starves
    infos = _infos([9, 9, 9, 0.5])
    infos[0].servers["me"] = ServerInfo(throughput=3.0)
    infos[1].servers["me"] = ServerInfo(throughput=3.0)
    assert should_choose_other_blocks("me", infos, balance_quality=0.9)
    # balanced swarm: no move
    infos2 = _infos([5, 5, 5, 5])
    infos2[0].servers["me"] = ServerInfo(throughput=3.0)
    assert not should_choose_other_blocks("me", infos2, balance_quality=0.75)


def test_measure_compute_rps_runs():
    from bloombee_amd.models.base import resolve_config

    rps = measure_compute_rps(resolve_config("llama-tiny"), "cpu",
                              n_tokens=1, n_steps=3)
    assert rps > 0


def test_stage_times_summary():
    st = StageTimes()
    with st.span("compute"):
        time.sleep(0.01)
    st.bump_step()
    s = st.summary()
    assert s["steps"] == 1 and s["compute"]["count"] == 1
    assert s["compute"]["mean_ms"] >= 10
    assert "compute" in st.table()


def test_reachability_probe():
    from bloombee_amd.net.rpc import RpcServer
    from bloombee_amd.server.reachability import (attach_reachability,
                                                  check_direct_reachability)

    async def run():
        target = RpcServer()

        async def info(meta, tensors):
            return {"ok": True}, []

        target.register("rpc_info", info)
        t_ep = await target.start()

        helper = RpcServer()
        attach_reachability(helper)
        h_ep = await helper.start()

        ok = await check_direct_reachability(t_ep, [h_ep])
        assert ok is True
        bad = await check_direct_reachability(("127.0.0.1", 1), [h_ep])
        assert bad is False
        none = await check_direct_reachability(t_ep, [("127.0.0.1", 1)])
        assert none is None
        await target.stop()
        await helper.stop()

    asyncio.run(run())


def test_per_module_log_levels(monkeypatch):
    """BBAMD_LOG per-module level overrides (ref utils/debug_config.py)."""
    import importlib
    import logging as stdlog

    from bloombee_amd.utils import logging as bl

    monkeypatch.setenv("BBAMD_LOG", "client.session=debug, server=warning")
    monkeypatch.setattr(bl, "_configured", False)
    bl.get_logger("client.session")
    assert stdlog.getLogger("bloombee_amd.client.session").level == stdlog.DEBUG
    assert stdlog.getLogger("bloombee_amd.server").level == stdlog.WARNING


def test_idle_session_reaper():
    """Sessions idle past the threshold are closed and their KV freed
    (crashed-client leak protection)."""
    import torch

    from bloombee_amd.models.base import resolve_config
    from bloombee_amd.server.backend import StackBackend

    cfg = resolve_config("llama-tiny")
    be = StackBackend(cfg, 0, 2, device="cpu", seed=0, kv_max_tokens=1 << 12)
    be.open_session("s1", batch_size=1, max_length=64)
    left_before = be.kv_pool.tokens_left
    assert be.reap_idle_sessions(max_idle_s=3600) == 0   # fresh: kept
    be.sessions["s1"].last_activity -= 7200
    assert be.reap_idle_sessions(max_idle_s=3600) == 1   # idle: reaped
    assert "s1" not in be.sessions
    assert be.kv_pool.tokens_left > left_before
    be.shutdown()


def test_long_prefill_chunking_matches_unchunked(monkeypatch):
    """Sequence-chunked prefill (BBAMD_MAX_CHUNK_TOKENS) must produce the
    same hidden states as one whole-sequence pass (ref max_chunk_size_bytes
    chunking, backend.py:525-531)."""
    import torch

    from bloombee_amd.models.base import resolve_config
    from bloombee_amd.server.backend import StackBackend

    cfg = resolve_config("llama-tiny")
    gen = torch.Generator().manual_seed(3)
    h = (torch.randn(2, 50, cfg.hidden_size, generator=gen) * 0.1).to(cfg.dtype)

    be1 = StackBackend(cfg, 0, 2, device="cpu", seed=0, kv_max_tokens=1 << 12)
    be1.open_session("a", batch_size=2, max_length=128)
    whole = be1.inference_step("a", h, 0)
    be1.shutdown()

    monkeypatch.setenv("BBAMD_MAX_CHUNK_TOKENS", "16")
    be2 = StackBackend(cfg, 0, 2, device="cpu", seed=0, kv_max_tokens=1 << 12)
    assert be2.max_chunk_tokens == 16
    be2.open_session("a", batch_size=2, max_length=128)
    chunked = be2.inference_step("a", h, 0)
    # decode continues correctly after a chunked prefill
    nxt = (torch.randn(2, 1, cfg.hidden_size, generator=gen) * 0.1).to(cfg.dtype)
    d1 = be2.inference_step("a", nxt, 50)
    be2.shutdown()

    assert torch.equal(whole, chunked)
    assert torch.isfinite(d1.float()).all()


def test_activation_dumper_roundtrip(tmp_path, monkeypatch):
    """BBAMD_DUMP_ACTIVATIONS captures real hidden states to disk (ref
    real_activation_dumper.py)."""
    import importlib

    import torch

    monkeypatch.setenv("BBAMD_DUMP_ACTIVATIONS", str(tmp_path))
    from bloombee_amd.utils import activation_dumper as ad
    importlib.reload(ad)
    assert ad.enabled()
    t = torch.randn(2, 3, 8).bfloat16()
    ad.capture_activation("blocks0_2", t)
    files = list(tmp_path.rglob("*.npy")) + list(tmp_path.rglob("*.pt"))
    assert files, list(tmp_path.rglob("*"))
    monkeypatch.delenv("BBAMD_DUMP_ACTIVATIONS")
    importlib.reload(ad)
    assert not ad.enabled()


def test_disk_cache_eviction(tmp_path):
    """free_disk_space_for evicts oldest cache entries until the requested
    room exists (ref utils/disk_cache.py:83)."""
    import os
    import time

    from bloombee_amd.utils.disk_cache import free_disk_space_for

    for i in range(3):
        p = tmp_path / f"blob{i}.bin"
        p.write_bytes(b"x" * 1000)
        os.utime(p, (time.time() - 100 + i, time.time() - 100 + i))
    # plenty of disk: nothing should be evicted
    assert free_disk_space_for(1, cache_dir=str(tmp_path))
    assert len(list(tmp_path.iterdir())) == 3


def test_ping_aggregator_unreachable_is_inf():
    """Unreachable endpoints report inf RTT instead of raising (ref
    utils/ping.py; live-endpoint RTTs covered by test_swarm_health_report)."""
    import math

    from bloombee_amd.utils.ping import PingAggregator

    agg = PingAggregator(timeout=0.3)
    out = agg.ping_many([("127.0.0.1", 1)])  # nothing listens on port 1
    assert math.isinf(out[("127.0.0.1", 1)])


def test_checkpoint_layout_roundtrip(tmp_path):
    """North-star per-block .npy layout (BASELINE.json): save a stack's
    blocks, reload into a fresh random-init server backend, bitwise-equal
    weights and identical decode output."""
    import torch

    from bloombee_amd.engine import BlockStack, LocalEngine
    from bloombee_amd.models.base import resolve_config
    from bloombee_amd.server.backend import StackBackend
    from bloombee_amd.server.from_pretrained import (SENTINEL,
                                                     is_converted,
                                                     save_block_weights)

    cfg = resolve_config("llama-tiny")
    src = BlockStack(cfg, 0, 4, device="cpu", seed=123)
    ck = tmp_path / "llama-tiny-np"
    for i, blk in enumerate(src.blocks):
        save_block_weights(blk, str(ck), i)
    (ck / SENTINEL).touch()
    assert is_converted(str(ck))
    assert any(p.suffix == ".npy" for p in ck.iterdir())

    # a backend seeded DIFFERENTLY must come out identical after loading
    be = StackBackend(cfg, 0, 4, device="cpu", seed=999,
                      kv_max_tokens=1 << 12, checkpoint_dir=str(ck))
    for b_src, b_dst in zip(src.blocks, be.stack.blocks):
        for (n1, p1), (n2, p2) in zip(b_src.named_parameters(),
                                      b_dst.named_parameters()):
            assert n1 == n2
            assert torch.equal(p1, p2), n1
    be.shutdown()


def test_client_weights_roundtrip(tmp_path):
    import torch

    from bloombee_amd.server.from_pretrained import (load_client_weights,
                                                     save_client_weights)

    e = torch.randn(100, 16).bfloat16()
    n = torch.randn(16).bfloat16()
    h = torch.randn(100, 16).bfloat16()
    save_client_weights(str(tmp_path), e, n, h)
    back = load_client_weights(str(tmp_path))
    assert torch.equal(back["embed"], e)
    assert torch.equal(back["final_norm"], n)
    assert torch.equal(back["lm_head"], h)


def test_mixed_attn_auto_policy_backend():
    """BBAMD_MIXED_ATTN=auto through the real backend: a swapped session
    RESTORES when the pool has room and decodes MIXED (host prefix,
    pos_offset > 0) under page scarcity."""
    from bloombee_amd.config import RuntimeConfig, KVConfig, get_config, set_config
    from bloombee_amd.models.base import resolve_config
    from bloombee_amd.server.backend import StackBackend

    cfg = resolve_config("llama-tiny")
    be = StackBackend(cfg, 0, 2, device="cpu", seed=0,
                            kv_max_tokens=16 * 16)
    old_cfg = get_config()
    set_config(RuntimeConfig(kv=KVConfig(mixed_attn="auto")))
    try:
        be.open_session("s", 1, 64)
        h = torch.randn(1, 20, cfg.hidden_size).to(cfg.dtype) * 0.1
        be.inference_step("s", h, 0)     # 20 tokens -> 2 pages
        handle = be.session_handle("s")

        # room in the pool -> auto restores
        handle.swap_out()
        assert handle.is_swapped and handle.swapped_pages_needed() == 2
        be.inference_step("s", h[:, :1], 20)
        assert not handle.is_swapped and handle.pos_offset == 0

        # page scarcity (1 free < 2 needed, but enough for the recent
        # segment) -> auto keeps the prefix host-side (mixed decode)
        handle.swap_out()
        held = be.kv_pool._take_pages(be.kv_pool.free_page_count() - 1)
        be.inference_step("s", h[:, :1], 21)
        assert handle.pos_offset > 0, "expected mixed-device decode"
        be.kv_pool._give_pages(held)
    finally:
        set_config(old_cfg)
        be.close_session("s")
