"""Device data plane (net/channels.py): mailbox + gloo-world exactness.

The serving stack's RCCL/xGMI path (SURVEY.md §7 hard-part 6) is validated
here on CPU: the same code runs backend "nccl" (= RCCL) on a GPU node, where
frames between same-node workers carry descriptors and payloads ride
isend/irecv. Three tiers:
  1. channel unit semantics (ordering, dtype round-trip, mailbox delivery)
  2. an in-process loopback swarm with channels on (mailbox path) matching
     the TCP-only swarm token-for-token
  3. a 3-process gloo world (client + 2 servers) where every activation hop
     — client→S1 step, S1→S2 push, S2→client reply — rides the pump.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from bloombee_amd.net.channels import channels

MODEL = "llama-tiny"
SEED = 0


def _free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


# ---------------------------------------------------------------------------
# tier 1: channel unit semantics (standalone world of one, mailbox)
# ---------------------------------------------------------------------------
def test_mailbox_send_then_recv_and_recv_then_send():
    channels.enable("cpu")
    try:
        t1 = torch.randn(3, 4)
        seq1 = channels.send(t1, 0)
        got = channels.recv([3, 4], torch.float32, 0, seq1).result(5)
        assert torch.equal(got, t1)
        # recv posted before the send arrives
        fut = channels.recv([2], torch.int64, 0, 1)
        assert not fut.done()
        t2 = torch.tensor([7, 9])
        seq2 = channels.send(t2, 0)
        assert seq2 == 1
        assert torch.equal(fut.result(5), t2)
    finally:
        channels.disable()


def test_mailbox_bool_rides_as_uint8():
    channels.enable("cpu")
    try:
        m = torch.tensor([[True, False], [False, True]])
        seq = channels.send(m, 0)
        got = channels.recv([2, 2], torch.bool, 0, seq).result(5)
        assert got.dtype == torch.uint8  # wire dtype; frame layer casts back
        assert torch.equal(got.bool(), m)
    finally:
        channels.disable()


def test_frame_descriptor_roundtrip():
    from bloombee_amd.net.tensors import (PendingTensor, pack_frame,
                                          unpack_frame)
    channels.enable("cpu")
    try:
        t = torch.randn(2, 5, dtype=torch.float32)
        m = torch.tensor([True, False])
        buf = pack_frame({"x": 1}, [t, m], dist_rank=0)
        # descriptor-only frame: far smaller than the payload
        assert len(buf) < 200
        meta, tensors = unpack_frame(buf)
        assert meta == {"x": 1}
        assert all(isinstance(x, PendingTensor) for x in tensors)
        got_t = tensors[0].resolve(5)
        got_m = tensors[1].resolve(5)
        assert torch.equal(got_t, t)
        assert got_m.dtype == torch.bool and torch.equal(got_m, m)
    finally:
        channels.disable()


# ---------------------------------------------------------------------------
# tier 2: in-process loopback swarm, channels on (mailbox delivery)
# ---------------------------------------------------------------------------
def _local_tokens(prompt, new_tokens):
    from bloombee_amd.engine import LocalEngine

    eng = LocalEngine(MODEL, device="cpu", seed=SEED, kv_max_tokens=1 << 14)
    kv = eng.kv_pool.allocate(prompt.shape[0], 64)
    toks = [eng.prefill(prompt, kv)]
    for _ in range(new_tokens - 1):
        toks.append(eng.decode_step(toks[-1], kv))
    kv.close()
    return torch.stack(toks, 1)


@pytest.mark.timeout(120)
def test_inprocess_swarm_over_channels_exact():
    from bloombee_amd.client import ClientConfig
    from bloombee_amd.models.auto import AutoDistributedModelForCausalLM
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server

    channels.enable("cpu")
    boot = Dht()
    servers = []
    try:
        for rng in [(0, 2), (2, 4)]:
            s = Server(MODEL, initial_peers=[boot.endpoint],
                       block_indices=rng, device="cpu", seed=SEED,
                       kv_max_tokens=1 << 14, update_period=2.0)
            s.run_in_background()
            servers.append(s)
        cfg = ClientConfig(initial_peers=[boot.endpoint])
        model = AutoDistributedModelForCausalLM.from_pretrained(
            MODEL, client_config=cfg, seed=SEED)
        # the swarm map must show dist ranks before the route is built
        assert all(
            si.dist_rank == 0
            for info in model.remote.manager.infos
            for si in info.servers.values())
        gen = torch.Generator().manual_seed(5)
        prompt = torch.randint(0, 1000, (2, 7), generator=gen)
        out = model.generate(prompt, max_new_tokens=6)
        expect = _local_tokens(prompt, 6)
        assert torch.equal(out[:, 7:], expect)
        model.remote.manager.shutdown()
    finally:
        for s in servers:
            s.shutdown()
        boot.shutdown()
        channels.disable()


# ---------------------------------------------------------------------------
# tier 3: 3-process gloo world — client rank 0, servers rank 1 and 2
# ---------------------------------------------------------------------------
def _world_worker(rank, world, port, q, done):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        channels.enable("cpu")  # collective: creates the channel group
        from bloombee_amd.net.dht import Dht

        if rank == 0:
            boot = Dht()
            ep = [boot.endpoint]
        else:
            ep = [None]
        dist.broadcast_object_list(ep, src=0)
        if rank in (1, 2):
            from bloombee_amd.server import Server

            rng = (0, 2) if rank == 1 else (2, 4)
            srv = Server(MODEL, initial_peers=[ep[0]], block_indices=rng,
                         device="cpu", seed=SEED, kv_max_tokens=1 << 14,
                         update_period=2.0)
            srv.run_in_background()
            dist.barrier()  # servers announced; no collectives after this
            # (a blocking barrier during decode would contend with the data
            # plane; lifecycle sync uses an mp.Event instead)
            assert done.wait(timeout=200)
            srv.shutdown()
        else:
            dist.barrier()  # wait for server announcements
            from bloombee_amd.client import ClientConfig
            from bloombee_amd.models.auto import AutoDistributedModelForCausalLM

            cfg = ClientConfig(initial_peers=[ep[0]])
            model = AutoDistributedModelForCausalLM.from_pretrained(
                MODEL, client_config=cfg, seed=SEED)
            ranks = sorted(si.dist_rank
                           for info in model.remote.manager.infos
                           for si in info.servers.values())
            gen = torch.Generator().manual_seed(5)
            prompt = torch.randint(0, 1000, (2, 7), generator=gen)
            out = model.generate(prompt, max_new_tokens=6)
            model.remote.manager.shutdown()
            q.put((out, ranks))
            done.set()
            boot.shutdown()
    finally:
        channels.disable()
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_gloo_world3_swarm_over_channels_exact():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    done = ctx.Event()
    port = _free_port()
    procs = [ctx.Process(target=_world_worker, args=(r, 3, port, q, done))
             for r in range(3)]
    for p in procs:
        p.start()
    out, ranks = q.get(timeout=240)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    # every advertised server carried its dist rank (1 and 2, x2 blocks each)
    assert ranks == [1, 1, 2, 2]
    gen = torch.Generator().manual_seed(5)
    prompt = torch.randint(0, 1000, (2, 7), generator=gen)
    expect = _local_tokens(prompt, 6)
    assert torch.equal(out[:, 7:], expect)
