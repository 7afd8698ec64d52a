"""Loopback swarm integration: 2 servers + DHT + client, exact-match decode
vs the single-process LocalEngine (mirrors the reference's
test_full_model.py / test_chained_calls.py tier, but self-contained — no
out-of-band swarm bootstrap required)."""
import time

import pytest
import torch

from bloombee_amd.client import ClientConfig
from bloombee_amd.engine import LocalEngine
from bloombee_amd.models.auto import AutoDistributedModelForCausalLM
from bloombee_amd.net.dht import Dht
from bloombee_amd.server import Server

MODEL = "llama-tiny"
SEED = 0


@pytest.fixture(scope="module")
def swarm():
    boot = Dht()
    servers = []
    s1 = Server(MODEL, initial_peers=[boot.endpoint], block_indices=(0, 2),
                device="cpu", seed=SEED, kv_max_tokens=1 << 14,
                update_period=2.0)
    s1.run_in_background()
    servers.append(s1)
    s2 = Server(MODEL, initial_peers=[boot.endpoint], block_indices=(2, 4),
                device="cpu", seed=SEED, kv_max_tokens=1 << 14,
                update_period=2.0)
    s2.run_in_background()
    servers.append(s2)
    yield boot, servers
    for s in servers:
        s.shutdown()
    boot.shutdown()


import contextlib


@contextlib.contextmanager
def _no_wire_gates():
    """Drop the codec min-size/min-gain gates so tiny test tensors actually
    exercise compression end-to-end."""
    import dataclasses

    from bloombee_amd import config as bconf
    cfg0 = bconf.get_config()
    comp = dataclasses.replace(cfg0.compression, min_size_bytes=1,
                               min_gain=0.0)
    bconf.set_config(dataclasses.replace(cfg0, compression=comp))
    try:
        yield
    finally:
        bconf.set_config(cfg0)


def _local_tokens(prompt, new_tokens):
    eng = LocalEngine(MODEL, device="cpu", seed=SEED, kv_max_tokens=1 << 14)
    kv = eng.kv_pool.allocate(prompt.shape[0], 64)
    toks = [eng.prefill(prompt, kv)]
    for _ in range(new_tokens - 1):
        toks.append(eng.decode_step(toks[-1], kv))
    kv.close()
    return torch.stack(toks, 1)


def _make_model(boot, **kw):
    cfg = ClientConfig(initial_peers=[boot.endpoint], **kw)
    return AutoDistributedModelForCausalLM.from_pretrained(
        MODEL, client_config=cfg, seed=SEED)


def test_swarm_greedy_matches_local(swarm):
    boot, _ = swarm
    model = _make_model(boot)
    gen = torch.Generator().manual_seed(5)
    prompt = torch.randint(0, 1000, (2, 7), generator=gen)
    out = model.generate(prompt, max_new_tokens=6)
    expect = _local_tokens(prompt, 6)
    assert torch.equal(out[:, 7:], expect), (out[:, 7:], expect)
    model.remote.manager.shutdown()


def test_swarm_greedy_no_push_matches_local(swarm):
    boot, _ = swarm
    model = _make_model(boot, use_server_to_server=False)
    gen = torch.Generator().manual_seed(5)
    prompt = torch.randint(0, 1000, (2, 7), generator=gen)
    out = model.generate(prompt, max_new_tokens=6)
    expect = _local_tokens(prompt, 6)
    assert torch.equal(out[:, 7:], expect)
    model.remote.manager.shutdown()


def test_forward_backward_grads(swarm):
    boot, _ = swarm
    model = _make_model(boot)
    model.transformer.pre_seq_len = 0
    gen = torch.Generator().manual_seed(6)
    ids = torch.randint(0, 1000, (1, 5), generator=gen)
    h = model.transformer.embed(ids).detach().requires_grad_(True)
    out = model.transformer.remote(h)
    loss = out.float().square().mean()
    loss.backward()
    assert h.grad is not None and torch.isfinite(h.grad).all()
    assert h.grad.abs().sum() > 0
    model.remote.manager.shutdown()


def test_failover_rebuilds_chain():
    """A mid-chain server dying mid-session must trigger history replay on a
    replacement (ref inference_session.py:802-831)."""
    boot = Dht()
    s1 = Server(MODEL, initial_peers=[boot.endpoint], block_indices=(0, 2),
                device="cpu", seed=SEED, kv_max_tokens=1 << 14, update_period=1.0)
    s1.run_in_background()
    dying = Server(MODEL, initial_peers=[boot.endpoint], block_indices=(2, 4),
                   device="cpu", seed=SEED, kv_max_tokens=1 << 14,
                   update_period=1.0)
    dying.run_in_background()
    backup = Server(MODEL, initial_peers=[boot.endpoint], block_indices=(2, 4),
                    device="cpu", seed=SEED, kv_max_tokens=1 << 14,
                    update_period=1.0)
    try:
        cfg = ClientConfig(initial_peers=[boot.endpoint],
                           blocked_servers=[backup.peer_id],
                           min_backoff=0.1, step_timeout=10.0)
        model = AutoDistributedModelForCausalLM.from_pretrained(
            MODEL, client_config=cfg, seed=SEED)
        gen = torch.Generator().manual_seed(5)
        prompt = torch.randint(0, 1000, (1, 7), generator=gen)
        session = model.remote.inference_session(40)
        h = model.embed(prompt)
        out1 = session.step(h)
        tok = model.lm_head(model.final_norm(out1[:, -1:]))[:, -1].argmax(-1)

        # kill the tail server, bring the backup online, unblock it
        dying.shutdown()
        backup.run_in_background()
        model.remote.manager.config.blocked_servers = None
        time.sleep(0.2)
        model.remote.manager.update()

        toks = [tok]
        for _ in range(3):
            h = model.embed(toks[-1].view(1, 1))
            out = session.step(h)
            toks.append(model.lm_head(model.final_norm(out[:, -1:]))[:, -1].argmax(-1))
        session.close()

        expect = _local_tokens(prompt, 4)
        got = torch.stack(toks, 1)
        assert torch.equal(got, expect), (got, expect)
        model.remote.manager.shutdown()
    finally:
        for s in (s1, backup):
            s.shutdown()
        boot.shutdown()


def test_bloom_swarm_matches_local():
    """BASELINE.json config 1: bloom-family plumbing — 2 CPU workers over
    loopback DHT, greedy decode exact-match vs local."""
    boot = Dht()
    s1 = Server("bloom-tiny", initial_peers=[boot.endpoint], block_indices=(0, 2),
                device="cpu", seed=SEED, kv_max_tokens=1 << 14, update_period=5.0)
    s2 = Server("bloom-tiny", initial_peers=[boot.endpoint], block_indices=(2, 4),
                device="cpu", seed=SEED, kv_max_tokens=1 << 14, update_period=5.0)
    s1.run_in_background()
    s2.run_in_background()
    try:
        cfg = ClientConfig(initial_peers=[boot.endpoint])
        model = AutoDistributedModelForCausalLM.from_pretrained(
            "bloom-tiny", client_config=cfg, seed=SEED)
        gen = torch.Generator().manual_seed(9)
        prompt = torch.randint(0, 1000, (2, 6), generator=gen)
        out = model.generate(prompt, max_new_tokens=5)

        eng = LocalEngine("bloom-tiny", device="cpu", seed=SEED,
                          kv_max_tokens=1 << 14)
        kv = eng.kv_pool.allocate(2, 64)
        toks = [eng.prefill(prompt, kv)]
        for _ in range(4):
            toks.append(eng.decode_step(toks[-1], kv))
        kv.close()
        expect = torch.stack(toks, 1)
        assert torch.equal(out[:, 6:], expect), (out[:, 6:], expect)
        model.remote.manager.shutdown()
    finally:
        s1.shutdown()
        s2.shutdown()
        boot.shutdown()


def test_swarm_compressed_wire_matches_local(swarm):
    boot, _ = swarm
    with _no_wire_gates():
        _compressed_roundtrip(boot, "bsplit+zlib")


def _compressed_roundtrip(boot, codec):
    model = _make_model(boot, wire_codec=codec)
    gen = torch.Generator().manual_seed(5)
    prompt = torch.randint(0, 1000, (2, 7), generator=gen)
    out = model.generate(prompt, max_new_tokens=6)
    expect = _local_tokens(prompt, 6)
    assert torch.equal(out[:, 7:], expect)  # codec is lossless
    model.remote.manager.shutdown()


def test_multi_call_generate_session_reuse(swarm):
    """Session resume across .generate() calls (ref RemotePastKeyValues
    multi-call resume, remote_generation.py:183-216)."""
    boot, _ = swarm
    model = _make_model(boot)
    gen = torch.Generator().manual_seed(5)
    prompt = torch.randint(0, 1000, (1, 7), generator=gen)
    sess = model.remote.inference_session(64)
    out1 = model.generate(prompt, max_new_tokens=3, session=sess)
    out2 = model.generate(out1[:, -1:], max_new_tokens=3, session=sess)
    sess.close()
    expect = _local_tokens(prompt, 6)
    got = torch.cat([out1[:, 7:], out2[:, 1:]], dim=1)
    assert torch.equal(got, expect), (got, expect)
    model.remote.manager.shutdown()


def test_swarm_microbatch_split_matches_local(swarm):
    """Batch >= split threshold: servers split into micro-batches and push
    slice-by-slice downstream (cross-stage overlap); outputs must still be
    exact (ref block_functions.py:2055-2460)."""
    boot, _ = swarm
    model = _make_model(boot)
    gen = torch.Generator().manual_seed(11)
    prompt = torch.randint(0, 1000, (8, 6), generator=gen)
    out = model.generate(prompt, max_new_tokens=5)
    expect = _local_tokens(prompt, 5)
    assert torch.equal(out[:, 6:], expect), (out[:, 6:], expect)
    model.remote.manager.shutdown()


@pytest.mark.parametrize("family_model", ["qwen3-tiny", "falcon-tiny",
                                          "mixtral-tiny", "gemma4-tiny"])
def test_family_swarm_matches_local(family_model):
    """Every registered family must decode exactly over a 2-server swarm
    (client model classes + server blocks + per-family KV geometry)."""
    from bloombee_amd.models.base import resolve_config
    n_layers = resolve_config(family_model).num_hidden_layers
    half = n_layers // 2
    boot = Dht()
    s1 = Server(family_model, initial_peers=[boot.endpoint],
                block_indices=(0, half),
                device="cpu", seed=SEED, kv_max_tokens=1 << 14, update_period=5.0)
    s2 = Server(family_model, initial_peers=[boot.endpoint],
                block_indices=(half, n_layers),
                device="cpu", seed=SEED, kv_max_tokens=1 << 14, update_period=5.0)
    s1.run_in_background()
    s2.run_in_background()
    try:
        cfg = ClientConfig(initial_peers=[boot.endpoint])
        model = AutoDistributedModelForCausalLM.from_pretrained(
            family_model, client_config=cfg, seed=SEED)
        V = model.config.vocab_size
        gen = torch.Generator().manual_seed(9)
        prompt = torch.randint(0, min(900, V), (2, 6), generator=gen)
        out = model.generate(prompt, max_new_tokens=4)

        eng = LocalEngine(family_model, device="cpu", seed=SEED,
                          kv_max_tokens=1 << 14)
        kv = eng.kv_pool.allocate(2, 64)
        toks = [eng.prefill(prompt, kv)]
        for _ in range(3):
            toks.append(eng.decode_step(toks[-1], kv))
        kv.close()
        expect = torch.stack(toks, 1)
        assert torch.equal(out[:, 6:], expect), (family_model, out[:, 6:], expect)
        model.remote.manager.shutdown()
    finally:
        s1.shutdown()
        s2.shutdown()
        boot.shutdown()


def test_deep_ptune_grads_and_parity(swarm):
    """Deep p-tuning: per-block prompts added at every block input must (a)
    produce the same hidden states as a local BlockStack composition and (b)
    receive gradients through rpc_backward (ref client/ptune.py deep mode)."""
    import torch.nn.functional as F
    from bloombee_amd.engine import BlockStack
    from bloombee_amd.models.base import resolve_config

    boot, _ = swarm
    cfg = ClientConfig(initial_peers=[boot.endpoint])
    model = AutoDistributedModelForCausalLM.from_pretrained(
        "llama-tiny", client_config=cfg, seed=SEED,
        pre_seq_len=4, deep_ptune=True)
    with torch.no_grad():
        torch.manual_seed(21)
        model.transformer.deep_prompts.normal_(0, 0.02)
    gen = torch.Generator().manual_seed(11)
    ids = torch.randint(0, 1000, (2, 6), generator=gen)

    logits = model(ids)
    # parity: same computation on a local stack built from the servers' seed
    mcfg = resolve_config("llama-tiny")
    stack = BlockStack(mcfg, 0, mcfg.num_hidden_layers, device="cpu", seed=SEED)
    h = F.embedding(ids, model.transformer.embed_tokens)
    p = model.transformer.prompt_embeds.detach().to(h.dtype)
    h = torch.cat([p.unsqueeze(0).expand(2, -1, -1), h], dim=1)
    ref = stack.forward_train(
        h, deep_prompts=model.transformer.deep_prompts.detach())
    ref_logits = model.lm_head(model.transformer.final_norm(ref))
    assert torch.equal(logits, ref_logits)

    loss = logits.float().square().mean()
    loss.backward()
    for name, par in (("prompt_embeds", model.transformer.prompt_embeds),
                      ("deep_prompts", model.transformer.deep_prompts)):
        assert par.grad is not None, name
        assert torch.isfinite(par.grad).all(), name
        assert par.grad.abs().sum() > 0, name
    # every block's prompt row must see gradient (not just span heads)
    per_block = model.transformer.deep_prompts.grad.flatten(1).abs().sum(1)
    assert (per_block > 0).all(), per_block
    model.remote.manager.shutdown()


def test_deep_ptune_generation_uses_prompts(swarm):
    """Generation with trained prompts must differ from the no-prompt decode
    and be deterministic across sessions (prefill carries deep prompts)."""
    boot, _ = swarm
    cfg = ClientConfig(initial_peers=[boot.endpoint])
    model = AutoDistributedModelForCausalLM.from_pretrained(
        "llama-tiny", client_config=cfg, seed=SEED,
        pre_seq_len=4, deep_ptune=True)
    with torch.no_grad():
        torch.manual_seed(22)
        model.transformer.deep_prompts.normal_(0, 0.5)
        model.transformer.prompt_embeds.normal_(0, 0.5)
    gen = torch.Generator().manual_seed(12)
    prompt = torch.randint(0, 1000, (1, 5), generator=gen)
    out1 = model.generate(prompt, max_new_tokens=4)
    out2 = model.generate(prompt, max_new_tokens=4)
    assert torch.equal(out1, out2)
    plain = _local_tokens(prompt, 4)
    # trained prompts shift the distribution; tiny chance of coincidence
    assert not torch.equal(out1[:, 5:], plain)
    model.remote.manager.shutdown()


def test_deep_ptune_microbatch_split_matches_small_batch(swarm):
    """Deep prompts must survive the server-side micro-batch split (B>=8
    prefill splits into slices; every slice re-applies the per-block
    prompts). Row 0 of a B=8 batch must match the B=1 result."""
    boot, _ = swarm
    cfg = ClientConfig(initial_peers=[boot.endpoint])
    model = AutoDistributedModelForCausalLM.from_pretrained(
        "llama-tiny", client_config=cfg, seed=SEED,
        pre_seq_len=4, deep_ptune=True)
    with torch.no_grad():
        torch.manual_seed(23)
        model.transformer.deep_prompts.normal_(0, 0.3)
        model.transformer.prompt_embeds.normal_(0, 0.3)
    gen = torch.Generator().manual_seed(13)
    prompt = torch.randint(0, 1000, (8, 5), generator=gen)
    out_b8 = model.generate(prompt, max_new_tokens=4)
    out_b1 = model.generate(prompt[:1], max_new_tokens=4)
    assert torch.equal(out_b8[:1], out_b1), (out_b8[:1], out_b1)
    model.remote.manager.shutdown()


def test_swarm_mt_compressed_wire_matches_local(swarm):
    """The native multithreaded chunked codec is lossless end-to-end."""
    boot, _ = swarm
    with _no_wire_gates():
        _compressed_roundtrip(boot, "bsplit+zlibmt")


def test_generate_exact_under_injected_rpc_faults(swarm):
    """With 20% of unary RPCs failing (deterministic injection), the retry
    + re-route machinery must still produce the exact greedy output."""
    from bloombee_amd.utils import fault_injection as fi

    boot, _ = swarm
    model = _make_model(boot, use_server_to_server=False,
                        min_backoff=0.05, max_retries=None)
    gen = torch.Generator().manual_seed(5)
    prompt = torch.randint(0, 1000, (2, 7), generator=gen)
    fi.configure(0.3, seed=7, max_faults=6)
    try:
        out = model.generate(prompt, max_new_tokens=6)
    finally:
        injected = fi.injected
        fi.configure(0.0)
    expect = _local_tokens(prompt, 6)
    assert torch.equal(out[:, 7:], expect)
    assert injected == 6, injected
    model.remote.manager.shutdown()


def test_concurrent_clients_exact(swarm):
    """4 clients decoding different prompts concurrently against the same
    2-server swarm must each match their single-process result (task-pool
    interleaving + per-session KV isolation)."""
    from concurrent.futures import ThreadPoolExecutor

    boot, _ = swarm

    def one(seed):
        model = _make_model(boot)
        gen = torch.Generator().manual_seed(100 + seed)
        prompt = torch.randint(0, 1000, (1, 5 + seed), generator=gen)
        out = model.generate(prompt, max_new_tokens=5)
        model.remote.manager.shutdown()
        return prompt, out

    with ThreadPoolExecutor(4) as ex:
        results = list(ex.map(one, range(4)))
    for seed, (prompt, out) in enumerate(results):
        expect = _local_tokens(prompt, 5)
        assert torch.equal(out[:, prompt.shape[1]:], expect), seed


def test_deep_ptune_optimizer_step_reduces_loss(swarm):
    """The p-tune trainer loop: optimizer over trainable_parameters()
    (shallow + deep prompts) reduces a fixed-target loss over the swarm."""
    boot, _ = swarm
    cfg = ClientConfig(initial_peers=[boot.endpoint])
    model = AutoDistributedModelForCausalLM.from_pretrained(
        "llama-tiny", client_config=cfg, seed=SEED,
        pre_seq_len=4, deep_ptune=True)
    params = model.trainable_parameters()
    assert len(params) == 2
    opt = torch.optim.Adam(params, lr=5e-2)
    gen = torch.Generator().manual_seed(31)
    ids = torch.randint(0, 1000, (1, 6), generator=gen)
    target = torch.randint(0, 1000, (1, 10), generator=gen)
    losses = []
    for _ in range(4):
        logits = model(ids)
        loss = torch.nn.functional.cross_entropy(
            logits.float().view(-1, logits.shape[-1]), target.view(-1))
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0], losses
    model.remote.manager.shutdown()


def test_sequence_classification_trains(swarm):
    """AutoDistributedModelForSequenceClassification: prompt-tuned
    classification head trains over the swarm (ref SST-2 notebook model)."""
    from bloombee_amd.models.auto import \
        AutoDistributedModelForSequenceClassification

    boot, _ = swarm
    cfg = ClientConfig(initial_peers=[boot.endpoint])
    model = AutoDistributedModelForSequenceClassification.from_pretrained(
        "llama-tiny", client_config=cfg, seed=SEED, num_labels=2,
        pre_seq_len=4, deep_ptune=True)
    params = model.trainable_parameters()
    assert len(params) == 3  # prompts + deep prompts + score head
    opt = torch.optim.Adam(params, lr=5e-2)
    gen = torch.Generator().manual_seed(17)
    ids = torch.randint(0, 1000, (4, 6), generator=gen)
    labels = torch.tensor([0, 1, 0, 1])
    losses = []
    for _ in range(4):
        logits = model(ids)
        assert logits.shape == (4, 2)
        loss = torch.nn.functional.cross_entropy(logits, labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0], losses
    model.remote.manager.shutdown()


def test_push_drop_recovers_exact():
    """Dropped server-to-server pushes (injected) must be absorbed by the
    client's step-timeout rebuild + history replay, still emitting the
    exact greedy tokens."""
    from bloombee_amd.utils import fault_injection as fi

    boot = Dht()
    s1 = Server(MODEL, initial_peers=[boot.endpoint], block_indices=(0, 2),
                device="cpu", seed=SEED, kv_max_tokens=1 << 14,
                update_period=2.0)
    s2 = Server(MODEL, initial_peers=[boot.endpoint], block_indices=(2, 4),
                device="cpu", seed=SEED, kv_max_tokens=1 << 14,
                update_period=2.0)
    s1.run_in_background()
    s2.run_in_background()
    try:
        cfg = ClientConfig(initial_peers=[boot.endpoint],
                           use_server_to_server=True,
                           step_timeout=3.0, min_backoff=0.1,
                           max_retries=None)
        model = AutoDistributedModelForCausalLM.from_pretrained(
            MODEL, client_config=cfg, seed=SEED)
        gen = torch.Generator().manual_seed(5)
        prompt = torch.randint(0, 1000, (2, 7), generator=gen)
        fi.configure(0.9, seed=3, max_faults=2,
                     methods=("s2s_push",))
        try:
            out = model.generate(prompt, max_new_tokens=6)
        finally:
            injected = fi.injected
            fi.configure(0.0)
        expect = _local_tokens(prompt, 6)
        assert torch.equal(out[:, 7:], expect)
        assert injected >= 1
        model.remote.manager.shutdown()
    finally:
        s1.shutdown()
        s2.shutdown()
        boot.shutdown()


def test_training_grads_exact_under_faults(swarm):
    """rpc_forward/rpc_backward retries (injected faults) must not change
    the computed gradients at all."""
    from bloombee_amd.utils import fault_injection as fi

    boot, _ = swarm
    model = _make_model(boot, min_backoff=0.05, max_retries=None,
                        ban_timeout=0.2)
    gen = torch.Generator().manual_seed(41)
    ids = torch.randint(0, 1000, (2, 5), generator=gen)

    h1 = model.transformer.embed(ids).detach().requires_grad_(True)
    model.transformer.remote(h1).float().square().mean().backward()

    h2 = model.transformer.embed(ids).detach().requires_grad_(True)
    fi.configure(0.8, seed=13, max_faults=3,
                 methods=("rpc_forward", "rpc_backward"))
    try:
        model.transformer.remote(h2).float().square().mean().backward()
    finally:
        injected = fi.injected
        fi.configure(0.0)
    assert injected >= 1
    assert torch.equal(h1.grad, h2.grad)
    model.remote.manager.shutdown()


def test_microbatch_push_drop_recovers_exact(swarm):
    """B>=8 prefill triggers the server-side micro-batch split with per-MB
    downstream pushes; dropped MB pushes must still converge to the exact
    greedy output (client-stream fallback + rebuild)."""
    from bloombee_amd.utils import fault_injection as fi

    boot, _ = swarm
    model = _make_model(boot, step_timeout=3.0, min_backoff=0.1,
                        max_retries=None, ban_timeout=0.2)
    gen = torch.Generator().manual_seed(51)
    prompt = torch.randint(0, 1000, (8, 6), generator=gen)
    fi.configure(0.5, seed=29, max_faults=2, methods=("s2s_push",))
    try:
        out = model.generate(prompt, max_new_tokens=4)
    finally:
        injected = fi.injected
        fi.configure(0.0)
    expect = _local_tokens(prompt, 4)
    assert torch.equal(out[:, 6:], expect)
    assert injected >= 1
    model.remote.manager.shutdown()


def test_sampled_generation_seedable(swarm):
    """do_sample with an explicit generator is reproducible (and actually
    samples: different seeds should usually diverge)."""
    boot, _ = swarm
    model = _make_model(boot)
    gen = torch.Generator().manual_seed(5)
    prompt = torch.randint(0, 1000, (1, 6), generator=gen)
    a = model.generate(prompt, max_new_tokens=8, do_sample=True,
                       temperature=1.5,
                       generator=torch.Generator().manual_seed(7))
    b = model.generate(prompt, max_new_tokens=8, do_sample=True,
                       temperature=1.5,
                       generator=torch.Generator().manual_seed(7))
    assert torch.equal(a, b)
    model.remote.manager.shutdown()


def test_chunked_stream_forward_backward(swarm, monkeypatch):
    """Payloads above the unary ceiling must round-trip through the chunked
    stream path (ref remote_forward_backward.py:46-118) with identical
    results and flowing gradients."""
    import bloombee_amd.net.streaming as streaming

    boot, _ = swarm
    model = _make_model(boot)
    model.transformer.pre_seq_len = 0
    gen = torch.Generator().manual_seed(6)
    ids = torch.randint(0, 1000, (2, 9), generator=gen)
    h = model.transformer.embed(ids).detach()
    # unary reference first
    out_unary = model.transformer.remote(h.clone().requires_grad_(True))
    # shrink the ceiling so this payload MUST stream in multiple parts
    monkeypatch.setattr(streaming, "MAX_UNARY_PAYLOAD_BYTES", 512)
    h2 = h.clone().requires_grad_(True)
    out_stream = model.transformer.remote(h2)
    assert torch.equal(out_stream, out_unary.detach()) or torch.allclose(
        out_stream.float(), out_unary.detach().float(), atol=1e-5)
    loss = out_stream.float().square().mean()
    loss.backward()
    assert h2.grad is not None and torch.isfinite(h2.grad).all()
    assert h2.grad.abs().sum() > 0
    model.remote.manager.shutdown()


@pytest.mark.timeout(120)
def test_rebalance_recovers_orphaned_blocks():
    """Killing the only server of blocks [2,4) must make an auto-placed
    neighbor (redundantly covering [0,2)) re-host the orphaned range
    (ref server.py:479-542 rebuild loop + should_choose_other_blocks —
    VERDICT r01 missing item 6)."""
    boot = Dht()
    # s_auto placed automatically (num_blocks=2 -> lands on [0,2) in an
    # empty swarm) and allowed to rebalance; the others are pinned
    s_auto = Server(MODEL, initial_peers=[boot.endpoint], num_blocks=2,
                    device="cpu", seed=SEED, kv_max_tokens=1 << 14,
                    update_period=1.0)
    s_auto.run_in_background()
    s_pin = Server(MODEL, initial_peers=[boot.endpoint], block_indices=(0, 2),
                   device="cpu", seed=SEED, kv_max_tokens=1 << 14,
                   update_period=1.0)
    s_pin.run_in_background()
    s_tail = Server(MODEL, initial_peers=[boot.endpoint], block_indices=(2, 4),
                    device="cpu", seed=SEED, kv_max_tokens=1 << 14,
                    update_period=1.0)
    s_tail.run_in_background()
    try:
        assert tuple(s_auto.block_range) == (0, 2)
        s_tail.shutdown()
        deadline = time.time() + 60
        while time.time() < deadline and tuple(s_auto.block_range) != (2, 4):
            time.sleep(0.5)
        assert tuple(s_auto.block_range) == (2, 4), \
            "auto-placed server did not re-cover the orphaned blocks"
        # the healed swarm must serve correct tokens again
        cfg = ClientConfig(initial_peers=[boot.endpoint], min_backoff=0.1)
        model = _make_model(boot, min_backoff=0.1)
        gen = torch.Generator().manual_seed(5)
        prompt = torch.randint(0, 1000, (2, 7), generator=gen)
        out = model.generate(prompt, max_new_tokens=4)
        expect = _local_tokens(prompt, 4)
        assert torch.equal(out[:, 7:], expect)
        model.remote.manager.shutdown()
    finally:
        for s in (s_auto, s_pin):
            s.shutdown()
        boot.shutdown()


@pytest.mark.timeout(180)
def test_kv_multiplex_swarm_exact():
    """BBAMD_KV_MULTIPLEX: a micro-batched session whose KV pool cannot hold
    the full batch (only the resident window is reserved) must still decode
    token-exactly — slices cycle through pinned host snapshots while the
    previous slice computes (ref memory_cache_manager.py:944-1371)."""
    import dataclasses

    from bloombee_amd import config as bconf

    cfg0 = bconf.get_config()
    mb = dataclasses.replace(cfg0.microbatch, enabled=True,
                             micro_batch_size=4, min_batch_to_split=8,
                             kv_multiplex=True)
    bconf.set_config(dataclasses.replace(cfg0, microbatch=mb))
    boot = Dht()
    servers = []
    try:
        for rng in [(0, 2), (2, 4)]:
            # pool: 13 pages = 208 tokens. The session (16 seqs x 13
            # tokens -> 1 page each) would need 256 reserved un-multiplexed
            # (> pool, refused outright); the 12-row resident window needs
            # 192 and fits.
            s = Server(MODEL, initial_peers=[boot.endpoint],
                       block_indices=rng, device="cpu", seed=SEED,
                       kv_max_tokens=208, update_period=2.0)
            s.run_in_background()
            servers.append(s)
        model = _make_model(boot)
        gen = torch.Generator().manual_seed(5)
        prompt = torch.randint(0, 1000, (16, 7), generator=gen)
        out = model.generate(prompt, max_new_tokens=6)
        expect = _local_tokens(prompt, 6)
        assert torch.equal(out[:, 7:], expect)
        model.remote.manager.shutdown()
    finally:
        bconf.set_config(cfg0)
        for s in servers:
            s.shutdown()
        boot.shutdown()
