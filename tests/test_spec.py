"""Speculative decoding tests (mirror reference test_spe_dec_tree.py,
test_spec_decoding_verify.py, test_spec_decoding_tree_shape.py,
test_speculative_pruner_manager.py, test_speculative_generation.py)."""
import pytest
import torch

from bloombee_amd.spec.pruner import (MidLMHead, PruningMethod,
                                      SimpleProbabilityPruner, create_pruner)
from bloombee_amd.spec.shape import AcceptanceStats, plan_tree_shape
from bloombee_amd.spec.tree import TokenTree
from bloombee_amd.spec.verify import verify_tree_greedy, verify_tree_sampling


def _chain_logits(V, mapping):
    """logits rows with argmax forced per position."""
    g = torch.zeros(len(mapping), V)
    for i, tok in enumerate(mapping):
        g[i, tok] = 10.0
    return g


def test_tree_structure_and_mask():
    t = TokenTree()
    a = t.add(5, -1)
    b = t.add(6, a)
    c = t.add(7, a)
    d = t.add(8, b)
    assert t.depth(d) == 2 and t.depth(a) == 0
    assert t.children(a) == [b, c]
    m = t.attention_mask()
    assert m[d, a] and m[d, b] and not m[d, c]
    assert (m.diagonal()).all()
    pos = t.position_ids(10)
    assert pos.tolist() == [10, 11, 11, 12]
    assert t.path_to(d) == [a, b, d]


def test_verify_greedy_full_path():
    # tree: root 5 -> {6, 9}; 6 -> 8
    t = TokenTree()
    r = t.add(5, -1)
    x = t.add(6, r)
    t.add(9, r)
    y = t.add(8, x)
    V = 16
    prefix = torch.zeros(V)
    prefix[5] = 9.0                       # target wants 5 -> accept root
    logits = torch.zeros(len(t), V)
    logits[r, 6] = 9.0                    # after 5, wants 6 -> accept x
    logits[x, 8] = 9.0                    # after 6, wants 8 -> accept y
    logits[y, 3] = 9.0                    # after 8, wants 3 -> bonus
    acc, bonus = verify_tree_greedy(t, logits, prefix)
    assert acc == [r, x, y] and bonus == 3


def test_verify_greedy_mismatch_is_bonus():
    t = TokenTree.chain([5, 6])
    V = 8
    prefix = torch.zeros(V)
    prefix[7] = 5.0                       # wants 7, tree offers 5
    logits = torch.zeros(len(t), V)
    acc, bonus = verify_tree_greedy(t, logits, prefix)
    assert acc == [] and bonus == 7


def test_verify_sampling_peaked_matches_greedy():
    gen = torch.Generator().manual_seed(0)
    t = TokenTree.chain([4, 2], probs=[0.9, 0.9])
    V = 8
    prefix = torch.full((V,), -20.0)
    prefix[4] = 20.0
    logits = torch.full((len(t), V), -20.0)
    logits[0, 2] = 20.0
    logits[1, 1] = 20.0
    acc, bonus = verify_tree_sampling(t, logits, prefix, generator=gen)
    assert acc == [0, 1] and bonus == 1


def test_tree_shape_budget():
    stats = AcceptanceStats()
    widths = plan_tree_shape(stats, budget=8, max_depth=5, max_width=3)
    assert sum(widths) <= 8 and widths[0] >= 1
    # high acceptance should grow depth
    for _ in range(50):
        stats.record(accepted_len=5, offered_depth=5)
    deep = plan_tree_shape(stats, budget=8, max_depth=5, max_width=3)
    assert len(deep) >= len(widths)


def test_probability_pruner_keeps_ancestors():
    head = MidLMHead(hidden_size=16, vocab_size=32, dtype=torch.float32)
    pruner = SimpleProbabilityPruner(head, threshold=2.0)  # prune everything
    tokens = [1, 2, 3]
    parents = [-1, 0, 1]
    keep = pruner.keep_indices(torch.randn(3, 16), tokens, parents)
    assert keep == [0]  # roots always survive
    assert create_pruner(PruningMethod.NONE, head) is None


def test_adaptive_pruner_trains():
    from bloombee_amd.spec.pruner import AdaptiveNeuralPruner

    head = MidLMHead(hidden_size=16, vocab_size=32, dtype=torch.float32)
    p = AdaptiveNeuralPruner(head)
    keep = p.keep_indices(torch.randn(4, 16), [1, 2, 3, 4], [-1, 0, 0, 1])
    assert 0 in keep
    loss = p.train_step(accepted=[0, 1])
    assert loss > 0


def test_paged_reorder_and_commit():
    from bloombee_amd.kv.paged import PagedKVCache
    from bloombee_amd.ops import reference as ref

    pool = PagedKVCache(num_layers=1, num_kv_heads=2, head_dim=8,
                        page_size=4, max_tokens=64)
    h = pool.allocate(1, 32)
    # committed prefix of 3 tokens + 5 speculative
    k = torch.randn(1, 2, 8, 8).to(torch.bfloat16)
    v = torch.randn(1, 2, 8, 8).to(torch.bfloat16)
    h.extend(3)
    ref.kv_write(k[:, :, :3], v[:, :, :3], h.k_pages(0), h.v_pages(0),
                 h.page_table(), torch.zeros(1, dtype=torch.int32))
    h.extend(5, speculative=True)
    ref.kv_write(k[:, :, 3:], v[:, :, 3:], h.k_pages(0), h.v_pages(0),
                 h.page_table(), torch.tensor([3], dtype=torch.int32))
    # accept spec nodes 1 and 3 (abs positions 4 and 6)
    h.reorder_and_commit([[1, 3]])
    assert h.seqs[0].l_acc == 5 and h.seqs[0].l_spec == 5
    kg, vg = ref.kv_gather(h.k_pages(0), h.v_pages(0), h.page_table(), 5, 0)
    want_k = torch.cat([k[0, :, :3], k[0, :, 4:5], k[0, :, 6:7]], dim=1)
    want_v = torch.cat([v[0, :, :3], v[0, :, 4:5], v[0, :, 6:7]], dim=1)
    assert torch.equal(kg, want_k)
    assert torch.equal(vg, want_v)
    h.close()


def test_speculative_swarm_matches_greedy():
    """End-to-end: greedy speculative generation over a 2-server loopback
    swarm must emit EXACTLY the plain greedy tokens (ref
    test_speculative_generation.py invariant)."""
    from bloombee_amd.client import ClientConfig
    from bloombee_amd.engine import LocalEngine
    from bloombee_amd.models.llama.speculative import \
        DistributedLlamaForSpeculativeGeneration
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server

    boot = Dht()
    s1 = Server("llama-tiny", initial_peers=[boot.endpoint], block_indices=(0, 2),
                device="cpu", seed=0, kv_max_tokens=1 << 14, update_period=5.0)
    s2 = Server("llama-tiny", initial_peers=[boot.endpoint], block_indices=(2, 4),
                device="cpu", seed=0, kv_max_tokens=1 << 14, update_period=5.0)
    s1.run_in_background()
    s2.run_in_background()
    try:
        cfg = ClientConfig(initial_peers=[boot.endpoint])
        model = DistributedLlamaForSpeculativeGeneration.from_pretrained(
            "llama-tiny", client_config=cfg, seed=0)
        # draft model: DIFFERENT seed — realistic partial-agreement drafting
        draft = LocalEngine("llama-tiny", device="cpu", seed=7,
                            kv_max_tokens=1 << 13)
        model.set_drafter(draft, node_budget=6, max_depth=3)

        gen = torch.Generator().manual_seed(5)
        prompt = torch.randint(0, 1000, (1, 7), generator=gen)
        out = model.generate_speculative(prompt, max_new_tokens=8)

        eng = LocalEngine("llama-tiny", device="cpu", seed=0,
                          kv_max_tokens=1 << 14)
        expect = eng.generate_greedy(prompt, 8)
        assert torch.equal(out, expect), (out, expect)
        model.remote.manager.shutdown()
    finally:
        s1.shutdown()
        s2.shutdown()
        boot.shutdown()


def test_speculative_with_pruner_still_greedy(monkeypatch):
    """Mid-network pruning at the last block: pruned subtrees must only
    REDUCE acceptance, never change emitted tokens (greedy invariant holds
    because pruned nodes are simply never accepted)."""
    import os

    from bloombee_amd.client import ClientConfig
    from bloombee_amd.engine import LocalEngine
    from bloombee_amd.models.llama.speculative import \
        DistributedLlamaForSpeculativeGeneration
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server

    monkeypatch.setenv("BBAMD_SPEC_PRUNING", "probability")
    boot = Dht()
    s1 = Server("llama-tiny", initial_peers=[boot.endpoint], block_indices=(0, 2),
                device="cpu", seed=0, kv_max_tokens=1 << 14, update_period=5.0)
    s2 = Server("llama-tiny", initial_peers=[boot.endpoint], block_indices=(2, 4),
                device="cpu", seed=0, kv_max_tokens=1 << 14, update_period=5.0)
    assert s2.backend.pruner is not None  # last-block server got the pruner
    s1.run_in_background()
    s2.run_in_background()
    try:
        cfg = ClientConfig(initial_peers=[boot.endpoint])
        model = DistributedLlamaForSpeculativeGeneration.from_pretrained(
            "llama-tiny", client_config=cfg, seed=0)
        draft = LocalEngine("llama-tiny", device="cpu", seed=7,
                            kv_max_tokens=1 << 13)
        model.set_drafter(draft, node_budget=6, max_depth=3)
        gen = torch.Generator().manual_seed(5)
        prompt = torch.randint(0, 1000, (1, 7), generator=gen)
        out = model.generate_speculative(prompt, max_new_tokens=8)
        eng = LocalEngine("llama-tiny", device="cpu", seed=0,
                          kv_max_tokens=1 << 14)
        expect = eng.generate_greedy(prompt, 8)
        assert torch.equal(out, expect), (out, expect)
        model.remote.manager.shutdown()
    finally:
        s1.shutdown()
        s2.shutdown()
        boot.shutdown()


def test_pruner_checkpoint_roundtrip(tmp_path):
    """save_pruner/load_pruner restore head + scorer + optimizer state
    (ref speculative_pruner/lm_head_trainer.py checkpoints)."""
    import torch

    from bloombee_amd.spec.pruner import (AdaptiveNeuralPruner, MidLMHead,
                                          load_pruner, save_pruner)

    head = MidLMHead(32, 100, seed=1, dtype=torch.float32)
    p = AdaptiveNeuralPruner(head)
    hidden = torch.randn(5, 32)
    p.keep_indices(hidden, [1, 2, 3, 4, 5], [-1, 0, 0, 1, 1])
    p.train_step([0, 1])
    path = str(tmp_path / "pruner.pt")
    save_pruner(p, path)

    head2 = MidLMHead(32, 100, seed=7, dtype=torch.float32)  # different init
    p2 = AdaptiveNeuralPruner(head2)
    load_pruner(p2, path)
    for a, b in zip(p.net.parameters(), p2.net.parameters()):
        assert torch.equal(a, b)
    assert torch.equal(p.head.weight, p2.head.weight)
    k1 = p.keep_indices(hidden, [1, 2, 3, 4, 5], [-1, 0, 0, 1, 1])
    k2 = p2.keep_indices(hidden, [1, 2, 3, 4, 5], [-1, 0, 0, 1, 1])
    assert k1 == k2


def test_speculative_survives_injected_faults():
    """Spec sessions must failover too: with injected stream faults the
    tree rounds retry on rebuilt chains (history now covers committed
    spec tokens) and the greedy invariant still holds exactly."""
    from bloombee_amd.client import ClientConfig
    from bloombee_amd.engine import LocalEngine
    from bloombee_amd.models.llama.speculative import \
        DistributedLlamaForSpeculativeGeneration
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server
    from bloombee_amd.utils import fault_injection as fi

    boot = Dht()
    s1 = Server("llama-tiny", initial_peers=[boot.endpoint],
                block_indices=(0, 2), device="cpu", seed=0,
                kv_max_tokens=1 << 14, update_period=2.0)
    s2 = Server("llama-tiny", initial_peers=[boot.endpoint],
                block_indices=(2, 4), device="cpu", seed=0,
                kv_max_tokens=1 << 14, update_period=2.0)
    s1.run_in_background()
    s2.run_in_background()
    try:
        cfg = ClientConfig(initial_peers=[boot.endpoint], min_backoff=0.1,
                           max_retries=None, step_timeout=5.0)
        model = DistributedLlamaForSpeculativeGeneration.from_pretrained(
            "llama-tiny", client_config=cfg, seed=0)
        draft = LocalEngine("llama-tiny", device="cpu", seed=7,
                            kv_max_tokens=1 << 13)
        model.set_drafter(draft, node_budget=6, max_depth=3)
        gen = torch.Generator().manual_seed(5)
        prompt = torch.randint(0, 1000, (1, 7), generator=gen)
        fi.configure(0.25, seed=11, max_faults=3,
                     methods=("inference_item", "open_inference", "spec_item"))
        try:
            out = model.generate_speculative(prompt, max_new_tokens=8)
        finally:
            injected = fi.injected
            fi.configure(0.0)
        eng = LocalEngine("llama-tiny", device="cpu", seed=0,
                          kv_max_tokens=1 << 14)
        expect = eng.generate_greedy(prompt, 8)
        assert torch.equal(out, expect), (out, expect, injected)
        assert injected >= 1
        model.remote.manager.shutdown()
    finally:
        s1.shutdown()
        s2.shutdown()
        boot.shutdown()
