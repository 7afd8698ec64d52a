"""End-to-end speculative decoding with a TRAINED target/draft pair.

Random-init models have near-flat logits, so draft/target agreement is
luck and spec decoding cannot be demonstrated honestly (profiles/
r02_serving_and_kernels.md §5). This harness closes that gap within the
offline environment: a synthetic first-order "grammar" (each token has one
dominant successor + noise) is learned by BOTH a target (llama-spec-target,
~8 layers x 2048) and a small draft (llama-spec-draft, 2 x 512); agreement
then reflects model quality — the real-checkpoint structure — and the
measured spec-vs-plain speedup is a genuine single-GPU number.

    python benchmarks/spec_trained.py [--steps-train 400] [--decode 256]

The trained weights are copied into LocalEngines (the serving stack's
compute path: HIP kernels, paged KV, tree-verify kernel) for measurement.
"""
from __future__ import annotations

import argparse
import json
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

DEV = "cuda:0" if torch.cuda.is_available() else "cpu"


# ---------------------------------------------------------------------------
# synthetic grammar: token t's successor is perm[t] with prob 0.9, else
# uniform; easy to learn, near-deterministic greedy continuations
# ---------------------------------------------------------------------------
class Grammar:
    def __init__(self, vocab: int, seed: int = 0, p_follow: float = 0.9):
        g = torch.Generator().manual_seed(seed)
        self.perm = torch.randperm(vocab, generator=g)
        self.vocab = vocab
        self.p = p_follow
        self._gen = torch.Generator().manual_seed(seed + 1)

    def sample(self, B: int, T: int) -> torch.Tensor:
        out = torch.empty(B, T, dtype=torch.long)
        out[:, 0] = torch.randint(0, self.vocab, (B,), generator=self._gen)
        for t in range(1, T):
            follow = torch.rand(B, generator=self._gen) < self.p
            rand = torch.randint(0, self.vocab, (B,), generator=self._gen)
            out[:, t] = torch.where(follow, self.perm[out[:, t - 1]], rand)
        return out


class TrainableLM(torch.nn.Module):
    """Embed + BlockStack (train path) + norm + head, all trainable."""

    def __init__(self, name: str, device: str, seed: int = 0):
        super().__init__()
        from bloombee_amd.engine import BlockStack
        from bloombee_amd.models.base import resolve_config

        cfg = resolve_config(name)
        self.cfg = cfg
        self.stack = BlockStack(cfg, 0, cfg.num_hidden_layers, device=device,
                                seed=seed)
        for p in self.stack.parameters():
            p.requires_grad_(True)
        gen = torch.Generator().manual_seed(seed)
        dt = cfg.dtype
        self.embed = torch.nn.Parameter(
            (torch.randn(cfg.vocab_size, cfg.hidden_size, generator=gen) * 0.02)
            .to(dt).to(device))
        self.norm_w = torch.nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=dt, device=device))
        self.head = torch.nn.Parameter(
            (torch.randn(cfg.vocab_size, cfg.hidden_size, generator=gen) * 0.02)
            .to(dt).to(device))

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        h = F.embedding(ids, self.embed)
        h = self.stack.forward_train(h)
        hf = h.float()
        h = (hf * torch.rsqrt(hf.pow(2).mean(-1, keepdim=True)
                              + self.cfg.rms_norm_eps)).to(h.dtype) * self.norm_w
        return F.linear(h, self.head).float()


def train_lm(model: TrainableLM, grammar: Grammar, steps: int, B: int = 32,
             T: int = 64, lr: float = 3e-4, tag: str = "") -> float:
    opt = torch.optim.Adam(model.parameters(), lr=lr)
    loss = float("nan")
    for s in range(steps):
        ids = grammar.sample(B, T + 1).to(model.embed.device)
        logits = model(ids[:, :-1])
        loss = F.cross_entropy(logits.reshape(-1, grammar.vocab),
                               ids[:, 1:].reshape(-1))
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        if s % 100 == 0:
            print(f"[train {tag}] step {s}: loss {loss.item():.3f}",
                  file=sys.stderr)
    return float(loss)


@torch.no_grad()
def to_engine(model: TrainableLM, name: str, kv_tokens: int,
              quantize_q4: bool = False):
    """Copy trained weights into a LocalEngine (the HIP/paged-KV path)."""
    from bloombee_amd.engine import LocalEngine

    eng = LocalEngine(name, device=DEV, seed=0, kv_max_tokens=kv_tokens)
    eng.stack.load_state_dict(model.stack.state_dict())
    eng.embed = model.embed.detach().clone()
    eng.final_norm_w = model.norm_w.detach().clone()
    eng.lm_head_w = model.head.detach().clone()
    if quantize_q4:
        for blk in eng.stack.blocks:
            blk.quantize_weights_q4()
    return eng


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps-train", type=int, default=400)
    ap.add_argument("--decode", type=int, default=256)
    ap.add_argument("--node-budget", type=int, default=8)
    ap.add_argument("--max-depth", type=int, default=6)
    ap.add_argument("--target", default="llama-spec-target")
    ap.add_argument("--draft", default="llama-spec-draft")
    ap.add_argument("--draft-q4", action="store_true",
                    help="additionally 4-bit-quantize the trained draft")
    ap.add_argument("--swarm", action="store_true",
                    help="run the spec loop THROUGH the serving stack: the "
                         "trained target is saved to the per-block npy "
                         "layout and served by two workers; the client "
                         "drafts locally and verifies via "
                         "InferenceSession.spec_step/spec_commit")
    args = ap.parse_args()

    if args.swarm:
        return run_swarm_spec(args)

    from bloombee_amd.spec.drafter import MultiDrafter
    from bloombee_amd.spec.tree import TokenTree
    from bloombee_amd.spec.verify import verify_tree_greedy

    from bloombee_amd.models.base import resolve_config
    grammar = Grammar(resolve_config(args.target).vocab_size, seed=0)
    tgt_m = TrainableLM(args.target, DEV, seed=0)
    drf_m = TrainableLM(args.draft, DEV, seed=1)
    t0 = time.monotonic()
    lt = train_lm(tgt_m, grammar, args.steps_train, tag="target")
    ld = train_lm(drf_m, grammar, args.steps_train, tag="draft")
    train_s = time.monotonic() - t0
    print(f"[train] done in {train_s:.0f}s: target loss {lt:.3f}, "
          f"draft loss {ld:.3f}", file=sys.stderr)

    tgt = to_engine(tgt_m, args.target, 1 << 14)
    draft = to_engine(drf_m, args.draft, 1 << 13, quantize_q4=args.draft_q4)
    del tgt_m, drf_m

    prompt = grammar.sample(1, 32)
    # ---- plain greedy baseline on the target ----
    kv = tgt.kv_pool.allocate(1, 32 + args.decode + 8)
    tok = tgt.prefill(prompt.to(DEV), kv)
    for _ in range(8):
        tok = tgt.decode_step(tok, kv)
    if DEV.startswith("cuda"):
        torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(args.decode):
        tok = tgt.decode_step(tok, kv)
    if DEV.startswith("cuda"):
        torch.cuda.synchronize()
    plain_tps = args.decode / (time.monotonic() - t0)
    kv.close()

    # ---- speculative decode (tree verify on the target's HIP kernels) ----
    drafter = MultiDrafter(draft, node_budget=args.node_budget,
                           max_depth=args.max_depth, cost_ratio=0.10)
    drafter.collect_dists = False  # greedy verify never reads them
    drafter.start_session(32 + args.decode + args.node_budget + 16)
    kv = tgt.kv_pool.allocate(1, 32 + args.decode + args.node_budget + 16)
    tok = tgt.prefill(prompt.to(DEV), kv)
    history = prompt[0].tolist()
    pending = int(tok)
    emitted, rounds = 0, 0
    draft_s = verify_s = prefill_s = 0.0

    def _sync():
        if DEV.startswith("cuda"):
            torch.cuda.synchronize()

    _sync()
    t0 = time.monotonic()
    while emitted < args.decode:
        td = time.monotonic()
        sub = drafter.build_tree_incremental(torch.tensor(history + [pending]))
        _sync()
        draft_s += time.monotonic() - td
        prefill_s += getattr(drafter, "t_prefill", 0.0)
        tv = time.monotonic()
        tree = TokenTree()
        tree.add(pending, -1, 1.0)
        for i in range(len(sub)):
            tree.add(sub.tokens[i],
                     0 if sub.parents[i] == -1 else sub.parents[i] + 1,
                     sub.probs[i])
        prefix = kv.seqs[0].l_acc
        toks = tree.token_tensor().view(1, -1).to(tgt.device)
        pos = tree.position_ids(prefix).view(1, -1)
        mask = tree.attention_mask().unsqueeze(0)
        kv.rollback()
        kv.extend(len(tree), speculative=True)
        hid = tgt._embed(toks)
        sp = torch.full((1,), prefix, dtype=torch.int32, device=tgt.device)
        h = tgt.stack.forward_inference(hid, kv, sp, pos.int().to(tgt.device),
                                        tree_mask=mask.to(tgt.device))
        logits = tgt.logits_for(h[0]).float().cpu()
        acc, bonus = verify_tree_greedy(tree, logits, logits[0], start=0)
        accepted = [0] + acc
        kv.reorder_and_commit([accepted])
        emit = [tree.tokens[i] for i in accepted]
        history += emit
        emitted += len(emit)
        pending = bonus
        rounds += 1
        _sync()
        verify_s += time.monotonic() - tv
        drafter.record_result(len(acc), offered_depth=args.max_depth)
    spec_tps = emitted / (time.monotonic() - t0)
    kv.close()
    drafter.close_session()

    print(json.dumps({
        "config": "speculative decode, TRAINED pair (synthetic grammar)"
                  + (" w4-draft" if args.draft_q4 else ""),
        "target": args.target, "draft": args.draft,
        "train_steps": args.steps_train,
        "final_losses": [round(lt, 3), round(ld, 3)],
        "tokens": emitted, "rounds": rounds,
        "tokens_per_round": round(emitted / rounds, 2),
        "plain_greedy_tokens_per_s": round(plain_tps, 1),
        "spec_tokens_per_s": round(spec_tps, 1),
        "spec_speedup": round(spec_tps / plain_tps, 2),
        "draft_ms_per_round": round(draft_s / rounds * 1e3, 2),
        "draft_prefill_ms_per_round": round(prefill_s / rounds * 1e3, 2),
        "verify_ms_per_round": round(verify_s / rounds * 1e3, 2),
    }))


def run_swarm_spec(args):
    """Speculative decoding over the REAL serving stack (ROUND3 item 5):
    trained target weights are saved to the npy layout and served by two
    workers (device data plane); the client drafts with its local trained
    draft (whole-chain hipGraph) and verifies via the session's spec
    protocol (tree rides the stream, paged KV commit/rollback per span)."""
    import tempfile

    import torch.nn.functional as F

    from bloombee_amd import ops
    from bloombee_amd.client.config import ClientConfig
    from bloombee_amd.client.routing import RemoteSequenceManager
    from bloombee_amd.client.session import InferenceSession
    from bloombee_amd.models.base import resolve_config
    from bloombee_amd.net.channels import channels
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server
    from bloombee_amd.server.from_pretrained import (save_block_weights,
                                                     save_client_weights)
    from bloombee_amd.spec.drafter import MultiDrafter
    from bloombee_amd.spec.tree import TokenTree
    from bloombee_amd.spec.verify import verify_tree_greedy

    def _sync():
        if DEV.startswith("cuda"):
            torch.cuda.synchronize()

    cfg = resolve_config(args.target)
    grammar = Grammar(cfg.vocab_size, seed=0)
    tgt_m = TrainableLM(args.target, DEV, seed=0)
    drf_m = TrainableLM(args.draft, DEV, seed=1)
    train_lm(tgt_m, grammar, args.steps_train, tag="target")
    train_lm(drf_m, grammar, args.steps_train, tag="draft")

    ckpt = tempfile.mkdtemp(suffix="-np")
    for i, blk in enumerate(tgt_m.stack.blocks):
        save_block_weights(blk, ckpt, i)
    save_client_weights(ckpt, tgt_m.embed.data, tgt_m.norm_w.data,
                        tgt_m.head.data)
    embed = tgt_m.embed.data.clone()
    norm_w = tgt_m.norm_w.data.clone()
    head = tgt_m.head.data.clone()
    draft = to_engine(drf_m, args.draft, 1 << 13,
                      quantize_q4=args.draft_q4)
    del tgt_m, drf_m

    channels.enable(DEV)
    boot = Dht()
    servers = []
    try:
        L = cfg.num_hidden_layers
        for rng in [(0, L // 2), (L // 2, L)]:
            srv = Server(args.target, initial_peers=[boot.endpoint],
                         block_indices=rng, device=DEV, seed=0,
                         kv_max_tokens=1 << 14, update_period=10.0,
                         checkpoint_dir=ckpt)
            srv.run_in_background()
            servers.append(srv)
        ccfg = ClientConfig(initial_peers=[boot.endpoint],
                            keep_history=False, step_timeout=120.0)
        mgr = RemoteSequenceManager(ccfg, args.target, L)
        maxlen = 32 + args.decode + args.node_budget + 32

        def lm_head(hidden_last):
            y = ops.rms_norm(hidden_last, norm_w, cfg.rms_norm_eps)
            return F.linear(y, head).float()

        prompt = grammar.sample(1, 32)
        # ---- plain greedy over the swarm (the baseline spec must beat) --
        session = InferenceSession(mgr, max_length=maxlen, config=ccfg)
        out = session.step(F.embedding(prompt.to(DEV), embed))
        tok = lm_head(out[:, -1]).argmax(-1)
        for _ in range(4):
            out = session.step(F.embedding(tok.view(1, 1), embed))
            tok = lm_head(out[:, -1]).argmax(-1)
        _sync()
        t0 = time.monotonic()
        for _ in range(args.decode):
            out = session.step(F.embedding(tok.view(1, 1), embed))
            tok = lm_head(out[:, -1]).argmax(-1)
        _sync()
        plain_tps = args.decode / (time.monotonic() - t0)
        session.close()

        # ---- speculative over the swarm ---------------------------------
        session = InferenceSession(mgr, max_length=maxlen, config=ccfg,
                                   allow_push=False)
        drafter = MultiDrafter(draft, node_budget=args.node_budget,
                               max_depth=args.max_depth, cost_ratio=0.10)
        drafter.collect_dists = False
        drafter.start_session(maxlen)
        out = session.step(F.embedding(prompt.to(DEV), embed))
        pending = int(lm_head(out[:, -1]).argmax(-1))
        history = prompt[0].tolist()
        emitted, rounds = 0, 0
        _sync()
        t0 = time.monotonic()
        while emitted < args.decode:
            sub = drafter.build_tree_incremental(
                torch.tensor(history + [pending]))
            tree = TokenTree()
            tree.add(pending, -1, 1.0)
            for i in range(len(sub)):
                tree.add(sub.tokens[i],
                         0 if sub.parents[i] == -1 else sub.parents[i] + 1,
                         sub.probs[i])
            toks = tree.token_tensor().view(1, -1)
            pos = tree.position_ids(session.position).view(1, -1)
            mask = tree.attention_mask().unsqueeze(0)
            hid = F.embedding(toks.to(DEV), embed)
            out, _keep = session.spec_step(hid, pos, mask)
            logits = lm_head(out[0]).cpu()
            acc, bonus = verify_tree_greedy(tree, logits, logits[0], start=0)
            accepted = [0] + acc
            session.spec_commit([accepted])
            emit = [tree.tokens[i] for i in accepted]
            history += emit
            emitted += len(emit)
            pending = bonus
            rounds += 1
            drafter.record_result(len(acc), offered_depth=args.max_depth)
        _sync()
        spec_tps = emitted / (time.monotonic() - t0)
        session.close()
        drafter.close_session()
        mgr.shutdown()
        print(json.dumps({
            "config": "speculative decode OVER THE SWARM (trained pair, "
                      "2 workers, device data plane)",
            "tokens": emitted, "rounds": rounds,
            "tokens_per_round": round(emitted / rounds, 2),
            "plain_swarm_tokens_per_s": round(plain_tps, 1),
            "spec_swarm_tokens_per_s": round(spec_tps, 1),
            "spec_speedup": round(spec_tps / plain_tps, 2),
        }))
    finally:
        for srv in servers:
            srv.shutdown()
        boot.shutdown()
        channels.disable()


if __name__ == "__main__":
    main()
