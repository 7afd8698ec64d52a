"""Swarm fine-tuning benchmark (parity: reference benchmarks/
benchmark_training.py:17-31 — fwd+bwd steps/s with p-tuned prompts, server
blocks frozen; seq_len 128, pre_seq_len 16 defaults)."""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from bloombee_amd.client import ClientConfig  # noqa: E402
from bloombee_amd.models.auto import AutoDistributedModelForCausalLM  # noqa: E402
from benchmarks.benchmark_inference import parse_endpoint  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", required=True)
    ap.add_argument("--initial-peers", nargs="+", required=True)
    ap.add_argument("--seq-len", type=int, default=128)
    ap.add_argument("--pre-seq-len", type=int, default=16)
    ap.add_argument("--batch", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--lr", type=float, default=1e-3)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    cfg = ClientConfig(initial_peers=[parse_endpoint(p) for p in args.initial_peers])
    model = AutoDistributedModelForCausalLM.from_pretrained(
        args.model, client_config=cfg, seed=args.seed,
        pre_seq_len=args.pre_seq_len)
    opt = torch.optim.Adam([model.transformer.prompt_embeds], lr=args.lr)
    V = model.config.vocab_size
    ids = torch.randint(0, V, (args.batch, args.seq_len),
                        generator=torch.Generator().manual_seed(args.seed))

    def step():
        opt.zero_grad()
        logits = model(ids)
        loss = torch.nn.functional.cross_entropy(
            logits[:, args.pre_seq_len:-1].reshape(-1, V).float(),
            ids[:, 1:].reshape(-1))
        loss.backward()
        opt.step()
        return float(loss)

    step()  # warmup
    t0 = time.monotonic()
    for _ in range(args.steps):
        loss = step()
    dt = time.monotonic() - t0
    print(f"training: {args.steps / dt:.3f} steps/sec (last loss {loss:.3f})")


if __name__ == "__main__":
    main()
