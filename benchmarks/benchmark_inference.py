"""Swarm inference benchmark (parity: reference
benchmarks/benchmark_inference.py:27-75 — tokens/sec over an InferenceSession
against a running swarm; defaults seq_len 2048, warmup 1).

    python benchmarks/benchmark_inference.py --model llama-tiny \
        --initial-peers 127.0.0.1:31337 --seq-len 128 --batch 1
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from bloombee_amd.client import ClientConfig  # noqa: E402
from bloombee_amd.models.auto import AutoDistributedModelForCausalLM  # noqa: E402


def parse_endpoint(s):
    h, p = s.rsplit(":", 1)
    return h, int(p)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", required=True)
    ap.add_argument("--initial-peers", nargs="+", required=True)
    ap.add_argument("--seq-len", type=int, default=2048)
    ap.add_argument("--prompt-len", type=int, default=16)
    ap.add_argument("--batch", type=int, default=1)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--client-device", default="cpu",
                    help="device for the client-side embed/LM-head (ref "
                         "README: use cuda for 150k+ vocab families)")
    args = ap.parse_args()

    cfg = ClientConfig(initial_peers=[parse_endpoint(p) for p in args.initial_peers])
    model = AutoDistributedModelForCausalLM.from_pretrained(
        args.model, client_config=cfg, seed=args.seed,
        device=args.client_device)
    V = model.config.vocab_size
    gen = torch.Generator().manual_seed(args.seed)
    prompt = torch.randint(0, V, (args.batch, args.prompt_len), generator=gen)
    new_tokens = args.seq_len - args.prompt_len

    for _ in range(args.warmup):
        model.generate(prompt, max_new_tokens=4)
    t0 = time.monotonic()
    out = model.generate(prompt, max_new_tokens=new_tokens)
    dt = time.monotonic() - t0
    toks = args.batch * new_tokens
    print(f"inference: {toks / dt:.2f} tokens/sec "
          f"({toks} tokens in {dt:.2f}s, batch {args.batch})")


if __name__ == "__main__":
    main()
