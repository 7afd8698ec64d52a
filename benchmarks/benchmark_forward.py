"""Swarm forward-pass benchmark (parity: reference benchmarks/
benchmark_forward.py — tokens/sec of full-sequence rpc_forward chains)."""
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
    ap.add_argument("--batch", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    cfg = ClientConfig(initial_peers=[parse_endpoint(p) for p in args.initial_peers])
    model = AutoDistributedModelForCausalLM.from_pretrained(
        args.model, client_config=cfg, seed=args.seed)
    V = model.config.vocab_size
    ids = torch.randint(0, V, (args.batch, args.seq_len),
                        generator=torch.Generator().manual_seed(args.seed))
    with torch.no_grad():
        model(ids)  # warmup
        t0 = time.monotonic()
        for _ in range(args.steps):
            model(ids)
        dt = time.monotonic() - t0
    toks = args.steps * args.batch * args.seq_len
    print(f"forward: {toks / dt:.2f} tokens/sec ({args.steps} steps, "
          f"batch {args.batch} x {args.seq_len})")


if __name__ == "__main__":
    main()
