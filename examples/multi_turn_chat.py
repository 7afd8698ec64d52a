"""Multi-turn generation with a reused inference session — the reference's
session-reuse pattern (README Python API): past tokens stay in the swarm's
KV cache across .generate() calls.

    python examples/multi_turn_chat.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from bloombee_amd.client import ClientConfig  # noqa: E402
from bloombee_amd.models.auto import AutoDistributedModelForCausalLM  # noqa: E402
from bloombee_amd.net.dht import Dht  # noqa: E402
from bloombee_amd.server import Server  # noqa: E402


def main():
    boot = Dht()
    servers = [
        Server("llama-tiny", initial_peers=[boot.endpoint],
               block_indices=(0, 2), device="cpu", seed=0,
               kv_max_tokens=1 << 14),
        Server("llama-tiny", initial_peers=[boot.endpoint],
               block_indices=(2, 4), device="cpu", seed=0,
               kv_max_tokens=1 << 14),
    ]
    for s in servers:
        s.run_in_background()
    try:
        model = AutoDistributedModelForCausalLM.from_pretrained(
            "llama-tiny",
            client_config=ClientConfig(initial_peers=[boot.endpoint]), seed=0)
        gen = torch.Generator().manual_seed(0)
        with model.remote.inference_session(max_length=128) as sess:
            for turn in range(3):
                user = torch.randint(0, 1000, (1, 6), generator=gen)
                out = model.generate(user, max_new_tokens=8, session=sess)
                print(f"turn {turn}: in={user.tolist()[0]} "
                      f"out={out[:, 6:].tolist()[0]} "
                      f"(cache at {sess.position} tokens)")
        model.remote.manager.shutdown()
    finally:
        for s in servers:
            s.shutdown()
        boot.shutdown()


if __name__ == "__main__":
    main()
