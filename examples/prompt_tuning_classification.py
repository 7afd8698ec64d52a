"""Prompt-tune a classifier over a swarm — the reference's
prompt-tuning-sst2.ipynb workflow, self-contained and offline: spins up a
2-worker loopback swarm on random-init llama-tiny, then trains shallow +
deep prompts and a classification head on a synthetic 2-class task while
the transformer blocks stay frozen on the servers.

    python examples/prompt_tuning_classification.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from bloombee_amd.client import ClientConfig  # noqa: E402
from bloombee_amd.models.auto import \
    AutoDistributedModelForSequenceClassification  # noqa: E402
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
        model = AutoDistributedModelForSequenceClassification.from_pretrained(
            "llama-tiny",
            client_config=ClientConfig(initial_peers=[boot.endpoint]),
            seed=0, num_labels=2, pre_seq_len=8, deep_ptune=True)
        opt = torch.optim.Adam(model.trainable_parameters(), lr=3e-2)

        # synthetic task: class = whether the first token is < 500
        gen = torch.Generator().manual_seed(0)
        ids = torch.randint(0, 1000, (32, 8), generator=gen)
        labels = (ids[:, 0] < 500).long()

        for step in range(12):
            logits = model(ids)
            loss = torch.nn.functional.cross_entropy(logits, labels)
            acc = (logits.argmax(-1) == labels).float().mean()
            opt.zero_grad()
            loss.backward()
            opt.step()
            print(f"step {step:2d}  loss {float(loss):.4f}  acc {float(acc):.2f}")
        model.remote.manager.shutdown()
    finally:
        for s in servers:
            s.shutdown()
        boot.shutdown()


if __name__ == "__main__":
    main()
