"""GPU serving-stack integration (no GPU test
exercised Server + handler + session — only kernels). Two workers + a client
on ONE MI355X, activations on the device data plane (mailbox delivery: all
three share the process and the GPU), greedy decode exact-match vs the GPU
LocalEngine. The graph-captured backend decode step runs on the hot path.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu

MODEL = "llama-tiny"
SEED = 0


def _local_tokens(prompt, new_tokens, device):
    from bloombee_amd.engine import LocalEngine

    eng = LocalEngine(MODEL, device=device, seed=SEED, kv_max_tokens=1 << 14)
    kv = eng.kv_pool.allocate(prompt.shape[0], 64)
    toks = [eng.prefill(prompt.to(device), kv)]
    for _ in range(new_tokens - 1):
        toks.append(eng.decode_step(toks[-1], kv))
    kv.close()
    return torch.stack([t.cpu() for t in toks], 1)


@pytest.mark.timeout(600)
def test_gpu_swarm_greedy_matches_local_engine():
    from bloombee_amd.client import ClientConfig
    from bloombee_amd.models.auto import AutoDistributedModelForCausalLM
    from bloombee_amd.net.channels import channels
    from bloombee_amd.net.dht import Dht
    from bloombee_amd.server import Server

    device = "cuda:0"
    channels.enable(device)
    boot = Dht()
    servers = []
    try:
        for rng in [(0, 2), (2, 4)]:
            s = Server(MODEL, initial_peers=[boot.endpoint],
                       block_indices=rng, device=device, seed=SEED,
                       kv_max_tokens=1 << 14, update_period=2.0)
            s.run_in_background()
            servers.append(s)
        cfg = ClientConfig(initial_peers=[boot.endpoint])
        model = AutoDistributedModelForCausalLM.from_pretrained(
            MODEL, client_config=cfg, seed=SEED, device=device)
        gen = torch.Generator().manual_seed(5)
        prompt = torch.randint(0, 1000, (2, 7), generator=gen)
        # enough steps that the backend's hipGraph capture (2 warmup steps)
        # kicks in and replays on the hot path
        out = model.generate(prompt.to(device), max_new_tokens=8)
        expect = _local_tokens(prompt, 8, device)
        assert torch.equal(out[:, 7:].cpu(), expect)
        # both servers must have captured their decode graph (lifetime
        # counter: the per-session graph state is dropped at session close)
        for s in servers:
            assert s.backend.graphs_captured > 0, \
                "backend decode was never graph-captured on the GPU path"
        model.remote.manager.shutdown()
    finally:
        for s in servers:
            s.shutdown()
        boot.shutdown()
        channels.disable()
