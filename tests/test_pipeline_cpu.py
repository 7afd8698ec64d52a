"""Distributed pipeline decode (gloo, world_size 2, loopback) must produce
exactly the tokens of the single-process LocalEngine — the multi-GPU RCCL
path is the same code with backend nccl (driver runs it at round end)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from bloombee_amd.engine import LocalEngine

def _free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank, world, q, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from bloombee_amd.parallel.pipeline import PipelineStage

        stage = PipelineStage("llama-tiny", "cpu", global_batch=4,
                              micro_batches=2, seed=0, kv_max_tokens=8192,
                              max_session_len=64)
        gen = torch.Generator().manual_seed(7)
        prompt = torch.randint(0, 1000, (4, 10), generator=gen)
        ids = stage.prefill_round(prompt if rank == 0 else None, 10)
        toks = []
        for _ in range(5):
            ids = stage.decode_round(ids if rank == 0 else None)
            if rank == 0:
                toks.append(ids.clone())
        if rank == 0:
            q.put(torch.stack(toks, 1))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_pipeline_matches_local_engine():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [ctx.Process(target=_worker, args=(r, 2, q, port)) for r in range(2)]
    for p in procs:
        p.start()
    got = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    eng = LocalEngine("llama-tiny", device="cpu", seed=0, kv_max_tokens=8192)
    gen = torch.Generator().manual_seed(7)
    prompt = torch.randint(0, 1000, (4, 10), generator=gen)
    kv = eng.kv_pool.allocate(4, 64)
    ids = eng.prefill(prompt, kv)
    expect = []
    for _ in range(5):
        ids = eng.decode_step(ids, kv)
        expect.append(ids.clone())
    kv.close()
    expect = torch.stack(expect, 1)
    assert torch.equal(got, expect), f"pipeline tokens {got} != local {expect}"


@pytest.mark.timeout(120)
def test_single_rank_pipeline_matches_local():
    """world=1 PipelineStage (no dist) == LocalEngine."""
    from bloombee_amd.parallel.pipeline import PipelineStage

    stage = PipelineStage("llama-tiny", "cpu", global_batch=4, seed=0,
                          kv_max_tokens=8192, max_session_len=64)
    gen = torch.Generator().manual_seed(7)
    prompt = torch.randint(0, 1000, (4, 10), generator=gen)
    ids = stage.prefill_round(prompt, 10)
    toks = [ids]

    eng = LocalEngine("llama-tiny", device="cpu", seed=0, kv_max_tokens=8192)
    kv = eng.kv_pool.allocate(4, 64)
    ids2 = eng.prefill(prompt, kv)
    assert torch.equal(toks[0], ids2)
    for _ in range(3):
        ids = stage.decode_round(ids)
        ids2 = eng.decode_step(ids2, kv)
        assert torch.equal(ids, ids2)
    kv.close()
