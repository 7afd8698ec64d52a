"""Driver benchmark contract: flagship decode throughput.

Measures BASELINE.json's metric — decode tokens/sec (+ p50 step latency) for
Llama-3-8B greedy decode, sharded over N MI355X workers as a micro-batched
pipeline over RCCL/xGMI (bloombee_amd/parallel/pipeline.py). Synthetic data,
random-init weights (no network in this environment), bf16.

Weak scaling: global_batch = batch_per_gpu * N (per-GPU work fixed as N grows).

  python bench.py --gpus N --steps K --warmup W
  (N>1 is launched by the driver via torch.distributed.run, one rank per GPU)
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch


def stage_max_pos(model: str) -> int:
    from bloombee_amd.models.base import resolve_config

    return resolve_config(model).max_position_embeddings


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=64)
    ap.add_argument("--warmup", type=int, default=16)
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--batch-per-gpu", type=int, default=32)
    ap.add_argument("--prompt", type=int, default=2048,
                    help="prompt length prefillled before the timed decode steps")
    ap.add_argument("--device", default=None, help="override (e.g. cpu for tests)")
    ap.add_argument("--micro-batches", type=int, default=0)
    ap.add_argument("--tp", type=int, default=1,
                    help="tensor-parallel degree within the pipeline (world %% tp == 0)")
    args = ap.parse_args()

    import torch.distributed as dist
    from bloombee_amd.parallel.pipeline import PipelineStage, init_distributed

    device = init_distributed("cpu" if args.device == "cpu" else "auto")
    if args.device:
        device = args.device
    rank = dist.get_rank() if dist.is_initialized() else 0
    world = dist.get_world_size() if dist.is_initialized() else 1
    assert world == args.gpus or not dist.is_initialized(), \
        f"world {world} != --gpus {args.gpus}"
    on_gpu = device.startswith("cuda")

    global_batch = args.batch_per_gpu * world
    session_len = args.prompt + args.steps + args.warmup + 8
    assert session_len <= stage_max_pos(args.model), (
        f"prompt+steps {session_len} exceeds the model's "
        f"max_position_embeddings — shorten --prompt or --steps")
    # KV budget: whole session for every sequence on every rank (page-rounded)
    kv_tokens = global_batch * ((session_len + 15) // 16 + 1) * 16 + 1024

    stage = PipelineStage(args.model, device, global_batch,
                          micro_batches=args.micro_batches,
                          kv_max_tokens=kv_tokens, max_session_len=session_len,
                          tp=args.tp)

    gen = torch.Generator().manual_seed(42)
    V = stage.config.vocab_size
    ids = None
    if rank == 0:
        prompt = torch.randint(0, V, (global_batch, args.prompt), generator=gen)
    # ---- prefill (untimed setup; chunked to bound activation memory) ----
    CH = 512
    for c0 in range(0, args.prompt, CH):
        chunk = None
        if rank == 0:
            chunk = prompt[:, c0:c0 + CH]
        ids = stage.prefill_round(chunk, min(CH, args.prompt - c0))

    # ---- warmup ----
    for _ in range(args.warmup):
        ids = stage.decode_round(ids if rank == 0 else None)

    if dist.is_initialized():
        dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()

    step_ms = []
    t0 = time.monotonic()
    if on_gpu and rank == 0:
        ev = [torch.cuda.Event(enable_timing=True) for _ in range(args.steps + 1)]
        ev[0].record()
    for s in range(args.steps):
        ids = stage.decode_round(ids if rank == 0 else None)
        if on_gpu and rank == 0:
            ev[s + 1].record()
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t1 = time.monotonic()
    elapsed = t1 - t0
    if dist.is_initialized():
        e = torch.tensor([elapsed])
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e)

    if rank == 0:
        if on_gpu:
            step_ms = [ev[i].elapsed_time(ev[i + 1]) for i in range(args.steps)]
        else:
            step_ms = [elapsed / args.steps * 1000] * args.steps
        p50 = sorted(step_ms)[len(step_ms) // 2]
        tokens_per_s = global_batch * args.steps / elapsed
        result = {
            "metric": "decode tokens/sec, Llama-3-8B greedy, pipeline over RCCL/xGMI",
            "value": round(tokens_per_s, 2),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed * 1000 / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bfloat16" if stage.config.torch_dtype == "bfloat16" else stage.config.torch_dtype,
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "seq_len": args.prompt,
                "parallelism": (f"pp{world // args.tp}xtp{args.tp}" if args.tp > 1
                                else f"pp{world}"),
                "micro_batches": stage.M,
                "p50_step_ms": round(p50, 3),
            },
        }
        print(json.dumps(result))
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
