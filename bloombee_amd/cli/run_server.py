"""Worker entrypoint (parity: reference cli/run_server.py).

    python -m bloombee_amd.cli.run_server llama-3-8b \
        --initial-peers 127.0.0.1:31337 --block-indices 0:16 --device cuda:0
"""
from __future__ import annotations

import argparse

from bloombee_amd.server import Server
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


def parse_endpoint(s: str):
    host, port = s.rsplit(":", 1)
    return host, int(port)


def _offload_policy(args):
    if (args.weight_gpu_percent >= 100.0 and not args.compress_weight
            and args.attn_sparsity >= 1.0):
        return None
    from bloombee_amd.offload import OffloadPolicy
    return OffloadPolicy(weight_gpu_percent=args.weight_gpu_percent,
                         compress_weight=args.compress_weight,
                         attn_sparsity=args.attn_sparsity)


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("model", help="model preset name or local checkpoint dir")
    ap.add_argument("--initial-peers", nargs="*", default=[])
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=0)
    ap.add_argument("--dht-port", type=int, default=0)
    ap.add_argument("--block-indices", default=None,
                    help="start:end block range (default: auto-select)")
    ap.add_argument("--num-blocks", type=int, default=None)
    ap.add_argument("--device", default=None,
                    help="cuda:0 / cpu (default: cuda if available)")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--attn-cache-tokens", type=int, default=1 << 18)
    ap.add_argument("--update-period", type=float, default=30.0)
    ap.add_argument("--checkpoint-dir", default=None,
                    help="per-block .npy weight dir (random init if absent)")
    ap.add_argument("--network-rps", type=float, default=None,
                    help="advertised NIC steps/s cap folded into throughput "
                         "(ref speedtest-based network_rps; offline -> operator-provided)")
    ap.add_argument("--throughput", type=float, default=None,
                    help="announced rps (default: measured)")
    ap.add_argument("--session-max-idle", type=float, default=600.0,
                    help="reap inference sessions idle past this (s)")
    ap.add_argument("--max-chunk-tokens", type=int, default=None,
                    help="prefill sequence-chunk bound (BBAMD_MAX_CHUNK_TOKENS)")
    ap.add_argument("--weight-gpu-percent", type=float, default=100.0,
                    help="FlexGen-style offload: %% of blocks resident in HBM")
    ap.add_argument("--compress-weight", action="store_true",
                    help="4-bit group-quantize the host weight tier")
    ap.add_argument("--attn-sparsity", type=float, default=1.0,
                    help="top-k sparse decode attention fraction (1.0 = dense)")
    ap.add_argument("--adapters", nargs="*", default=[], metavar="NAME=DIR",
                    help="preload LoRA adapters: per-block subdirs block{i}/")
    ap.add_argument("--announce-host", default=None,
                    help="advertise this address instead of the bind host "
                         "(NAT, ref --announce_maddrs)")
    ap.add_argument("--announce-port", type=int, default=None)
    ap.add_argument("--identity-path", default=None,
                    help="persistent peer identity file (ref --identity_path)")
    ap.add_argument("--max-batch-size", type=int, default=2048,
                    help="max sequences per inference session")
    ap.add_argument("--torch-dtype", default=None,
                    choices=["bfloat16", "float32"],
                    help="serve dtype (kernels are bf16; float32 = CPU path)")
    ap.add_argument("--quant-type", default="none",
                    choices=["none", "nf4"],
                    help="nf4 maps to the 4-bit host weight tier "
                         "(compress_weight); LLM.int8 is not supported")
    args = ap.parse_args()

    if args.max_chunk_tokens is not None:
        import os
        os.environ["BBAMD_MAX_CHUNK_TOKENS"] = str(args.max_chunk_tokens)
    if args.quant_type == "nf4":
        args.compress_weight = True

    import torch

    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    if args.torch_dtype:
        import os
        os.environ["BBAMD_TORCH_DTYPE"] = args.torch_dtype
    block_indices = None
    if args.block_indices:
        a, b = args.block_indices.split(":")
        block_indices = (int(a), int(b))

    throughput = args.throughput
    if throughput is None:
        from bloombee_amd.models.base import resolve_config
        from bloombee_amd.server.throughput import get_server_throughput

        cfg = resolve_config(args.model)
        nb = (block_indices[1] - block_indices[0]) if block_indices \
            else (args.num_blocks or cfg.num_hidden_layers)
        throughput = get_server_throughput(
            cfg, device, nb, network_rps=args.network_rps)["throughput"]

    server = Server(
        args.model,
        initial_peers=[parse_endpoint(p) for p in args.initial_peers],
        host=args.host, port=args.port, dht_port=args.dht_port,
        block_indices=block_indices, num_blocks=args.num_blocks,
        device=device, seed=args.seed, kv_max_tokens=args.attn_cache_tokens,
        update_period=args.update_period, checkpoint_dir=args.checkpoint_dir,
        throughput=throughput, network_rps=args.network_rps,
        session_max_idle=args.session_max_idle,
        offload_policy=_offload_policy(args),
        adapters=dict(a.split("=", 1) for a in args.adapters) or None,
        identity_path=args.identity_path,
        max_batch_size=args.max_batch_size,
        announce_host=args.announce_host,
        announce_port=args.announce_port,
    )
    server.run()


if __name__ == "__main__":
    main()
