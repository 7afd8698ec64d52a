"""Swarm health monitor (parity: the reference's health website + DHT
models registry, server.py:978-984 — here a CLI that joins the DHT, walks
the block announcements and prints per-block coverage and per-server
state, plus measured RTTs).

    python -m bloombee_amd.cli.health llama-3-8b \
        --initial-peers 127.0.0.1:31337
"""
from __future__ import annotations

import argparse
from typing import Optional

from bloombee_amd.data_structures import (compute_spans,
                                          get_remote_module_infos,
                                          module_uids)
from bloombee_amd.models.base import resolve_config
from bloombee_amd.net.dht import Dht
from bloombee_amd.utils.ping import PingAggregator


def swarm_health(model: str, initial_peers, num_blocks: Optional[int] = None,
                 dht: Optional[Dht] = None, ping: bool = True) -> dict:
    """-> {"blocks": [n_servers per block], "servers": {peer: {...}},
    "complete": bool}. Importable (tests); the CLI prints it."""
    cfg = resolve_config(model)
    nb = num_blocks or cfg.num_hidden_layers
    own = dht is None
    node = dht or Dht(initial_peers=list(initial_peers))
    try:
        infos = get_remote_module_infos(node, module_uids(model, nb))
        spans = compute_spans(infos)
        coverage = [len(i.servers) for i in infos]
        servers = {}
        rtts = {}
        if ping and spans:
            agg = PingAggregator(timeout=2.0)
            eps = {(s.server_info.host, s.server_info.port)
                   for s in spans.values()}
            rtts = agg.ping_many(sorted(eps))
        for peer, span in spans.items():
            si = span.server_info
            servers[peer] = {
                "blocks": f"{span.start}:{span.end}",
                "host": f"{si.host}:{si.port}",
                "state": si.state.name,
                "throughput": si.throughput,
                "cache_tokens_left": si.cache_tokens_left,
                "rtt_ms": (round(rtts.get((si.host, si.port), float("inf")) * 1e3, 2)
                           if rtts else None),
                "version": si.version,
                "dtype": si.torch_dtype,
            }
        return {"model": model, "blocks": coverage,
                "complete": all(c > 0 for c in coverage), "servers": servers}
    finally:
        if own:
            node.shutdown()


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("model")
    ap.add_argument("--initial-peers", nargs="+", required=True)
    ap.add_argument("--num-blocks", type=int, default=None)
    ap.add_argument("--no-ping", action="store_true")
    ap.add_argument("--json", action="store_true", help="machine-readable")
    ap.add_argument("--watch", type=float, default=0.0, metavar="SECONDS",
                    help="refresh every N seconds until interrupted")
    args = ap.parse_args()

    peers = []
    for p in args.initial_peers:
        host, _, port = p.rpartition(":")
        peers.append((host or "127.0.0.1", int(port)))
    import json as _json
    import time as _time

    def report():
        h = swarm_health(args.model, peers, args.num_blocks,
                         ping=not args.no_ping)
        if args.json:
            print(_json.dumps(h))
            return
        print(f"model: {h['model']}  complete: {h['complete']}")
        cov = h["blocks"]
        print("block coverage:", " ".join(str(c) for c in cov),
              f"({sum(1 for c in cov if c == 0)} uncovered)")
        for peer, s in sorted(h["servers"].items()):
            print(f"  {peer[:12]:12s} blocks {s['blocks']:>7s} {s['state']:8s} "
                  f"rps {s['throughput']:<8.3g} rtt {s['rtt_ms']} ms "
                  f"kv_left {s['cache_tokens_left']} @ {s['host']}")

    report()
    while args.watch > 0:
        _time.sleep(args.watch)
        report()


if __name__ == "__main__":
    main()
