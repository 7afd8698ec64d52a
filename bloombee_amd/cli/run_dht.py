"""Bootstrap DHT node (parity: reference cli/run_dht.py).

    python -m bloombee_amd.cli.run_dht --host 0.0.0.0 --port 31337
"""
from __future__ import annotations

import argparse
import time

from bloombee_amd.net.dht import Dht
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


def parse_endpoint(s: str):
    host, port = s.rsplit(":", 1)
    return host, int(port)


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=0)
    ap.add_argument("--initial-peers", nargs="*", default=[],
                    help="host:port of existing DHT nodes")
    args = ap.parse_args()

    dht = Dht(initial_peers=[parse_endpoint(p) for p in args.initial_peers],
              host=args.host, port=args.port)
    logger.info("DHT bootstrap node at %s:%d — pass this as --initial-peers "
                "to servers and clients", *dht.endpoint)
    try:
        while True:
            time.sleep(5)
    except KeyboardInterrupt:
        dht.shutdown()


if __name__ == "__main__":
    main()
