"""CLI server entry point (reference parity: `hivemind-server`-style —
SURVEY.md §1 L3 "operator / CLI"):

    python -m learning_at_home_amd.server --num-experts 4 --expert-cls ffn \
        --hidden-dim 256 --expert-prefix ffn --initial-peer 127.0.0.1:1234

Runs a Server (DHT peer + Runtime + ConnectionHandler + announcer) until
SIGINT; prints the endpoints other peers need.
"""
import argparse
import signal
import time

import torch

from . import Server
from ..utils import get_logger

logger = get_logger("lah.server_cli")


def main():
    p = argparse.ArgumentParser(description="learning_at_home_amd expert server")
    p.add_argument("--num-experts", type=int, default=4)
    p.add_argument("--expert-cls", default="ffn", choices=["ffn", "transformer", "nop"])
    p.add_argument("--hidden-dim", type=int, default=1024)
    p.add_argument("--expert-prefix", default="expert")
    p.add_argument("--expert-offset", type=int, default=0)
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=0)
    p.add_argument("--dht-port", type=int, default=0)
    p.add_argument("--initial-peer", action="append", default=[],
                   help="host:port of an existing DHT peer (repeatable)")
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--max-batch-size", type=int, default=16384)
    p.add_argument("--announce-period", type=float, default=30.0)
    p.add_argument("--checkpoint-dir", default=None)
    p.add_argument("--device", default="cuda" if torch.cuda.is_available() else "cpu")
    args = p.parse_args()

    peers = []
    for peer in args.initial_peer:
        host, port = peer.rsplit(":", 1)
        peers.append((host, int(port)))

    server = Server.create(
        args.num_experts, args.expert_cls, args.hidden_dim,
        expert_prefix=args.expert_prefix, expert_offset=args.expert_offset,
        optim_kwargs={"lr": args.lr}, max_batch_size=args.max_batch_size,
        device=torch.device(args.device) if args.device != "cpu" else None,
        listen_on=(args.host, args.port), initial_peers=peers,
        dht_port=args.dht_port, announce_period=args.announce_period,
        checkpoint_dir=args.checkpoint_dir, start=True,
    )
    logger.info(f"serving {args.num_experts} '{args.expert_cls}' experts "
                f"'{args.expert_prefix}.{args.expert_offset}..' at "
                f"{server.addr}:{server.port}; DHT peer at {server.dht.endpoint}")
    logger.info(f"join with: --initial-peer {server.dht.endpoint[0]}:{server.dht.endpoint[1]}")

    stop = [False]
    signal.signal(signal.SIGINT, lambda *a: stop.__setitem__(0, True))
    signal.signal(signal.SIGTERM, lambda *a: stop.__setitem__(0, True))
    while not stop[0]:
        time.sleep(0.5)
    logger.info("shutting down")
    server.shutdown()
    if server.dht is not None:
        server.dht.shutdown()


if __name__ == "__main__":
    main()
