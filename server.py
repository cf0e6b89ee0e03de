#!/usr/bin/env python3
"""Server entrypoint (reference-compatible CLI: `python server.py [--config config.yaml]`).

Modes:
* standalone (default when transport.kind == "loopback"): server AND all
  clients in this process — the reference's multi-process RabbitMQ deployment
  collapsed onto one host/GPU;
* distributed (transport.kind == "rccl"): this process runs only the control-
  plane server (TCPStore master + round state machine); clients join via
  `python client.py --layer_id N` (reference client CLI, src/client.py:13-17)
  or via torchrun (one rank per GPU).
"""

import argparse
import signal
import sys

from split_learning_amd.config import load_config
from split_learning_amd.parallel.control import StoreControl
from split_learning_amd.parallel.launch import run_loopback
from split_learning_amd.parallel.server import Server
from split_learning_amd.utils import Logger, print_with_color


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="config.yaml")
    ap.add_argument("--device", default=None,
                    help="loopback mode device (default: cuda:0 if available)")
    args = ap.parse_args()

    cfg = load_config(args.config)
    logger = Logger(f"{cfg['log_path']}/app.log", cfg["debug_mode"])

    def handle_sigint(_sig, _frm):
        print_with_color("Interrupted; shutting down.", "yellow")
        sys.exit(1)

    signal.signal(signal.SIGINT, handle_sigint)

    if cfg["transport"]["kind"] == "loopback":
        import torch
        device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
        print_with_color(f"[standalone] server + clients in-process on {device}", "green")
        server, _ = run_loopback(cfg, device=device, logger=logger)
        print_with_color("Training complete.", "green")
    else:
        addr = cfg["transport"]["master-addr"]
        port = int(cfg["transport"]["master-port"])
        print_with_color(f"[server] control plane at {addr}:{port}; waiting for "
                         f"{cfg['server']['clients']} clients", "green")
        control = StoreControl.create(addr, port, is_server=True)
        server = Server(cfg, control, logger=logger)
        server.run()


if __name__ == "__main__":
    main()
