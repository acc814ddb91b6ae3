"""CLI for the nerrfd -> gRPC bridge (used by deploy/tracker-daemonset.yaml).

    python -m nerrf_amd.serve.daemon_bridge_main --daemon 127.0.0.1:50052 \
        --listen 0.0.0.0:50051
"""
from __future__ import annotations

import argparse
import signal
import threading

from .daemon_bridge import GrpcBridge


def main(argv=None) -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--daemon", required=True, help="nerrfd host:port")
    ap.add_argument("--listen", default="127.0.0.1:50051")
    args = ap.parse_args(argv)
    host, port = args.daemon.rsplit(":", 1)
    bridge = GrpcBridge(host, int(port), address=args.listen)
    bridge.start()
    print(f"bridge: nerrfd {args.daemon} -> gRPC {bridge.address}", flush=True)
    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *_: stop.set())
    signal.signal(signal.SIGINT, lambda *_: stop.set())
    stop.wait()
    bridge.stop()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
