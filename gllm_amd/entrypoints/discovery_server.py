"""Discovery server CLI (encoder disaggregation).

Parity: reference entrypoints/discovery_server.py.

    python -m gllm_amd.entrypoints.discovery_server --port 29800
"""

import argparse
import time


def main():
    p = argparse.ArgumentParser(description="gllm_amd discovery server")
    p.add_argument("--host", type=str, default="0.0.0.0")
    p.add_argument("--port", type=int, default=29800)
    p.add_argument("--ttl", type=float, default=10.0)
    args = p.parse_args()
    from gllm_amd.disagg.discovery import DiscoveryServer
    srv = DiscoveryServer(args.host, args.port, ttl_s=args.ttl).start()
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        srv.stop()


if __name__ == "__main__":
    main()
