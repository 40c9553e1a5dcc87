"""podgetter — debug dump of the kubelet read-only /pods endpoint.

Reference counterpart: cmd/podgetter/main.go (builds a kubelet client
against 127.0.0.1:10250 and prints the pod list).
"""

from __future__ import annotations

import argparse
import json
import sys

from ..cluster.kubeclient import KubeletClient


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="gpushare-podgetter")
    p.add_argument("--kubelet-address", default="127.0.0.1")
    p.add_argument("--kubelet-port", type=int, default=10250)
    p.add_argument("--token", default=None)
    p.add_argument("--timeout", type=int, default=10)
    args = p.parse_args(argv)

    client = KubeletClient(
        address=args.kubelet_address,
        port=args.kubelet_port,
        token=args.token,
        timeout=args.timeout,
    )
    try:
        pods = client.get_node_running_pods()
    except Exception as e:  # noqa: BLE001
        print(f"error: {e}", file=sys.stderr)
        return 1
    finally:
        client.close()
    json.dump(pods, sys.stdout, indent=2)
    print()
    return 0


if __name__ == "__main__":
    sys.exit(main())
