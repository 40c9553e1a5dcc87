"""gpushare-top — node-side live view of shared-GPU usage.

The inspect CLI (cluster-side) shows *allocated* gpu-mem from pod
annotations; this tool runs ON a node and shows *actual* usage straight
from amdsmi: per-GPU VRAM used/total, per-process VRAM and engine time
(host-namespace pids; container_name where the driver resolves the
cgroup).  The operator's `nvidia-smi` analogue for gpushare nodes — the
reference ships nothing comparable (its debug tool, cmd/podgetter, dumps
the kubelet pod list only).

Usage:  gpushare-top [-w SECONDS]   (watch mode; default: one snapshot)
"""

from __future__ import annotations

import argparse
import sys
import time


def _fmt_bytes(n: int) -> str:
    if n >= 1 << 30:
        return f"{n / (1 << 30):.1f}GiB"
    if n >= 1 << 20:
        return f"{n / (1 << 20):.0f}MiB"
    return f"{n}B"


def snapshot(source, out) -> None:
    gpus = source.devices()
    usage = source.vram_usage()
    procs = source.process_usage()
    print(
        f"{'GPU':>3} {'UUID':<22} {'VRAM used/total':<20} "
        f"{'procs':>5}  render",
        file=out,
    )
    for g in gpus:
        used = usage.get(g.index, 0)
        plist = procs.get(g.index, [])
        print(
            f"{g.index:>3} {g.uuid[:22]:<22} "
            f"{_fmt_bytes(used)+'/'+_fmt_bytes(g.memory_bytes):<20} "
            f"{len(plist):>5}  {g.render_path or '-'}",
            file=out,
        )
    rows = [
        (g.index, p)
        for g in gpus
        for p in procs.get(g.index, [])
        if p.get("vram_bytes", 0) > 0 or p.get("gfx_engine_ns", 0) > 0
    ]
    if rows:
        print(file=out)
        print(
            f"{'GPU':>3} {'PID(host)':>10} {'VRAM':>10} {'GTT':>10} "
            f"{'CUs':>4}  container",
            file=out,
        )
        for idx, p in rows:
            print(
                f"{idx:>3} {p['pid']:>10} {_fmt_bytes(p['vram_bytes']):>10} "
                f"{_fmt_bytes(p['gtt_bytes']):>10} "
                f"{p.get('cu_occupancy', 0):>4}  "
                f"{p.get('container_name') or '-'}",
                file=out,
            )


def main(argv=None, source=None, out=sys.stdout) -> int:
    p = argparse.ArgumentParser(
        prog="gpushare-top",
        description="live shared-GPU usage on this node (amdsmi)",
    )
    p.add_argument("-w", "--watch", type=float, default=0.0, metavar="SECONDS",
                   help="refresh every N seconds (0 = one snapshot)")
    p.add_argument("-o", "--output", choices=("table", "json"),
                   default="table", help="output format")
    args = p.parse_args(argv)

    if source is None:
        from ..device.amdsmi_source import AmdSmiSource

        try:
            source = AmdSmiSource()
        except RuntimeError as e:
            print(f"error: {e}", file=sys.stderr)
            return 1
    try:
        while True:
            if args.output == "json":
                import json

                print(
                    json.dumps(
                        {
                            "gpus": [
                                {
                                    "index": g.index,
                                    "uuid": g.uuid,
                                    "vram_total_bytes": g.memory_bytes,
                                    "vram_used_bytes": source.vram_usage().get(
                                        g.index, 0
                                    ),
                                    "render_path": g.render_path,
                                    "processes": source.process_usage().get(
                                        g.index, []
                                    ),
                                }
                                for g in source.devices()
                            ]
                        }
                    ),
                    file=out,
                )
            else:
                snapshot(source, out)
            if args.watch <= 0:
                return 0
            time.sleep(args.watch)
            if args.output == "table":
                print("\x1b[2J\x1b[H", end="", file=out)  # clear screen
    except KeyboardInterrupt:
        return 0


if __name__ == "__main__":
    sys.exit(main())
