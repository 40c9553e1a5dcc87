#!/usr/bin/env python3
"""Capture the real /sys KFD topology + DRM layout as a test fixture.

Copies exactly the files gpushare_amd/device/kfd_topology.py consumes
(gpu_id, properties, mem_banks/*/properties, io_links/*/properties, the
renderD*/card* device symlink structure and numa_node) into a plain
directory tree that kfd_topology.read_topology/resolve can walk directly.

Run on a GPU box:  python tools/capture_kfd_snapshot.py gpurun_out/kfd_snapshot
"""

from __future__ import annotations

import os
import sys
from pathlib import Path

KFD = Path("/sys/class/kfd/kfd/topology/nodes")
DRM = Path("/sys/class/drm")


def _copy(src: Path, dst: Path) -> None:
    try:
        data = src.read_text()
    except OSError:
        return
    dst.parent.mkdir(parents=True, exist_ok=True)
    dst.write_text(data)


def main() -> int:
    out = Path(sys.argv[1] if len(sys.argv) > 1 else "kfd_snapshot")
    nodes_out = out / "kfd" / "topology" / "nodes"
    drm_out = out / "drm"

    if not KFD.is_dir():
        print(f"no KFD topology at {KFD}", file=sys.stderr)
        return 1

    for node in sorted(KFD.iterdir()):
        if not node.name.isdigit():
            continue
        dst = nodes_out / node.name
        _copy(node / "gpu_id", dst / "gpu_id")
        _copy(node / "properties", dst / "properties")
        for sub in ("mem_banks", "io_links"):
            d = node / sub
            if d.is_dir():
                for entry in sorted(d.iterdir()):
                    _copy(
                        entry / "properties",
                        dst / sub / entry.name / "properties",
                    )

    # DRM: preserve the shared-PCI-parent structure via relative symlinks
    device_dirs: dict[str, Path] = {}
    for entry in sorted(DRM.iterdir()):
        if not (
            entry.name.startswith("renderD") or entry.name.startswith("card")
        ):
            continue
        dev = entry / "device"
        if not dev.exists():
            continue
        real = os.path.realpath(dev)
        if real not in device_dirs:
            pci_dir = drm_out / "pci" / os.path.basename(real)
            pci_dir.mkdir(parents=True, exist_ok=True)
            _copy(Path(real) / "numa_node", pci_dir / "numa_node")
            device_dirs[real] = pci_dir
        d = drm_out / entry.name
        d.mkdir(parents=True, exist_ok=True)
        link = d / "device"
        if not link.exists():
            os.symlink(
                os.path.relpath(device_dirs[real], d),
                link,
                target_is_directory=True,
            )

    n_gpu = sum(
        1
        for n in nodes_out.iterdir()
        if (n / "gpu_id").exists()
        and (n / "gpu_id").read_text().strip() not in ("", "0")
    )
    print(f"captured {n_gpu} GPU node(s) under {out}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
