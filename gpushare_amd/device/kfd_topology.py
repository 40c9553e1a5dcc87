"""KFD sysfs topology walker.

Maps each GPU (by KFD ``gpu_id``) to its DRM render node, card node, NUMA
node and xGMI peers by reading ``/sys/class/kfd/kfd/topology/nodes/*``.
This is the ROCm-native replacement for the reference's ``/dev/nvidia%d``
minor-number parsing (nvidia.go:60-71), which assumed sequential minors —
KFD's ``drm_render_minor`` property is authoritative even when other DRM
devices shift the numbering (SURVEY §7 hard part 3).

No amdsmi dependency: also usable standalone and in tests via a fake sysfs
root (``topology_root`` argument).
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from pathlib import Path
from typing import Optional

KFD_TOPOLOGY = "/sys/class/kfd/kfd/topology/nodes"
DRM_CLASS = "/sys/class/drm"

# HSA IO link types (kfd_topology uapi): 2 = PCIe, 11 = xGMI.
IOLINK_PCIE = 2
IOLINK_XGMI = 11


@dataclass
class KFDNode:
    node_id: int
    gpu_id: int                      # 0 for CPU nodes
    properties: dict = field(default_factory=dict)
    render_minor: int = -1
    xgmi_peer_nodes: list = field(default_factory=list)   # KFD node ids
    pcie_peer_nodes: list = field(default_factory=list)
    vram_bytes: int = 0

    @property
    def is_gpu(self) -> bool:
        return self.gpu_id != 0


def _read_properties(path: Path) -> dict:
    props: dict = {}
    try:
        for line in path.read_text().splitlines():
            parts = line.split()
            if len(parts) == 2:
                key, val = parts
                try:
                    props[key] = int(val)
                except ValueError:
                    props[key] = val
    except OSError:
        pass
    return props


def read_topology(topology_root: str = KFD_TOPOLOGY) -> list[KFDNode]:
    root = Path(topology_root)
    nodes: list[KFDNode] = []
    if not root.is_dir():
        return nodes
    for entry in sorted(root.iterdir(), key=lambda p: int(p.name) if p.name.isdigit() else 1 << 30):
        if not entry.name.isdigit():
            continue
        node = KFDNode(node_id=int(entry.name), gpu_id=0)
        try:
            node.gpu_id = int((entry / "gpu_id").read_text().strip() or "0")
        except (OSError, ValueError):
            node.gpu_id = 0
        node.properties = _read_properties(entry / "properties")
        node.render_minor = int(node.properties.get("drm_render_minor", -1))

        # VRAM from mem_banks (heap_type 1 = FB public, 2 = FB private)
        banks = entry / "mem_banks"
        if banks.is_dir():
            for b in banks.iterdir():
                props = _read_properties(b / "properties")
                if props.get("heap_type") in (1, 2):
                    node.vram_bytes += int(props.get("size_in_bytes", 0))

        links = entry / "io_links"
        if links.is_dir():
            for l in links.iterdir():
                props = _read_properties(l / "properties")
                peer = int(props.get("node_to", -1))
                ltype = int(props.get("type", -1))
                if ltype == IOLINK_XGMI:
                    node.xgmi_peer_nodes.append(peer)
                elif ltype == IOLINK_PCIE:
                    node.pcie_peer_nodes.append(peer)
        nodes.append(node)
    return nodes


def gpu_nodes(topology_root: str = KFD_TOPOLOGY) -> dict[int, KFDNode]:
    """KFD gpu_id -> node, GPUs only."""
    return {n.gpu_id: n for n in read_topology(topology_root) if n.is_gpu}


def render_to_card_map(drm_root: str = DRM_CLASS) -> dict[int, str]:
    """renderD minor -> /dev/dri/card<n> path, matched via shared PCI parent."""
    root = Path(drm_root)
    if not root.is_dir():
        return {}
    dev_of: dict[str, str] = {}     # realpath(device) -> card name
    for entry in root.iterdir():
        if entry.name.startswith("card") and (entry / "device").exists():
            dev_of[os.path.realpath(entry / "device")] = entry.name
    out: dict[int, str] = {}
    for entry in root.iterdir():
        if entry.name.startswith("renderD") and (entry / "device").exists():
            minor = int(entry.name[len("renderD"):])
            card = dev_of.get(os.path.realpath(entry / "device"))
            if card:
                out[minor] = f"/dev/dri/{card}"
    return out


def numa_node_of_render(minor: int, drm_root: str = DRM_CLASS) -> int:
    p = Path(drm_root) / f"renderD{minor}" / "device" / "numa_node"
    try:
        return int(p.read_text().strip())
    except (OSError, ValueError):
        return -1


@dataclass
class GPUTopology:
    """Resolved per-GPU topology, keyed by KFD gpu_id."""

    render_path: Optional[str]
    card_path: Optional[str]
    numa_node: int
    kfd_node_id: int
    xgmi_peer_gpu_ids: list
    vram_bytes: int
    gfx_target_version: int
    unique_id: int = 0              # KFD unique_id — ROCr UUID is GPU-<hex16>


def resolve(topology_root: str = KFD_TOPOLOGY, drm_root: str = DRM_CLASS
            ) -> dict[int, GPUTopology]:
    """Full topology: KFD gpu_id -> GPUTopology."""
    nodes = read_topology(topology_root)
    by_node_id = {n.node_id: n for n in nodes}
    cards = render_to_card_map(drm_root)
    out: dict[int, GPUTopology] = {}
    for n in nodes:
        if not n.is_gpu:
            continue
        render = f"/dev/dri/renderD{n.render_minor}" if n.render_minor >= 0 else None
        peers = [
            by_node_id[p].gpu_id
            for p in n.xgmi_peer_nodes
            if p in by_node_id and by_node_id[p].is_gpu
        ]
        out[n.gpu_id] = GPUTopology(
            render_path=render,
            card_path=cards.get(n.render_minor),
            numa_node=numa_node_of_render(n.render_minor, drm_root),
            kfd_node_id=n.node_id,
            xgmi_peer_gpu_ids=peers,
            vram_bytes=n.vram_bytes,
            gfx_target_version=int(n.properties.get("gfx_target_version", 0)),
            unique_id=int(n.properties.get("unique_id", 0)),
        )
    return out
