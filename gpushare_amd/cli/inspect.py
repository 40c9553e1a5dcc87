"""kubectl-inspect-gpushare — cluster-wide GPU-share utilization viewer.

Reference counterpart: cmd/inspect (shipped as kubectl-inspect-gpushare-v2).
Same data model and output shape (summary table per node with per-GPU
Allocated/Total columns, optional -d per-pod details, cluster totals), same
annotation readers: the per-container allocation map
``scheduler.framework.gpushare.allocation`` is preferred
(nodeinfo.go:244-271), falling back to the single-GPU
``ALIYUN_COM_GPU_MEM_IDX`` annotation; pods with gpu-mem but no usable
annotation land in the "pending" bucket (devs[-1], nodeinfo.go:136-139).

Deviation from the reference, deliberate: memory-unit inference used
``per-GPU mem > 100 ⇒ MiB`` (nodeinfo.go:227-243), which misclassifies a
288-GiB MI355X as MiB; the threshold here is 1024.
"""

from __future__ import annotations

import argparse
import sys
import time

from .. import consts
from ..cluster import podutils
from ..cluster.kubeclient import RestKubeClient


# --------------------------------------------------------------------------- #
# data model (reference: cmd/inspect/nodeinfo.go)
# --------------------------------------------------------------------------- #

PENDING = -1


def get_allocation(pod: dict) -> dict[int, int]:
    """Per-GPU memory map for a pod: the allocation annotation if present,
    else {IDX: total}, else {-1: total} (pending)."""
    alloc_map = podutils.allocation_map_from_annotation(pod)
    if alloc_map:
        out: dict[int, int] = {}
        for _container, per_gpu in alloc_map.items():
            if not isinstance(per_gpu, dict):
                continue
            for idx_str, mem in per_gpu.items():
                try:
                    out[int(idx_str)] = out.get(int(idx_str), 0) + int(mem)
                except (ValueError, TypeError):
                    continue
        if out:
            return out
    total = podutils.gpu_memory_of_pod(pod)
    idx = podutils.gpu_id_from_annotation(pod)
    return {idx if idx >= 0 else PENDING: total}


class NodeInfo:
    def __init__(self, node: dict, pods: list[dict]):
        self.node = node
        self.name = node.get("metadata", {}).get("name", "?")
        alloc = node.get("status", {}).get("allocatable", {})
        self.gpu_count = int(alloc.get(consts.RESOURCE_COUNT, 0) or 0)
        self.total_mem = int(alloc.get(consts.RESOURCE_NAME, 0) or 0)
        self.address = "unknown"
        for addr in node.get("status", {}).get("addresses", []):
            if addr.get("type") == "InternalIP":
                self.address = addr.get("address", "unknown")
                break
        # real per-GPU capacities from the plugin's topology annotation
        # (heterogeneous nodes; reference assumed total/count everywhere)
        self.per_gpu_units: list[int] = []
        raw = (node.get("metadata", {}).get("annotations") or {}).get(
            consts.ANN_NODE_TOPOLOGY
        )
        if raw:
            import json as _json

            try:
                units = _json.loads(raw).get("per_gpu_units", [])
                if len(units) == self.gpu_count:
                    self.per_gpu_units = [int(u) for u in units]
            except (ValueError, TypeError):
                pass
        # devs: idx -> {"used": int, "pods": [pod]}
        self.devs: dict[int, dict] = {
            i: {"used": 0, "pods": []} for i in range(self.gpu_count)
        }
        for pod in pods:
            if podutils.gpu_memory_of_pod(pod) <= 0:
                continue
            for idx, mem in get_allocation(pod).items():
                dev = self.devs.setdefault(idx, {"used": 0, "pods": []})
                dev["used"] += mem
                dev["pods"].append(pod)

    @property
    def per_gpu_total(self) -> int:
        return self.total_mem // self.gpu_count if self.gpu_count else 0

    def gpu_total(self, idx: int) -> int:
        if 0 <= idx < len(self.per_gpu_units):
            return self.per_gpu_units[idx]
        return self.per_gpu_total

    @property
    def used_mem(self) -> int:
        return sum(d["used"] for d in self.devs.values())

    @property
    def has_pending(self) -> bool:
        return PENDING in self.devs


def infer_memory_unit(infos: list[NodeInfo]) -> str:
    for info in infos:
        if info.gpu_count and info.per_gpu_total >= 1024:
            return consts.MIB
    return consts.GIB


# --------------------------------------------------------------------------- #
# rendering (reference: cmd/inspect/display.go)
# --------------------------------------------------------------------------- #

def _table(rows: list[list[str]]) -> str:
    if not rows:
        return ""
    widths = [max(len(r[i]) for r in rows if i < len(r)) for i in range(max(map(len, rows)))]
    return "\n".join(
        "  ".join(c.ljust(widths[i]) for i, c in enumerate(r)).rstrip()
        for r in rows
    )


def display_summary(infos: list[NodeInfo], out=sys.stdout) -> None:
    infos = [n for n in infos if n.total_mem > 0]
    unit = infer_memory_unit(infos)
    max_gpus = max((n.gpu_count for n in infos), default=0)
    has_pending = any(n.has_pending for n in infos)

    header = ["NAME", "IPADDRESS"]
    header += [f"GPU{i}(Allocated/Total)" for i in range(max_gpus)]
    if has_pending:
        header.append("PENDING(Allocated)")
    header.append(f"GPU Memory({unit})")

    rows = [header]
    used_cluster = total_cluster = 0
    for n in infos:
        row = [n.name, n.address]
        for i in range(max_gpus):
            dev = n.devs.get(i)
            row.append(
                f"{dev['used']}/{n.gpu_total(i)}" if dev is not None and i < n.gpu_count
                else "0/0"
            )
        if has_pending:
            row.append(str(n.devs.get(PENDING, {"used": 0})["used"]))
        row.append(f"{n.used_mem}/{n.total_mem}")
        rows.append(row)
        used_cluster += n.used_mem
        total_cluster += n.total_mem

    print(_table(rows), file=out)
    print("-" * 72, file=out)
    pct = int(100 * used_cluster / total_cluster) if total_cluster else 0
    print("Allocated/Total GPU Memory In Cluster:", file=out)
    print(f"{used_cluster}/{total_cluster} ({pct}%)", file=out)


def display_details(infos: list[NodeInfo], out=sys.stdout) -> None:
    used_cluster = total_cluster = 0
    for n in infos:
        if n.total_mem <= 0:
            continue
        print(f"\nNAME:       {n.name}", file=out)
        print(f"IPADDRESS:  {n.address}\n", file=out)
        header = ["NAME", "NAMESPACE"]
        header += [f"GPU{i}(Allocated)" for i in range(n.gpu_count)]
        if n.has_pending:
            header.append("Pending(Allocated)")
        rows = [header]
        seen = set()
        for idx in sorted(n.devs, key=lambda k: (k == PENDING, k)):
            for pod in n.devs[idx]["pods"]:
                uid = podutils.pod_uid(pod)
                if uid in seen:
                    continue
                seen.add(uid)
                alloc = get_allocation(pod)
                row = [podutils.pod_name(pod), podutils.pod_namespace(pod)]
                cols = list(range(n.gpu_count)) + ([PENDING] if n.has_pending else [])
                row += [str(alloc.get(k, 0)) for k in cols]
                rows.append(row)
        print(_table(rows), file=out)
        pct = int(100 * n.used_mem / n.total_mem) if n.total_mem else 0
        print(f"Allocated : {n.used_mem} ({pct}%)", file=out)
        print(f"Total :     {n.total_mem}", file=out)
        print("-" * 72, file=out)
        used_cluster += n.used_mem
        total_cluster += n.total_mem
    pct = int(100 * used_cluster / total_cluster) if total_cluster else 0
    print(f"\n\nAllocated/Total GPU Memory In Cluster: "
          f"{used_cluster}/{total_cluster} ({pct}%)", file=out)


# --------------------------------------------------------------------------- #
# assembly
# --------------------------------------------------------------------------- #

def is_shared_gpu_node(node: dict) -> bool:
    alloc = node.get("status", {}).get("allocatable", {})
    return int(alloc.get(consts.RESOURCE_NAME, 0) or 0) > 0


def active_pods(pods: list[dict]) -> list[dict]:
    return [
        p for p in pods
        if podutils.pod_phase(p) not in ("Succeeded", "Failed")
    ]


def _with_retries(fn, retries: int = 5, interval: float = 0.1):
    """Client-side retry budget for transient apiserver errors — the
    kubectl plugin must not fail on first touch (reference:
    cmd/inspect/podinfo.go:24,64-70 — 5 × 100 ms)."""
    last: Exception = None
    for attempt in range(retries):
        try:
            return fn()
        except Exception as e:  # noqa: BLE001 — transport and API errors
            last = e
            if attempt < retries - 1:
                time.sleep(interval)
    raise last


def build_node_infos(kube, node_name: str = "") -> list[NodeInfo]:
    nodes = _with_retries(kube.list_nodes).get("items", [])
    if node_name:
        nodes = [n for n in nodes if n.get("metadata", {}).get("name") == node_name]
    nodes = [n for n in nodes if is_shared_gpu_node(n)]
    infos = []
    for node in nodes:
        name = node.get("metadata", {}).get("name", "")
        pods = _with_retries(
            lambda n=name: kube.list_pods(field_selector=f"spec.nodeName={n}")
        ).get("items", [])
        infos.append(NodeInfo(node, active_pods(pods)))
    return infos


def display_json(infos: list[NodeInfo], out=sys.stdout) -> None:
    """Machine-readable dump (no reference counterpart — the reference's
    tabwriter output is awkward to script against)."""
    import json

    nodes = []
    for n in infos:
        devs = {}
        for idx, dev in sorted(n.devs.items()):
            devs[str(idx)] = {
                "total": n.gpu_total(idx) if idx >= 0 else None,
                "used": dev["used"],
                "pods": [
                    {
                        "namespace": podutils.pod_namespace(p),
                        "name": podutils.pod_name(p),
                        "gpu_mem": get_allocation(p).get(idx, 0),
                    }
                    for p in dev["pods"]
                ],
            }
        nodes.append(
            {
                "name": n.name,
                "address": n.address,
                "gpu_count": n.gpu_count,
                "gpu_mem_total": n.total_mem,
                "gpu_mem_used": n.used_mem,
                "per_gpu_units": n.per_gpu_units or None,
                "devices": devs,
            }
        )
    json.dump(
        {
            "unit": infer_memory_unit(infos),
            "nodes": nodes,
            "cluster": {
                "gpu_mem_total": sum(n.total_mem for n in infos),
                "gpu_mem_used": sum(n.used_mem for n in infos),
            },
        },
        out,
        indent=1,
    )
    out.write("\n")


def main(argv=None, kube=None, out=sys.stdout) -> int:
    p = argparse.ArgumentParser(prog="kubectl-inspect-gpushare")
    p.add_argument("-d", "--details", action="store_true",
                   help="per-pod allocation details")
    p.add_argument("-o", "--output", choices=("table", "json"),
                   default="table",
                   help="output format (json: machine-readable dump)")
    p.add_argument("node", nargs="?", default="",
                   help="restrict to one node")
    p.add_argument("--api-url", default=None, help=argparse.SUPPRESS)
    args = p.parse_args(argv)

    if kube is None:
        kube = RestKubeClient(base_url=args.api_url) if args.api_url else RestKubeClient()
    infos = sorted(build_node_infos(kube, args.node), key=lambda n: n.name)
    if not infos:
        print("No shared-GPU nodes found", file=out)
        return 1
    if args.output == "json":
        display_json(infos, out=out)
    elif args.details:
        display_details(infos, out=out)
    else:
        display_summary(infos, out=out)
    return 0


if __name__ == "__main__":
    sys.exit(main())
