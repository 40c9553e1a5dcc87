"""Binpack placement policy — pure state + selection, no I/O.

Policy (matches the gpushare extender's documented binpack behavior, demo:
3×2 GiB pods land on ONE GPU): among GPUs with enough free memory, choose
the one with the *least* remaining free memory (best-fit), breaking ties by
lower GPU index.  Node choice: the node whose best-fit GPU leaves the least
slack.
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class NodeGPUState:
    node: str
    per_gpu_units: list[int]                 # capacity per GPU
    allocated: list[int] = field(default_factory=list)

    def __post_init__(self):
        if not self.allocated:
            self.allocated = [0] * len(self.per_gpu_units)

    def free(self, idx: int) -> int:
        return self.per_gpu_units[idx] - self.allocated[idx]

    def best_fit(self, request: int) -> Optional[int]:
        best, best_free = None, None
        for i in range(len(self.per_gpu_units)):
            f = self.free(i)
            if f >= request and (best_free is None or f < best_free):
                best, best_free = i, f
        return best

    @property
    def total_allocated(self) -> int:
        return sum(self.allocated)

    @property
    def total_capacity(self) -> int:
        return sum(self.per_gpu_units)


class BinpackState:
    """Thread-safe multi-node allocation ledger."""

    def __init__(self):
        self._lock = threading.Lock()
        self.nodes: dict[str, NodeGPUState] = {}

    def set_node(self, node: str, per_gpu_units: list[int],
                 allocated: Optional[list[int]] = None) -> None:
        with self._lock:
            self.nodes[node] = NodeGPUState(
                node, list(per_gpu_units), list(allocated or [])
            )

    def filter_nodes(self, request: int, candidates: list[str]) -> list[str]:
        with self._lock:
            out = []
            for name in candidates:
                st = self.nodes.get(name)
                if st is not None and st.best_fit(request) is not None:
                    out.append(name)
            return out

    def assume(self, node: str, request: int) -> Optional[int]:
        """Reserve `request` units on the best-fit GPU; returns GPU index."""
        with self._lock:
            st = self.nodes.get(node)
            if st is None:
                return None
            idx = st.best_fit(request)
            if idx is None:
                return None
            st.allocated[idx] += request
            return idx

    def release(self, node: str, gpu_idx: int, units: int) -> None:
        with self._lock:
            st = self.nodes.get(node)
            if st is None or not (0 <= gpu_idx < len(st.allocated)):
                return
            st.allocated[gpu_idx] = max(0, st.allocated[gpu_idx] - units)

    def packing(self) -> dict:
        """Utilization report: allocated/capacity overall and per node."""
        with self._lock:
            total_cap = sum(s.total_capacity for s in self.nodes.values())
            total_alloc = sum(s.total_allocated for s in self.nodes.values())
            return {
                "allocated_units": total_alloc,
                "capacity_units": total_cap,
                "packing_pct": (100.0 * total_alloc / total_cap) if total_cap else 0.0,
                "per_node": {
                    n: list(s.allocated) for n, s in self.nodes.items()
                },
            }
