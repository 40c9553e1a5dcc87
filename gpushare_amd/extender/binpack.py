"""Binpack placement policy — pure state + selection, no I/O.

Single-GPU policy (matches the gpushare extender's documented binpack
behavior, demo: 3×2 GiB pods land on ONE GPU): among GPUs with enough free
memory, choose the one with the *least* remaining free memory (best-fit),
breaking ties by lower GPU index.  Node choice: the node whose best-fit GPU
leaves the least slack.

Multi-GPU (MI355X-native; no reference counterpart — the reference rejects
any request larger than one GPU): a request that exceeds every single GPU's
free memory is split over the SMALLEST set of GPUs that fits, and among
same-size sets the one with the most pairwise xGMI links wins (ties: least
total free, i.e. tightest pack) — so RCCL inside the co-located containers
runs over direct xGMI point-to-point links (7×~153 GB/s per MI355X) rather
than bouncing through PCIe.  The split fills the least-free chosen GPUs
first, preserving large contiguous holes for future single-GPU pods.
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from itertools import combinations
from typing import Optional


@dataclass
class NodeGPUState:
    node: str
    per_gpu_units: list[int]                 # capacity per GPU
    allocated: list[int] = field(default_factory=list)
    xgmi: list[tuple] = field(default_factory=list)   # adjacency per GPU
    numa: list[int] = field(default_factory=list)     # NUMA node per GPU (-1 unknown)

    def __post_init__(self):
        if not self.allocated:
            self.allocated = [0] * len(self.per_gpu_units)
        if len(self.xgmi) != len(self.per_gpu_units):
            self.xgmi = [tuple(p) for p in self.xgmi] + [()] * (
                len(self.per_gpu_units) - len(self.xgmi)
            )
        if len(self.numa) != len(self.per_gpu_units):
            self.numa = list(self.numa) + [-1] * (
                len(self.per_gpu_units) - len(self.numa)
            )

    def free(self, idx: int) -> int:
        return self.per_gpu_units[idx] - self.allocated[idx]

    def best_fit(self, request: int, spread: bool = False) -> Optional[int]:
        """binpack (default): least feasible free — co-locate tightly.
        spread: MOST free — isolate tenants (trades packing for less
        interference; see profiles/colocation_fairness_gpu_box.md for the
        measured trade)."""
        best, best_free = None, None
        for i in range(len(self.per_gpu_units)):
            f = self.free(i)
            if f < request:
                continue
            if (
                best_free is None
                or (f > best_free if spread else f < best_free)
            ):
                best, best_free = i, f
        return best

    def _xgmi_edges(self, idxs: tuple) -> int:
        s = set(idxs)
        return sum(1 for i in idxs for p in self.xgmi[i] if p in s) // 2

    def _numa_domains(self, idxs: tuple) -> int:
        """Distinct NUMA domains a GPU set spans (unknown counts as its
        own domain, so an all-unknown node is unaffected).  On an 8-OAM
        full-mesh MI355X every pair ties on xGMI edges — host-memory
        locality (4 GPUs per socket) is the honest second key."""
        return len({
            self.numa[i] if self.numa[i] >= 0 else -(i + 2) for i in idxs
        })

    def best_fit_multi(
        self, request: int, spread: bool = False
    ) -> Optional[dict[int, int]]:
        """Placement map {gpu_idx: units}.  Single GPU when possible;
        otherwise the smallest, most-xGMI-connected, tightest set."""
        idx = self.best_fit(request, spread=spread)
        if idx is not None:
            return {idx: request}
        n = len(self.per_gpu_units)
        frees = [self.free(i) for i in range(n)]
        usable = [i for i in range(n) if frees[i] > 0]
        if sum(frees[i] for i in usable) < request:
            return None
        for k in range(2, len(usable) + 1):
            best = None  # (-edges, total_free, idxs)
            for combo in combinations(usable, k):
                total = sum(frees[i] for i in combo)
                if total < request:
                    continue
                key = (
                    -self._xgmi_edges(combo),
                    self._numa_domains(combo),
                    total,
                    combo,
                )
                if best is None or key < best:
                    best = key
            if best is not None:
                combo = best[-1]
                # fill least-free first; the last GPU takes the remainder
                remaining = request
                split: dict[int, int] = {}
                for i in sorted(combo, key=lambda i: (frees[i], i)):
                    take = min(frees[i], remaining)
                    split[i] = take
                    remaining -= take
                    if remaining == 0:
                        break
                return split
        return None

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
                 allocated: Optional[list[int]] = None,
                 xgmi: Optional[list] = None,
                 numa: Optional[list] = None) -> None:
        with self._lock:
            self.nodes[node] = NodeGPUState(
                node,
                list(per_gpu_units),
                list(allocated or []),
                [tuple(p) for p in (xgmi or [])],
                [int(n) for n in (numa or [])],
            )

    def filter_nodes(self, request: int, candidates: list[str]) -> list[str]:
        with self._lock:
            out = []
            for name in candidates:
                st = self.nodes.get(name)
                if st is not None and st.best_fit_multi(request) is not None:
                    out.append(name)
            return out

    def score_nodes(
        self, request: int, candidates: list[str], max_score: int = 10,
        spread: bool = False,
    ) -> dict[str, int]:
        """Binpack node scoring for the scheduler `prioritize` webhook:
        higher for the node whose placement leaves the least slack — pods
        funnel onto the fullest feasible node, keeping whole GPUs (and
        whole nodes) free for large/multi-GPU pods."""
        with self._lock:
            slacks: dict[str, int] = {}
            for name in candidates:
                st = self.nodes.get(name)
                if st is None:
                    continue
                split = st.best_fit_multi(request)
                if split is None:
                    slacks[name] = -1  # infeasible: score 0
                    continue
                slacks[name] = sum(st.free(i) for i in split) - request
            feasible = {n: s for n, s in slacks.items() if s >= 0}
            scores = {n: 0 for n in slacks}  # infeasible: 0
            if not feasible:
                return scores
            best = min(feasible.values())
            worst = max(feasible.values())
            for name, slack in feasible.items():
                if worst == best:
                    scores[name] = max_score
                else:
                    # binpack: tightest fit -> max_score; spread: loosest
                    frac = (worst - slack) / (worst - best)
                    if spread:
                        frac = 1.0 - frac
                    scores[name] = 1 + round((max_score - 1) * frac)
            return scores

    def assume(self, node: str, request: int) -> Optional[int]:
        """Reserve `request` units on the best-fit single GPU; returns the
        GPU index (legacy single-GPU path; multi: assume_multi)."""
        with self._lock:
            st = self.nodes.get(node)
            if st is None:
                return None
            idx = st.best_fit(request)
            if idx is None:
                return None
            st.allocated[idx] += request
            return idx

    def assume_multi(self, node: str, request: int,
                     spread: bool = False) -> Optional[dict[int, int]]:
        """Reserve `request` units across one or more xGMI-adjacent GPUs;
        returns the placement map {gpu_idx: units}."""
        with self._lock:
            st = self.nodes.get(node)
            if st is None:
                return None
            split = st.best_fit_multi(request, spread=spread)
            if split is None:
                return None
            for idx, units in split.items():
                st.allocated[idx] += units
            return split

    def release(self, node: str, gpu_idx: int, units: int) -> None:
        with self._lock:
            st = self.nodes.get(node)
            if st is None or not (0 <= gpu_idx < len(st.allocated)):
                return
            st.allocated[gpu_idx] = max(0, st.allocated[gpu_idx] - units)

    def release_multi(self, node: str, split: dict[int, int]) -> None:
        with self._lock:
            st = self.nodes.get(node)
            if st is None:
                return
            for gpu_idx, units in split.items():
                if 0 <= gpu_idx < len(st.allocated):
                    st.allocated[gpu_idx] = max(
                        0, st.allocated[gpu_idx] - units
                    )

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
