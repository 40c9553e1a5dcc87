"""Node-scoped pod informer: LIST once, then WATCH — the client-go informer
pattern, stdlib-only.

Why: the reference performs 1–2 remote pod LISTs inside *every* Allocate
under a global mutex (allocate.go:59-62, podmanager.go:125-160) — O(pods)
work and a full apiserver round-trip on the hot path.  With an informer the
steady-state Allocate touches only process memory; the apiserver sees one
long-lived watch stream per node instead of a LIST storm (on an 8×MI355X
node at 10 pods/s churn that is ~40 LISTs/s saved).

Correctness: the watch is opened BEFORE the LIST, so no event between the
two is lost; buffered events are reconciled against the LIST snapshot by
``metadata.resourceVersion`` (higher wins; DELETED beats an older stored
version).  The consumer (PodManager) keeps an authoritative direct-LIST
fallback for its final matching attempt, so scheduling never *depends* on
the informer being caught up — it is an accelerator, not a correctness
dependency.  Store objects are replaced whole, never mutated, so readers
may hold snapshots without locks.
"""

from __future__ import annotations

import json
import logging
import threading
import time
from typing import Optional

from .. import metrics

log = logging.getLogger(__name__)


def _rv(pod: dict) -> int:
    try:
        return int(pod.get("metadata", {}).get("resourceVersion", 0))
    except (TypeError, ValueError):
        return 0


def _uid(pod: dict) -> str:
    md = pod.get("metadata", {})
    return md.get("uid") or f"{md.get('namespace','default')}/{md.get('name','')}"


class PodInformer:
    """Watches all pods bound to one node (fieldSelector spec.nodeName=...)."""

    def __init__(
        self,
        kube,                       # RestKubeClient
        node_name: Optional[str] = None,   # None ⇒ watch all nodes' pods
        resync_interval: float = 300.0,
        reconnect_backoff: float = 1.0,
        on_event=None,              # callback(etype, pod) after each apply
    ):
        self.kube = kube
        self.node_name = node_name
        self.on_event = on_event
        self.resync_interval = resync_interval
        self.reconnect_backoff = reconnect_backoff
        self._store: dict[str, dict] = {}
        # deletion tombstones uid -> (rv, monotonic time): a re-LIST served
        # from a stale apiserver cache may still contain a pod whose
        # DELETED event the watch already delivered; without a tombstone
        # the merge would resurrect it and the allocator could match a
        # dead pod.  Pruned by age (uids are never reused in k8s, so age
        # only bounds memory, not correctness).
        self._tombstones: dict[str, tuple[int, float]] = {}
        self._tombstone_ttl = 600.0
        self._tombstone_cap = 10_000
        self._lock = threading.Lock()
        self._cv = threading.Condition(self._lock)
        self._version = 0            # bumped on every applied event / relist
        self._synced = threading.Event()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        # counters (metrics / tests)
        self.events_seen = 0
        self.relists = 0
        self.reconnects = 0

    # ------------------------------------------------------------------ #
    # lifecycle
    # ------------------------------------------------------------------ #
    def start(self) -> "PodInformer":
        self._thread = threading.Thread(
            target=self._run, name=f"pod-informer-{self.node_name}", daemon=True
        )
        self._thread.start()
        return self

    def stop(self) -> None:
        self._stop.set()
        self._synced.clear()

    @property
    def synced(self) -> bool:
        return self._synced.is_set()

    def wait_synced(self, timeout: float = 10.0) -> bool:
        return self._synced.wait(timeout)

    # ------------------------------------------------------------------ #
    # read side
    # ------------------------------------------------------------------ #
    @property
    def version(self) -> int:
        with self._lock:
            return self._version

    def wait_newer(self, version: int, timeout: float) -> int:
        """Block until the store has changed past ``version`` (an event was
        applied or a relist ran), or until timeout.  Returns the current
        version.  Lets the allocator retry the instant the watch catches up
        instead of sleeping a fixed backoff."""
        deadline = time.monotonic() + timeout
        with self._cv:
            while self._version <= version:
                remaining = deadline - time.monotonic()
                if remaining <= 0 or not self._cv.wait(remaining):
                    break
            return self._version

    def pods(self) -> list[dict]:
        with self._lock:
            return list(self._store.values())

    def pending_pods(self) -> list[dict]:
        with self._lock:
            return [
                p
                for p in self._store.values()
                if p.get("status", {}).get("phase") == "Pending"
            ]

    # ------------------------------------------------------------------ #
    # sync loop
    # ------------------------------------------------------------------ #
    def _run(self) -> None:
        selector = (
            f"spec.nodeName={self.node_name}" if self.node_name else ""
        )
        backoff = self.reconnect_backoff
        while not self._stop.is_set():
            conn = None
            try:
                conn, resp = self.kube.watch_pods_stream(
                    field_selector=selector
                )
                # LIST after the watch is open: an event raced between the
                # two waits in the stream's socket buffer and is reconciled
                # by resourceVersion when read below — nothing is lost.
                snapshot = self.kube.list_pods(
                    field_selector=selector
                ).get("items", [])
                with self._lock:
                    self._store = {
                        _uid(p): p
                        for p in snapshot
                        if not (
                            (tomb := self._tombstones.get(_uid(p)))
                            and _rv(p) <= tomb[0]
                        )
                    }
                    self._version += 1
                    self._cv.notify_all()
                self.relists += 1
                self._synced.set()
                metrics.observe_informer_state(True)
                backoff = self.reconnect_backoff
                last_list = time.monotonic()

                while not self._stop.is_set():
                    line = resp.readline()
                    if not line:
                        raise ConnectionError("watch stream closed by server")
                    line = line.strip()
                    if line:
                        evt = json.loads(line)
                        etype, obj = evt.get("type", ""), evt.get("object", {})
                        self._apply(etype, obj)
                        self.events_seen += 1
                        metrics.observe_informer_event()
                        if self.on_event is not None:
                            try:
                                self.on_event(etype, obj)
                            except Exception as e:  # noqa: BLE001
                                log.warning("informer on_event failed: %s", e)
                    if (
                        self.resync_interval > 0
                        and time.monotonic() - last_list > self.resync_interval
                    ):
                        snapshot = self.kube.list_pods(
                            field_selector=selector
                        ).get("items", [])
                        self._merge_snapshot(snapshot)
                        self.relists += 1
                        last_list = time.monotonic()
            except Exception as e:  # noqa: BLE001 — any stream failure
                if self._stop.is_set():
                    break
                self._synced.clear()
                self.reconnects += 1
                metrics.observe_informer_state(False, reconnected=True)
                log.warning(
                    "pod informer stream failed (%s); reconnecting in %.1fs",
                    e,
                    backoff,
                )
                self._stop.wait(backoff)
                backoff = min(backoff * 2, 30.0)
            finally:
                if conn is not None:
                    try:
                        conn.close()
                    except OSError:
                        pass

    def _apply(self, etype: str, pod: dict) -> None:
        uid = _uid(pod)
        with self._lock:
            cur = self._store.get(uid)
            if etype == "DELETED":
                if cur is None or _rv(cur) <= _rv(pod):
                    self._store.pop(uid, None)
                    now = time.monotonic()
                    self._tombstones[uid] = (
                        max(_rv(pod), _rv(cur) if cur else 0),
                        now,
                    )
                    # evict from the FRONT (dict insertion order == time
                    # order): age-expired entries, plus a hard size cap —
                    # O(evicted) per event, NEVER a full-dict rebuild (a
                    # rebuild per DELETED melted the informer thread in
                    # the 256k-pod soak: >10k young tombstones meant every
                    # delete paid O(n) for zero evictions)
                    cutoff = now - self._tombstone_ttl
                    while self._tombstones:
                        first = next(iter(self._tombstones))
                        _rv0, t0 = self._tombstones[first]
                        if (
                            t0 <= cutoff
                            or len(self._tombstones) > self._tombstone_cap
                        ):
                            del self._tombstones[first]
                        else:
                            break
            elif etype in ("ADDED", "MODIFIED"):
                tomb = self._tombstones.get(uid)
                if tomb is not None and _rv(pod) <= tomb[0]:
                    pass  # stale replay of a pod we saw deleted
                elif cur is None or _rv(cur) <= _rv(pod):
                    self._store[uid] = pod
                    self._tombstones.pop(uid, None)
            elif etype == "BOOKMARK":
                pass
            else:
                log.warning("unknown watch event type %r", etype)
            self._version += 1
            self._cv.notify_all()

    def _merge_snapshot(self, snapshot: list[dict]) -> None:
        """Periodic anti-entropy re-LIST: adopt newer objects, drop pods the
        snapshot no longer contains *and* that have not changed since (a
        fresher stored version means a watch event raced the LIST — keep it)."""
        snap = {_uid(p): p for p in snapshot}
        max_snap_rv = max((_rv(p) for p in snapshot), default=0)
        with self._lock:
            for uid, pod in snap.items():
                tomb = self._tombstones.get(uid)
                if tomb is not None and _rv(pod) <= tomb[0]:
                    continue  # stale list still carries a deleted pod
                if uid not in self._store or _rv(self._store[uid]) < _rv(pod):
                    self._store[uid] = pod
                    self._tombstones.pop(uid, None)
            for uid in list(self._store):
                if uid not in snap and _rv(self._store[uid]) <= max_snap_rv:
                    del self._store[uid]
            self._version += 1
            self._cv.notify_all()
