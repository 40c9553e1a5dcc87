"""Kubernetes Event emission.

The reference's RBAC grants ``events create/patch`` but no code ever emits
one.  Here health flips and failed allocations produce real,
node-scoped Events so `kubectl describe node` / `kubectl get events` show
what the plugin did — the observability a cluster operator actually uses.
"""

from __future__ import annotations

import logging
import time
from datetime import datetime, timezone

log = logging.getLogger(__name__)


def _now_iso() -> str:
    return datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


class EventRecorder:
    def __init__(self, kube_client, node_name: str, namespace: str = "default",
                 component: str = "gpushare-device-plugin"):
        self.kube = kube_client
        self.node_name = node_name
        self.namespace = namespace
        self.component = component
        self._seq = 0

    def emit(self, reason: str, message: str, etype: str = "Warning") -> bool:
        """Best-effort create; never raises into the caller's hot path."""
        self._seq += 1
        event = {
            "metadata": {
                "name": f"{self.node_name}.gpushare.{int(time.time()*1e3)}.{self._seq}",
                "namespace": self.namespace,
            },
            "involvedObject": {
                "kind": "Node",
                "name": self.node_name,
                "uid": self.node_name,
            },
            "reason": reason,
            "message": message,
            "type": etype,
            "source": {"component": self.component, "host": self.node_name},
            "firstTimestamp": _now_iso(),
            "lastTimestamp": _now_iso(),
            "count": 1,
        }
        try:
            self.kube.create_event(self.namespace, event)
            return True
        except Exception as e:  # noqa: BLE001 — events are best-effort
            log.debug("event emit failed (%s: %s)", reason, e)
            return False


class NullRecorder:
    """No-op recorder for harnesses without an apiserver."""

    def emit(self, reason: str, message: str, etype: str = "Warning") -> bool:
        return False
