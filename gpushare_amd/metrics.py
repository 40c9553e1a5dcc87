"""Prometheus metrics — observability the reference lacked entirely
(glog only; RBAC granted events create/patch but no code used
it).  Exposed on ``--metrics-port`` by the daemon; scrape-friendly for a
ServiceMonitor.

All metric objects live in an optional registry so the package imports
cleanly if prometheus_client is absent (it is present in the target image).
"""

from __future__ import annotations

import logging
from typing import Optional

log = logging.getLogger(__name__)

try:
    from prometheus_client import (
        Counter,
        Gauge,
        Histogram,
        start_http_server,
    )

    AVAILABLE = True
except ImportError:  # pragma: no cover
    AVAILABLE = False

if AVAILABLE:
    ALLOCATE_TOTAL = Counter(
        "gpushare_allocate_total",
        "Allocate() RPCs",
        ["outcome"],  # ok | poisoned
    )
    ALLOCATE_LATENCY = Histogram(
        "gpushare_allocate_seconds",
        "Allocate() end-to-end latency",
        buckets=(0.0005, 0.001, 0.002, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25,
                 0.5, 1.0, 2.5),
    )
    ALLOCATE_STAGE = Histogram(
        "gpushare_allocate_stage_seconds",
        "Allocate() stage latency",
        ["stage"],  # list | patch
        buckets=(0.0005, 0.001, 0.002, 0.005, 0.01, 0.025, 0.05, 0.1, 0.5),
    )
    DEVICES_ADVERTISED = Gauge(
        "gpushare_fake_devices", "Fake devices advertised (memory grains)"
    )
    GPUS_UNHEALTHY = Gauge(
        "gpushare_unhealthy_gpus", "Physical GPUs currently Unhealthy"
    )
    HEALTH_EVENTS = Counter(
        "gpushare_health_events_total", "Health events", ["kind"]
    )
    LISTANDWATCH_SENDS = Counter(
        "gpushare_listandwatch_sends_total", "ListAndWatch payloads sent"
    )
    INFORMER_EVENTS = Counter(
        "gpushare_informer_events_total", "Pod watch events applied"
    )
    INFORMER_RECONNECTS = Counter(
        "gpushare_informer_reconnects_total", "Pod watch stream reconnects"
    )
    INFORMER_SYNCED = Gauge(
        "gpushare_informer_synced", "Pod informer synced (1) / degraded (0)"
    )
    VRAM_USED = Gauge(
        "gpushare_vram_used_bytes", "Live VRAM usage per GPU", ["gpu"]
    )


def observe_allocate(total_s: float, list_s: float, patch_s: float, ok: bool) -> None:
    if not AVAILABLE:
        return
    ALLOCATE_TOTAL.labels("ok" if ok else "poisoned").inc()
    ALLOCATE_LATENCY.observe(total_s)
    ALLOCATE_STAGE.labels("list").observe(list_s)
    ALLOCATE_STAGE.labels("patch").observe(patch_s)


def observe_health_event(kind: str, unhealthy_count: int) -> None:
    if not AVAILABLE:
        return
    HEALTH_EVENTS.labels(kind or "unknown").inc()
    GPUS_UNHEALTHY.set(unhealthy_count)


def observe_inventory(n_devices: int) -> None:
    if AVAILABLE:
        DEVICES_ADVERTISED.set(n_devices)


def observe_law_send() -> None:
    if AVAILABLE:
        LISTANDWATCH_SENDS.inc()


def observe_vram_usage(usage: dict) -> None:
    if not AVAILABLE:
        return
    for gpu_idx, used in usage.items():
        VRAM_USED.labels(str(gpu_idx)).set(used)


def observe_informer_event() -> None:
    if AVAILABLE:
        INFORMER_EVENTS.inc()


def observe_informer_state(synced: bool, reconnected: bool = False) -> None:
    if not AVAILABLE:
        return
    INFORMER_SYNCED.set(1 if synced else 0)
    if reconnected:
        INFORMER_RECONNECTS.inc()


def serve(port: int) -> Optional[object]:
    """Start the /metrics HTTP endpoint; returns the server or None."""
    if not AVAILABLE:
        log.warning("prometheus_client not installed; metrics disabled")
        return None
    server, _thread = start_http_server(port)
    log.info("metrics on :%d/metrics", port)
    return server
