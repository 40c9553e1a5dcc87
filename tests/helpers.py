"""Synthetic k8s objects for tests and bench."""

from __future__ import annotations

import itertools
import time

from gpushare_amd import consts

_counter = itertools.count()


def make_pod(
    name: str,
    mem: int,
    gpu_idx: int = 0,
    node: str = "node-a",
    namespace: str = "default",
    assume_time_ns: int | None = None,
    assigned: str = "false",
    phase: str = "Pending",
    containers: list[int] | None = None,
    extra_annotations: dict | None = None,
) -> dict:
    """An extender-assumed gpushare pod.  ``containers`` optionally splits
    ``mem`` into per-container limits (default: one container)."""
    if containers is None:
        containers = [mem]
    assert sum(containers) == mem
    anns = {
        consts.ENV_RESOURCE_INDEX: str(gpu_idx),
        consts.ENV_RESOURCE_ASSUME_TIME: str(
            time.time_ns() if assume_time_ns is None else assume_time_ns
        ),
        consts.ENV_ASSIGNED_FLAG: assigned,
    }
    if extra_annotations:
        anns.update(extra_annotations)
    return {
        "metadata": {
            "name": name,
            "namespace": namespace,
            "uid": f"uid-{name}-{next(_counter)}",
            "annotations": anns,
        },
        "spec": {
            "nodeName": node,
            "containers": [
                {
                    "name": f"c{i}",
                    "resources": {
                        "limits": {consts.RESOURCE_NAME: str(m)}
                    },
                }
                for i, m in enumerate(containers)
            ],
        },
        "status": {"phase": phase},
    }


def plain_pod(name: str, node: str = "node-a", phase: str = "Running") -> dict:
    return {
        "metadata": {"name": name, "namespace": "default", "uid": f"uid-{name}"},
        "spec": {"nodeName": node, "containers": [{"name": "c0"}]},
        "status": {"phase": phase},
    }
