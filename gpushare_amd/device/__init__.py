"""Device layer — MI355X GPU abstraction.

The AMD-native counterpart of the reference's L1 (pkg/gpu/nvidia/nvidia.go):
enumeration, per-GPU HBM capacity, fake-device expansion, and health events.
Two sources implement one interface:

- :class:`~gpushare_amd.device.amdsmi_source.AmdSmiSource` — real GPUs via the
  in-tree ``_amdsmi`` dlopen shim + KFD sysfs topology (render nodes, xGMI).
- :class:`~gpushare_amd.device.mock_source.MockSource` — CI / no-GPU hosts
  (BASELINE config 1), driven by ``GPUSHARE_MOCK_SPEC``.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Iterable, Optional, Protocol

from .. import consts


@dataclass(frozen=True)
class PhysicalGPU:
    """One physical MI355X as seen by the plugin.

    ``uuid`` is the stable short identifier embedded in fake-device IDs
    (total ID must stay ≤63 chars, api.proto:84 — validated in fakedev).
    """

    index: int                      # plugin device index (== HIP ordinal order)
    uuid: str
    memory_bytes: int               # VRAM total (MI355X: 288 GiB HBM3E)
    render_path: Optional[str] = None   # /dev/dri/renderD<minor>
    card_path: Optional[str] = None     # /dev/dri/card<n>
    bdf: Optional[str] = None           # PCI domain:bus:dev.fn
    kfd_node: Optional[int] = None      # KFD topology node index
    numa_node: int = -1
    xgmi_peers: tuple = ()          # plugin indices of xGMI-linked GPUs
    extras: dict = field(default_factory=dict, compare=False, hash=False)

    def mem_units(self, unit: str) -> int:
        """Schedulable memory grains (fake devices) for this GPU.

        GiB default: 288 for MI355X (reference equivalent: nvidia.go:34-45,
        which floors MiB/1024 for GiB mode).
        """
        shift = 30 if unit == consts.GIB else 20
        return self.memory_bytes >> shift


@dataclass
class HealthEvent:
    """A health transition for one physical GPU (or all, if index is None)."""

    gpu_index: Optional[int]        # None ⇒ applies to every GPU
    healthy: bool
    kind: str = ""                  # e.g. "thermal_throttle", "gpu_pre_reset"
    message: str = ""


class DeviceSource(Protocol):
    """What the plugin needs from a GPU backend."""

    def devices(self) -> list[PhysicalGPU]: ...

    def watch_health(self, stop_event) -> Iterable[HealthEvent]:
        """Blocking generator of health transitions; returns when
        ``stop_event`` is set.  May be a no-op generator (mock)."""
        ...

    def close(self) -> None: ...


def create_source(mock_spec: Optional[str] = None) -> DeviceSource:
    """Pick the device source.

    Explicit ``mock_spec`` (or env ``GPUSHARE_MOCK_SPEC``) forces mock mode;
    otherwise real amdsmi enumeration is attempted and a clear error raised if
    no AMD GPUs are present (mirrors the reference's refusal to run without
    GPUs, gpumanager.go:36-47 — but failing loudly instead of blocking on an
    empty select{}).
    """
    spec = mock_spec or os.environ.get("GPUSHARE_MOCK_SPEC")
    if spec:
        from .mock_source import MockSource

        return MockSource.from_spec(spec)
    from .amdsmi_source import AmdSmiSource

    return AmdSmiSource()
