"""Mock device source — CI / no-GPU hosts (BASELINE config 1).

The reference has no mock backend at all (its tests cannot run without a
node); this one makes every control-plane code path testable on
CPU.  Spec grammar:

- ``"<count>x<mem><GiB|MiB>"`` — homogeneous, e.g. ``"1x8GiB"`` (config 1),
  ``"8x288GiB"`` (an 8×MI355X node);
- ``"<m1>+<m2>+...<GiB|MiB>"`` — heterogeneous per-GPU capacities, e.g.
  ``"288+288+96GiB"`` (the reference assumes homogeneous nodes,
  nvidia.go:70-72; this build does not).

Health events can be injected programmatically for watcher tests.
"""

from __future__ import annotations

import queue
import re
from typing import Iterable

from . import HealthEvent, PhysicalGPU

_SPEC_RE = re.compile(r"^(\d+)x(\d+)(GiB|MiB)$")
_HETERO_RE = re.compile(r"^(\d+(?:\+\d+)+)(GiB|MiB)$")


class MockSource:
    def __init__(self, gpus: list[PhysicalGPU]):
        self._gpus = gpus
        self._events: "queue.Queue[Optional[HealthEvent]]" = queue.Queue()

    @classmethod
    def from_spec(cls, spec: str) -> "MockSource":
        spec = spec.strip()
        m = _SPEC_RE.match(spec)
        if m:
            count, mem, unit = int(m.group(1)), int(m.group(2)), m.group(3)
            mems = [mem] * count
        else:
            h = _HETERO_RE.match(spec)
            if not h:
                raise ValueError(
                    f"bad GPUSHARE_MOCK_SPEC {spec!r}; expected e.g. "
                    f"'1x8GiB' or '288+288+96GiB'"
                )
            mems, unit = [int(x) for x in h.group(1).split("+")], h.group(2)
            count = len(mems)
        shift = 30 if unit == "GiB" else 20
        gpus = [
            PhysicalGPU(
                index=i,
                uuid=f"mock-{i:02d}",
                memory_bytes=mems[i] << shift,
                render_path=f"/dev/dri/renderD{128 + i}",
                card_path=f"/dev/dri/card{i}",
                bdf=f"0000:{0x10 + i:02x}:00.0",
                numa_node=i // 4,
                xgmi_peers=tuple(j for j in range(count) if j != i),
                extras={"rocr_uuid": f"GPU-{0xA0C0DE00 + i:016x}"},
            )
            for i in range(count)
        ]
        return cls(gpus)

    def devices(self) -> list[PhysicalGPU]:
        return list(self._gpus)

    # -- health injection (tests) -------------------------------------------
    def inject_health_event(self, ev: HealthEvent) -> None:
        self._events.put(ev)

    def watch_health(self, stop_event) -> Iterable[HealthEvent]:
        while not stop_event.is_set():
            try:
                ev = self._events.get(timeout=0.05)
            except queue.Empty:
                continue
            if ev is not None:
                yield ev

    def close(self) -> None:
        self._events.put(None)


def single_mock(mem_gib: int = 8) -> MockSource:
    return MockSource.from_spec(f"1x{mem_gib}GiB")
