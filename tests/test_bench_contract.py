"""The bench.py driver contract: one JSON line, required fields, sane types.

The round driver invokes `python bench.py --gpus N --steps K --warmup W`
and parses rank 0's single JSON line — this test pins that contract so a
refactor cannot silently break the measurement harness.
"""

from __future__ import annotations

import importlib.util
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.skipif(
    importlib.util.find_spec("torch") is None,
    reason="bench.py needs torch (absent in the lint-only CI image)",
)
def test_bench_emits_contract_json_line():
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "2",
         "--warmup", "1", "--mock", "1x8GiB", "--pods-per-gpu", "2"],
        cwd=REPO, capture_output=True, text=True, timeout=180,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line: {out.stdout!r}"
    d = json.loads(lines[0])

    assert d["metric"].startswith("pods/sec")
    assert isinstance(d["value"], (int, float)) and d["value"] > 0
    assert d["unit"] == "pods/s"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert isinstance(d["ms_per_step"], (int, float)) and d["ms_per_step"] > 0
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["vs_baseline"] is None     # reference publishes no numbers
    assert d["data"] == "synthetic"
    cfg = d["config"]
    assert cfg["pods_allocated"] == 2 * 2 and cfg["pods_failed"] == 0
    for key in ("allocate_p50_ms", "allocate_p99_ms", "packing_pct_peak",
                "server_allocate_p50_ms", "plugin_rank_rss_mb"):
        assert isinstance(cfg[key], (int, float)), key
