#!/usr/bin/env python3
"""Run all five BASELINE.json benchmark configs and write a JSON report.

  1. mock-GPU plumbing (1 fake 8 GiB device, stub kubelet, CPU only)
  2. single MI355X, 4 × 72 GiB pods sharing one device
  3. 8×MI355X, 32 × 72 GiB bin-packed (torchrun, 8 ranks)
  4. mixed-size stress: random {8,16,32,64,128} GiB, packing %
  5. churn: 10 pods/s sustained, Allocate p50/p99 + extender RTT

Configs 2/3 use real GPUs when present (amdsmi source), mock otherwise —
each result records which.  Usage:
  python benchmarks/run_all.py [--out report.json] [--quick]
"""

from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(extra: list[str], ranks: int = 1, timeout: int = 900) -> dict:
    if ranks > 1:
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", f"--nproc-per-node={ranks}",
            "--master-addr", "127.0.0.1", "--master-port", "29771",
            "bench.py", *extra,
        ]
    else:
        cmd = [sys.executable, "bench.py", *extra]
    proc = subprocess.run(
        cmd, cwd=REPO, capture_output=True, text=True, timeout=timeout
    )
    for line in proc.stdout.splitlines():
        if line.startswith("{"):
            return json.loads(line)
    return {
        "error": f"no result line (rc={proc.returncode})",
        "stderr": proc.stderr[-2000:],
    }


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="benchmarks/report.json")
    p.add_argument("--quick", action="store_true", help="fewer steps")
    p.add_argument("--gpus-available", type=int, default=0,
                   help="real GPUs on this node (0 = probe)")
    args = p.parse_args()

    steps = "5" if args.quick else "20"
    churn_steps = "3" if args.quick else "15"
    n_gpus = args.gpus_available
    if n_gpus == 0:
        try:
            sys.path.insert(0, REPO)
            from gpushare_amd.device.amdsmi_source import AmdSmiSource

            n_gpus = len(AmdSmiSource().devices())
        except Exception:
            n_gpus = 0

    report = {"gpus_detected": n_gpus, "results": {}}

    configs = {
        "1_mock_plumbing": dict(
            extra=["--mock", "1x8GiB", "--gpus", "1", "--steps", steps,
                   "--warmup", "2", "--pods-per-gpu", "4"],
            ranks=1,
        ),
        "2_single_mi355x_4x72": dict(
            extra=["--gpus", "1", "--steps", steps, "--warmup", "2",
                   "--pods-per-gpu", "4"],
            ranks=1,
        ),
        "3_8gpu_binpack_32x72": dict(
            extra=["--gpus", "8", "--steps", steps, "--warmup", "2",
                   "--pods-per-gpu", "4"],
            ranks=8,
        ),
        "4_mixed_stress": dict(
            extra=["--gpus", "8", "--steps", steps, "--warmup", "2",
                   "--pods-per-gpu", "8", "--mixed"],
            ranks=8,
        ),
        "5_churn_10pps": dict(
            extra=["--gpus", "8", "--steps", churn_steps, "--warmup", "1",
                   "--pods-per-gpu", "4", "--rate", "10"],
            ranks=8,
        ),
        # MI355X-native extension (not a reference config): pods larger
        # than one GPU, split over xGMI-adjacent pairs by the extender
        "6_multigpu_span_5x400": dict(
            extra=["--gpus", "8", "--mock", "8x288GiB", "--steps", steps,
                   "--warmup", "2", "--pods-per-gpu", "5",
                   "--pod-gib", "400"],
            ranks=1,
        ),
    }
    for name, cfg in configs.items():
        t0 = time.time()
        print(f"== {name} ==", flush=True)
        try:
            result = run_bench(cfg["extra"], ranks=cfg["ranks"])
        except subprocess.TimeoutExpired:
            result = {"error": "timeout"}
        result["wall_s"] = round(time.time() - t0, 1)
        report["results"][name] = result
        print(json.dumps(result.get("config", result), indent=None)[:400],
              flush=True)

    os.makedirs(os.path.dirname(os.path.join(REPO, args.out)), exist_ok=True)
    with open(os.path.join(REPO, args.out), "w") as f:
        json.dump(report, f, indent=2)
    print(f"report -> {args.out}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
