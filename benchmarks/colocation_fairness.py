#!/usr/bin/env python3
"""Co-location fairness on one MI355X: what sharing actually costs.

gpushare's premise is N pods space/time-sharing one GPU.  This experiment
measures per-tenant GEMM throughput as co-location density grows — the
number a capacity planner needs when deciding pods-per-GPU.

For N in {1, 2, 4, 8}: launch N identical tenant processes on GPU 0
(each under a memguard budget as Allocate would inject), each running
bf16 GEMMs for a fixed window; report per-tenant and aggregate TFLOP/s
plus the fairness spread (min/max across tenants).

Usage (on an MI355X node):  python benchmarks/colocation_fairness.py
Prints one JSON line per density.
"""

from __future__ import annotations

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TENANT = r"""
import json, os, time
import torch
n = 8192
x = torch.randn(n, n, device="cuda:0", dtype=torch.bfloat16)
w = torch.randn(n, n, device="cuda:0", dtype=torch.bfloat16)
for _ in range(3):
    y = x @ w
torch.cuda.synchronize()
WINDOW = float(os.environ.get("T_WINDOW", "8"))
t0 = time.perf_counter()
iters = 0
while time.perf_counter() - t0 < WINDOW:
    y = x @ w
    iters += 1
    if iters % 8 == 0:
        torch.cuda.synchronize()
torch.cuda.synchronize()
dt = time.perf_counter() - t0
tflops = iters * (2 * n**3) / dt / 1e12
print(json.dumps({"iters": iters, "secs": round(dt, 3),
                  "tflops": round(tflops, 1)}), flush=True)
"""


def run_density(n_tenants: int, budget_gib: int = 30) -> dict:
    import gpushare_amd

    lib = os.path.join(
        os.path.dirname(gpushare_amd.__file__), "libgpushare_memguard.so"
    )
    procs = []
    for i in range(n_tenants):
        env = dict(os.environ)
        if os.path.exists(lib):
            env["LD_PRELOAD"] = lib
            env["GPUSHARE_MEM_LIMIT_BYTES"] = str(budget_gib << 30)
            env["GPUSHARE_POD_UID"] = f"fair{n_tenants}-{i}"
        procs.append(
            subprocess.Popen(
                [sys.executable, "-c", TENANT],
                env=env, stdout=subprocess.PIPE,
                stderr=subprocess.PIPE, text=True, cwd=REPO,
            )
        )
    results = []
    for p in procs:
        out, err = p.communicate(timeout=300)
        if p.returncode != 0:
            raise RuntimeError(f"tenant failed: {err[-800:]}")
        results.append(json.loads(out.strip().splitlines()[-1]))
    for i in range(n_tenants):
        shm = f"/dev/shm/gpushare.memguard.fair{n_tenants}-{i}"
        if os.path.exists(shm):
            os.unlink(shm)
    tfl = [r["tflops"] for r in results]
    return {
        "tenants": n_tenants,
        "per_tenant_tflops": tfl,
        "aggregate_tflops": round(sum(tfl), 1),
        "min_tflops": min(tfl),
        "max_tflops": max(tfl),
        "fairness_min_over_max": round(min(tfl) / max(tfl), 3),
    }


BW_TENANT = r"""
import json, os, time
import torch
n_bytes = 4 << 30
src = torch.empty(n_bytes, dtype=torch.uint8, device="cuda:0")
dst = torch.empty(n_bytes, dtype=torch.uint8, device="cuda:0")
dst.copy_(src); torch.cuda.synchronize()
WINDOW = float(os.environ.get("T_WINDOW", "8"))
t0 = time.perf_counter()
iters = 0
while time.perf_counter() - t0 < WINDOW:
    dst.copy_(src)
    iters += 1
    if iters % 4 == 0:
        torch.cuda.synchronize()
torch.cuda.synchronize()
dt = time.perf_counter() - t0
gbps = iters * 2 * n_bytes / dt / 1e9   # read + write traffic
print(json.dumps({"iters": iters, "secs": round(dt, 3),
                  "hbm_gbps": round(gbps, 1)}), flush=True)
"""


def run_mixed() -> dict:
    """Cross-workload interference: a compute-bound GEMM tenant and an
    HBM-bandwidth-bound copy tenant, solo then co-located."""
    def run(scripts):
        procs = [
            subprocess.Popen([sys.executable, "-c", s],
                             stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                             text=True, cwd=REPO)
            for s in scripts
        ]
        out = []
        for p in procs:
            o, e = p.communicate(timeout=300)
            if p.returncode != 0:
                raise RuntimeError(e[-800:])
            out.append(json.loads(o.strip().splitlines()[-1]))
        return out

    gemm_solo = run([TENANT])[0]
    bw_solo = run([BW_TENANT])[0]
    gemm_co, bw_co = run([TENANT, BW_TENANT])
    return {
        "mixed_interference": {
            "gemm_solo_tflops": gemm_solo["tflops"],
            "gemm_colocated_tflops": gemm_co["tflops"],
            "gemm_retained": round(gemm_co["tflops"] / gemm_solo["tflops"], 3),
            "bw_solo_gbps": bw_solo["hbm_gbps"],
            "bw_colocated_gbps": bw_co["hbm_gbps"],
            "bw_retained": round(bw_co["hbm_gbps"] / bw_solo["hbm_gbps"], 3),
        }
    }


def _with_retry(fn, attempts: int = 2):
    """A tenant can die to transient box state (observed once: a GPU
    memory-fault flake that a fresh run could not reproduce with or
    without memguard) — retry once before failing the whole experiment."""
    last = None
    for _ in range(attempts):
        try:
            return fn()
        except (RuntimeError, subprocess.TimeoutExpired) as e:
            last = e
            print(f"# tenant run failed, retrying: {e}", file=sys.stderr)
    raise last


def main() -> int:
    import argparse

    p = argparse.ArgumentParser()
    p.add_argument("--densities", default="1,2,4,8",
                   help="comma-separated tenant counts")
    p.add_argument("--window", type=float, default=8.0,
                   help="seconds of GEMM per tenant (T_WINDOW)")
    p.add_argument("--skip-mixed", action="store_true")
    args = p.parse_args()
    os.environ["T_WINDOW"] = str(args.window)

    sys.path.insert(0, REPO)
    for n in [int(x) for x in args.densities.split(",") if x]:
        print(json.dumps(_with_retry(lambda n=n: run_density(n))), flush=True)
    if not args.skip_mixed:
        print(json.dumps(_with_retry(run_mixed)), flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
