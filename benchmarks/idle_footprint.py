#!/usr/bin/env python3
"""Idle footprint of the production daemon.

Starts the real daemon (amdsmi source on a GPU box, mock source
elsewhere) against a fake apiserver + stub kubelet, with health-check and
active canary probing enabled, then samples /proc/<pid>/status VmRSS and
utime+stime over an observation window.  The round-1 daemon idled at
~525 MB RSS because the in-process canary mapped the HIP runtime; the r2
default (--probe-mode subprocess) must bring the daemon back to
control-plane size (reference pod: 300 Mi Guaranteed,
device-plugin-ds.yaml:34-40).

Usage: python benchmarks/idle_footprint.py [--seconds 45]
       [--probe-mode subprocess|inproc|both] [--mock] [-o out.json]
"""

from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from gpushare_amd import consts  # noqa: E402
from gpushare_amd.cluster.fakeapiserver import FakeApiServer  # noqa: E402
from gpushare_amd.cluster.kubeclient import FakeKubeClient  # noqa: E402
from gpushare_amd.deviceplugin.stubkubelet import StubKubelet  # noqa: E402

NODE = "footprint-node"


def read_rss_mb(pid: int) -> float:
    with open(f"/proc/{pid}/status") as fh:
        for line in fh:
            if line.startswith("VmRSS:"):
                return int(line.split()[1]) / 1024.0
    return 0.0


def read_cpu_ticks(pid: int) -> int:
    with open(f"/proc/{pid}/stat") as fh:
        parts = fh.read().split()
    return int(parts[13]) + int(parts[14])  # utime + stime


def measure(probe_mode: str, seconds: float, mock: bool) -> dict:
    store = FakeKubeClient(node_name=NODE)
    api = FakeApiServer(store=store).start()
    tmp = tempfile.mkdtemp(prefix="gpushare-fp-")
    kubeconfig = os.path.join(tmp, "kubeconfig")
    with open(kubeconfig, "w") as fh:
        json.dump(
            {
                "current-context": "fp",
                "contexts": [
                    {"name": "fp", "context": {"cluster": "c", "user": "u"}}
                ],
                "clusters": [{"name": "c", "cluster": {"server": api.url}}],
                "users": [{"name": "u", "user": {"token": "t"}}],
            },
            fh,
        )
    sockdir = os.path.join(tmp, "dp")
    os.makedirs(sockdir)
    kubelet = StubKubelet(sockdir)
    kubelet.start()

    env = dict(os.environ)
    env.update(NODE_NAME=NODE, KUBECONFIG=kubeconfig, PYTHONPATH=REPO)
    cmd = [
        sys.executable, "-m", "gpushare_amd.cli.daemon",
        "--socket-dir", sockdir,
        "--health-check", "--deep-probe", "10",
        "--probe-mode", probe_mode,
    ]
    if mock:
        cmd += ["--mock-spec", "1x288GiB"]
    proc = subprocess.Popen(
        cmd, env=env, cwd=REPO,
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
    )
    result = {"probe_mode": probe_mode, "mock": mock, "seconds": seconds}
    try:
        client = kubelet.wait_for_plugin(consts.RESOURCE_NAME, timeout=60)
        inv = client.wait_for_devices(min_count=1, timeout=30)
        result["fake_devices"] = len(inv)
        # settle, then sample
        time.sleep(5)
        samples = []
        t0, c0 = time.monotonic(), read_cpu_ticks(proc.pid)
        deadline = time.monotonic() + seconds
        while time.monotonic() < deadline:
            samples.append(read_rss_mb(proc.pid))
            time.sleep(2)
        t1, c1 = time.monotonic(), read_cpu_ticks(proc.pid)
        hz = os.sysconf("SC_CLK_TCK")
        result.update(
            rss_mb_mean=round(sum(samples) / len(samples), 1),
            rss_mb_max=round(max(samples), 1),
            cpu_pct=round(100.0 * (c1 - c0) / hz / (t1 - t0), 2),
            n_samples=len(samples),
        )
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=10)
        kubelet.stop()
        api.stop()
    return result


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--seconds", type=float, default=45)
    p.add_argument("--probe-mode", default="both",
                   choices=("subprocess", "inproc", "both"))
    p.add_argument("--mock", action="store_true",
                   help="mock source (no GPU box)")
    p.add_argument("-o", "--out", default="")
    args = p.parse_args()

    modes = (
        ["subprocess", "inproc"]
        if args.probe_mode == "both"
        else [args.probe_mode]
    )
    rows = [measure(m, args.seconds, args.mock) for m in modes]
    out = {"host": os.uname().nodename, "rows": rows}
    text = json.dumps(out, indent=2)
    print(text)
    if args.out:
        with open(args.out, "w") as fh:
            fh.write(text + "\n")
    return 0


if __name__ == "__main__":
    sys.exit(main())
