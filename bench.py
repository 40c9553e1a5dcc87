#!/usr/bin/env python3
"""Flagship benchmark: GPU-share scheduling throughput on MI355X.

Measures the BASELINE.json headline metric — Allocate() p50/p99 latency,
gpu-mem packing %, and pods/sec scheduled — on synthetic pod specs against
the full stack: fake apiserver (real HTTP), in-tree scheduler extender
(binpack webhook, real HTTP), and the device plugin (real unix-socket gRPC,
real MI355X device enumeration through the native amdsmi shim when GPUs are
present; mock device source otherwise, reported in `data`).

Topology per run (same shape as a real node):
  rank 0 hosts apiserver + extender + ONE plugin managing N GPUs;
  every rank is a churn generator playing scheduler+kubelet for its share
  of pods (weak scaling: pods per GPU fixed as N grows).

One step = one churn wave: each rank creates/binds/allocates
`--pods-per-gpu` pods (filling its GPU at --pod-gib 72 × 4 = 288 GiB),
records peak packing, then deletes/releases them.  The timed region is the
control plane — that IS the workload of this framework (SURVEY: the
reference schedules containers, it does not execute tensors).

Usage (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nproc-per-node N bench.py --gpus N --steps K --warmup W

Rank 0 prints exactly ONE JSON line with the aggregate result.
"""

from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
import torch.distributed as dist

from gpushare_amd import consts
from gpushare_amd.allocator import Allocator
from gpushare_amd.cluster.kubeclient import RestKubeClient
from gpushare_amd.cluster.podmanager import PodManager
from gpushare_amd.deviceplugin.stubkubelet import DevicePluginClient
from gpushare_amd.deviceplugin.server import GPUSharePlugin
from gpushare_amd.extender.server import ExtenderClient

NODE = "bench-node"


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--pods-per-gpu", type=int, default=4)
    p.add_argument("--pod-gib", type=int, default=0,
               help="0 = derive as advertised_units // pods_per_gpu "
                    "(287//4=71 on real MI355X: 288 GiB minus reserved)")
    p.add_argument("--rate", type=float, default=0.0,
                   help="cap total pod churn at this many pods/s across all "
                        "ranks (BASELINE config 5: 10); 0 = unthrottled")
    p.add_argument(
        "--mixed",
        action="store_true",
        help="BASELINE config 4: random {8,16,32,64,128} GiB pods; report packing%%",
    )
    p.add_argument(
        "--mock",
        default=None,
        help="force mock device source, e.g. 8x288GiB (default: real GPUs, "
        "mock fallback when none present)",
    )
    return p.parse_args()


def get_gpus(n: int, mock_spec):
    """First n GPUs from the real amdsmi source, else mock."""
    if mock_spec is None and os.environ.get("GPUSHARE_MOCK_SPEC"):
        mock_spec = os.environ["GPUSHARE_MOCK_SPEC"]
    if mock_spec is None:
        try:
            from gpushare_amd.device.amdsmi_source import AmdSmiSource

            gpus = AmdSmiSource().devices()
            if len(gpus) >= n:
                return gpus[:n], "amdsmi"
            print(
                f"# only {len(gpus)} real GPUs for --gpus {n}; using mock",
                file=sys.stderr,
            )
        except (RuntimeError, ImportError) as e:
            print(f"# no real GPUs ({e}); using mock", file=sys.stderr)
        mock_spec = f"{n}x288GiB"
    from gpushare_amd.device.mock_source import MockSource

    return MockSource.from_spec(mock_spec).devices()[:n], "mock"


def make_pod_spec(name: str, mem: int) -> dict:
    return {
        "metadata": {"name": name, "namespace": "default"},
        "spec": {
            "nodeName": NODE,
            "containers": [
                {
                    "name": "main",
                    "resources": {"limits": {consts.RESOURCE_NAME: str(mem)}},
                }
            ],
        },
        "status": {"phase": "Pending"},
    }


MIXED_SIZES = [8, 16, 32, 64, 128]


class RateLimiter:
    """Paces one rank's pod submissions at rate/world pods/s."""

    def __init__(self, per_rank_rate: float):
        self.interval = 1.0 / per_rank_rate
        self.next_t = time.perf_counter()

    def wait(self):
        now = time.perf_counter()
        if now < self.next_t:
            time.sleep(self.next_t - now)
        self.next_t = max(self.next_t + self.interval, now - 5 * self.interval)


def run_wave(
    rank: int,
    step: int,
    api: RestKubeClient,
    ext: ExtenderClient,
    plugin_client: DevicePluginClient,
    args,
    lat_allocate: list,
    lat_extender: list,
    grains: list[str],
    rate_limiter=None,
    barrier=None,
) -> tuple[int, int, float]:
    """One churn wave for this rank.  Returns (allocated, failed, peak_packing)."""
    n_pods = args.pods_per_gpu
    pods = []
    # deterministic "random" mixed sizes (no global RNG: reproducible per rank/step)
    for i in range(n_pods):
        mem = (
            MIXED_SIZES[(rank * 131 + step * 17 + i * 7) % len(MIXED_SIZES)]
            if args.mixed
            else args.pod_gib
        )
        pods.append((f"bench-r{rank}-s{step}-p{i}", mem))

    allocated = failed = 0
    placed: dict = {}
    # create + bind + allocate
    for name, mem in pods:
        if rate_limiter is not None:
            rate_limiter.wait()
        spec = make_pod_spec(name, mem)
        api.create_pod(spec)
        t0 = time.perf_counter()
        # real scheduler flow: filter (carries the full pod) then bind
        ext.filter(spec, [NODE])
        err = ext.bind("default", name, NODE)
        lat_extender.append(time.perf_counter() - t0)
        if err:
            failed += 1
            api.delete_pod("default", name)
            continue
        t0 = time.perf_counter()
        resp = plugin_client.allocate([grains[:mem]])
        lat_allocate.append(time.perf_counter() - t0)
        envs = resp.container_responses[0].envs
        if envs[consts.ENV_RESOURCE_INDEX] == "-1":
            failed += 1
        else:
            allocated += 1
            placed[name] = (mem, envs[consts.ENV_RESOURCE_INDEX])
    peak = ext.packing()["packing_pct"]
    if barrier is not None:
        barrier()  # all ranks fully allocated -> peak sample is honest
        peak = max(peak, ext.packing()["packing_pct"])
    # delete + release (release stub built locally — the informer's delete
    # event carries the pod object, no extra apiserver GET needed)
    for name, mem in pods:
        api.delete_pod("default", name)
        if name in placed:
            mem_placed, idx = placed[name]
            stub = make_pod_spec(name, mem_placed)
            stub["metadata"]["annotations"] = {consts.ENV_RESOURCE_INDEX: idx}
            ext.release(stub, NODE)
    return allocated, failed, peak


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    distributed = world > 1
    if distributed:
        # gloo: the measured path is the k8s control plane (gRPC/HTTP); the
        # GPUs are exercised by device enumeration + injection, not collectives
        dist.init_process_group("gloo", rank=rank, world_size=world)

    sync = []
    if rank == 0:
        # --- rank 0: apiserver proc + extender proc + in-proc plugin ------
        # apiserver and extender run as their own processes (as in a real
        # cluster) so the plugin's GIL is not shared with them.
        gpus, source_kind = get_gpus(args.gpus, args.mock)
        repo = os.path.dirname(os.path.abspath(__file__))
        api_proc = subprocess.Popen(
            [sys.executable, "-m", "gpushare_amd.cluster.fakeapiserver",
             "--node", NODE],
            stdout=subprocess.PIPE, text=True, cwd=repo,
        )
        api_url = api_proc.stdout.readline().split()[1]

        kube0 = RestKubeClient(base_url=api_url)
        units = [g.mem_units(consts.GIB) for g in gpus]
        # advertise node resources (what kubelet would do from ListAndWatch +
        # the plugin's patch_gpu_count) so the extender daemon discovers them
        kube0.patch_node_status(NODE, {"status": {
            "capacity": {consts.RESOURCE_COUNT: str(len(gpus)),
                         consts.RESOURCE_NAME: str(sum(units))},
            "allocatable": {consts.RESOURCE_COUNT: str(len(gpus)),
                            consts.RESOURCE_NAME: str(sum(units))},
        }})
        # --no-watch: the bench releases explicitly (synchronous wave
        # teardown); the daemon's informer auto-release would race it
        ext_proc = subprocess.Popen(
            [sys.executable, "-m", "gpushare_amd.extender",
             "--api-url", api_url, "--port", "0",
             "--resync-interval", "3600", "--no-watch"],
            stdout=subprocess.PIPE, text=True, cwd=repo,
        )
        ext_url = ext_proc.stdout.readline().split()[1]

        from gpushare_amd.cluster.informer import PodInformer

        informer = PodInformer(RestKubeClient(base_url=api_url), NODE)
        informer.start()
        informer.wait_synced(timeout=10)
        pm = PodManager(
            RestKubeClient(base_url=api_url),
            NODE,
            kubelet_client=None,
            query_kubelet=False,
            cache_ttl=0.05,
            apiserver_retries=0,
            informer=informer,
        )
        sockdir = tempfile.mkdtemp(prefix="gpushare-bench-")
        plugin = GPUSharePlugin(
            gpus, Allocator(gpus, pm), socket_dir=sockdir
        )
        plugin.start()
        sync = [
            {
                "api_url": api_url,
                "ext_url": ext_url,
                "socket": plugin.socket_path,
                "source": source_kind,
                "units_per_gpu": gpus[0].mem_units(consts.GIB),
            }
        ]
    if distributed:
        if rank != 0:
            sync = [None]
        dist.broadcast_object_list(sync, src=0)
    info = sync[0]

    api = RestKubeClient(base_url=info["api_url"])
    ext = ExtenderClient(info["ext_url"])
    plugin_client = DevicePluginClient(info["socket"], consts.RESOURCE_NAME)
    if args.pod_gib <= 0:
        args.pod_gib = info["units_per_gpu"] // args.pods_per_gpu
    total_grains = args.gpus * info["units_per_gpu"]
    plugin_client.wait_for_devices(min_count=total_grains, timeout=30)
    grains = sorted(plugin_client.devices)

    lat_allocate: list[float] = []
    lat_extender: list[float] = []
    limiter = RateLimiter(args.rate / world) if args.rate > 0 else None

    # --- warmup -----------------------------------------------------------
    for step in range(args.warmup):
        if distributed:
            dist.barrier()
        run_wave(rank, -1 - step, api, ext, plugin_client, args,
                 lat_allocate, lat_extender, grains, limiter,
                 dist.barrier if distributed else None)
    lat_allocate.clear()
    lat_extender.clear()
    if rank == 0:
        plugin.allocator.stats = type(plugin.allocator.stats)()

    # --- timed region -----------------------------------------------------
    if distributed:
        dist.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    allocated = failed = 0
    packing_samples = []
    for step in range(args.steps):
        if distributed:
            dist.barrier()  # align waves: capacity invariant needs all ranks'
                            # deletes from wave N-1 done before wave N creates
        a, f, peak = run_wave(rank, step, api, ext, plugin_client, args,
                              lat_allocate, lat_extender, grains, limiter,
                              dist.barrier if distributed else None)
        allocated += a
        failed += f
        packing_samples.append(peak)
        # stderr heartbeat so a soak killed by an outer timeout still
        # leaves rate/latency evidence in its log (never on stdout — the
        # one-JSON-line contract stays intact)
        if rank == 0 and step and step % 2000 == 0:
            dt = time.perf_counter() - t0
            lat = sorted(lat_allocate)
            p50 = lat[len(lat) // 2] * 1e3 if lat else 0.0
            print(
                f"# progress step={step}/{args.steps} "
                f"pods={allocated} failed={failed} "
                f"rate={allocated / dt:.0f}/s/rank0 "
                f"alloc_p50={p50:.2f}ms",
                file=sys.stderr,
                flush=True,
            )
    if distributed:
        dist.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # --- aggregate --------------------------------------------------------
    if distributed:
        gathered: list = [None] * world if rank == 0 else None
        dist.gather_object(
            {
                "elapsed": elapsed,
                "allocated": allocated,
                "failed": failed,
                "lat_allocate": lat_allocate,
                "lat_extender": lat_extender,
                "packing": packing_samples,
            },
            gathered,
            dst=0,
        )
    else:
        gathered = [
            {
                "elapsed": elapsed,
                "allocated": allocated,
                "failed": failed,
                "lat_allocate": lat_allocate,
                "lat_extender": lat_extender,
                "packing": packing_samples,
            }
        ]

    if rank == 0:
        elapsed_max = max(g["elapsed"] for g in gathered)
        total_alloc = sum(g["allocated"] for g in gathered)
        total_failed = sum(g["failed"] for g in gathered)
        all_lat = sorted(x for g in gathered for x in g["lat_allocate"])
        all_ext = sorted(x for g in gathered for x in g["lat_extender"])
        pct = lambda v, p: v[min(len(v) - 1, int(p * len(v)))] * 1e3 if v else 0.0  # noqa: E731
        packing_peak = max(
            (p for g in gathered for p in g["packing"]), default=0.0
        )
        import resource

        server_stats = plugin.allocator.stats.snapshot()
        pods_per_sec = total_alloc / elapsed_max if elapsed_max > 0 else 0.0
        rss_mb = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1024

        result = {
            "metric": "pods/sec scheduled (gpu-mem share Allocate pipeline)",
            "value": round(pods_per_sec, 2),
            "unit": "pods/s",
            "n_gpus": args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed_max / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "n/a",
            "data": "synthetic",
            "config": {
                "model": "gpushare-device-plugin (MI355X)",
                "global_batch": args.pods_per_gpu * world,
                "seq_len": args.pod_gib,
                "parallelism": f"1 plugin + {world} churn ranks",
                "baseline_config": (
                    "mixed-size stress {8,16,32,64,128}GiB"
                    if args.mixed
                    else f"{args.pods_per_gpu}/GPU × {args.pod_gib}GiB binpack"
                ),
                "device_source": info["source"],
                "rate_cap_pods_per_s": args.rate or None,
                "pods_allocated": total_alloc,
                "pods_failed": total_failed,
                "allocate_p50_ms": round(pct(all_lat, 0.50), 3),
                "allocate_p99_ms": round(pct(all_lat, 0.99), 3),
                "extender_rtt_p50_ms": round(pct(all_ext, 0.50), 3),
                "extender_rtt_p99_ms": round(pct(all_ext, 0.99), 3),
                "packing_pct_peak": round(packing_peak, 2),
                "server_allocate_p50_ms": round(server_stats["p50_ms"], 3),
                "server_allocate_p99_ms": round(server_stats["p99_ms"], 3),
                "plugin_rank_rss_mb": round(rss_mb, 1),
            },
        }
        print(json.dumps(result), flush=True)

    # --- teardown ---------------------------------------------------------
    # the measurement is already printed; a wedged teardown (grpc stream,
    # informer reconnect loop, subprocess) must never hang the run
    import threading as _threading

    _watchdog = _threading.Timer(20.0, lambda: os._exit(0))
    _watchdog.daemon = True
    _watchdog.start()
    sys.stdout.flush()
    plugin_client.close()
    ext.close()
    api.close()
    if distributed:
        dist.barrier()
        dist.destroy_process_group()
    if rank == 0:
        informer.stop()
        plugin.stop()
        ext_proc.terminate()
        api_proc.terminate()
        ext_proc.wait(timeout=5)
        api_proc.wait(timeout=5)
    os._exit(0)


if __name__ == "__main__":
    main()
