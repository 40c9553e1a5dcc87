"""Workload profiled by rocprofv3: the in-tree gfx950 canary kernels.

torch is deliberately absent: any torch kernel launch under rocprofv3
segfaults on this stack (torch 2.10+rocm7.0 / rocprofv3 7.2), so the
profile covers the framework's own HIP kernels (mfma_canary_kernel,
vram_write/check_pattern, bw_copy).
"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import gpushare_amd._canary as canary

for rep in range(5):
    r = canary.probe(0, vram_probe_mb=256, bandwidth=True)
    assert r["ok"], r
print("final:", r, flush=True)
