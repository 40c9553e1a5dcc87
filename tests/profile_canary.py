"""Workload profiled by rocprofv3: canary probe + tenant GEMM on MI355X."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import gpushare_amd._canary as canary
print("probe:", canary.probe(0, vram_probe_mb=256, bandwidth=True), flush=True)
import torch
# randn's Philox kernel segfaults under rocprofv3 on this stack; use fills
x = torch.empty(4096, 4096, device="cuda:0", dtype=torch.bfloat16).fill_(0.01)
w = torch.empty(4096, 4096, device="cuda:0", dtype=torch.bfloat16).fill_(0.02)
for _ in range(20):
    x = (x @ w).clamp_(-3, 3)
torch.cuda.synchronize()
print("tenant gemm done", flush=True)
