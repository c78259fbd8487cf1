#!/usr/bin/env python3
"""Hash an 8 GiB HBM buffer with the CDNA4 SHA-256 chunk kernel (for
rocprofv3 kernel-trace / PMC runs)."""
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch

from modelx_amd import _core

size = int(float(os.environ.get("HASH_GIB", "8")) * (1 << 30))
cs = int(os.environ.get("HASH_CHUNK", str(128 << 10)))
eng = _core.GpuEngine(device=0, num_slots=2, slot_bytes=1 << 20, num_streams=1)
buf = torch.empty(size, dtype=torch.uint8, device="cuda:0")
buf.random_(0, 256)
torch.cuda.synchronize()
eng.sha256_chunk_leaves(buf.data_ptr(), 1 << 20, cs)  # warm
t0 = time.monotonic()
reps = int(os.environ.get("HASH_REPS", "3"))
for _ in range(reps):
    eng.sha256_chunk_leaves(buf.data_ptr(), size, cs)
dt = (time.monotonic() - t0) / reps
print(f"sha256_chunk_leaves: {size / dt / (1 << 30):.1f} GiB/s (chunk={cs}, reps={reps})")

# small-blob regime A/B: 64 MiB spans (the per-slot streaming-hash shape)
small = 64 << 20
t0 = time.monotonic()
for _ in range(20):
    eng.sha256_chunk_leaves(buf.data_ptr(), small, cs)
dt = (time.monotonic() - t0) / 20
print(f"sha256_chunk_leaves 64MiB: {small / dt / (1 << 30):.1f} GiB/s")
