"""Staged-sleep D2H throughput at real model tensor mix."""
import sys
import time

import torch

from fma_amd.models.llama import LlamaConfig
from fma_amd.ops import actuation
from fma_amd.ops.actuation import require_native

C = require_native()
gib = float(sys.argv[1]) if len(sys.argv) > 1 else 16.0
cfg = LlamaConfig.from_total_gib(gib)
specs = cfg.param_specs(0, 1)
layout, total, _ = actuation.plan_layout(specs)
params = {n: torch.empty(s, dtype=d, device="cuda")
          for n, (o, s, d) in layout.items()}
for t in params.values():
    t.view(torch.uint8).fill_(1)
names = sorted(params)
offsets, off = [], 0
for n in names:
    offsets.append(off)
    off += actuation.align_up(max(params[n].nbytes, 1))
ts = [params[n] for n in names]
host = torch.empty(off, dtype=torch.uint8, pin_memory=True)
g = off / (1 << 30)
sizes = sorted(t.nbytes for t in ts)
print(f"{len(ts)} tensors, {g:.1f} GiB, median {sizes[len(sizes)//2]>>20} MiB,"
      f" min {sizes[0]} B, max {sizes[-1]>>20} MiB")

def t(label, fn, n=2):
    best = 1e9
    for _ in range(n):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        best = min(best, time.perf_counter() - t0)
    print(f"{label:16s} {best:7.3f}s  {g / best:7.1f} GiB/s")

for ns in (1, 2, 4):
    t(f"sleep_staged/s{ns}",
      lambda ns=ns: C.pack_to_host(ts, offsets, host, 0, 0, ns))
for ns in (1, 2, 4):
    t(f"sleep_pt/s{ns}",
      lambda ns=ns: C.pack_to_host(ts, offsets, host, 2, 0, ns))
for cmb in (512, 1024):
    t(f"sleep_staged/c{cmb}",
      lambda c=cmb: C.pack_to_host(ts, offsets, host, 0, c << 20, 1))
dev = torch.empty(1 << 30, dtype=torch.uint8, device="cuda")
hb = torch.empty(1 << 30, dtype=torch.uint8, pin_memory=True)
def raw():
    hb.copy_(dev, non_blocking=True)
torch.cuda.synchronize()
t0 = time.perf_counter(); raw(); torch.cuda.synchronize()
print(f"raw_d2h 1GiB      {1/( time.perf_counter()-t0):7.1f} GiB/s")
