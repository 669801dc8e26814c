"""Why is staged sleep slow AFTER an overlapped wake in the bench flow?"""
import sys
import time

import torch

from fma_amd.models.llama import LlamaConfig
from fma_amd.ops import actuation
from fma_amd.ops.actuation import PackActuator, require_native

C = require_native()
gib = float(sys.argv[1]) if len(sys.argv) > 1 else 16.0
cfg = LlamaConfig.from_total_gib(gib)
specs = cfg.param_specs(0, 1)
layout, total, _ = actuation.plan_layout(specs)
params = {n: torch.empty(s, dtype=d, device="cuda")
          for n, (o, s, d) in layout.items()}
for t in params.values():
    t.view(torch.uint8).fill_(1)
packer = PackActuator(params, mode=0)
host = torch.empty(packer.total_bytes, dtype=torch.uint8, pin_memory=True)
g = packer.total_bytes / (1 << 30)

def serial_wake():
    for n in packer.names:
        t = packer.tensors[n]
        t.untyped_storage().resize_(t.numel() * t.element_size())
    t = C.restore_from_host(packer._tensor_list(), packer.offsets, host,
                            packer.mode, packer.chunk_bytes, 1)
    packer.asleep = False
    return t

def cycle(label, wake_fn, drop_staging=False, n=3):
    for i in range(n):
        torch.cuda.synchronize()
        ts = packer.sleep(host)
        torch.cuda.synchronize()
        if drop_staging:
            C._release_staging(0)
        t0 = time.perf_counter()
        wake_fn()
        torch.cuda.synchronize()
        tw = time.perf_counter() - t0
        print(f"{label:22s} cycle {i}: sleep {g/ts:5.1f} GiB/s "
              f"wake {g/tw:5.1f} GiB/s")

cycle("A overlapped+staging", lambda: packer.wake(host))
cycle("B overlapped, fresh stg", lambda: packer.wake(host), drop_staging=True)
cycle("C serial wake", serial_wake)
cycle("A2 overlapped again", lambda: packer.wake(host))

# kernel-read vs SDMA-read on re-allocated memory (run with arg2="kernel")
if len(sys.argv) > 2 and sys.argv[2] == "kernel":
    ts_l = packer._tensor_list()
    out = torch.empty(packer.total_bytes, dtype=torch.uint8, device="cuda")
    d = C.gather_d2d(ts_l, packer.offsets, out, 3)
    print(f"gather_d2d on re-allocated tensors: {g/d:6.1f} GiB/s")
    t0 = time.perf_counter()
    C.pack_to_host(ts_l, packer.offsets, host, 2, 0, 1)
    torch.cuda.synchronize()
    print(f"sdma per-tensor D2H same tensors:   "
          f"{g/(time.perf_counter()-t0):6.1f} GiB/s")
