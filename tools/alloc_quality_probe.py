"""Does allocating (caching allocator or hipMalloc) WHILE SDMA copies are
in flight produce memory that later reads back slower over PCIe?"""
import threading
import time

import torch

from fma_amd.ops.actuation import require_native

C = require_native()
N = 8 << 30  # 8 GiB of tensors
host = torch.empty(N, dtype=torch.uint8, pin_memory=True)

def d2h_rate(ts, label):
    offs, off = [], 0
    for t in ts:
        offs.append(off)
        off += (t.nbytes + 255) // 256 * 256
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    C.pack_to_host(ts, offs, host, 2, 0, 1)  # per-tensor, 1 stream
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"{label:28s} D2H {off/ (1<<30) / dt:6.1f} GiB/s")

def alloc_tensors():
    ts = [torch.empty(256 << 20, dtype=torch.uint8, device="cuda")
          for _ in range(32)]
    for t in ts:
        t.fill_(1)
    return ts

# 1) idle allocation
ts = alloc_tensors()
d2h_rate(ts, "alloc idle")
del ts
torch.cuda.empty_cache()

# 2) allocation under H2D traffic
traffic_dev = torch.empty(4 << 30, dtype=torch.uint8, device="cuda")
traffic_host = torch.empty(4 << 30, dtype=torch.uint8, pin_memory=True)
stop = threading.Event()
def traffic():
    s = torch.cuda.Stream()
    with torch.cuda.stream(s):
        while not stop.is_set():
            traffic_dev.copy_(traffic_host, non_blocking=True)
            s.synchronize()
th = threading.Thread(target=traffic)
th.start()
time.sleep(0.2)
ts = alloc_tensors()
torch.cuda.synchronize()
stop.set(); th.join()
d2h_rate(ts, "alloc under H2D traffic")
del ts, traffic_dev
torch.cuda.empty_cache()

# 3) re-alloc same sizes AFTER traffic stopped (fresh hipMalloc, idle)
ts = alloc_tensors()
d2h_rate(ts, "re-alloc idle")
