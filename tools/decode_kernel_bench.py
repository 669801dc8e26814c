"""attn_decode kernel microbench: time vs chunks at a given context."""
import os
import sys
import time

import torch

from fma_amd.ops.actuation import require_native

C = require_native()
t = int(sys.argv[1]) if len(sys.argv) > 1 else 4000
qh, kvh, hd = 32, 8, 128
torch.manual_seed(1)
K = torch.randn(8192, kvh, hd, dtype=torch.bfloat16, device="cuda")
V = torch.randn(8192, kvh, hd, dtype=torch.bfloat16, device="cuda")
q = torch.randn(qh, hd, dtype=torch.bfloat16, device="cuda")
bytes_eff = 2 * t * kvh * hd * 2  # distinct K+V bytes

for ch in ("", "8", "16", "32", "64", "128"):
    if ch:
        os.environ["FMA_DECODE_CHUNKS"] = ch
    else:
        os.environ.pop("FMA_DECODE_CHUNKS", None)
    for _ in range(20):
        C.attn_decode_bf16(q, K, V, t)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    N = 200
    for _ in range(N):
        C.attn_decode_bf16(q, K, V, t)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / N * 1e6
    print(f"chunks={ch or 'auto':>4}  {us:7.1f} us  "
          f"{bytes_eff / us / 1e3:7.1f} GB/s distinct "
          f"({4 * bytes_eff / us / 1e3:7.1f} with 4x GQA)")
