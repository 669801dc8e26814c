"""Isolated microbench of the batched gather kernel (D2D, no PCIe)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

import fma_amd._C as C  # noqa: E402
from fma_amd.ops.actuation import align_up  # noqa: E402


def run(total_mb=1024, ntensors=40, repeats=20):
    per = (total_mb << 20) // ntensors // 2
    ts = [torch.randn(per, device="cuda:0", dtype=torch.bfloat16)
          for _ in range(ntensors)]
    offsets, off = [], 0
    for t in ts:
        offsets.append(off)
        off += align_up(t.nbytes)
    out = torch.empty(off, dtype=torch.uint8, device="cuda:0")
    sec = C.gather_d2d(ts, offsets, out, repeats)
    moved = 2 * off  # read + write
    print(f"gather_d2d: {off/2**20:.0f} MiB x{ntensors} tensors: "
          f"{sec*1e6:.0f} us/iter = {moved/sec/1e12:.2f} TB/s (r+w)")
    # verify
    ref = torch.cat([t.view(torch.uint8).view(-1) for t in ts])
    got = torch.cat([out[o:o + t.nbytes] for o, t in zip(offsets, ts)])
    assert torch.equal(ref, got), "gather_d2d corrupted data"


if __name__ == "__main__":
    for mb in (256, 1024, 4096):
        run(mb)
