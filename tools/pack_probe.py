"""Decompose pack-mode wake cost on a GPU box.

Phases timed separately (16 GiB default, 70B-like tensor sizes):
  alloc_fresh   caching-allocator resize of every storage after empty_cache
  restore_hot   staged restore with storages already allocated
  restore_cold  plain restore preceded by the resize loop (old wake path)
  restore_ovl   restore_from_host_overlapped from released storages
  pt_hot/ovl    same for per-tensor mode
"""

import argparse
import time

import torch

from fma_amd.ops.actuation import align_up, require_native


def make_tensors(total_gib: float, shard_mib: int = 1664):
    ts = []
    remaining = int(total_gib * (1 << 30))
    shard = shard_mib << 20
    i = 0
    while remaining > 0:
        n = min(shard, remaining) // 2
        ts.append(torch.empty(n, dtype=torch.bfloat16, device="cuda"))
        ts[-1].uniform_(-1, 1)
        remaining -= n * 2
        i += 1
    offsets, off = [], 0
    for t in ts:
        offsets.append(off)
        off += align_up(t.nbytes)
    return ts, offsets, off


def release(ts):
    for t in ts:
        t.untyped_storage().resize_(0)
    torch.cuda.empty_cache()


def alloc_all(ts):
    for t in ts:
        t.untyped_storage().resize_(t.numel() * t.element_size())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gib", type=float, default=16.0)
    ap.add_argument("--chunk-mb", type=int, default=256)
    args = ap.parse_args()
    C = require_native()
    chunk = args.chunk_mb << 20

    ts, offsets, total = make_tensors(args.gib)
    host = torch.empty(total, dtype=torch.uint8, pin_memory=True)
    gib = total / (1 << 30)
    print(f"{len(ts)} tensors, {gib:.2f} GiB, chunk {args.chunk_mb} MiB")

    def t(label, fn):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        print(f"{label:14s} {dt:7.3f}s  {gib / dt:7.1f} GiB/s")
        return dt

    # park the bytes once
    for ns in (1, 2):
        t(f"sleep_staged/s{ns}",
          lambda ns=ns: C.pack_to_host(ts, offsets, host, 0, chunk, ns))
        release(ts)
        t(f"restore_ovl/s{ns}", lambda ns=ns: C.restore_from_host_overlapped(
            ts, offsets, host, 0, chunk, ns))

    for cmb in (128, 512, 1024):
        ch = cmb << 20
        C.pack_to_host(ts, offsets, host, 0, ch, 1)
        release(ts)
        t(f"restore_ovl/c{cmb}", lambda ch=ch: C.restore_from_host_overlapped(
            ts, offsets, host, 0, ch, 1))

    C.pack_to_host(ts, offsets, host, 0, chunk, 1)
    release(ts)
    t("pt_ovl/s1", lambda: C.restore_from_host_overlapped(
        ts, offsets, host, 2, chunk, 1))

    C.pack_to_host(ts, offsets, host, 2, chunk, 1)
    t("pt_hot/s1", lambda: C.restore_from_host(ts, offsets, host, 2, chunk, 1))

    # raw link for comparison
    dev = torch.empty(total, dtype=torch.uint8, device="cuda")
    t("raw_h2d", lambda: dev.copy_(host, non_blocking=True))
    del dev
    torch.cuda.empty_cache()
    t("raw_d2h", lambda: host.copy_(
        torch.empty(0)) if False else None) if False else None


if __name__ == "__main__":
    main()
