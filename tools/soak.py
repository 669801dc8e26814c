"""Long-soak bit-exactness: N sleep/wake cycles with per-cycle checksums.

The strongest correctness statement for the headline path: every cycle
must restore every byte. Checksums are fp64 sums over stratified slices
of the arena (cheap, catches any corruption pattern the debug tool saw).

GPU:  python tools/soak.py --gib 64 --cycles 40
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from fma_amd.models.llama import LlamaConfig  # noqa: E402
from fma_amd.runtime.engine import ActuationEngine  # noqa: E402


def checksums(eng):
    out = []
    for i, (name, p) in enumerate(sorted(eng.params.items())):
        if i % 7 == 0 or p.numel() < 1024:
            out.append(p.view(-1)[:4096].double().sum().item())
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gib", type=float, default=64.0)
    ap.add_argument("--cycles", type=int, default=40)
    args = ap.parse_args()
    eng = ActuationEngine(LlamaConfig.from_total_gib(args.gib), 0, seed=2024)
    ref = checksums(eng)
    worst_sleep = worst_wake = 0.0
    t0 = time.perf_counter()
    for c in range(args.cycles):
        ts = eng.sleep()
        tw = eng.wake_up()
        worst_sleep = max(worst_sleep, ts)
        worst_wake = max(worst_wake, tw)
        got = checksums(eng)
        if got != ref:
            bad = sum(1 for a, b in zip(got, ref) if a != b)
            print(f"CORRUPTION at cycle {c}: {bad}/{len(ref)} checksums "
                  f"differ", flush=True)
            sys.exit(1)
        if c % 10 == 9:
            print(f"cycle {c+1}/{args.cycles} clean "
                  f"(sleep<= {worst_sleep:.3f}s wake<= {worst_wake:.3f}s)",
                  flush=True)
    dt = time.perf_counter() - t0
    print(f"SOAK OK: {args.cycles} cycles x {eng.total_bytes/2**30:.1f} GiB "
          f"bit-exact in {dt:.0f}s; worst sleep {worst_sleep:.3f}s, "
          f"worst wake {worst_wake:.3f}s")


if __name__ == "__main__":
    main()
