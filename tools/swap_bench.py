"""Model-swap benchmark (BASELINE config #3): two models on one MI355X.

Measures time-to-ready after swap between two random-init models held by
one launcher-style process: the sleeping model's weights sit in pinned
host DRAM, the active model occupies HBM. Two strategies:

- sequential: sleep(A) then wake(B) — the reference's flow;
- concurrent: wake(B) while A's sleep D2H is still draining (PCIe is
  full-duplex and 288 GB HBM fits both models transiently), gated so B
  reports ready as soon as its own H2D completes.

Run on a GPU box:  python tools/swap_bench.py --gib 15 --cycles 3
"""

import argparse
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from fma_amd.models.llama import LlamaConfig  # noqa: E402
from fma_amd.runtime.engine import ActuationEngine  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gib", type=float, default=15.0,
                    help="~GiB per model (15 ~= llama-3-8b bf16)")
    ap.add_argument("--cycles", type=int, default=3)
    args = ap.parse_args()

    cfg_a = LlamaConfig.from_total_gib(args.gib)
    cfg_b = LlamaConfig.from_total_gib(args.gib)
    a = ActuationEngine(cfg_a, 0, seed=1)
    b = ActuationEngine(cfg_b, 0, seed=2)
    print(f"two models of {a.total_bytes/2**30:.1f} GiB each", file=sys.stderr)
    b.sleep()

    seq, conc = [], []
    for _ in range(args.cycles):
        # sequential: sleep(A); wake(B)
        t0 = time.perf_counter()
        a.sleep()
        b.wake_up()
        seq.append(time.perf_counter() - t0)
        # sequential back
        t0 = time.perf_counter()
        b.sleep()
        a.wake_up()
        seq.append(time.perf_counter() - t0)

    for _ in range(args.cycles):
        # concurrent: wake(B) immediately; A drains D2H in parallel
        t0 = time.perf_counter()
        th = threading.Thread(target=a.sleep)
        th.start()
        b.wake_up()
        ready_b = time.perf_counter() - t0
        th.join()
        conc.append(ready_b)
        t0 = time.perf_counter()
        th = threading.Thread(target=b.sleep)
        th.start()
        a.wake_up()
        ready_a = time.perf_counter() - t0
        th.join()
        conc.append(ready_a)

    out = {
        "metric": "time-to-ready after swap (s)",
        "gib_per_model": round(a.total_bytes / 2**30, 2),
        "sequential_mean_s": round(sum(seq) / len(seq), 4),
        "concurrent_mean_s": round(sum(conc) / len(conc), 4),
        "speedup": round(sum(seq) / sum(conc), 3),
        "cycles": args.cycles,
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
