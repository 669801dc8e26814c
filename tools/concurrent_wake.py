"""Concurrent wake benchmark (BASELINE config #5): N models, one MI355X.

The launcher-populator scenario: several launchers on one node each hold a
sleeping model; all wake at once and contend for the GPU's single PCIe
link and host DRAM. Measures per-model wake plus the makespan, with and
without admission serialization.

Run:  python tools/concurrent_wake.py --n 4 --gib 15
To fill 288 GB HBM:  --n 4 --gib 60
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
    ap.add_argument("--n", type=int, default=4)
    ap.add_argument("--gib", type=float, default=15.0)
    args = ap.parse_args()

    engines = [ActuationEngine(LlamaConfig.from_total_gib(args.gib), 0,
                               seed=i) for i in range(args.n)]
    total_gib = sum(e.total_bytes for e in engines) / 2**30
    print(f"{args.n} engines, {total_gib:.0f} GiB total on HBM",
          file=sys.stderr)

    def cycle(mode):
        for e in engines:
            e.sleep()
        times = [0.0] * args.n
        t0 = time.perf_counter()
        if mode == "concurrent":
            threads = []
            for i, e in enumerate(engines):
                def w(i=i, e=e):
                    s = time.perf_counter()
                    e.wake_up()
                    times[i] = time.perf_counter() - s
                th = threading.Thread(target=w)
                th.start()
                threads.append(th)
            for th in threads:
                th.join()
        else:
            for i, e in enumerate(engines):
                s = time.perf_counter()
                e.wake_up()
                times[i] = time.perf_counter() - s
        makespan = time.perf_counter() - t0
        return makespan, times

    # warmup one cycle
    cycle("serial")
    serial_mk, serial_t = cycle("serial")
    conc_mk, conc_t = cycle("concurrent")
    print(json.dumps({
        "metric": "concurrent wake of N models (s)",
        "n": args.n,
        "gib_each": round(engines[0].total_bytes / 2**30, 2),
        "serial_makespan_s": round(serial_mk, 3),
        "concurrent_makespan_s": round(conc_mk, 3),
        "concurrent_first_ready_s": round(min(conc_t), 3),
        "concurrent_last_ready_s": round(max(conc_t), 3),
    }))


if __name__ == "__main__":
    main()
