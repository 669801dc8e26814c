"""GPU presence/count probes that DO NOT initialize the HIP runtime.

The launcher must stay HIP-clean: it forks serving instances, and a child
forked from a HIP-initialized parent degrades catastrophically (measured
~0.3 s per kernel launch -> 100+ s engine init for a 15 GiB model).
``torch.cuda.is_available()``/``device_count()`` initialize HIP, so any
process that will fork GPU children must use these probes instead.
"""

from __future__ import annotations

import os
import subprocess
from typing import Optional

_CACHED_COUNT: Optional[int] = None


def gpu_present() -> bool:
    """AMD GPUs on this machine? (KFD device node; no HIP init)."""
    return os.path.exists("/dev/kfd")


def gpu_count() -> int:
    """Number of visible GPUs without initializing HIP."""
    global _CACHED_COUNT
    if _CACHED_COUNT is not None:
        return _CACHED_COUNT
    if not gpu_present():
        _CACHED_COUNT = 0
        return 0
    vis = os.environ.get("HIP_VISIBLE_DEVICES")
    if vis is not None:
        _CACHED_COUNT = len([v for v in vis.split(",") if v != ""])
        return _CACHED_COUNT
    try:
        out = subprocess.run(
            ["rocm-smi", "--showid", "--json"], capture_output=True,
            text=True, timeout=15).stdout
        import json
        cards = [k for k in json.loads(out or "{}") if k.startswith("card")]
        _CACHED_COUNT = len(cards)
    except Exception:
        # KFD topology fallback: nodes with a GPU id
        n = 0
        base = "/sys/class/kfd/kfd/topology/nodes"
        try:
            for d in os.listdir(base):
                try:
                    with open(os.path.join(base, d, "gpu_id")) as f:
                        if int(f.read().strip() or 0) != 0:
                            n += 1
                except OSError:
                    continue
        except OSError:
            pass
        _CACHED_COUNT = n
    return _CACHED_COUNT
