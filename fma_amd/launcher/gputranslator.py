"""GPU UUID <-> index translation for the launcher (MI355X edition).

The dual-pods controller addresses accelerators by UUID (the requester
reports them — reference pkg/spi/interface.go:34-44); a serving process
needs *indices* for ``HIP_VISIBLE_DEVICES``. The reference translates with
pynvml (reference inference_server/launcher/gputranslator.py:173-195) and
offers two GPU-less mock modes (ConfigMap gpu-map / naive enumeration,
gputranslator.py:80-171). Here:

- real mode: amdsmi python bindings, falling back to ``rocm-smi
  --showuniqueid`` / ``amd-smi`` subprocess parsing;
- gpu-map mode: JSON {uuid: index} from FMA_GPU_MAP_JSON or a file
  (FMA_GPU_MAP_FILE) — how the store-backed e2e injects fake GPUs;
- naive mode: GPU-0..N-1.
"""

from __future__ import annotations

import json
import os
import re
import subprocess
from typing import Dict, List, Optional


class GpuTranslationError(Exception):
    pass


def _amdsmi_map() -> Optional[Dict[str, int]]:
    try:
        import amdsmi  # type: ignore
    except Exception:
        return None
    try:
        amdsmi.amdsmi_init()
        out = {}
        for idx, h in enumerate(amdsmi.amdsmi_get_processor_handles()):
            try:
                info = amdsmi.amdsmi_get_gpu_asic_info(h)
                uid = info.get("asic_serial") or f"GPU-{idx}"
            except Exception:
                uid = f"GPU-{idx}"
            out[str(uid)] = idx
        return out or None
    except Exception:
        return None


def _rocm_smi_map() -> Optional[Dict[str, int]]:
    try:
        res = subprocess.run(["rocm-smi", "--showuniqueid", "--json"],
                             capture_output=True, text=True, timeout=20)
        data = json.loads(res.stdout or "{}")
        out = {}
        for card, fields in data.items():
            m = re.match(r"card(\d+)", card)
            if not m:
                continue
            uid = fields.get("Unique ID") or fields.get("Serial Number")
            if uid:
                out[str(uid)] = int(m.group(1))
        return out or None
    except Exception:
        return None


class GpuTranslator:
    MODE_REAL = "real"
    MODE_GPU_MAP = "gpu-map"
    MODE_NAIVE = "naive"

    def __init__(self, mode: Optional[str] = None,
                 gpu_map: Optional[Dict[str, int]] = None):
        if mode is None:
            mode = os.environ.get("FMA_GPU_MODE")
        if mode is None:
            if os.environ.get("FMA_GPU_MAP_JSON") or \
                    os.environ.get("FMA_GPU_MAP_FILE"):
                mode = self.MODE_GPU_MAP
            else:
                # must not initialize HIP (the launcher forks GPU children)
                from fma_amd.utils.gpus import gpu_present
                mode = self.MODE_REAL if gpu_present() else self.MODE_NAIVE
        self.mode = mode
        self._map = gpu_map or self._load_map()

    def _load_map(self) -> Dict[str, int]:
        if self.mode == self.MODE_REAL:
            m = _amdsmi_map() or _rocm_smi_map()
            if m:
                return m
            # no SMI identity source: positional ids (no HIP init — the
            # launcher forks GPU children)
            from fma_amd.utils.gpus import gpu_count
            n = gpu_count()
            if n == 0:
                raise GpuTranslationError("real mode but no GPUs visible")
            return {f"GPU-{i}": i for i in range(n)}
        if self.mode == self.MODE_GPU_MAP:
            blob = os.environ.get("FMA_GPU_MAP_JSON")
            if not blob:
                path = os.environ.get("FMA_GPU_MAP_FILE")
                if path and os.path.exists(path):
                    with open(path) as f:
                        blob = f.read()
            if not blob:
                raise GpuTranslationError("gpu-map mode but no map provided")
            return {str(k): int(v) for k, v in json.loads(blob).items()}
        n = int(os.environ.get("FMA_MOCK_GPU_COUNT", "8"))
        return {f"GPU-{i}": i for i in range(n)}

    def refresh(self) -> None:
        self._map = self._load_map()

    def uuids(self) -> List[str]:
        return sorted(self._map, key=lambda u: self._map[u])

    def to_indices(self, uuids: List[str]) -> List[int]:
        missing = [u for u in uuids if u not in self._map]
        if missing:
            self.refresh()
            missing = [u for u in uuids if u not in self._map]
        if missing:
            raise GpuTranslationError(f"unknown GPU uuids: {missing}")
        return [self._map[u] for u in uuids]

    def visible_devices_value(self, uuids: List[str]) -> str:
        return ",".join(str(i) for i in self.to_indices(uuids))
