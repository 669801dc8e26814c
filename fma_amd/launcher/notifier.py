"""Notifier sidecar: reflect launcher instance state into Pod metadata.

The controller's informer only sees Pod updates, so launcher-internal
state changes must be converted into Pod events: poll the launcher LIST
every interval, compute a SHA-256 signature over the sorted (id, status)
pairs, and patch it onto the launcher Pod's
``dual-pods.llm-d.ai/vllm-instance-signature`` annotation when it changes
(reference inference_server/launcher/launcher_pod_notifier.py:31, 100-103,
135-194; sidecar injection utils/pod-helper.go:356-400).
"""

from __future__ import annotations

import hashlib
import json
import time
from typing import Callable, Optional

import httpx

from fma_amd.api import contracts


def instances_signature(list_body: dict) -> str:
    pairs = sorted((i.get("instance_id", ""), i.get("status", ""))
                   for i in list_body.get("instances", []))
    blob = json.dumps(pairs, separators=(",", ":")).encode()
    return hashlib.sha256(blob).hexdigest()


class PodNotifier:
    def __init__(self, launcher_url: str,
                 patch_annotation: Callable[[str, str], None],
                 interval: float = 2.0):
        """patch_annotation(key, value) applies one annotation to the
        launcher Pod (store client in production, a recorder in tests)."""
        self.launcher_url = launcher_url.rstrip("/")
        self.patch_annotation = patch_annotation
        self.interval = interval
        self.last_signature: Optional[str] = None
        self._stop = False

    def poll_once(self, client: Optional[httpx.Client] = None) -> Optional[str]:
        own = client is None
        c = client or httpx.Client(timeout=10)
        try:
            r = c.get(self.launcher_url + contracts.LAUNCHER_API_ROOT)
            r.raise_for_status()
            sig = instances_signature(r.json())
            if sig != self.last_signature:
                self.patch_annotation(
                    contracts.INSTANCE_SIGNATURE_ANNOTATION, sig)
                self.last_signature = sig
            return sig
        finally:
            if own:
                c.close()

    def run(self) -> None:
        with httpx.Client(timeout=10) as c:
            while not self._stop:
                try:
                    self.poll_once(c)
                except Exception:
                    pass  # launcher restarting; retry next tick
                time.sleep(self.interval)

    def stop(self) -> None:
        self._stop = True
