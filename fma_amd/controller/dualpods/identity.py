"""Deterministic identities: instance IDs and nominal-provider hashes.

Mirrors the reference's scheme (reference pkg/controller/dual-pods/
inference-server.go:1016-1058): an instance ID is derived from the full
ModelServerConfig plus the exact GPU set, so the same ISC on the same GPUs
always maps to the same instance — that is what makes the hot-start lookup
(find a launcher already holding the sleeping instance) work. Format:
``"I" + base64url(sha256(canonical_config + ";gpus=" + uuids)) + "i"``
(we canonicalize with sorted-key JSON where the reference uses Go yaml —
self-consistency is what matters).

The nominal hash (direct path) fingerprints the rendered server-providing
Pod spec + GPUs + node (reference inference-server.go:1843-1947) so a
sleeping provider can be matched to a new requester wanting the identical
server.
"""

from __future__ import annotations

import base64
import hashlib
import json
from typing import Any, Dict, List


def _canonical(obj: Any) -> str:
    return json.dumps(obj, sort_keys=True, separators=(",", ":"),
                      default=str)


def instance_id(model_server_config: Dict[str, Any],
                gpu_uuids: List[str]) -> str:
    blob = _canonical(model_server_config) + ";gpus=" + ",".join(gpu_uuids)
    digest = hashlib.sha256(blob.encode()).digest()
    return "I" + base64.urlsafe_b64encode(digest).decode().rstrip("=") + "i"


def nominal_hash(pod_spec: Dict[str, Any], gpu_uuids: List[str],
                 node_name: str) -> str:
    blob = _canonical({"spec": pod_spec, "gpus": gpu_uuids,
                       "node": node_name})
    return hashlib.sha256(blob.encode()).hexdigest()


def template_hash(template: Dict[str, Any]) -> str:
    """Node-independent launcher template hash (reference
    utils/pod-helper.go:143-197 canonicalizes before hashing)."""
    return hashlib.sha256(_canonical(template).encode()).hexdigest()
