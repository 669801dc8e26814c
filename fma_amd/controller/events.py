"""Kubernetes-style Events in the cluster store.

The reference emits Events `LauncherStuck` and `OutdatedRoutingMetadata`
(reference docs/dual-pods.md:687-717, populator reportStuckLaunchers
populator.go:579-626). Events here are ordinary store objects of kind
``Event`` with k8s-shaped fields; repeats bump ``count`` instead of
creating duplicates, like an EventRecorder's aggregation.
"""

from __future__ import annotations

import hashlib
from typing import Any, Dict

from fma_amd.store import objects as ob
from fma_amd.store.memstore import Conflict, MemStore, NotFound

EVENT_TYPE_WARNING = "Warning"
EVENT_TYPE_NORMAL = "Normal"

REASON_LAUNCHER_STUCK = "LauncherStuck"
REASON_OUTDATED_ROUTING = "OutdatedRoutingMetadata"


def record_event(store: MemStore, involved: Dict[str, Any], reason: str,
                 message: str, type_: str = EVENT_TYPE_WARNING,
                 actor: str = "system", namespace: str = "default") -> None:
    key = hashlib.sha256(
        f"{ob.uid_of(involved)}/{reason}/{message}".encode()
    ).hexdigest()[:10]
    name = f"{ob.name_of(involved)}.{key}"
    existing = store.try_get("Event", name, namespace)
    if existing is not None:
        existing["count"] = int(existing.get("count", 1)) + 1
        existing["lastTimestamp"] = ob.now()
        try:
            store.update(existing, actor=actor)
        except (Conflict, NotFound):
            pass
        return
    ev = ob.new_object("Event", name, namespace=namespace)
    ev.update({
        "type": type_,
        "reason": reason,
        "message": message,
        "count": 1,
        "firstTimestamp": ob.now(),
        "lastTimestamp": ob.now(),
        "involvedObject": {
            "kind": involved.get("kind"),
            "name": ob.name_of(involved),
            "uid": ob.uid_of(involved),
        },
    })
    try:
        store.create(ev, actor=actor)
    except Conflict:
        pass
