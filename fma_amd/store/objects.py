"""Object model for the in-process cluster store.

The reference coordinates everything through kube-apiserver objects: Pod
metadata is the ACID record of bindings (reference docs/dual-pods.md:729-737)
and controllers rely on resourceVersion-preconditioned updates
(reference pkg/controller/dual-pods/inference-server.go:268-270). This module
gives the MI355X stack the same object semantics without requiring a
Kubernetes cluster: objects are plain dicts with the standard metadata
fields, manipulated through helpers that enforce the same invariants.

When a real cluster is present, the same controllers can run against it by
swapping the store client; nothing in the controllers depends on this module
being in-process.
"""

from __future__ import annotations

import copy
import time
import uuid as _uuid
from typing import Any, Dict, List, Optional


def new_object(
    kind: str,
    name: str,
    namespace: str = "default",
    labels: Optional[Dict[str, str]] = None,
    annotations: Optional[Dict[str, str]] = None,
    spec: Optional[Dict[str, Any]] = None,
    status: Optional[Dict[str, Any]] = None,
) -> Dict[str, Any]:
    return {
        "kind": kind,
        "metadata": {
            "name": name,
            "namespace": namespace,
            "uid": "",
            "resourceVersion": "",
            "generation": 0,
            "creationTimestamp": None,
            "deletionTimestamp": None,
            "labels": dict(labels or {}),
            "annotations": dict(annotations or {}),
            "finalizers": [],
            "ownerReferences": [],
        },
        "spec": copy.deepcopy(spec or {}),
        "status": copy.deepcopy(status or {}),
    }


def meta(obj: Dict[str, Any]) -> Dict[str, Any]:
    return obj.setdefault("metadata", {})


def name_of(obj: Dict[str, Any]) -> str:
    return meta(obj).get("name", "")


def namespace_of(obj: Dict[str, Any]) -> str:
    return meta(obj).get("namespace", "default")


def uid_of(obj: Dict[str, Any]) -> str:
    return meta(obj).get("uid", "")


def rv_of(obj: Dict[str, Any]) -> str:
    return meta(obj).get("resourceVersion", "")


def labels_of(obj: Dict[str, Any]) -> Dict[str, str]:
    return meta(obj).setdefault("labels", {})


def annotations_of(obj: Dict[str, Any]) -> Dict[str, str]:
    return meta(obj).setdefault("annotations", {})


def finalizers_of(obj: Dict[str, Any]) -> List[str]:
    return meta(obj).setdefault("finalizers", [])


def is_deleting(obj: Dict[str, Any]) -> bool:
    return meta(obj).get("deletionTimestamp") is not None


def generate_uid() -> str:
    return str(_uuid.uuid4())


def now() -> float:
    return time.time()


def key_of(obj: Dict[str, Any]) -> str:
    return f"{namespace_of(obj)}/{name_of(obj)}"


def deepcopy(obj: Dict[str, Any]) -> Dict[str, Any]:
    return copy.deepcopy(obj)


# ----------------------------------------------------------------------------
# Pod helpers (the subset of corev1.Pod the controllers consume)
# ----------------------------------------------------------------------------

def pod_node_name(pod: Dict[str, Any]) -> str:
    return pod.get("spec", {}).get("nodeName", "")


def pod_phase(pod: Dict[str, Any]) -> str:
    return pod.get("status", {}).get("phase", "Pending")


def pod_ip(pod: Dict[str, Any]) -> str:
    return pod.get("status", {}).get("podIP", "")


def pod_is_ready(pod: Dict[str, Any]) -> bool:
    for cond in pod.get("status", {}).get("conditions", []):
        if cond.get("type") == "Ready":
            return cond.get("status") == "True"
    return False


def set_pod_ready(pod: Dict[str, Any], ready: bool) -> None:
    conds = pod.setdefault("status", {}).setdefault("conditions", [])
    for cond in conds:
        if cond.get("type") == "Ready":
            cond["status"] = "True" if ready else "False"
            return
    conds.append({"type": "Ready", "status": "True" if ready else "False"})


def pod_containers(pod: Dict[str, Any]) -> List[Dict[str, Any]]:
    return pod.get("spec", {}).get("containers", [])


def pod_container_start_time(pod: Dict[str, Any]) -> Optional[float]:
    """Start time of the Pod's running container(s): the origin for the
    fma_actuation_seconds histogram (reference inference-server.go:574-591
    reads ContainerStatuses state.running.startedAt). Falls back to
    status.startTime, then creationTimestamp."""
    best = None
    for cs in pod.get("status", {}).get("containerStatuses", []):
        t = ((cs.get("state") or {}).get("running") or {}).get("startedAt")
        if isinstance(t, (int, float)):
            best = t if best is None else min(best, t)
    if best is not None:
        return best
    t = pod.get("status", {}).get("startTime")
    if isinstance(t, (int, float)):
        return t
    t = meta(pod).get("creationTimestamp")
    return t if isinstance(t, (int, float)) else None


def find_container(pod: Dict[str, Any], name: str) -> Optional[Dict[str, Any]]:
    for c in pod_containers(pod):
        if c.get("name") == name:
            return c
    return None


def container_env_set(container: Dict[str, Any], name: str, value: str) -> None:
    env = container.setdefault("env", [])
    for e in env:
        if e.get("name") == name:
            e["value"] = value
            return
    env.append({"name": name, "value": value})


def container_restart_count(pod: Dict[str, Any], container_name: str) -> int:
    for cs in pod.get("status", {}).get("containerStatuses", []):
        if cs.get("name") == container_name:
            return int(cs.get("restartCount", 0))
    return 0
