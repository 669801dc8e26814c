"""Standard Pod index functions for O(1) controller lookups.

The reference registers six Pod indexers on its shared informer
(reference pkg/controller/dual-pods/controller.go:129-159, index funcs
:595-636): ``inferenceserverconfig``, ``launcherconfighash``,
``requester``, ``nodeName``, ``nominal``, ``gpu``. Our informer substrate
is the store itself (in-process MemStore, or the store server), so the
same indexes live store-side: they are maintained incrementally on every
mutation under the store lock, and ``index_get`` is a dict hit instead of
a full-namespace scan.

Each index function maps an object to the list of keys it should be
findable under (empty list = not indexed).
"""

from __future__ import annotations

from typing import Any, Dict, List

from fma_amd.api import contracts
from fma_amd.store import objects as ob


def by_requester(pod: Dict[str, Any]) -> List[str]:
    """Providers keyed by their '<uid> <name>' requester annotation."""
    v = ob.annotations_of(pod).get(contracts.REQUESTER_ANNOTATION)
    return [v] if v else []


def by_isc(pod: Dict[str, Any]) -> List[str]:
    """Requesters keyed by the InferenceServerConfig they reference."""
    v = ob.annotations_of(pod).get(contracts.INFERENCE_SERVER_CONFIG_ANNOTATION)
    return [v] if v else []


def by_nominal(pod: Dict[str, Any]) -> List[str]:
    """Direct providers keyed by nominal hash (sleeper lookup)."""
    v = ob.annotations_of(pod).get(contracts.NOMINAL_ANNOTATION)
    return [v] if v else []


def by_launcher_config_hash(pod: Dict[str, Any]) -> List[str]:
    v = ob.annotations_of(pod).get(contracts.LAUNCHER_CONFIG_HASH_ANNOTATION)
    return [v] if v else []


def by_node(pod: Dict[str, Any]) -> List[str]:
    """Any Pod keyed by the node it runs on (or is labeled for)."""
    v = ob.pod_node_name(pod) or ob.labels_of(pod).get(
        contracts.NODE_NAME_LABEL)
    return [v] if v else []


def by_gpu(pod: Dict[str, Any]) -> List[str]:
    """Providers keyed by each GPU UUID they occupy."""
    v = ob.annotations_of(pod).get(contracts.ACCELERATORS_ANNOTATION, "")
    return [u for u in v.split(",") if u]


def by_launcher_node(pod: Dict[str, Any]) -> List[str]:
    """Launcher Pods keyed by node (the launchers-on-node working set)."""
    if ob.labels_of(pod).get(contracts.COMPONENT_LABEL) != \
            contracts.LAUNCHER_COMPONENT:
        return []
    v = ob.pod_node_name(pod) or ob.labels_of(pod).get(
        contracts.NODE_NAME_LABEL)
    return [v] if v else []


POD_INDEXES = {
    "requester": by_requester,
    "inferenceserverconfig": by_isc,
    "nominal": by_nominal,
    "launcherconfighash": by_launcher_config_hash,
    "nodeName": by_node,
    "gpu": by_gpu,
    "launcherNode": by_launcher_node,
}


def install_pod_indexes(store) -> None:
    """Idempotently register the standard Pod indexes on a store that
    supports ``add_index`` (MemStore; the store server installs them at
    boot so StoreClient.index_get hits them over HTTP)."""
    for name, fn in POD_INDEXES.items():
        store.add_index("Pod", name, fn)
