#!/usr/bin/env python3
"""Ensure the ``gpu-map`` ConfigMap has an entry for this node.

Analog of the reference's scripts/ensure-nodes-mapped.sh:1-60 (which
spawns per-node pods running nvidia-smi and patches the results into the
``gpu-map`` ConfigMap the dual-pods controller's direct path reads). On
the AMD stack each node runs this tool once (as a DaemonSet init step or
by hand): it enumerates the local GPUs through GpuTranslator (amdsmi ->
rocm-smi -> naive fallback) and merge-PATCHes ``{node: {uuid: index}}``
into the ConfigMap via any store backend.

    python tools/ensure_nodes_mapped.py --store-url http://...   # own store
    python tools/ensure_nodes_mapped.py --backend kube --store-url https://...
    python tools/ensure_nodes_mapped.py --backend in-cluster     # in a Pod
"""
import argparse
import json
import socket
import sys

sys.path.insert(0, ".")

from fma_amd.api import contracts  # noqa: E402
from fma_amd.launcher.gputranslator import GpuTranslator  # noqa: E402
from fma_amd.store import objects as ob  # noqa: E402
from fma_amd.store.memstore import NotFound  # noqa: E402


def make_store(backend: str, url: str):
    if backend == "store":
        from fma_amd.store.client import StoreClient
        return StoreClient(url, actor="node-agent")
    from fma_amd.store.kubestore import KubeStore
    if backend == "kube":
        return KubeStore(url, actor="node-agent")
    return KubeStore.in_cluster(actor="node-agent")


def ensure_mapped(store, node: str, namespace: str = "default",
                  translator: GpuTranslator = None) -> dict:
    """Idempotent: creates the ConfigMap if needed, fills data[node] if
    absent, leaves existing entries alone. Returns the node's map."""
    tr = translator or GpuTranslator()
    try:
        cm = store.get("ConfigMap", contracts.GPU_MAP_CONFIGMAP, namespace)
    except NotFound:
        cm = store.create(ob.new_object(
            "ConfigMap", contracts.GPU_MAP_CONFIGMAP, namespace=namespace))
    existing = (cm.get("data") or {}).get(node)
    if existing:
        return json.loads(existing)
    mapping = {u: tr.to_indices([u])[0] for u in tr.uuids()}
    store.patch("ConfigMap", contracts.GPU_MAP_CONFIGMAP,
                {"data": {node: json.dumps(mapping)}}, namespace,
                actor="node-agent")
    return mapping


def main():
    ap = argparse.ArgumentParser("fma-ensure-nodes-mapped")
    ap.add_argument("--store-url", default="http://127.0.0.1:8081")
    ap.add_argument("--backend", default="store",
                    choices=("store", "kube", "in-cluster"))
    ap.add_argument("--namespace", default="default")
    ap.add_argument("--node", default=socket.gethostname())
    args = ap.parse_args()
    store = make_store(args.backend, args.store_url)
    mapping = ensure_mapped(store, args.node, args.namespace)
    print(f"gpu-map[{args.node}] = {json.dumps(mapping)}")


if __name__ == "__main__":
    main()
