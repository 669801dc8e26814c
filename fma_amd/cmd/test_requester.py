"""test-requester: fake scheduler+kubelet GPU assignment for e2e.

Equivalent of the reference's cmd/test-requester (main.go:56-144,
gpu-allocation.go:41-257): emulates how the device plugin would attribute
GPUs to a requesting Pod. The allocation ledger is a ConfigMap
``gpu-allocs`` in the cluster store mapping GPU UUID -> {node, podUID};
this process claims free GPUs for its Pod (honoring a HIP_VISIBLE_DEVICES
pin when present), sweeps allocations whose holder Pod is gone, and then
serves the normal requester SPI with the claimed UUIDs.
"""

from __future__ import annotations

import argparse
import json
import os
import random
from typing import Dict, List

from fma_amd.api import contracts
from fma_amd.store import objects as ob
from fma_amd.store.client import StoreClient
from fma_amd.store.memstore import Conflict, NotFound

ALLOCS_CONFIGMAP = "gpu-allocs"


def claim_gpus(store, node: str, pod_uid: str, count: int,
               namespace: str = "default",
               pinned: List[str] = None) -> List[str]:
    """Claim `count` GPUs on `node` for `pod_uid`; retries on conflicts
    (the ConfigMap is the ACID ledger, like the reference's gpu-allocs)."""
    for _ in range(50):
        cm = store.try_get("ConfigMap", ALLOCS_CONFIGMAP, namespace)
        if cm is None:
            cm = ob.new_object("ConfigMap", ALLOCS_CONFIGMAP,
                               namespace=namespace)
            cm["data"] = {}
            try:
                cm = store.create(cm)
            except Conflict:
                continue
        data: Dict[str, Dict] = {k: json.loads(v)
                                 for k, v in cm.get("data", {}).items()}
        # sweep dead holders
        live_uids = {ob.uid_of(p) for p in store.list("Pod", namespace)}
        for uuid in list(data):
            if data[uuid].get("podUID") not in live_uids:
                del data[uuid]
        mine = [u for u, h in data.items() if h.get("podUID") == pod_uid]
        if len(mine) >= count:
            return sorted(mine)[:count]
        gpu_map_cm = store.try_get("ConfigMap", contracts.GPU_MAP_CONFIGMAP,
                                   namespace)
        node_map = json.loads(
            (gpu_map_cm or {}).get("data", {}).get(node, "{}"))
        candidates = [u for u in node_map if u not in data]
        if pinned:
            candidates = [u for u in pinned if u in node_map and
                          u not in data]
        random.shuffle(candidates)
        for u in candidates[: count - len(mine)]:
            data[u] = {"node": node, "podUID": pod_uid}
            mine.append(u)
        if len(mine) < count:
            raise RuntimeError(
                f"not enough free GPUs on {node}: have {len(mine)}, "
                f"need {count}")
        cm["data"] = {k: json.dumps(v) for k, v in data.items()}
        try:
            store.update(cm)
            return sorted(mine)[:count]
        except (Conflict, NotFound):
            continue
    raise RuntimeError("could not commit GPU allocation after retries")


def main() -> None:
    ap = argparse.ArgumentParser("fma-test-requester")
    ap.add_argument("--store-url", default="http://127.0.0.1:8081")
    ap.add_argument("--gpus", type=int, default=1)
    args = ap.parse_args()
    store = StoreClient(args.store_url, actor="test-requester")
    node = os.environ.get("NODE_NAME", "node-1")
    pod_name = os.environ.get("POD_NAME", "")
    pod = store.get("Pod", pod_name)
    pinned = None
    if os.environ.get(contracts.VISIBLE_DEVICES_ENV):
        pinned = os.environ[contracts.VISIBLE_DEVICES_ENV].split(",")
    uuids = claim_gpus(store, node, ob.uid_of(pod), args.gpus, pinned=pinned)
    os.environ["FMA_ACCELERATORS"] = ",".join(uuids)
    from fma_amd.requester import server
    server.main()


if __name__ == "__main__":
    main()
