"""test-server + test-requester (GPU allocation ledger) + store HTTP."""

import threading

import pytest
from fastapi.testclient import TestClient

from fma_amd.cmd.test_requester import claim_gpus
from fma_amd.cmd.test_server import create_app as mk_test_server
from fma_amd.store import objects as ob
from fma_amd.store.memstore import MemStore
from fma_amd.store.server import create_app as mk_store_app


def test_test_server_delayed_health_and_sleep():
    c = TestClient(mk_test_server(startup_delay=9999))
    assert c.get("/health").status_code == 503
    c2 = TestClient(mk_test_server(startup_delay=0))
    assert c2.get("/health").status_code == 200
    assert c2.get("/is_sleeping").json() == {"is_sleeping": False}
    c2.post("/sleep", params={"level": 1})
    assert c2.get("/is_sleeping").json() == {"is_sleeping": True}
    assert c2.post("/v1/completions", json={}).status_code == 409
    c2.post("/wake_up")
    assert c2.get("/is_sleeping").json() == {"is_sleeping": False}


def mk_gpu_world():
    store = MemStore()
    cm = ob.new_object("ConfigMap", "gpu-map")
    cm["data"] = {"node-a": '{"GPU-0": 0, "GPU-1": 1, "GPU-2": 2}'}
    store.create(cm)
    pods = [store.create(ob.new_object("Pod", f"p{i}",
                                       spec={"nodeName": "node-a"}))
            for i in range(3)]
    return store, pods


def test_claim_gpus_exclusive():
    store, pods = mk_gpu_world()
    a = claim_gpus(store, "node-a", ob.uid_of(pods[0]), 2)
    b = claim_gpus(store, "node-a", ob.uid_of(pods[1]), 1)
    assert len(a) == 2 and len(b) == 1
    assert not set(a) & set(b)
    # idempotent for the same pod
    assert claim_gpus(store, "node-a", ob.uid_of(pods[0]), 2) == a
    with pytest.raises(RuntimeError):
        claim_gpus(store, "node-a", ob.uid_of(pods[2]), 1)


def test_claim_gpus_sweeps_dead_holders():
    store, pods = mk_gpu_world()
    claim_gpus(store, "node-a", ob.uid_of(pods[0]), 3)
    store.delete("Pod", "p0")
    got = claim_gpus(store, "node-a", ob.uid_of(pods[1]), 3)
    assert len(got) == 3


def test_claim_gpus_honors_pin():
    store, pods = mk_gpu_world()
    got = claim_gpus(store, "node-a", ob.uid_of(pods[0]), 1,
                     pinned=["GPU-2"])
    assert got == ["GPU-2"]


def test_store_http_roundtrip():
    store = MemStore()
    c = TestClient(mk_store_app(store))
    r = c.post("/apis/Pod", json={"metadata": {"name": "x"}},
               headers={"X-FMA-Actor": "user"})
    assert r.status_code == 201
    pod = r.json()
    assert c.get("/apis/Pod/default/x").status_code == 200
    lst = c.get("/apis/Pod").json()
    assert len(lst["items"]) == 1 and lst["revision"] >= 1
    pod["spec"] = {"nodeName": "n"}
    r = c.put("/apis/Pod/default/x", json=pod)
    assert r.status_code == 200
    # stale RV conflict over HTTP
    r = c.put("/apis/Pod/default/x", json=pod)
    assert r.status_code == 409
    assert c.delete("/apis/Pod/default/x").status_code == 200
    assert c.get("/apis/Pod/default/x").status_code == 404


def test_ensure_nodes_mapped_fills_gpu_map():
    """tools/ensure_nodes_mapped.py (the ensure-nodes-mapped.sh analog):
    idempotently fills gpu-map[node] through BOTH store backends using
    the GPU-less naive translator."""
    import json
    import sys
    sys.path.insert(0, "tools")
    from ensure_nodes_mapped import ensure_mapped

    from fma_amd.api import contracts as C
    from fma_amd.launcher.gputranslator import GpuTranslator
    from fma_amd.store.kubeapiserver import create_app as kube_app
    from fma_amd.store.kubestore import KubeStore
    from fma_amd.store.memstore import MemStore

    from tests.test_live_servers import ServerThread, free_port

    tr = GpuTranslator(mode=GpuTranslator.MODE_NAIVE,
                       gpu_map={"GPU-0": 0, "GPU-1": 1})

    # in-process MemStore
    st = MemStore()
    m = ensure_mapped(st, "node-x", translator=tr)
    assert m == {"GPU-0": 0, "GPU-1": 1}
    # idempotent: second call returns the stored entry untouched
    assert ensure_mapped(st, "node-x", translator=tr) == m
    cm = st.get("ConfigMap", C.GPU_MAP_CONFIGMAP)
    assert json.loads(cm["data"]["node-x"]) == m

    # over the Kubernetes wire protocol (merge PATCH per node)
    port = free_port()
    with ServerThread(kube_app(MemStore()), port):
        ks = KubeStore(f"http://127.0.0.1:{port}", actor="node-agent")
        assert ensure_mapped(ks, "node-a", translator=tr) == m
        assert ensure_mapped(ks, "node-b", translator=tr) == m
        data = ks.get("ConfigMap", C.GPU_MAP_CONFIGMAP)["data"]
        assert set(data) == {"node-a", "node-b"}
