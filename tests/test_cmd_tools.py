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
