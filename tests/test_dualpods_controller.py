"""Dual-pods controller reconcile tests (CPU; fake store + fake HTTP).

Covers the actuation paths the reference exercises only in e2e (its
dual-pods package has no unit tests — SURVEY §4 notes the gap): hot, warm,
cold, reclaim, unbind-on-delete, direct path, restart recovery.
"""

import copy

import pytest

from fma_amd.api import contracts as C
from fma_amd.controller.dualpods.controller import (DualPodsController,
                                                    ControllerConfig,
                                                    PROVIDER_FINALIZER,
                                                    REQUESTER_FINALIZER)
from fma_amd.controller.dualpods.identity import instance_id
from fma_amd.controller.httpadapter import FakeHttp
from fma_amd.store import objects as ob
from fma_amd.store.memstore import MemStore


class FakeInstanceServer:
    """Mimics the serving runtime's sleep/wake endpoints."""

    def __init__(self, sleeping=True):
        self.sleeping = sleeping
        self.sleeps = 0
        self.wakes = 0

    def __call__(self, method, path, json, params):
        if method == "GET" and path == "/is_sleeping":
            return 200, {"is_sleeping": self.sleeping}
        if method == "POST" and path == "/wake_up":
            self.sleeping = False
            self.wakes += 1
            return 200, {"status": "ok"}
        if method == "POST" and path == "/sleep":
            self.sleeping = True
            self.sleeps += 1
            return 200, {"status": "ok"}
        return 404, {}


class FakeLauncher:
    """Dict-backed launcher API; each created instance gets a
    FakeInstanceServer registered at the right port by the test."""

    def __init__(self, http, ip, on_create=None):
        self.instances = {}
        self.http = http
        self.ip = ip
        self.on_create = on_create

    def __call__(self, method, path, json, params):
        root = "/v2/vllm/instances"
        if path == root and method == "GET":
            return 200, {
                "total_instances": len(self.instances),
                "running_instances": sum(
                    1 for i in self.instances.values()
                    if i["status"] == "running"),
                "instances": list(self.instances.values()),
                "revision": 1,
            }
        if path.startswith(root + "/"):
            iid = path[len(root) + 1:]
            if method == "GET":
                if iid not in self.instances:
                    return 404, {}
                return 200, self.instances[iid]
            if method == "PUT":
                if iid in self.instances:
                    return 409, {}
                inst = {"instance_id": iid, "status": "running"}
                inst.update(json or {})
                self.instances[iid] = inst
                if self.on_create:
                    self.on_create(iid, inst)
                return 201, inst
            if method == "DELETE":
                if iid not in self.instances:
                    return 404, {}
                return 200, self.instances.pop(iid)
        return 404, {}


class Stub:
    def __init__(self, gpus):
        self.gpus = gpus
        self.ready_calls = 0
        self.unready_calls = 0
        self.proxy_target = None

    def __call__(self, method, path, json, params):
        if path == "/v1/dual-pods/accelerators":
            return 200, self.gpus
        if path == "/v1/become-ready":
            self.ready_calls += 1
            return 200, {}
        if path == "/v1/become-unready":
            self.unready_calls += 1
            return 200, {}
        if path == "/v1/dual-pods/accelerator-memory-usage":
            return 200, {g: 0 for g in self.gpus}
        if path == "/v1/proxy/config" and method == "PUT":
            if self.proxy_target is not None:
                return 409, {}
            self.proxy_target = json
            return 200, {"status": "ok"}
        return 404, {}


MSC = {"port": 8000, "options": "--model tiny",
       "env_vars": {}, "labels": {"llm-d.ai/model": "tiny"},
       "annotations": {}}


def mk_world(msc=None, launcher_ready=True, with_launcher=True,
             sleeping_target=True, max_instances=2):
    msc = copy.deepcopy(msc or MSC)
    store = MemStore()
    http = FakeHttp()
    ctl = DualPodsController(store, http, ControllerConfig())

    isc = ob.new_object("InferenceServerConfig", "isc1",
                        spec={"modelServerConfig": msc,
                              "launcherConfigName": "lc1"})
    store.create(isc)
    lc = ob.new_object("LauncherConfig", "lc1",
                       spec={"maxInstances": max_instances, "podTemplate": {}})
    store.create(lc)

    req = ob.new_object(
        "Pod", "req1",
        annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"},
        spec={"nodeName": "node-a", "containers": [{"name": "stub"}]})
    req = store.create(req)
    req["status"] = {"phase": "Running", "podIP": "10.0.0.1"}
    req = store.update(req)

    stub = Stub(["GPU-0"])
    http.register("10.0.0.1:8081", stub)

    world = {"store": store, "http": http, "ctl": ctl, "req": req,
             "stub": stub, "msc": msc,
             "iid": instance_id(msc, ["GPU-0"])}

    if with_launcher:
        lp = ob.new_object(
            "Pod", "launcher1",
            labels={C.COMPONENT_LABEL: C.LAUNCHER_COMPONENT,
                    C.LAUNCHER_CONFIG_NAME_LABEL: "lc1",
                    C.SLEEPING_LABEL: "true"},
            annotations={"dual-pods.llm-d.ai/max-instances":
                         str(max_instances)},
            spec={"nodeName": "node-a", "containers": [{"name": "launcher"}]})
        lp = store.create(lp)
        lp["status"] = {"phase": "Running", "podIP": "10.0.0.2"}
        ob.set_pod_ready(lp, launcher_ready)
        lp = store.update(lp)
        fl = FakeLauncher(http, "10.0.0.2")
        http.register("10.0.0.2:8001", fl)
        world["launcher_pod"] = lp
        world["launcher"] = fl
        if sleeping_target:
            inst_srv = FakeInstanceServer(sleeping=True)
            fl.instances[world["iid"]] = {
                "instance_id": world["iid"], "status": "running",
                "options": "--model tiny --port 8000",
                "gpu_uuids": ["GPU-0"],
                "annotations": {"isc-name": "isc1", "inference-port": "8000"},
            }
            http.register("10.0.0.2:8000", inst_srv)
            world["inst_srv"] = inst_srv
    return world


def drive(ctl, item, max_iters=25):
    for _ in range(max_iters):
        retry = ctl._process(item)
        if not retry:
            return
    raise AssertionError(f"item {item} did not converge")


def infsvr_item(store, name="req1"):
    pod = store.get("Pod", name)
    return ("infsvr", ob.pod_node_name(pod), ob.uid_of(pod), name)


def test_hot_start_binds_and_wakes():
    w = mk_world()
    drive(w["ctl"], infsvr_item(w["store"]))
    lp = w["store"].get("Pod", "launcher1")
    anns = ob.annotations_of(lp)
    assert anns[C.REQUESTER_ANNOTATION].endswith(" req1")
    assert anns[C.INSTANCE_ID_ANNOTATION] == w["iid"]
    assert anns[C.SERVER_PORT_ANNOTATION] == "8000"
    assert PROVIDER_FINALIZER in ob.finalizers_of(lp)
    assert ob.labels_of(lp)[C.SLEEPING_LABEL] == "false"
    assert ob.labels_of(lp)["llm-d.ai/model"] == "tiny"  # ISC routing label
    assert w["inst_srv"].wakes == 1
    assert not w["inst_srv"].sleeping
    assert w["stub"].ready_calls >= 1
    # proxy pointed at the serving endpoint (release-0.7 feature)
    assert w["stub"].proxy_target == {"address": "10.0.0.2", "port": 8000}
    req = w["store"].get("Pod", "req1")
    assert REQUESTER_FINALIZER in ob.finalizers_of(req)
    assert ob.labels_of(req)[C.DUAL_LABEL] == "launcher1"
    assert ob.labels_of(req)[C.INSTANCE_LABEL] == w["iid"][:63]


def test_warm_start_creates_instance():
    w = mk_world(sleeping_target=False)

    # when the controller creates the instance, register its server
    def on_create(iid, inst):
        http_srv = FakeInstanceServer(sleeping=False)
        w["http"].register("10.0.0.2:8000", http_srv)
        w["inst_srv"] = http_srv

    w["launcher"].on_create = on_create
    drive(w["ctl"], infsvr_item(w["store"]))
    assert w["iid"] in w["launcher"].instances
    created = w["launcher"].instances[w["iid"]]
    assert "--port 8000" in created["options"]
    assert created["gpu_uuids"] == ["GPU-0"]
    assert created["annotations"]["isc-name"] == "isc1"
    assert w["stub"].ready_calls >= 1


def test_cold_start_creates_launcher_pod():
    w = mk_world(with_launcher=False)
    ctl = w["ctl"]
    item = infsvr_item(w["store"])
    ctl._process(item)  # one pass: should create a launcher pod
    launchers = [p for p in w["store"].list("Pod")
                 if ob.labels_of(p).get(C.COMPONENT_LABEL) ==
                 C.LAUNCHER_COMPONENT]
    assert len(launchers) == 1
    lp = launchers[0]
    assert ob.pod_node_name(lp) == "node-a"
    assert ob.labels_of(lp)[C.LAUNCHER_CONFIG_NAME_LABEL] == "lc1"
    assert ob.annotations_of(lp)["dual-pods.llm-d.ai/max-instances"] == "2"
    # only one launcher created even if reconcile repeats
    ctl._process(item)
    launchers = [p for p in w["store"].list("Pod")
                 if ob.labels_of(p).get(C.COMPONENT_LABEL) ==
                 C.LAUNCHER_COMPONENT]
    assert len(launchers) <= 2  # second pass may create another while unready


def test_reclaim_deletes_port_conflict_victim():
    w = mk_world(sleeping_target=False, max_instances=1)
    # occupy the only slot with a conflicting instance on the same port
    other_srv = FakeInstanceServer(sleeping=True)
    w["launcher"].instances["Iotheri"] = {
        "instance_id": "Iotheri", "status": "running",
        "options": "--model other --port 8000",
        "annotations": {"inference-port": "8000"},
    }

    def on_create(iid, inst):
        w["http"].register("10.0.0.2:8000", FakeInstanceServer(sleeping=False))

    w["launcher"].on_create = on_create
    w["http"].register("10.0.0.2:8000", other_srv)
    drive(w["ctl"], infsvr_item(w["store"]))
    assert "Iotheri" not in w["launcher"].instances
    assert w["iid"] in w["launcher"].instances


def test_unbind_on_requester_delete():
    w = mk_world()
    drive(w["ctl"], infsvr_item(w["store"]))
    item = infsvr_item(w["store"])
    # delete the requester: finalizer holds it, controller unbinds
    w["store"].delete("Pod", "req1")
    assert ob.is_deleting(w["store"].get("Pod", "req1"))
    drive(w["ctl"], item)
    assert w["store"].try_get("Pod", "req1") is None  # finalizer released
    lp = w["store"].get("Pod", "launcher1")
    anns = ob.annotations_of(lp)
    assert C.REQUESTER_ANNOTATION not in anns
    assert C.INSTANCE_ID_ANNOTATION not in anns
    assert ob.labels_of(lp)[C.SLEEPING_LABEL] == "true"
    assert "llm-d.ai/model" not in ob.labels_of(lp)  # de-routed
    assert PROVIDER_FINALIZER not in ob.finalizers_of(lp)
    assert w["inst_srv"].sleeping
    assert w["inst_srv"].sleeps == 1


def test_second_requester_hot_starts_after_unbind():
    w = mk_world()
    item = infsvr_item(w["store"])
    drive(w["ctl"], item)
    w["store"].delete("Pod", "req1")
    drive(w["ctl"], item)
    # same ISC, new requester, same GPU -> hot start on sleeping instance
    req2 = ob.new_object(
        "Pod", "req2",
        annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"},
        spec={"nodeName": "node-a", "containers": [{"name": "stub"}]})
    req2 = w["store"].create(req2)
    req2["status"] = {"phase": "Running", "podIP": "10.0.0.3"}
    w["store"].update(req2)
    w["http"].register("10.0.0.3:8081", Stub(["GPU-0"]))
    drive(w["ctl"], infsvr_item(w["store"], "req2"))
    lp = w["store"].get("Pod", "launcher1")
    assert ob.annotations_of(lp)[C.REQUESTER_ANNOTATION].endswith(" req2")
    assert w["inst_srv"].wakes == 2  # woken again, same instance
    assert len(w["launcher"].instances) == 1


def test_stopped_bound_instance_deletes_requester():
    w = mk_world()
    drive(w["ctl"], infsvr_item(w["store"]))
    item = infsvr_item(w["store"])
    w["launcher"].instances[w["iid"]]["status"] = "stopped"
    w["ctl"]._process(item)
    req = w["store"].try_get("Pod", "req1")
    # requester deleted (or deleting, held by finalizer)
    assert req is None or ob.is_deleting(req)


def test_missing_isc_reports_status():
    w = mk_world()
    req = w["store"].get("Pod", "req1")
    ob.annotations_of(req)[C.INFERENCE_SERVER_CONFIG_ANNOTATION] = "nope"
    w["store"].update(req)
    w["ctl"]._process(infsvr_item(w["store"]))
    req = w["store"].get("Pod", "req1")
    assert "not found" in ob.annotations_of(req).get(C.STATUS_ANNOTATION, "")


def test_restart_recovery_from_annotations():
    """A fresh controller instance reconstructs the binding from Pod
    metadata alone (reference controller.go:64-99)."""
    w = mk_world()
    drive(w["ctl"], infsvr_item(w["store"]))
    ctl2 = DualPodsController(w["store"], w["http"], ControllerConfig())
    # the new controller processes the same item without cached state
    drive(ctl2, infsvr_item(w["store"]))
    lp = w["store"].get("Pod", "launcher1")
    assert ob.annotations_of(lp)[C.REQUESTER_ANNOTATION].endswith(" req1")
    assert w["inst_srv"].wakes == 1  # not re-woken; state observed as awake


# ---------------------------------------------------------------------------
# direct (launcher-less) path
# ---------------------------------------------------------------------------

PATCH = """
spec:
  containers:
  - name: inference-server
    image: fma-amd/server:latest
    ports:
    - containerPort: 8200
"""


def mk_direct_world():
    store = MemStore()
    http = FakeHttp()
    ctl = DualPodsController(store, http, ControllerConfig(sleeper_limit=1))
    cm = ob.new_object("ConfigMap", C.GPU_MAP_CONFIGMAP)
    cm["data"] = {"node-a": '{"GPU-0": 0}'}
    store.create(cm)
    req = ob.new_object(
        "Pod", "dreq",
        annotations={C.SERVER_PATCH_ANNOTATION: PATCH},
        spec={"nodeName": "node-a", "containers": [
            {"name": "stub"},
            {"name": "inference-server", "image": "placeholder"}]})
    req = store.create(req)
    req["status"] = {"phase": "Running", "podIP": "10.1.0.1"}
    req = store.update(req)
    stub = Stub(["GPU-0"])
    http.register("10.1.0.1:8081", stub)
    return {"store": store, "http": http, "ctl": ctl, "stub": stub}


def test_direct_creates_provider_then_relays():
    w = mk_direct_world()
    item = infsvr_item(w["store"], "dreq")
    w["ctl"]._process(item)
    provider = w["store"].try_get("Pod", "dreq-server")
    assert provider is not None
    c = ob.find_container(provider, "inference-server")
    env = {e["name"]: e["value"] for e in c["env"]}
    assert env[C.VISIBLE_DEVICES_ENV] == "0"
    assert c["resources"]["limits"][C.GPU_RESOURCE_NAME] == "0"
    assert ob.pod_node_name(provider) == "node-a"

    # "kubelet" brings the provider up with an awake server
    provider["status"] = {"phase": "Running", "podIP": "10.1.0.2"}
    ob.set_pod_ready(provider, True)
    w["store"].update(provider)
    w["http"].register("10.1.0.2:8200", FakeInstanceServer(sleeping=False))
    drive(w["ctl"], item)
    provider = w["store"].get("Pod", "dreq-server")
    assert ob.annotations_of(provider)[C.REQUESTER_ANNOTATION].endswith(
        " dreq")
    assert w["stub"].ready_calls >= 1


def test_direct_sleeper_reuse():
    w = mk_direct_world()
    item = infsvr_item(w["store"], "dreq")
    w["ctl"]._process(item)
    provider = w["store"].get("Pod", "dreq-server")
    provider["status"] = {"phase": "Running", "podIP": "10.1.0.2"}
    ob.set_pod_ready(provider, True)
    w["store"].update(provider)
    srv = FakeInstanceServer(sleeping=False)
    w["http"].register("10.1.0.2:8200", srv)
    drive(w["ctl"], item)

    # requester goes away: provider slept + unbound, but kept as sleeper
    w["store"].delete("Pod", "dreq")
    drive(w["ctl"], item)
    provider = w["store"].get("Pod", "dreq-server")
    assert srv.sleeping
    assert ob.labels_of(provider)[C.SLEEPING_LABEL] == "true"

    # identical new requester finds the sleeper by nominal hash
    req2 = ob.new_object(
        "Pod", "dreq",  # same name/spec => same nominal hash
        annotations={C.SERVER_PATCH_ANNOTATION: PATCH},
        spec={"nodeName": "node-a", "containers": [
            {"name": "stub"},
            {"name": "inference-server", "image": "placeholder"}]})
    req2 = w["store"].create(req2)
    req2["status"] = {"phase": "Running", "podIP": "10.1.0.3"}
    w["store"].update(req2)
    w["http"].register("10.1.0.3:8081", Stub(["GPU-0"]))
    drive(w["ctl"], infsvr_item(w["store"], "dreq"))
    provider = w["store"].get("Pod", "dreq-server")
    assert ob.annotations_of(provider)[C.REQUESTER_ANNOTATION].endswith(
        " dreq")
    assert not srv.sleeping  # woken, not re-created
    providers = [p for p in w["store"].list("Pod")
                 if ob.annotations_of(p).get(
                     C.NOMINAL_ANNOTATION)]
    assert len(providers) == 1
