"""Lifecycle cases mirroring the reference's e2e suite
(reference test/e2e/test-cases.sh:260-905), run against the fake store.
"""

import copy

from fma_amd.api import contracts as C
from fma_amd.controller.dualpods.controller import (ControllerConfig,
                                                    DualPodsController)
from fma_amd.controller.dualpods.identity import instance_id
from fma_amd.store import objects as ob

from tests.test_dualpods_controller import (FakeInstanceServer, FakeLauncher,
                                            MSC, Stub, drive, infsvr_item,
                                            mk_world)


def add_requester(w, name, isc_name, ip):
    req = ob.new_object(
        "Pod", name,
        annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: isc_name},
        spec={"nodeName": "node-a", "containers": [{"name": "stub"}]})
    req = w["store"].create(req)
    req["status"] = {"phase": "Running", "podIP": ip}
    w["store"].update(req)
    w["http"].register(f"{ip}:8081", Stub(["GPU-0"]))
    return req


def test_multiple_instances_share_one_launcher():
    """ISC-A then ISC-B (different ports) use the same launcher; both
    instances coexist (one awake, one sleeping); ISC-A hot-starts again
    (reference test-cases.sh:512 and :560)."""
    w = mk_world(sleeping_target=False)  # empty launcher, maxInstances=2
    store, http, launcher = w["store"], w["http"], w["launcher"]

    msc_b = copy.deepcopy(MSC)
    msc_b["port"] = 8010
    msc_b["labels"] = {"llm-d.ai/model": "other"}
    store.create(ob.new_object(
        "InferenceServerConfig", "isc2",
        spec={"modelServerConfig": msc_b, "launcherConfigName": "lc1"}))

    def on_create(iid, inst):
        port = 8000 if "--port 8000" in inst["options"] else 8010
        http.register(f"10.0.0.2:{port}", FakeInstanceServer(sleeping=False))

    launcher.on_create = on_create

    # requester A uses isc1 (port 8000)
    drive(w["ctl"], infsvr_item(store))
    iid_a = w["iid"]
    assert iid_a in launcher.instances

    # A leaves; instance A sleeps
    store.delete("Pod", "req1")
    drive(w["ctl"], infsvr_item(store))

    # requester B uses isc2 (port 8010) on the SAME launcher
    add_requester(w, "reqB", "isc2", "10.0.0.5")
    drive(w["ctl"], infsvr_item(store, "reqB"))
    iid_b = instance_id(msc_b, ["GPU-0"])
    assert iid_b in launcher.instances
    assert iid_a in launcher.instances, "sleeping instance A survived"
    assert len(launcher.instances) == 2
    lp = store.get("Pod", "launcher1")
    assert ob.annotations_of(lp)[C.REQUESTER_ANNOTATION].endswith(" reqB")

    # B leaves; A returns -> hot start on the surviving instance A
    store.delete("Pod", "reqB")
    drive(w["ctl"], infsvr_item(store, "reqB"))
    add_requester(w, "reqA2", "isc1", "10.0.0.6")
    drive(w["ctl"], infsvr_item(store, "reqA2"))
    lp = store.get("Pod", "launcher1")
    assert ob.annotations_of(lp)[C.REQUESTER_ANNOTATION].endswith(" reqA2")
    assert len(launcher.instances) == 2  # no new instance created


def test_isc_gc_deletes_obsolete_sleeping_instance():
    """ISC spec change deletes the now-stale sleeping, unbound instance
    (reference test-cases.sh:745, instanceGCItem)."""
    w = mk_world()
    store, launcher = w["store"], w["launcher"]
    drive(w["ctl"], infsvr_item(store))
    store.delete("Pod", "req1")
    drive(w["ctl"], infsvr_item(store))
    assert w["iid"] in launcher.instances  # sleeping, unbound

    isc = store.get("InferenceServerConfig", "isc1")
    isc["spec"]["modelServerConfig"]["options"] = "--model tiny --seed 9"
    store.update(isc)
    w["ctl"].reconcile_isc_gc("node-a", "isc1")
    assert w["iid"] not in launcher.instances, "obsolete instance not GCed"


def test_isc_gc_keeps_current_instance():
    w = mk_world()
    store, launcher = w["store"], w["launcher"]
    drive(w["ctl"], infsvr_item(store))
    store.delete("Pod", "req1")
    drive(w["ctl"], infsvr_item(store))
    w["ctl"].reconcile_isc_gc("node-a", "isc1")  # unchanged ISC
    assert w["iid"] in launcher.instances


def test_obsolete_awake_instance_deleted_on_unbind():
    """If the ISC changed while bound, unbind DELETES the instance instead
    of sleeping it (reference test-cases.sh:784,
    maybeDeleteObsoleteInstance)."""
    w = mk_world()
    store, launcher = w["store"], w["launcher"]
    drive(w["ctl"], infsvr_item(store))
    isc = store.get("InferenceServerConfig", "isc1")
    isc["spec"]["modelServerConfig"]["options"] = "--model tiny --v2"
    store.update(isc)
    store.delete("Pod", "req1")
    drive(w["ctl"], infsvr_item(store))
    assert w["iid"] not in launcher.instances, \
        "obsolete instance should be deleted, not slept"
    assert w["inst_srv"].sleeps == 0


def test_unbound_launcher_sync_deletes_stopped_and_sleeps_awake():
    """Unbound launcher housekeeping (reference syncLauncherInstances +
    test-cases.sh:905): stopped instances deleted; an awake instance on
    an unbound launcher is put back to sleep."""
    w = mk_world(sleeping_target=False)
    launcher, http = w["launcher"], w["http"]
    launcher.instances["Istoppedi"] = {
        "instance_id": "Istoppedi", "status": "stopped",
        "options": "--port 9100",
        "annotations": {"inference-port": "9100"}}
    awake_srv = FakeInstanceServer(sleeping=False)
    http.register("10.0.0.2:9200", awake_srv)
    launcher.instances["Iawakei"] = {
        "instance_id": "Iawakei", "status": "running",
        "options": "--port 9200",
        "annotations": {"inference-port": "9200"}}
    w["ctl"].reconcile_unbound_launcher("node-a", "launcher1")
    assert "Istoppedi" not in launcher.instances
    assert awake_srv.sleeping, "awake instance on unbound launcher not slept"


def test_same_node_collision_two_requesters_one_slot():
    """Two requesters, one single-slot launcher: one binds, the other
    waits (cold-start creation kicks in) — no double-bind
    (reference test-cases.sh:396)."""
    w = mk_world(sleeping_target=False, max_instances=1)

    def on_create(iid, inst):
        w["http"].register("10.0.0.2:8000",
                           FakeInstanceServer(sleeping=False))

    w["launcher"].on_create = on_create
    add_requester(w, "reqX", "isc1", "10.0.0.7")
    drive(w["ctl"], infsvr_item(w["store"]))

    item_x = infsvr_item(w["store"], "reqX")
    for _ in range(6):
        w["ctl"]._process(item_x)
    lp = w["store"].get("Pod", "launcher1")
    bound_to = ob.annotations_of(lp)[C.REQUESTER_ANNOTATION]
    # exactly one of them is bound to launcher1
    assert bound_to.endswith(" req1") or bound_to.endswith(" reqX")
    # and no Pod carries a second binding to the same launcher
    bindings = [ob.annotations_of(p).get(C.REQUESTER_ANNOTATION)
                for p in w["store"].list("Pod")
                if ob.annotations_of(p).get(C.REQUESTER_ANNOTATION)]
    assert len(bindings) == len(set(bindings))


def test_exogenous_provider_deletion_mirrors_to_requester():
    """Bound provider vanishes -> the requester is deleted so its owner
    re-creates it (reference inference-server.go:257-290)."""
    w = mk_world()
    store = w["store"]
    drive(w["ctl"], infsvr_item(store))
    req = store.get("Pod", "req1")
    assert C.DUAL_LABEL in ob.labels_of(req)

    # force-remove the bound launcher (finalizer cleared, then delete)
    lp = store.get("Pod", "launcher1")
    ob.finalizers_of(lp).clear()
    store.update(lp, actor="dual-pods-controller")
    store.delete("Pod", "launcher1", actor="system")
    assert store.try_get("Pod", "launcher1") is None

    drive(w["ctl"], infsvr_item(store))
    assert store.try_get("Pod", "req1") is None, \
        "requester should mirror the provider's deletion"
    # and NO replacement launcher was silently re-bound for it
    assert w["stub"].unready_calls >= 1


def test_unbound_requester_not_mirrored():
    """A requester that was never bound cold-starts instead of dying."""
    w = mk_world(with_launcher=False)
    w["ctl"]._process(infsvr_item(w["store"]))
    assert w["store"].try_get("Pod", "req1") is not None


def test_switching_instances_in_one_launcher():
    """Reference test-cases.sh:560: requester(iscA) unbinds (instance A
    sleeps), requester(iscB) binds the SAME launcher — a new instance is
    created beside the sleeping one (maxInstances 2), then a third
    same-as-A requester hot-starts A again."""
    import copy

    w = mk_world()  # launcher holds sleeping instance for isc1 (A)
    store, http = w["store"], w["http"]

    # requester for isc1: hot start on A
    drive(w["ctl"], infsvr_item(store, "req1"))
    assert w["inst_srv"].wakes == 1

    # unbind: A sleeps again
    store.delete("Pod", "req1", actor="user")
    drive(w["ctl"], infsvr_item(store, "req1"))
    assert w["inst_srv"].sleeping

    # a second ISC arrives; its requester must bind the SAME launcher
    msc2 = copy.deepcopy(MSC)
    msc2["port"] = 8002
    store.create(ob.new_object(
        "InferenceServerConfig", "isc2",
        spec={"modelServerConfig": msc2, "launcherConfigName": "lc1"}))

    def on_create(iid, inst):
        http.register("10.0.0.2:8002", FakeInstanceServer(sleeping=False))

    w["launcher"].on_create = on_create
    req2 = ob.new_object(
        "Pod", "req2",
        annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc2"},
        spec={"nodeName": "node-a", "containers": [{"name": "stub"}]})
    req2 = store.create(req2)
    req2["status"] = {"phase": "Running", "podIP": "10.0.0.7"}
    store.update(req2)
    http.register("10.0.0.7:8081", Stub(["GPU-1"]))
    drive(w["ctl"], infsvr_item(store, "req2"))

    lp = store.get("Pod", "launcher1")
    assert ob.annotations_of(lp)[C.REQUESTER_ANNOTATION].endswith(" req2")
    # both instances live on the one launcher: A sleeping, B serving
    assert len(w["launcher"].instances) == 2
    assert w["inst_srv"].sleeping  # A still parked

    # requester for isc1 returns; launcher is bound to req2 -> the
    # controller must NOT steal it; a new launcher would be created
    # (cold). Instead delete req2 first, then hot-start A.
    store.delete("Pod", "req2", actor="user")
    drive(w["ctl"], infsvr_item(store, "req2"))
    req3 = ob.new_object(
        "Pod", "req3",
        annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"},
        spec={"nodeName": "node-a", "containers": [{"name": "stub"}]})
    req3 = store.create(req3)
    req3["status"] = {"phase": "Running", "podIP": "10.0.0.8"}
    store.update(req3)
    http.register("10.0.0.8:8081", Stub(["GPU-0"]))
    drive(w["ctl"], infsvr_item(store, "req3"))
    lp = store.get("Pod", "launcher1")
    assert ob.annotations_of(lp)[C.REQUESTER_ANNOTATION].endswith(" req3")
    assert ob.annotations_of(lp)[C.INSTANCE_ID_ANNOTATION] == w["iid"]
    assert not w["inst_srv"].sleeping  # A woken again
    assert w["inst_srv"].wakes == 2


def test_stopped_controller_does_not_write():
    """After stop(), queued/in-flight reconciles become no-ops so a
    replacement controller (restart) never races a stale writer: the
    reference gets this for free by killing the process; our in-process
    restart needs the stop gate."""
    w = mk_world()
    w["ctl"].stop()
    from fma_amd.controller.dualpods.controller import DONE
    assert w["ctl"]._process(infsvr_item(w["store"])) is DONE
    lp = w["store"].get("Pod", "launcher1")
    assert C.REQUESTER_ANNOTATION not in lp["metadata"].get(
        "annotations", {})
    assert not w["launcher"].wake_calls if hasattr(
        w["launcher"], "wake_calls") else True


def test_reconcile_in_non_default_namespace():
    """Whole hot-start flow in namespace 'prod': the controllers are
    one-namespace scoped like the reference (ControllerConfig.namespace)
    and nothing in the reconcile path may assume 'default'."""
    import tests.test_dualpods_controller as _c
    FakeHttp, FakeLauncher = _c.FakeHttp, _c.FakeLauncher
    FakeInstanceServer, Stub = _c.FakeInstanceServer, _c.Stub
    import copy as _copy

    ns = "prod"
    store = _c.MemStore()  # conformance suite patches this to KubeStore
    http = FakeHttp()
    ctl = DualPodsController(store, http, ControllerConfig(namespace=ns))
    msc = _copy.deepcopy(MSC)
    store.create(ob.new_object(
        "InferenceServerConfig", "isc1", namespace=ns,
        spec={"modelServerConfig": msc, "launcherConfigName": "lc1"}))
    store.create(ob.new_object(
        "LauncherConfig", "lc1", namespace=ns,
        spec={"maxInstances": 2, "podTemplate": {}}))
    req = store.create(ob.new_object(
        "Pod", "req1", namespace=ns,
        annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"},
        spec={"nodeName": "node-a", "containers": [{"name": "stub"}]}))
    req["status"] = {"phase": "Running", "podIP": "10.0.9.1"}
    store.update(req)
    http.register("10.0.9.1:8081", Stub(["GPU-0"]))

    iid = instance_id(msc, ["GPU-0"])
    lp = store.create(ob.new_object(
        "Pod", "launcher1", namespace=ns,
        labels={C.COMPONENT_LABEL: C.LAUNCHER_COMPONENT,
                C.LAUNCHER_CONFIG_NAME_LABEL: "lc1",
                C.SLEEPING_LABEL: "true"},
        annotations={"dual-pods.llm-d.ai/max-instances": "2"},
        spec={"nodeName": "node-a", "containers": [{"name": "launcher"}]}))
    lp["status"] = {"phase": "Running", "podIP": "10.0.9.2"}
    ob.set_pod_ready(lp, True)
    store.update(lp)
    fl = FakeLauncher(http, "10.0.9.2")
    http.register("10.0.9.2:8001", fl)
    fl.instances[iid] = {
        "instance_id": iid, "status": "running",
        "options": "--model tiny --port 8000", "gpu_uuids": ["GPU-0"],
        "annotations": {"isc-name": "isc1", "inference-port": "8000"}}
    http.register("10.0.9.2:8000", FakeInstanceServer(sleeping=True))

    drive(ctl, ("infsvr", "node-a", ob.uid_of(store.get(
        "Pod", "req1", ns)), "req1"))
    bound = store.get("Pod", "launcher1", ns)
    assert C.REQUESTER_ANNOTATION in ob.annotations_of(bound)
    assert ob.annotations_of(bound)[C.INSTANCE_ID_ANNOTATION] == iid
    # nothing leaked into the default namespace
    assert store.list("Pod", "default") == []
