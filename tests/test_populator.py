"""Launcher-populator tests (fake store, synchronous queue processing)."""

import time

from fma_amd.api import contracts as C
from fma_amd.controller.populator.populator import (HANDS_OFF,
                                                    LauncherPopulator)
from fma_amd.store import objects as ob
from fma_amd.store.memstore import MemStore


class FakeClock:
    def __init__(self, t=1000.0):
        self.t = t

    def time(self):
        return self.t


def mk_node(store, name, labels=None, allocatable=None):
    node = ob.new_object("Node", name, labels=labels or {})
    node["status"] = {"allocatable": allocatable or {}}
    return store.create(node)


def mk_lc(store, name="lc1", max_instances=2):
    return store.create(ob.new_object(
        "LauncherConfig", name,
        spec={"maxInstances": max_instances, "podTemplate": {}}))


def mk_lpp(store, name, lc_name="lc1", count=2, match_labels=None,
           resources=None):
    spec = {
        "enhancedNodeSelector": {
            "labelSelector": {"matchLabels": match_labels or {}},
        },
        "countForLauncher": [
            {"launcherConfigName": lc_name, "launcherCount": count}],
    }
    if resources:
        spec["enhancedNodeSelector"]["allocatableResources"] = resources
    return store.create(ob.new_object("LauncherPopulationPolicy", name,
                                      spec=spec))


def mk_pop(store, clock=None):
    pop = LauncherPopulator(store, clock=clock or FakeClock())
    # synchronous digest of everything present
    for lc in store.list("LauncherConfig"):
        pop._process_digest(("lc", ob.name_of(lc)))
    for lpp in store.list("LauncherPopulationPolicy"):
        pop._process_digest(("lpp", ob.name_of(lpp)))
    pop._keys_started.set()
    return pop


def launcher_pods(store, node=None):
    out = []
    for p in store.list("Pod"):
        if ob.labels_of(p).get(C.COMPONENT_LABEL) == C.LAUNCHER_COMPONENT:
            if node is None or ob.pod_node_name(p) == node:
                out.append(p)
    return out


def drain_key(pop, key, iters=10):
    for _ in range(iters):
        if not pop._process_key(key):
            return
        pop.clock.t += 6  # skip past expectation timeout
    raise AssertionError("key did not settle")


def test_desired_is_max_over_lpps():
    store = MemStore()
    mk_node(store, "n1", labels={"gpu": "mi355x"})
    mk_lc(store)
    mk_lpp(store, "p1", count=1, match_labels={"gpu": "mi355x"})
    mk_lpp(store, "p2", count=3, match_labels={"gpu": "mi355x"})
    pop = mk_pop(store)
    desired, _ = pop.policy.snapshot_for_key("n1", "lc1")
    assert desired == 3


def test_hands_off_when_lc_missing():
    store = MemStore()
    mk_node(store, "n1")
    mk_lpp(store, "p1", lc_name="ghost", count=2)
    pop = mk_pop(store)
    pop._process_digest(("lpp", "p1"))
    desired, _ = pop.policy.snapshot_for_key("n1", "ghost")
    assert desired == HANDS_OFF
    drain_key(pop, ("n1", "ghost"))
    assert launcher_pods(store) == []


def test_resource_range_matching():
    store = MemStore()
    mk_node(store, "big", allocatable={"amd.com/gpu": 8, "memory": "512Gi"})
    mk_node(store, "small", allocatable={"amd.com/gpu": 1, "memory": "64Gi"})
    mk_lc(store)
    mk_lpp(store, "p1", count=2,
           resources={"amd.com/gpu": {"min": 4}})
    pop = mk_pop(store)
    assert pop.policy.snapshot_for_key("big", "lc1")[0] == 2
    assert pop.policy.snapshot_for_key("small", "lc1")[0] == 0


def test_creates_launchers_to_desired_count():
    store = MemStore()
    mk_node(store, "n1", labels={"gpu": "x"})
    mk_lc(store)
    mk_lpp(store, "p1", count=2, match_labels={"gpu": "x"})
    pop = mk_pop(store)
    drain_key(pop, ("n1", "lc1"))
    pods = launcher_pods(store, "n1")
    assert len(pods) == 2
    for p in pods:
        assert ob.labels_of(p)[C.LAUNCHER_CONFIG_NAME_LABEL] == "lc1"
        assert ob.labels_of(p)[C.SLEEPING_LABEL] == "true"
        assert ob.annotations_of(p)[C.LAUNCHER_TEMPLATE_HASH_ANNOTATION]
        # owner-ref for GC on LC deletion
        assert ob.meta(p)["ownerReferences"][0]["kind"] == "LauncherConfig"
    # idempotent
    drain_key(pop, ("n1", "lc1"))
    assert len(launcher_pods(store, "n1")) == 2


def test_scale_down_deletes_only_unbound():
    store = MemStore()
    mk_node(store, "n1", labels={"gpu": "x"})
    mk_lc(store)
    lpp = mk_lpp(store, "p1", count=3, match_labels={"gpu": "x"})
    pop = mk_pop(store)
    drain_key(pop, ("n1", "lc1"))
    pods = launcher_pods(store, "n1")
    assert len(pods) == 3
    # bind one
    bound = pods[0]
    ob.annotations_of(bound)[C.REQUESTER_ANNOTATION] = "uid req"
    store.update(bound)
    # scale policy down to 1
    lpp = store.get("LauncherPopulationPolicy", "p1")
    lpp["spec"]["countForLauncher"][0]["launcherCount"] = 1
    store.update(lpp)
    pop._process_digest(("lpp", "p1"))
    drain_key(pop, ("n1", "lc1"))
    left = launcher_pods(store, "n1")
    # the bound one must survive; total live = desired(1) incl. bound
    assert any(ob.annotations_of(p).get(C.REQUESTER_ANNOTATION)
               for p in left)
    assert len(left) == 1


def test_template_change_replaces_stale_launchers():
    store = MemStore()
    mk_node(store, "n1", labels={"gpu": "x"})
    lc = mk_lc(store)
    mk_lpp(store, "p1", count=1, match_labels={"gpu": "x"})
    pop = mk_pop(store)
    drain_key(pop, ("n1", "lc1"))
    old = launcher_pods(store, "n1")[0]
    old_hash = ob.annotations_of(old)[C.LAUNCHER_TEMPLATE_HASH_ANNOTATION]

    lc = store.get("LauncherConfig", "lc1")
    lc["spec"]["podTemplate"] = {
        "spec": {"containers": [{"name": "launcher", "image": "new:v2"}]}}
    store.update(lc)
    pop._process_digest(("lc", "lc1"))
    drain_key(pop, ("n1", "lc1"))
    pods = launcher_pods(store, "n1")
    assert len(pods) == 1
    new_hash = ob.annotations_of(pods[0])[C.LAUNCHER_TEMPLATE_HASH_ANNOTATION]
    assert new_hash != old_hash


def test_lpp_status_reports_errors():
    store = MemStore()
    mk_node(store, "n1")
    mk_lc(store)
    lpp = store.create(ob.new_object(
        "LauncherPopulationPolicy", "bad",
        spec={"enhancedNodeSelector": {"labelSelector": {}},
              "countForLauncher": [
                  {"launcherConfigName": "lc1", "launcherCount": 1},
                  {"launcherConfigName": "lc1", "launcherCount": 2}]}))
    pop = mk_pop(store)
    pop._process_digest(("lpp", "bad"))
    lpp = store.get("LauncherPopulationPolicy", "bad")
    assert lpp["status"]["errors"]
    assert "unique" in lpp["status"]["errors"][0]


def test_stuck_phases():
    store = MemStore()
    clock = FakeClock(1000.0)
    mk_lc(store)
    pop = mk_pop(store, clock)
    # unscheduled young pod
    pod = ob.new_object("Pod", "young",
                        labels={C.COMPONENT_LABEL: C.LAUNCHER_COMPONENT})
    pod = store.create(pod)
    ob.meta(pod)["creationTimestamp"] = 950.0
    assert pop.launcher_phase(pod) == "unbound"
    ob.meta(pod)["creationTimestamp"] = 1000.0 - 200
    assert pop.launcher_phase(pod) == "stuck_scheduling"
    pod["spec"] = {"nodeName": "n1"}
    assert pop.launcher_phase(pod) == "unbound"  # scheduled, young enough
    ob.meta(pod)["creationTimestamp"] = 1000.0 - 500
    assert pop.launcher_phase(pod) == "stuck_starting"
    ob.set_pod_ready(pod, True)
    assert pop.launcher_phase(pod) == "unbound"
    ob.annotations_of(pod)[C.REQUESTER_ANNOTATION] = "u r"
    assert pop.launcher_phase(pod) == "bound"


def test_stuck_label_applied():
    store = MemStore()
    clock = FakeClock(2000.0)
    mk_node(store, "n1", labels={"gpu": "x"})
    mk_lc(store)
    mk_lpp(store, "p1", count=1, match_labels={"gpu": "x"})
    pop = mk_pop(store, clock)
    drain_key(pop, ("n1", "lc1"))
    pod = launcher_pods(store, "n1")[0]
    # jump the clock past the starting threshold (creationTimestamp is
    # server-side immutable), pod never becomes Ready
    created = ob.meta(store.get("Pod", ob.name_of(pod)))["creationTimestamp"]
    clock.t = created + 500
    drain_key(pop, ("n1", "lc1"))
    cur = store.get("Pod", ob.name_of(pod))
    assert ob.labels_of(cur)[C.LAUNCHER_STUCK_LABEL] == "stuck_starting"


def test_lc_deletion_garbage_collects_launchers():
    store = MemStore()
    mk_node(store, "n1", labels={"gpu": "x"})
    mk_lc(store)
    mk_lpp(store, "p1", count=2, match_labels={"gpu": "x"})
    pop = mk_pop(store)
    drain_key(pop, ("n1", "lc1"))
    assert len(launcher_pods(store)) == 2
    store.delete("LauncherConfig", "lc1")
    # owner-reference GC in the store removes the launcher pods
    assert launcher_pods(store) == []


def test_failed_launcher_replaced():
    store = MemStore()
    mk_node(store, "n1", labels={"gpu": "x"})
    mk_lc(store)
    mk_lpp(store, "p1", count=1, match_labels={"gpu": "x"})
    pop = mk_pop(store)
    drain_key(pop, ("n1", "lc1"))
    pod = launcher_pods(store, "n1")[0]
    cur = store.get("Pod", ob.name_of(pod))
    cur.setdefault("status", {})["phase"] = "Failed"
    store.update(cur, actor="node-agent", subresource="status")
    drain_key(pop, ("n1", "lc1"))
    pods = launcher_pods(store, "n1")
    assert len(pods) == 1
    assert ob.uid_of(pods[0]) != ob.uid_of(pod), "failed launcher not replaced"
