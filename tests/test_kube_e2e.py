"""Full-stack e2e over the KUBERNETES wire protocol.

Same real-process actuation flow as tests/test_e2e_single_node.py, but
every component (both controllers + the node agent) talks to the
apiserver double through KubeStore — i.e., through the Kubernetes REST
protocol with the shipped CEL admission policies enforced. This is the
analog of the reference's kind e2e (reference
test/e2e/run-launcher-based.sh driving test-cases.sh): scenarios covered
here map to launcher-based pod creation (:260), wake fast path (:465),
controller restart recovery (:720) and deletion/unbinding (:836).
"""

import os
import sys
import time

import httpx
import pytest

from fma_amd.api import contracts as C
from fma_amd.controller.dualpods.controller import (ControllerConfig,
                                                    DualPodsController)
from fma_amd.controller.httpadapter import HttpAdapter
from fma_amd.controller.populator.populator import LauncherPopulator
from fma_amd.node.agent import NodeAgent
from fma_amd.store import objects as ob
from fma_amd.store.kubeapiserver import create_app
from fma_amd.store.kubestore import KubeStore
from fma_amd.store.memstore import Invalid, MemStore

from tests.test_e2e_single_node import (ISC_PORT, launcher_pod,
                                        mk_isc_lc_lpp, mk_requester,
                                        requester_ready, wait_for)
from tests.test_live_servers import ServerThread, free_port

pytestmark = pytest.mark.timeout(700)


@pytest.fixture()
def kube_cluster(tmp_path):
    mem = MemStore()
    port = free_port()
    with ServerThread(create_app(mem), port):
        base = f"http://127.0.0.1:{port}"
        admin = KubeStore(base, actor="dual-pods-controller")
        node = ob.new_object("Node", "node-a", labels={"gpu": "mi355x"})
        node["status"] = {"allocatable": {C.GPU_RESOURCE_NAME: 8}}
        admin.create(node)

        env = {
            "PYTHONPATH": os.path.dirname(os.path.dirname(
                os.path.abspath(__file__))),
            "FMA_FAKE_GPU": "1",
            "FMA_MOCK_GPU_COUNT": "4",
            "FMA_GPU_MODE": "naive",
            "FMA_ACCELERATORS": "GPU-0",
        }
        agent = NodeAgent(KubeStore(base, actor="node-agent"), "node-a",
                          node_index=9, log_dir=str(tmp_path),
                          extra_env=env)
        agent.start()
        ctl = DualPodsController(KubeStore(base, actor="dual-pods-controller"),
                                 HttpAdapter(), ControllerConfig())
        ctl.start()
        pop = LauncherPopulator(KubeStore(base, actor="launcher-populator"))
        pop.start()
        user = KubeStore(base, actor="user")
        try:
            yield {"base": base, "store": user, "agent": agent,
                   "ctl": ctl, "pop": pop}
        finally:
            ctl.stop()
            pop.stop()
            agent.stop()


def test_kube_full_actuation_hot_start_and_vap(kube_cluster):
    store, agent = kube_cluster["store"], kube_cluster["agent"]
    mk_isc_lc_lpp(store)

    lp = wait_for(lambda: launcher_pod(store), 100, desc="launcher pod")
    wait_for(lambda: ob.pod_is_ready(store.get("Pod", ob.name_of(lp))),
             90, desc="launcher Ready")

    mk_requester(store, "kreq1")
    wait_for(lambda: requester_ready(store, agent, "kreq1"), 200,
             desc="requester /ready 200 over the kube protocol")

    lp = store.get("Pod", ob.name_of(lp))
    anns = ob.annotations_of(lp)
    assert anns[C.REQUESTER_ANNOTATION].endswith(" kreq1")
    assert ob.labels_of(lp)[C.SLEEPING_LABEL] == "false"
    iid = anns[C.INSTANCE_ID_ANNOTATION]

    # the serving instance answers on the launcher's IP
    lp_ip = lp["status"]["podIP"]
    r = httpx.post(f"http://{lp_ip}:{ISC_PORT}/v1/completions",
                   json={"prompt": "hi", "max_tokens": 2}, timeout=10)
    assert r.status_code == 200

    # notifier signature reflected onto the launcher Pod over the kube
    # protocol (the informer-visible launcher-state channel)
    wait_for(lambda: C.INSTANCE_SIGNATURE_ANNOTATION in ob.annotations_of(
        store.get("Pod", ob.name_of(lp))), 30, desc="notifier signature")

    # CEL VAP: a user cannot clear the binding annotation on the provider
    hacked = store.get("Pod", ob.name_of(lp))
    hacked["metadata"]["annotations"][C.REQUESTER_ANNOTATION] = "u hack"
    with pytest.raises(Invalid):
        store.update(hacked)

    # delete requester -> sleep + unbind; launcher survives with the
    # sleeping instance
    store.delete("Pod", "kreq1")
    wait_for(lambda: store.try_get("Pod", "kreq1") is None, 150,
             desc="requester gone")
    lp = store.get("Pod", ob.name_of(lp))
    assert C.REQUESTER_ANNOTATION not in ob.annotations_of(lp)
    assert ob.labels_of(lp)[C.SLEEPING_LABEL] == "true"

    # hot start: same-ISC requester rebinds the sleeping instance
    mk_requester(store, "kreq2")
    wait_for(lambda: requester_ready(store, agent, "kreq2"), 150,
             desc="kreq2 hot start")
    lp = store.get("Pod", ob.name_of(lp))
    assert ob.annotations_of(lp)[C.INSTANCE_ID_ANNOTATION] == iid
    r = httpx.get(f"http://{lp_ip}:{ISC_PORT}/is_sleeping", timeout=5)
    assert r.json() == {"is_sleeping": False}


@pytest.mark.timeout(700)
def test_kube_controller_restart_recovery(kube_cluster):
    """Controller restart recovers bindings purely from Pod metadata read
    back over the kube protocol (reference test-cases.sh:720). Deadlines
    are the suite's widest: this test runs late in the full suite when
    a loaded host is slowest (it has never failed in isolation)."""
    store, agent = kube_cluster["store"], kube_cluster["agent"]
    base = kube_cluster["base"]
    mk_isc_lc_lpp(store)
    lp = wait_for(lambda: launcher_pod(store), 150, desc="launcher pod")
    wait_for(lambda: ob.pod_is_ready(store.get("Pod", ob.name_of(lp))),
             120, desc="launcher Ready")
    mk_requester(store, "rreq1")
    wait_for(lambda: requester_ready(store, agent, "rreq1"), 300,
             desc="requester ready")

    kube_cluster["ctl"].stop()
    for th in kube_cluster["ctl"].workers.threads:
        th.join(timeout=35)  # drain in-flight reconciles before handover
    ctl2 = DualPodsController(
        KubeStore(base, actor="dual-pods-controller"), HttpAdapter(),
        ControllerConfig())
    ctl2.start()
    try:
        store.delete("Pod", "rreq1")

        def _gone():
            return store.try_get("Pod", "rreq1") is None

        def _state():  # diagnostic on timeout
            pods = [(ob.name_of(p), ob.labels_of(p).get(C.SLEEPING_LABEL),
                     ob.is_deleting(p)) for p in store.list("Pod")]
            return f"requester gone via recovered controller; pods={pods}"

        try:
            wait_for(_gone, 250, desc="requester gone")
        except AssertionError:
            raise AssertionError(f"timed out: {_state()}") from None
        lp2 = store.get("Pod", ob.name_of(lp))
        assert C.REQUESTER_ANNOTATION not in ob.annotations_of(lp2)
        assert ob.labels_of(lp2)[C.SLEEPING_LABEL] == "true"
    finally:
        ctl2.stop()


def test_kube_instance_cap_and_second_launcher(kube_cluster):
    """Per-launcher instance cap over the kube protocol: with
    maxInstances=1 and two different-ISC requesters, the stack creates a
    second launcher rather than over-filling the first (reference
    test-cases.sh:633 'per-launcher instance cap' + :396 same-node
    collision handling)."""
    store, agent = kube_cluster["store"], kube_cluster["agent"]
    store.create(ob.new_object(
        "LauncherConfig", "lc1",
        spec={"maxInstances": 1, "podTemplate": {"spec": {"containers": [{
            "name": "launcher",
            "command": [__import__("sys").executable, "-m",
                        "fma_amd.launcher.service"],
        }]}}}))
    for i, port in enumerate((8372, 8373)):
        store.create(ob.new_object(
            "InferenceServerConfig", f"isc-cap{i}",
            spec={"modelServerConfig": {"port": port,
                                        "options": "--model tiny"},
                  "launcherConfigName": "lc1"}))
    store.create(ob.new_object(
        "LauncherPopulationPolicy", "lpp-cap",
        spec={"enhancedNodeSelector": {"labelSelector": {}},
              "countForLauncher": [
                  {"launcherConfigName": "lc1", "launcherCount": 2}]}))
    wait_for(lambda: sum(
        1 for p in store.list("Pod")
        if ob.labels_of(p).get(C.COMPONENT_LABEL) == C.LAUNCHER_COMPONENT
        and ob.pod_is_ready(p)) >= 2, 120, desc="2 launchers ready")

    for i in range(2):
        store.create(ob.new_object(
            "Pod", f"capreq{i}",
            annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: f"isc-cap{i}"},
            spec={"nodeName": "node-a", "containers": [{
                "name": "requester",
                "command": [__import__("sys").executable, "-m",
                            "fma_amd.requester.server"]}]}))
    for i in range(2):
        wait_for(lambda i=i: requester_ready(store, agent, f"capreq{i}"),
                 120, desc=f"capreq{i} ready")
    bound = {ob.annotations_of(p)[C.REQUESTER_ANNOTATION]
             for p in store.list("Pod")
             if ob.labels_of(p).get(C.COMPONENT_LABEL) ==
             C.LAUNCHER_COMPONENT
             and ob.annotations_of(p).get(C.REQUESTER_ANNOTATION)}
    assert len(bound) == 2  # two launchers, one instance each (cap=1)


def test_kube_stopped_instance_recovers(kube_cluster):
    """Stopped-instance recovery over the kube protocol (reference
    test-cases.sh:905): kill the serving instance process; the sentinel
    exit surfaces through the launcher, the controller deletes the
    requester Pod (its owner would re-create it in a Deployment)."""
    import signal

    store, agent = kube_cluster["store"], kube_cluster["agent"]
    mk_isc_lc_lpp(store)
    lp = wait_for(lambda: launcher_pod(store), 100, desc="launcher pod")
    wait_for(lambda: ob.pod_is_ready(store.get("Pod", ob.name_of(lp))),
             90, desc="launcher Ready")
    mk_requester(store, "sreq1")
    wait_for(lambda: requester_ready(store, agent, "sreq1"), 200,
             desc="requester ready")

    # SIGKILL the forked serving-instance process: it is the launcher
    # process's child (the wire state carries no pid, as in the
    # reference; the e2e reaches around the API like test-cases.sh does
    # with kubectl exec)
    import psutil

    lname = ob.name_of(lp)
    launcher_proc = psutil.Process(agent.pods[lname].proc.pid)
    kids = wait_for(lambda: launcher_proc.children() or None, 50,
                    desc="instance child process")
    os.kill(kids[0].pid, signal.SIGKILL)

    # controller reacts: requester Pod deleted (reference deletes it so
    # the owning ReplicaSet re-creates a fresh one)
    wait_for(lambda: store.try_get("Pod", "sreq1") is None, 200,
             desc="requester deleted after instance death")


def test_kube_two_node_population_and_binding(tmp_path):
    """Two node agents + both controllers all over the kube protocol:
    the populator spans nodes (one launcher each), and a requester on
    node-b binds node-b's launcher (binding locality) — the kind
    multi-node analog running against the apiserver double."""
    mem = MemStore()
    port = free_port()
    with ServerThread(create_app(mem), port):
        base = f"http://127.0.0.1:{port}"
        admin = KubeStore(base, actor="dual-pods-controller")
        env = {
            "PYTHONPATH": os.path.dirname(os.path.dirname(
                os.path.abspath(__file__))),
            "FMA_FAKE_GPU": "1",
            "FMA_GPU_MODE": "naive",
            "FMA_ACCELERATORS": "GPU-0",
        }
        agents = {}
        for i, node in enumerate(("node-a", "node-b"), start=1):
            n = ob.new_object("Node", node, labels={"gpu": "mi355x"})
            n["status"] = {"allocatable": {C.GPU_RESOURCE_NAME: 8}}
            admin.create(n)
            agents[node] = NodeAgent(
                KubeStore(base, actor="node-agent"), node,
                node_index=30 + i, log_dir=str(tmp_path), extra_env=env)
            agents[node].start()
        ctl = DualPodsController(
            KubeStore(base, actor="dual-pods-controller"), HttpAdapter(),
            ControllerConfig())
        ctl.start()
        pop = LauncherPopulator(KubeStore(base, actor="launcher-populator"))
        pop.start()
        user = KubeStore(base, actor="user")
        try:
            mk_isc_lc_lpp(user)  # LPP matches both nodes
            wait_for(lambda: len([
                p for p in user.list("Pod")
                if ob.labels_of(p).get(C.COMPONENT_LABEL) ==
                C.LAUNCHER_COMPONENT]) == 2, 90, desc="two launchers")
            by_node = {}
            for p in user.list("Pod"):
                if ob.labels_of(p).get(C.COMPONENT_LABEL) == \
                        C.LAUNCHER_COMPONENT:
                    by_node[ob.pod_node_name(p)] = p
            assert set(by_node) == {"node-a", "node-b"}
            for node, lp in by_node.items():
                wait_for(lambda lp=lp: ob.pod_is_ready(
                    user.get("Pod", ob.name_of(lp))), 150,
                    desc=f"launcher on {node} ready")

            pod = ob.new_object(
                "Pod", "breq",
                annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"},
                spec={"nodeName": "node-b", "containers": [{
                    "name": "requester",
                    "command": [sys.executable, "-m",
                                "fma_amd.requester.server"]}]})
            user.create(pod)
            wait_for(lambda: requester_ready(user, agents["node-b"],
                                             "breq"), 200,
                     desc="node-b requester ready")
            lp_b = user.get("Pod", ob.name_of(by_node["node-b"]))
            assert ob.annotations_of(lp_b)[C.REQUESTER_ANNOTATION].endswith(
                " breq")
            lp_a = user.get("Pod", ob.name_of(by_node["node-a"]))
            assert C.REQUESTER_ANNOTATION not in ob.annotations_of(lp_a)
        finally:
            ctl.stop()
            pop.stop()
            for a in agents.values():
                a.stop()
