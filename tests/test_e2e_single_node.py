"""End-to-end single-node actuation test (CPU, real processes, real HTTP).

The analog of the reference's kind-without-GPUs e2e
(reference test/e2e/run-launcher-based.sh + test-cases.sh): an in-process
cluster store + node agent play apiserver + kubelet; the launcher, serving
runtime and requester stub run as REAL separate processes on per-Pod
loopback IPs; the dual-pods controller and launcher-populator run their
real threads and reconcile over real HTTP.

Flow exercised: LPP/LC -> populator creates a launcher Pod -> node agent
spawns the launcher process -> requester Pod (ISC annotation) -> controller
discovers GPUs via the stub SPI, binds the launcher, creates the instance
(launcher forks the serving runtime), relays readiness -> requester /ready
goes 200. Then deletion -> sleep + unbind, and a second requester hot-
starts on the sleeping instance.
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
from fma_amd.store.admission import install_policies
from fma_amd.store.memstore import MemStore

pytestmark = pytest.mark.timeout(180)

ISC_PORT = 8355


#: widen every e2e deadline when the machine is oversubscribed (e.g.
#: two suites in parallel): FMA_TEST_WAIT_SCALE=2. The deadlines are
#: sized for a solo run; the only flakes ever observed were concurrent
#: full-suite runs on one 8-core box.
_WAIT_SCALE = float(os.environ.get("FMA_TEST_WAIT_SCALE", "1"))


def wait_for(cond, timeout=60, interval=0.25, desc="condition"):
    deadline = time.time() + timeout * _WAIT_SCALE
    while time.time() < deadline:
        v = cond()
        if v:
            return v
        time.sleep(interval)
    raise AssertionError(f"timed out waiting for {desc}")


@pytest.fixture()
def cluster(tmp_path):
    store = MemStore()
    install_policies(store)
    node = ob.new_object("Node", "node-a", labels={"gpu": "mi355x"})
    node["status"] = {"allocatable": {C.GPU_RESOURCE_NAME: 8}}
    store.create(node)

    env = {
        "PYTHONPATH": os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))),
        "FMA_FAKE_GPU": "1",
        "FMA_MOCK_GPU_COUNT": "4",
        "FMA_GPU_MODE": "naive",
        "FMA_ACCELERATORS": "GPU-0",
    }
    agent = NodeAgent(store, "node-a", node_index=7,
                      log_dir=str(tmp_path), extra_env=env)
    agent.start()

    ctl = DualPodsController(store, HttpAdapter(), ControllerConfig())
    ctl.start()
    pop = LauncherPopulator(store)
    pop.start()
    yield {"store": store, "agent": agent, "ctl": ctl, "pop": pop}
    ctl.stop()
    pop.stop()
    agent.stop()


def mk_isc_lc_lpp(store):
    store.create(ob.new_object(
        "LauncherConfig", "lc1",
        spec={"maxInstances": 2, "podTemplate": {"spec": {"containers": [{
            "name": "launcher",
            "command": [sys.executable, "-m", "fma_amd.launcher.service"],
        }]}}}))
    store.create(ob.new_object(
        "InferenceServerConfig", "isc1",
        spec={"modelServerConfig": {
            "port": ISC_PORT,
            "options": "--model tiny",
            "labels": {"llm-d.ai/model": "tiny"}},
            "launcherConfigName": "lc1"}))
    store.create(ob.new_object(
        "LauncherPopulationPolicy", "lpp1",
        spec={"enhancedNodeSelector": {
            "labelSelector": {"matchLabels": {"gpu": "mi355x"}}},
            "countForLauncher": [
                {"launcherConfigName": "lc1", "launcherCount": 1}]}))


def mk_requester(store, name):
    pod = ob.new_object(
        "Pod", name,
        annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"},
        spec={"nodeName": "node-a", "containers": [{
            "name": "requester",
            "command": [sys.executable, "-m", "fma_amd.requester.server"],
        }]})
    return store.create(pod, actor="user")


def launcher_pod(store):
    pods = [p for p in store.list("Pod")
            if ob.labels_of(p).get(C.COMPONENT_LABEL) == C.LAUNCHER_COMPONENT]
    return pods[0] if pods else None


def requester_ready(store, agent, name):
    pod = store.try_get("Pod", name)
    if pod is None:
        return False
    pp = agent.pods.get(name)
    if pp is None:
        return False
    try:
        r = httpx.get(f"http://{pp.ip}:8080/ready", timeout=2)
        return r.status_code == 200
    except httpx.HTTPError:
        return False


def test_full_launcher_based_actuation(cluster):
    store, agent = cluster["store"], cluster["agent"]
    mk_isc_lc_lpp(store)

    # populator creates a launcher; agent runs it and it becomes Ready
    lp = wait_for(lambda: launcher_pod(store), 30, desc="launcher pod")
    wait_for(lambda: ob.pod_is_ready(store.get("Pod", ob.name_of(lp))),
             60, desc="launcher Ready")

    # requester arrives; full actuation to readiness
    mk_requester(store, "req1")
    wait_for(lambda: requester_ready(store, agent, "req1"), 90,
             desc="requester /ready 200")

    lp = store.get("Pod", ob.name_of(lp))
    anns = ob.annotations_of(lp)
    assert anns[C.REQUESTER_ANNOTATION].endswith(" req1")
    assert ob.labels_of(lp)[C.SLEEPING_LABEL] == "false"
    assert ob.labels_of(lp)["llm-d.ai/model"] == "tiny"
    iid = anns[C.INSTANCE_ID_ANNOTATION]

    # the serving instance answers completions on the launcher's IP
    lp_ip = lp["status"]["podIP"]
    r = httpx.post(f"http://{lp_ip}:{ISC_PORT}/v1/completions",
                   json={"prompt": "hi", "max_tokens": 2}, timeout=10)
    assert r.status_code == 200

    # notifier signature reflected onto the Pod
    wait_for(lambda: C.INSTANCE_SIGNATURE_ANNOTATION in ob.annotations_of(
        store.get("Pod", ob.name_of(lp))), 20, desc="notifier signature")

    # server log relayed to the requester's stdout via /v1/set-log
    req_log = os.path.join(agent.log_dir, "pod-req1.log")
    wait_for(lambda: os.path.exists(req_log) and
             "[kickoff]" in open(req_log, errors="replace").read(), 30,
             desc="server log relayed to requester")

    # delete the requester: unbind + sleep, launcher survives
    store.delete("Pod", "req1", actor="user")
    wait_for(lambda: store.try_get("Pod", "req1") is None, 60,
             desc="requester gone")
    lp = store.get("Pod", ob.name_of(lp))
    assert C.REQUESTER_ANNOTATION not in ob.annotations_of(lp)
    assert ob.labels_of(lp)[C.SLEEPING_LABEL] == "true"
    assert "llm-d.ai/model" not in ob.labels_of(lp)
    r = httpx.get(f"http://{lp_ip}:{ISC_PORT}/is_sleeping", timeout=5)
    assert r.json() == {"is_sleeping": True}

    # second requester with the same ISC: hot start on the same instance
    mk_requester(store, "req2")
    wait_for(lambda: requester_ready(store, agent, "req2"), 60,
             desc="req2 ready (hot start)")
    lp = store.get("Pod", ob.name_of(lp))
    assert ob.annotations_of(lp)[C.REQUESTER_ANNOTATION].endswith(" req2")
    assert ob.annotations_of(lp)[C.INSTANCE_ID_ANNOTATION] == iid
    r = httpx.get(f"http://{lp_ip}:{ISC_PORT}/is_sleeping", timeout=5)
    assert r.json() == {"is_sleeping": False}


DIRECT_PATCH = """
spec:
  containers:
  - name: inference-server
    command: ["{python}", "-m", "fma_amd.runtime.server",
              "--model", "tiny", "--port", "8361"]
    ports:
    - containerPort: 8361
"""


def test_direct_path_actuation(cluster):
    """Direct (launcher-less) provider: server-patch -> nominal Pod ->
    node agent runs the serving runtime -> readiness relay + proxy config
    (the reference's Milestone-2 flow, test/e2e/run.sh)."""
    store, agent = cluster["store"], cluster["agent"]
    cm = ob.new_object("ConfigMap", C.GPU_MAP_CONFIGMAP)
    cm["data"] = {"node-a": '{"GPU-0": 0}'}
    store.create(cm)

    patch = DIRECT_PATCH.format(python=sys.executable)
    pod = ob.new_object(
        "Pod", "dreq1",
        annotations={C.SERVER_PATCH_ANNOTATION: patch},
        spec={"nodeName": "node-a", "containers": [
            {"name": "requester",
             "command": [sys.executable, "-m", "fma_amd.requester.server"]},
            {"name": "inference-server"}]})
    store.create(pod, actor="user")

    wait_for(lambda: requester_ready(store, agent, "dreq1"), 90,
             desc="direct requester ready")
    provider = store.get("Pod", "dreq1-server")
    assert ob.annotations_of(provider)[C.REQUESTER_ANNOTATION].endswith(
        " dreq1")
    prov_ip = provider["status"]["podIP"]
    r = httpx.get(f"http://{prov_ip}:8361/is_sleeping", timeout=5)
    assert r.json() == {"is_sleeping": False}

    # proxy on the requester points at the provider's serving endpoint
    stub_ip = agent.pods["dreq1"].ip
    r = httpx.get(f"http://{stub_ip}:8081/v1/proxy/config", timeout=5)
    assert r.status_code == 200
    assert r.json() == {"address": prov_ip, "port": 8361}

    # delete requester -> provider slept, kept as sleeper
    store.delete("Pod", "dreq1", actor="user")
    wait_for(lambda: store.try_get("Pod", "dreq1") is None, 60,
             desc="direct requester gone")
    provider = store.get("Pod", "dreq1-server")
    assert ob.labels_of(provider)[C.SLEEPING_LABEL] == "true"
    r = httpx.get(f"http://{prov_ip}:8361/is_sleeping", timeout=5)
    assert r.json() == {"is_sleeping": True}


def test_controller_restart_recovers_live_binding(cluster):
    """Kill the controller mid-flight and start a fresh one: the binding
    and serving state recover purely from Pod metadata
    (reference test-cases.sh:720)."""
    store, agent = cluster["store"], cluster["agent"]
    mk_isc_lc_lpp(store)
    lp = wait_for(lambda: launcher_pod(store), 30, desc="launcher pod")
    wait_for(lambda: ob.pod_is_ready(store.get("Pod", ob.name_of(lp))),
             60, desc="launcher Ready")
    mk_requester(store, "req1")
    wait_for(lambda: requester_ready(store, agent, "req1"), 90,
             desc="requester ready")

    cluster["ctl"].stop()
    for th in cluster["ctl"].workers.threads:
        th.join(timeout=35)  # drain in-flight reconciles before handover
    ctl2 = DualPodsController(store, HttpAdapter(), ControllerConfig())
    ctl2.start()
    try:
        # the recovered controller can still unbind cleanly...
        store.delete("Pod", "req1", actor="user")
        wait_for(lambda: store.try_get("Pod", "req1") is None, 60,
                 desc="requester gone via recovered controller")
        lp2 = store.get("Pod", ob.name_of(lp))
        assert C.REQUESTER_ANNOTATION not in ob.annotations_of(lp2)
        assert ob.labels_of(lp2)[C.SLEEPING_LABEL] == "true"
        # ...and hot-start the next requester
        mk_requester(store, "req9")
        wait_for(lambda: requester_ready(store, agent, "req9"), 60,
                 desc="req9 hot start via recovered controller")
    finally:
        ctl2.stop()


def test_four_concurrent_actuations_on_one_node(cluster):
    """Config #5 shape at the control-plane level: the populator holds 4
    launchers on the node; 4 requesters with 4 different ISCs all reach
    Ready concurrently."""
    store, agent = cluster["store"], cluster["agent"]
    store.create(ob.new_object(
        "LauncherConfig", "lc1",
        spec={"maxInstances": 2, "podTemplate": {"spec": {"containers": [{
            "name": "launcher",
            "command": [sys.executable, "-m", "fma_amd.launcher.service"],
        }]}}}))
    for i in range(4):
        store.create(ob.new_object(
            "InferenceServerConfig", f"isc-c{i}",
            spec={"modelServerConfig": {
                "port": 8380 + i, "options": "--model tiny"},
                "launcherConfigName": "lc1"}))
    store.create(ob.new_object(
        "LauncherPopulationPolicy", "lpp4",
        spec={"enhancedNodeSelector": {"labelSelector": {}},
              "countForLauncher": [
                  {"launcherConfigName": "lc1", "launcherCount": 4}]}))

    wait_for(lambda: sum(
        1 for p in store.list("Pod")
        if ob.labels_of(p).get(C.COMPONENT_LABEL) == C.LAUNCHER_COMPONENT
        and ob.pod_is_ready(p)) >= 4, 90, desc="4 launchers ready")

    for i in range(4):
        pod = ob.new_object(
            "Pod", f"creq{i}",
            annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: f"isc-c{i}"},
            spec={"nodeName": "node-a", "containers": [{
                "name": "requester",
                "command": [sys.executable, "-m",
                            "fma_amd.requester.server"]}]})
        store.create(pod, actor="user")

    for i in range(4):
        wait_for(lambda i=i: requester_ready(store, agent, f"creq{i}"), 120,
                 desc=f"creq{i} ready")
    # four distinct launchers are bound
    bound = [ob.annotations_of(p).get(C.REQUESTER_ANNOTATION)
             for p in store.list("Pod")
             if ob.labels_of(p).get(C.COMPONENT_LABEL) ==
             C.LAUNCHER_COMPONENT
             and ob.annotations_of(p).get(C.REQUESTER_ANNOTATION)]
    assert len(bound) == 4 and len(set(bound)) == 4


def test_launcher_crash_recovery(cluster):
    """Chaos: SIGKILL the real launcher process mid-service. The node
    agent reaps the pod's whole process tree (orphaned instances would
    otherwise squat on the server port) and restarts the container in
    place — restartPolicy Always, the k8s semantics the reference relies
    on (utils/pod-helper.go:44 treats restarts+unready as 'in trouble').
    The dual-pods controller then re-creates the missing instance on the
    restarted launcher and the SAME requester becomes ready again."""
    store, agent = cluster["store"], cluster["agent"]
    mk_isc_lc_lpp(store)
    lp0 = wait_for(lambda: launcher_pod(store), 30, desc="launcher pod")
    wait_for(lambda: ob.pod_is_ready(store.get("Pod", ob.name_of(lp0))),
             60, desc="launcher Ready")
    mk_requester(store, "cr1")
    wait_for(lambda: requester_ready(store, agent, "cr1"), 90,
             desc="first actuation ready")
    lname = ob.name_of(launcher_pod(store))
    uid0 = ob.uid_of(store.get("Pod", lname))

    agent.pods[lname].proc.kill()  # chaos

    # the launcher container restarts in place (same Pod object)...
    wait_for(lambda: agent.pods[lname].restarts >= 1, 30,
             desc="launcher restarted by the agent")
    # ...and the controller re-creates the instance on it: the SAME
    # requester's model answers completions again. (The requester stub
    # keeps its last readiness through the gap — the relay is one-way,
    # like the reference's — so the proof of recovery is the server
    # actually serving, not the stale stub bit.)
    lp = store.get("Pod", lname)
    assert ob.uid_of(lp) == uid0, "pod was replaced, not restarted"
    lp_ip = lp["status"]["podIP"]

    def completion_ok():
        try:
            r = httpx.post(f"http://{lp_ip}:{ISC_PORT}/v1/completions",
                           json={"prompt": "hi", "max_tokens": 2},
                           timeout=5)
            return r.status_code == 200
        except httpx.HTTPError:
            return False

    wait_for(completion_ok, 120,
             desc="completions served again after crash recovery")
    lp = store.get("Pod", lname)
    assert ob.annotations_of(lp)[C.REQUESTER_ANNOTATION].endswith(" cr1")
    assert store.try_get("Pod", "cr1") is not None
    assert requester_ready(store, agent, "cr1")
    # kubelet-style restart accounting surfaced on the Pod
    lp = store.get("Pod", lname)
    assert lp["status"]["containerStatuses"][0]["restartCount"] >= 1


def test_direct_provider_crash_recovery(cluster):
    """Chaos on the direct path: SIGKILL the serving-runtime process.
    The agent restarts the container in place; the restarted server
    boots fresh and the controller's bound reconcile re-drives it to
    serving (is_sleeping False) for the same requester."""
    store, agent = cluster["store"], cluster["agent"]
    cm = ob.new_object("ConfigMap", C.GPU_MAP_CONFIGMAP)
    cm["data"] = {"node-a": '{"GPU-0": 0}'}
    store.create(cm)
    patch = DIRECT_PATCH.format(python=sys.executable)
    pod = ob.new_object(
        "Pod", "xreq1",
        annotations={C.SERVER_PATCH_ANNOTATION: patch},
        spec={"nodeName": "node-a", "containers": [
            {"name": "requester",
             "command": [sys.executable, "-m", "fma_amd.requester.server"]},
            {"name": "inference-server"}]})
    store.create(pod, actor="user")
    wait_for(lambda: requester_ready(store, agent, "xreq1"), 90,
             desc="direct requester ready")
    provider = store.get("Pod", "xreq1-server")
    prov_ip = provider["status"]["podIP"]
    uid0 = ob.uid_of(provider)

    agent.pods["xreq1-server"].proc.kill()  # chaos

    wait_for(lambda: agent.pods["xreq1-server"].restarts >= 1, 30,
             desc="provider restarted by the agent")

    def serving_again():
        try:
            r = httpx.get(f"http://{prov_ip}:8361/is_sleeping", timeout=5)
            return r.status_code == 200 and r.json()["is_sleeping"] is False
        except httpx.HTTPError:
            return False

    wait_for(serving_again, 120, desc="direct provider serving again")
    provider = store.get("Pod", "xreq1-server")
    assert ob.uid_of(provider) == uid0, "provider replaced, not restarted"
    assert ob.annotations_of(provider)[C.REQUESTER_ANNOTATION].endswith(
        " xreq1")
