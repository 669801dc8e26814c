"""Two-node behavior: per-node population and binding locality
(the reference gets multi-node from kind's containerized nodes,
docs/local-test.md:38-48; here two node agents share one store)."""

import os
import sys

import pytest

from fma_amd.api import contracts as C
from fma_amd.controller.dualpods.controller import (ControllerConfig,
                                                    DualPodsController)
from fma_amd.controller.httpadapter import HttpAdapter
from fma_amd.controller.populator.populator import LauncherPopulator
from fma_amd.node.agent import NodeAgent
from fma_amd.store import objects as ob
from fma_amd.store.memstore import MemStore

from tests.test_e2e_single_node import (ISC_PORT, launcher_pod, mk_isc_lc_lpp,
                                        requester_ready, wait_for)

pytestmark = pytest.mark.timeout(180)


def test_population_and_binding_stay_per_node(tmp_path):
    store = MemStore()
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
        store.create(n)
        agents[node] = NodeAgent(store, node, node_index=20 + i,
                                 log_dir=str(tmp_path), extra_env=env)
        agents[node].start()
    ctl = DualPodsController(store, HttpAdapter(), ControllerConfig())
    ctl.start()
    pop = LauncherPopulator(store)
    pop.start()
    try:
        mk_isc_lc_lpp(store)  # LPP matches all nodes: 1 launcher per node
        wait_for(lambda: len([
            p for p in store.list("Pod")
            if ob.labels_of(p).get(C.COMPONENT_LABEL) ==
            C.LAUNCHER_COMPONENT]) == 2, 60, desc="two launchers")
        by_node = {}
        for p in store.list("Pod"):
            if ob.labels_of(p).get(C.COMPONENT_LABEL) == C.LAUNCHER_COMPONENT:
                by_node[ob.pod_node_name(p)] = p
        assert set(by_node) == {"node-a", "node-b"}
        for node, lp in by_node.items():
            wait_for(lambda lp=lp: ob.pod_is_ready(
                store.get("Pod", ob.name_of(lp))), 60,
                desc=f"launcher on {node} ready")

        # requester scheduled to node-b must bind node-b's launcher
        pod = ob.new_object(
            "Pod", "req-b",
            annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"},
            spec={"nodeName": "node-b", "containers": [{
                "name": "requester",
                "command": [sys.executable, "-m",
                            "fma_amd.requester.server"]}]})
        store.create(pod, actor="user")
        wait_for(lambda: requester_ready(store, agents["node-b"], "req-b"),
                 90, desc="req-b ready")
        lp_b = store.get("Pod", ob.name_of(by_node["node-b"]))
        assert ob.annotations_of(lp_b)[C.REQUESTER_ANNOTATION].endswith(
            " req-b")
        lp_a = store.get("Pod", ob.name_of(by_node["node-a"]))
        assert C.REQUESTER_ANNOTATION not in ob.annotations_of(lp_a), \
            "binding crossed nodes"
    finally:
        ctl.stop()
        pop.stop()
        for a in agents.values():
            a.stop()
