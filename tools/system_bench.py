"""Full-stack actuation benchmark: T_actuation through the whole stack.

Runs the real single-node stack (store + node agent + both controllers,
real launcher/requester processes) and measures requester-created ->
requester-Ready for the cold first actuation and subsequent hot starts of
a real-weight model — the T_actuation / Hot_hit metrics of the
reference's benchmark.md, with actual tensor movement underneath.

GPU box:  python tools/system_bench.py --gib 15 --hot-cycles 3
CPU box:  works too (fake arena; numbers reflect control plane only).
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import httpx  # noqa: E402
import torch  # noqa: E402

from fma_amd.api import contracts as C  # noqa: E402
from fma_amd.controller.dualpods.controller import (  # noqa: E402
    ControllerConfig, DualPodsController)
from fma_amd.controller.httpadapter import HttpAdapter  # noqa: E402
from fma_amd.controller.populator.populator import LauncherPopulator  # noqa: E402
from fma_amd.node.agent import NodeAgent  # noqa: E402
from fma_amd.store import objects as ob  # noqa: E402
from fma_amd.store.admission import install_policies  # noqa: E402
from fma_amd.store.memstore import MemStore  # noqa: E402

ISC_PORT = 8371


def wait_for(cond, timeout, desc):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if cond():
            return True
        time.sleep(0.05)
    raise TimeoutError(desc)


def requester_ready(agent, name):
    pp = agent.pods.get(name)
    if pp is None:
        return False
    try:
        return httpx.get(f"http://{pp.ip}:8080/ready",
                         timeout=2).status_code == 200
    except httpx.HTTPError:
        return False


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gib", type=float, default=15.0)
    ap.add_argument("--hot-cycles", type=int, default=3)
    ap.add_argument("--log-dir", default="/tmp/fma-sysbench")
    args = ap.parse_args()
    os.makedirs(args.log_dir, exist_ok=True)

    store = MemStore()
    install_policies(store)
    node = ob.new_object("Node", "node-a", labels={"gpu": "mi355x"})
    node["status"] = {"allocatable": {C.GPU_RESOURCE_NAME: 8}}
    store.create(node)
    env = {
        "PYTHONPATH": os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))),
        "FMA_GPU_MODE": "naive",
        "FMA_ACCELERATORS": "GPU-0",
    }
    if not torch.cuda.is_available():
        env["FMA_FAKE_GPU"] = "1"
    agent = NodeAgent(store, "node-a", node_index=9, log_dir=args.log_dir,
                      extra_env=env)
    agent.start()
    ctl = DualPodsController(store, HttpAdapter(), ControllerConfig())
    ctl.start()
    pop = LauncherPopulator(store)
    pop.start()

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
            "options": f"--model synthetic-{args.gib:g}gib"},
            "launcherConfigName": "lc1"}))
    store.create(ob.new_object(
        "LauncherPopulationPolicy", "lpp1",
        spec={"enhancedNodeSelector": {"labelSelector": {}},
              "countForLauncher": [
                  {"launcherConfigName": "lc1", "launcherCount": 1}]}))

    def mk_requester(name):
        pod = ob.new_object(
            "Pod", name,
            annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"},
            spec={"nodeName": "node-a", "containers": [{
                "name": "requester",
                "command": [sys.executable, "-m",
                            "fma_amd.requester.server"]}]})
        store.create(pod, actor="bench")

    try:
        # wait for the populated launcher
        wait_for(lambda: any(
            ob.pod_is_ready(p) for p in store.list("Pod")
            if ob.labels_of(p).get(C.COMPONENT_LABEL) ==
            C.LAUNCHER_COMPONENT), 120, "launcher ready")

        t0 = time.perf_counter()
        mk_requester("bench-req-0")
        wait_for(lambda: requester_ready(agent, "bench-req-0"), 600,
                 "cold actuation")
        t_cold = time.perf_counter() - t0

        hot = []
        for i in range(args.hot_cycles):
            store.delete("Pod", f"bench-req-{i}", actor="bench")
            wait_for(lambda: store.try_get("Pod", f"bench-req-{i}") is None,
                     120, "unbind")
            t0 = time.perf_counter()
            mk_requester(f"bench-req-{i+1}")
            wait_for(lambda: requester_ready(agent, f"bench-req-{i+1}"),
                     300, "hot actuation")
            hot.append(time.perf_counter() - t0)

        print(json.dumps({
            "metric": "T_actuation through the full stack (s)",
            "gib": args.gib,
            "gpu": torch.cuda.is_available(),
            "t_cold_s": round(t_cold, 3),
            "t_hot_s": [round(h, 3) for h in hot],
            "t_hot_mean_s": round(sum(hot) / len(hot), 3) if hot else None,
        }))
    finally:
        ctl.stop()
        pop.stop()
        agent.stop()


if __name__ == "__main__":
    main()
