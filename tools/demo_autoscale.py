"""Autoscaling demo: replica count follows a load curve; FMA actuates.

Analog of the reference's test/e2e/demo-fma-hpa scripts (there: a k8s
HPA scales a requester ReplicaSet under load and FMA binds/wakes
instances per replica). Here a minimal autoscaler drives the same FMA
behavior against the in-process single-node stack: replicas follow a
load curve, every scale-up is a real actuation (launcher bind + wake or
cold start), every scale-down is a real unbind+sleep.

Usage:  python tools/demo_autoscale.py [--curve 1,3,4,2,1] [--max 4]
Runs with the fake arena by default (FMA_FAKE_GPU=1, works anywhere);
real-GPU actuation of the same control flow is covered by the e2e suite
(tests/test_e2e_single_node.py on an MI355X box). On a single-GPU box
keep --max 1 — the naive GPU translator assigns one device per replica.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--curve", default="1,3,4,2,1",
                    help="desired replica count per tick")
    ap.add_argument("--max", type=int, default=4)
    ap.add_argument("--log-dir", default="/tmp/fma-autoscale")
    args = ap.parse_args()
    os.environ.setdefault("FMA_FAKE_GPU", "1")
    os.makedirs(args.log_dir, exist_ok=True)

    from fma_amd.api import contracts as C
    from fma_amd.benchmark.harness import LiveClusterOps
    from fma_amd.controller.dualpods.controller import (ControllerConfig,
                                                        DualPodsController)
    from fma_amd.controller.httpadapter import HttpAdapter
    from fma_amd.controller.populator.populator import LauncherPopulator
    from fma_amd.node.agent import NodeAgent
    from fma_amd.store import objects as ob
    from fma_amd.store.admission import install_policies
    from fma_amd.store.memstore import MemStore

    store = MemStore()
    install_policies(store)
    node = ob.new_object("Node", "node-a", labels={"gpu": "mi355x"})
    node["status"] = {"allocatable": {C.GPU_RESOURCE_NAME: 8}}
    store.create(node)
    env = {"PYTHONPATH": os.path.dirname(os.path.dirname(
               os.path.abspath(__file__))),
           "FMA_FAKE_GPU": os.environ.get("FMA_FAKE_GPU", "1"),
           "FMA_MOCK_GPU_COUNT": str(args.max),
           "FMA_GPU_MODE": "naive"}
    agent = NodeAgent(store, "node-a", node_index=23,
                      log_dir=args.log_dir, extra_env=env)
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
            "port": 8390, "options": "--model tiny",
            "labels": {"llm-d.ai/model": "tiny"}},
            "launcherConfigName": "lc1"}))
    store.create(ob.new_object(
        "LauncherPopulationPolicy", "lpp1",
        spec={"enhancedNodeSelector": {
            "labelSelector": {"matchLabels": {"gpu": "mi355x"}}},
            "countForLauncher": [
                {"launcherConfigName": "lc1",
                 "launcherCount": min(2, args.max)}]}))

    ops = LiveClusterOps(store, agent)
    live = {}   # name -> True
    events = []
    counter = 0
    try:
        for tick, want in enumerate(
                int(x) for x in args.curve.split(",")):
            want = min(want, args.max)
            # scale up: each new replica is an actuation we time
            while len(live) < want:
                counter += 1
                name = f"hpa-req-{counter}"
                t0 = time.perf_counter()
                ops.create_requester(name, "isc1")
                # fresh envs for each pod so GPUs differ
                ops.wait_ready(name, timeout=120)
                dt = time.perf_counter() - t0
                live[name] = True
                events.append({"tick": tick, "event": "scale-up",
                               "replica": name,
                               "t_actuation_s": round(dt, 3)})
                print(f"[tick {tick}] +{name} ready in {dt:.2f}s",
                      file=sys.stderr, flush=True)
            # scale down: newest first (HPA-ish)
            while len(live) > want:
                name = sorted(live)[-1]
                ops.delete_requester(name)
                del live[name]
                events.append({"tick": tick, "event": "scale-down",
                               "replica": name})
                print(f"[tick {tick}] -{name}", file=sys.stderr, flush=True)
        ups = [e["t_actuation_s"] for e in events if e["event"] == "scale-up"]
        print(json.dumps({
            "metric": "autoscale actuation latency (s)",
            "curve": args.curve,
            "scale_ups": len(ups),
            "first_s": ups[0] if ups else None,
            "rest_mean_s": round(sum(ups[1:]) / len(ups[1:]), 3)
            if len(ups) > 1 else None,
            "events": events,
        }))
    finally:
        for name in list(live):
            ops.delete_requester(name)
        ctl.stop()
        pop.stop()
        agent.stop()


if __name__ == "__main__":
    main()
