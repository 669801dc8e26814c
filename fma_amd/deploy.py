"""All-in-one single-node deployment.

  python -m fma_amd.deploy [--node-name node-1] [--store-port 8081]

Starts, in one process, everything a node needs (the analog of the
reference's Helm chart + kubelet on one machine):

- the cluster store with admission policies (HTTP on --store-port)
- the dual-pods controller and the launcher-populator
- the node agent (runs launcher / requester / provider Pods as local
  processes with per-Pod loopback IPs)
- Prometheus metrics on --metrics-port

Then drive it with the CLI:
  python -m fma_amd.cli apply -f manifests/example.yaml
  python -m fma_amd.cli get pods
"""

from __future__ import annotations

import argparse
import threading

from fma_amd.controller import metrics
from fma_amd.controller.dualpods.controller import (ControllerConfig,
                                                    DualPodsController)
from fma_amd.controller.httpadapter import HttpAdapter
from fma_amd.controller.populator.populator import LauncherPopulator
from fma_amd.node.agent import NodeAgent
from fma_amd.store import objects as ob
from fma_amd.store.admission import install_policies
from fma_amd.store.memstore import MemStore
from fma_amd.store.server import create_app


def main() -> None:
    import torch
    import uvicorn

    ap = argparse.ArgumentParser("fma-deploy")
    ap.add_argument("--node-name", default="node-1")
    ap.add_argument("--store-port", type=int, default=8081)
    ap.add_argument("--metrics-port", type=int, default=8002)
    ap.add_argument("--debug-port", type=int, default=8003)
    ap.add_argument("--sleeper-limit", type=int, default=1)
    ap.add_argument("--gpus", type=int, default=None,
                    help="GPUs to advertise on the Node (default: detect)")
    args = ap.parse_args()

    store = MemStore()
    install_policies(store)

    ngpus = args.gpus if args.gpus is not None else (
        torch.cuda.device_count() if torch.cuda.is_available() else 0)
    node = ob.new_object("Node", args.node_name,
                         labels={"fma.llm-d.ai/node-type": "mi355x"
                                 if ngpus else "cpu"})
    node["status"] = {"allocatable": {"amd.com/gpu": ngpus}}
    store.create(node)

    agent = NodeAgent(store, args.node_name, node_index=1)
    agent.start()

    # mini-scheduler: in the single-node stack, Pods created without a
    # nodeName are bound to this node (kube-scheduler's role)
    stop = threading.Event()

    def schedule_loop():
        for ev in store.watch(kinds=["Pod"], stop=stop):
            if ev.type in ("ADDED", "MODIFIED"):
                pod = store.try_get("Pod", ev.obj["metadata"]["name"])
                if pod is not None and not pod.get("spec", {}).get("nodeName"):
                    pod.setdefault("spec", {})["nodeName"] = args.node_name
                    try:
                        store.update(pod, actor="system")
                    except Exception:
                        pass

    threading.Thread(target=schedule_loop, daemon=True).start()
    ctl = DualPodsController(
        store, HttpAdapter(observe=metrics.observe_http),
        ControllerConfig(sleeper_limit=args.sleeper_limit))
    ctl.start()
    pop = LauncherPopulator(store)
    pop.start()
    metrics.serve_metrics(args.metrics_port)
    metrics.serve_debug(args.debug_port)

    app = create_app(store)
    print(f"fma-amd single-node stack up: store http://127.0.0.1:"
          f"{args.store_port}, node {args.node_name} ({ngpus} GPUs), "
          f"metrics :{args.metrics_port}")
    try:
        uvicorn.run(app, host="127.0.0.1", port=args.store_port,
                    log_level="warning")
    finally:
        stop.set()
        ctl.stop()
        pop.stop()
        agent.stop()


if __name__ == "__main__":
    main()
