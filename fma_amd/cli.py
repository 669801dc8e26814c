"""fma CLI: kubectl-style client for the cluster store.

  python -m fma_amd.cli apply -f manifests/example.yaml
  python -m fma_amd.cli get isc [NAME]
  python -m fma_amd.cli delete pod NAME
  python -m fma_amd.cli watch

Kinds accept the reference's short names (isc/lcfg/lpp —
reference CRD shortNames) plus pod/node/configmap.
"""

from __future__ import annotations

import argparse
import json
import sys
from typing import List

import yaml

from fma_amd.store.client import StoreClient

KIND_ALIASES = {
    "isc": "InferenceServerConfig",
    "inferenceserverconfig": "InferenceServerConfig",
    "lcfg": "LauncherConfig",
    "launcherconfig": "LauncherConfig",
    "lpp": "LauncherPopulationPolicy",
    "launcherpopulationpolicy": "LauncherPopulationPolicy",
    "pod": "Pod", "pods": "Pod",
    "node": "Node", "nodes": "Node",
    "configmap": "ConfigMap", "cm": "ConfigMap",
}


def resolve_kind(k: str) -> str:
    return KIND_ALIASES.get(k.lower(), k)


def cmd_apply(client: StoreClient, files: List[str]) -> None:
    from fma_amd.store.memstore import AlreadyExists, Conflict
    for path in files:
        with open(path) as f:
            docs = list(yaml.safe_load_all(f))
        for doc in docs:
            if not doc:
                continue
            doc["kind"] = resolve_kind(doc.get("kind", ""))
            name = doc.get("metadata", {}).get("name", "?")
            try:
                client.create(doc)
                print(f"{doc['kind']}/{name} created")
            except (AlreadyExists, Conflict):
                cur = client.get(doc["kind"], name,
                                 doc.get("metadata", {}).get("namespace",
                                                             "default"))
                cur["spec"] = doc.get("spec", cur.get("spec"))
                for key in ("labels", "annotations"):
                    if key in doc.get("metadata", {}):
                        cur["metadata"][key] = doc["metadata"][key]
                client.update(cur)
                print(f"{doc['kind']}/{name} configured")


def cmd_get(client: StoreClient, kind: str, name: str, output: str) -> None:
    kind = resolve_kind(kind)
    if name:
        objs = [client.get(kind, name)]
    else:
        objs = client.list(kind)
    if output == "json":
        print(json.dumps(objs, indent=2, default=str))
        return
    if output == "yaml":
        print(yaml.safe_dump_all(objs))
        return
    print(f"{'NAME':40} {'KIND':26} {'LABELS'}")
    for o in objs:
        meta = o.get("metadata", {})
        labels = ",".join(f"{k}={v}"
                          for k, v in (meta.get("labels") or {}).items())
        print(f"{meta.get('name', ''):40} {o.get('kind', ''):26} {labels}")


def main(argv=None) -> None:
    ap = argparse.ArgumentParser("fma")
    ap.add_argument("--store-url", default="http://127.0.0.1:8081")
    ap.add_argument("--kube", default="",
                    help="Kubernetes apiserver base URL: talk the k8s "
                         "wire protocol (KubeStore) instead of the fma "
                         "store server")
    ap.add_argument("--actor", default="user")
    sub = ap.add_subparsers(dest="cmd", required=True)
    p_apply = sub.add_parser("apply")
    p_apply.add_argument("-f", "--filename", action="append", required=True)
    p_get = sub.add_parser("get")
    p_get.add_argument("kind")
    p_get.add_argument("name", nargs="?", default="")
    p_get.add_argument("-o", "--output", default="table",
                       choices=["table", "json", "yaml"])
    p_del = sub.add_parser("delete")
    p_del.add_argument("kind")
    p_del.add_argument("name")
    sub.add_parser("watch")
    args = ap.parse_args(argv)

    if args.kube:
        from fma_amd.store.kubestore import KubeStore
        client = KubeStore(args.kube, actor=args.actor)
    else:
        client = StoreClient(args.store_url, actor=args.actor)
    if args.cmd == "apply":
        cmd_apply(client, args.filename)
    elif args.cmd == "get":
        cmd_get(client, args.kind, args.name, args.output)
    elif args.cmd == "delete":
        client.delete(resolve_kind(args.kind), args.name)
        print(f"{args.kind}/{args.name} deleted")
    elif args.cmd == "watch":
        since = client.list_revision() if args.kube else 0
        for ev in client.watch(since=since):
            meta = ev.obj.get("metadata", {})
            print(f"{ev.revision}\t{ev.type}\t{ev.kind}/{meta.get('name')}")
            sys.stdout.flush()


if __name__ == "__main__":
    main()
