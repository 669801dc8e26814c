"""CLI apply/get/delete against the HTTP store."""

import threading

from fastapi.testclient import TestClient

from fma_amd import cli
from fma_amd.store.memstore import MemStore
from fma_amd.store.server import create_app


class TCStoreClient(cli.StoreClient):
    """StoreClient routed through a FastAPI TestClient (no sockets)."""

    def __init__(self, app):
        self.base = ""
        self.actor = "user"
        self._client = TestClient(app, headers={"X-FMA-Actor": "user"})


def test_apply_get_delete(tmp_path, capsys):
    store = MemStore()
    app = create_app(store)
    client = TCStoreClient(app)
    cli.cmd_apply(client, ["manifests/example.yaml"])
    out = capsys.readouterr().out
    assert "LauncherConfig/lc-mi355x created" in out
    assert "Pod/my-model-request created" in out
    # idempotent re-apply configures
    cli.cmd_apply(client, ["manifests/example.yaml"])
    assert "configured" in capsys.readouterr().out
    cli.cmd_get(client, "isc", "", "table")
    assert "isc-tiny" in capsys.readouterr().out
    cli.cmd_get(client, "pod", "my-model-request", "json")
    assert "my-model-request" in capsys.readouterr().out
    client.delete("Pod", "my-model-request")
    assert store.try_get("Pod", "my-model-request") is None


def test_kubernetes_manifests_parse_and_cover_rbac():
    """manifests/kubernetes/controllers.yaml (the Helm-chart analog,
    reference charts/fma-controllers) must stay parseable and keep the
    RBAC surface the controllers need."""
    import os

    import yaml
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    docs = [d for d in yaml.safe_load_all(
        open(os.path.join(root, "manifests/kubernetes/controllers.yaml")))
        if d]
    kinds = sorted(d["kind"] for d in docs)
    assert kinds == ["ClusterRole", "ClusterRoleBinding", "Deployment",
                     "Deployment", "Role", "RoleBinding", "ServiceAccount"]
    role = next(d for d in docs if d["kind"] == "Role")
    pod_rule = next(r for r in role["rules"]
                    if "pods" in r.get("resources", []))
    assert set(pod_rule["verbs"]) >= {"create", "delete", "patch", "watch"}
    crd_rule = next(r for r in role["rules"]
                    if "inferenceserverconfigs" in r.get("resources", []))
    assert "watch" in crd_rule["verbs"]


def test_cli_against_kube_apiserver(tmp_path, capsys):
    """fma --kube drives a Kubernetes apiserver (the double here) through
    the k8s wire protocol: apply / get / delete round-trip."""
    from fma_amd.cli import main as cli_main
    from fma_amd.store.kubeapiserver import create_app
    from fma_amd.store.memstore import MemStore

    from tests.test_live_servers import ServerThread, free_port

    port = free_port()
    with ServerThread(create_app(MemStore()), port):
        base = f"http://127.0.0.1:{port}"
        f = tmp_path / "isc.yaml"
        f.write_text(
            "kind: InferenceServerConfig\n"
            "metadata: {name: cli-isc}\n"
            "spec:\n"
            "  modelServerConfig: {port: 8000, options: '--model tiny'}\n"
            "  launcherConfigName: lc1\n")
        cli_main(["--kube", base, "apply", "-f", str(f)])
        cli_main(["--kube", base, "get", "isc", "cli-isc", "-o", "json"])
        out = capsys.readouterr().out
        assert "cli-isc" in out and "launcherConfigName" in out
        cli_main(["--kube", base, "delete", "isc", "cli-isc"])
        out = capsys.readouterr().out
        assert "deleted" in out
