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
