"""Dual-pods reconcile suite re-run over the KUBERNETES wire protocol.

Backend conformance: every reconcile/lifecycle/budget test in
tests/test_dualpods_controller.py and tests/test_dualpods_lifecycle.py
was written against the in-process MemStore; here the same tests run
with the store swapped for a KubeStore talking to the in-tree apiserver
double over HTTP — proving the controller logic depends only on the
store CONTRACT (create/update/delete with preconditions, watch,
indexes), not on MemStore's in-process behavior. This is the conformance
level a real cluster sees.
"""

import threading

import pytest

from fma_amd.store.kubeapiserver import create_app
from fma_amd.store.kubestore import KubeStore
from fma_amd.store.memstore import MemStore

from tests.test_live_servers import ServerThread, free_port

# import the suites; their test functions are re-collected in this module
from tests.test_dualpods_controller import *  # noqa: F401,F403
from tests.test_dualpods_lifecycle import *  # noqa: F401,F403
import tests.test_dualpods_controller as _ctl_mod
import tests.test_dualpods_lifecycle as _life_mod

pytestmark = pytest.mark.timeout(300)

_server = None
_lock = threading.Lock()


class _KubeBackedStore(KubeStore):
    """KubeStore posing as the suite's MemStore factory: each call gets a
    fresh backing MemStore inside the shared apiserver double."""

    def __init__(self):
        global _server
        with _lock:
            if _server is None:
                port = free_port()
                app = create_app(MemStore())
                st = ServerThread(app, port)
                st.__enter__()
                _server = {"thread": st, "port": port, "app": app}
        _reset_backing(_server["app"])
        super().__init__(f"http://127.0.0.1:{_server['port']}",
                         actor="dual-pods-controller")


def _reset_backing(app):
    """Fresh cluster state per test: the double's handlers close over one
    MemStore, so reset it in place (objects, history, indexes)."""
    st = app.state.store
    with st._lock:
        st._objects.clear()
        st._history.clear()
        st._revision = 0
        st._indexes.clear()
        st._indexed_keys.clear()
        st._admission.clear()
    from fma_amd.store.admission import crd_schema_policy
    from fma_amd.store.indexes import install_pod_indexes
    install_pod_indexes(st)
    st.add_admission_hook(crd_schema_policy)


@pytest.fixture(autouse=True)
def _kube_backend(monkeypatch):
    monkeypatch.setattr(_ctl_mod, "MemStore", _KubeBackedStore)
    if hasattr(_life_mod, "MemStore"):
        monkeypatch.setattr(_life_mod, "MemStore", _KubeBackedStore)
    yield
