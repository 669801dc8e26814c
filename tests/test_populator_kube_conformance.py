"""Launcher-populator suite re-run over the Kubernetes wire protocol
(same conformance seam as tests/test_dualpods_kube_conformance.py)."""

import pytest

from tests.test_dualpods_kube_conformance import _KubeBackedStore
from tests.test_populator import *  # noqa: F401,F403
import tests.test_populator as _pop_mod

pytestmark = pytest.mark.timeout(300)


@pytest.fixture(autouse=True)
def _kube_backend(monkeypatch):
    monkeypatch.setattr(_pop_mod, "MemStore", _KubeBackedStore)
    yield
