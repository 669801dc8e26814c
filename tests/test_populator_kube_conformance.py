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


# Over the kube backend the apiserver-level CRD schema admission rejects
# the duplicate-key LPP at create (the reference behavior,
# test-cases.sh:266-296) — the populator-internal defense that test
# exercises is covered on the MemStore path, and the rejection itself in
# test_kube_backend/test_store_and_admission.
test_lpp_status_reports_errors = pytest.mark.skip(
    reason="kube backend rejects the bad LPP at create (CRD admission)")(
        _pop_mod.test_lpp_status_reports_errors)
