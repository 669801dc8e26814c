"""Kubernetes backend: wire-protocol, CEL admission, watch semantics.

The FMA stack's deployment story on a real cluster: KubeStore speaks the
Kubernetes REST protocol; here it runs against the in-tree apiserver
double (fma_amd/store/kubeapiserver.py), which evaluates the SHIPPED
ValidatingAdmissionPolicy YAML (manifests/validating-admission-policies/)
through the CEL evaluator — the analog of the reference's kind e2e + CEL
policy checks (reference test/e2e/test-cases.sh:317,
config/validating-admission-policies/*.yaml).
"""

import threading
import time

import httpx
import pytest

from fma_amd.api import contracts as C
from fma_amd.store import objects as ob
from fma_amd.store.kubeapiserver import create_app
from fma_amd.store.kubestore import KubeStore
from fma_amd.store.memstore import (AlreadyExists, Conflict, Invalid,
                                    MemStore, NotFound, RevisionTooOld)

from tests.test_live_servers import ServerThread, free_port

pytestmark = pytest.mark.timeout(120)


@pytest.fixture()
def kube():
    store = MemStore()
    port = free_port()
    with ServerThread(create_app(store), port):
        base = f"http://127.0.0.1:{port}"
        yield {
            "base": base,
            "mem": store,
            "user": KubeStore(base, actor="user"),
            "ctl": KubeStore(base, actor="dual-pods-controller"),
        }


def mk_pod(name, annotations=None, labels=None):
    return ob.new_object("Pod", name, annotations=annotations or {},
                         labels=labels or {},
                         spec={"nodeName": "node-a", "containers": []})


# ---------------------------------------------------------------------------
# wire protocol
# ---------------------------------------------------------------------------


def test_crud_roundtrip_core_and_crd(kube):
    ks = kube["user"]
    pod = ks.create(mk_pod("p1", labels={"a": "b"}))
    assert ob.uid_of(pod) and ob.rv_of(pod)
    assert ks.get("Pod", "p1")["metadata"]["labels"] == {"a": "b"}
    with pytest.raises(AlreadyExists):
        ks.create(mk_pod("p1"))

    isc = ks.create(ob.new_object(
        "InferenceServerConfig", "isc1",
        spec={"modelServerConfig": {"port": 8000},
              "launcherConfigName": "lc1"}))
    assert isc["spec"]["launcherConfigName"] == "lc1"
    # raw path sanity: the CRD lives under the k8s group path
    r = httpx.get(kube["base"] + "/apis/fma.llm-d.ai/v1alpha1/namespaces/"
                  "default/inferenceserverconfigs/isc1")
    assert r.status_code == 200

    node = ks.create(ob.new_object("Node", "node-a"))
    r = httpx.get(kube["base"] + "/api/v1/nodes/node-a")
    assert r.status_code == 200 and ob.uid_of(r.json()) == ob.uid_of(node)

    ks.delete("Pod", "p1")
    with pytest.raises(NotFound):
        ks.get("Pod", "p1")


def test_optimistic_concurrency_conflict(kube):
    ks = kube["user"]
    pod = ks.create(mk_pod("p2"))
    stale_rv = ob.rv_of(pod)
    pod2 = ks.get("Pod", "p2")
    pod2["metadata"]["labels"] = {"x": "1"}
    ks.update(pod2)
    # stale update loses
    pod["metadata"]["labels"] = {"x": "2"}
    with pytest.raises(Conflict):
        ks.update(pod, expect_rv=stale_rv)


def test_delete_preconditions(kube):
    ks = kube["user"]
    pod = ks.create(mk_pod("p3"))
    with pytest.raises(Conflict):
        ks.delete("Pod", "p3", expect_uid="wrong-uid")
    ks.delete("Pod", "p3", expect_uid=ob.uid_of(pod))
    assert ks.try_get("Pod", "p3") is None


def test_finalizer_defers_deletion(kube):
    ks, ctl = kube["user"], kube["ctl"]
    pod = mk_pod("p4")
    pod["metadata"]["finalizers"] = ["dual-pods.llm-d.ai/test"]
    pod = ks.create(pod)
    ks.delete("Pod", "p4")
    cur = ks.get("Pod", "p4")
    assert ob.is_deleting(cur)
    cur["metadata"]["finalizers"] = []
    ctl.update(cur)
    assert ks.try_get("Pod", "p4") is None


def test_label_selector_list(kube):
    ks = kube["user"]
    ks.create(mk_pod("l1", labels={C.COMPONENT_LABEL: C.LAUNCHER_COMPONENT}))
    ks.create(mk_pod("l2", labels={C.COMPONENT_LABEL: "other"}))
    got = ks.list("Pod", label_selector={
        C.COMPONENT_LABEL: C.LAUNCHER_COMPONENT})
    assert [ob.name_of(p) for p in got] == ["l1"]


def test_status_subresource(kube):
    ks = kube["user"]
    pod = ks.create(mk_pod("p5"))
    pod["status"] = {"phase": "Running", "podIP": "10.0.0.9"}
    updated = ks.update(pod, subresource="status")
    assert updated["status"]["podIP"] == "10.0.0.9"
    # generation untouched by status writes
    assert updated["metadata"]["generation"] == \
        pod["metadata"]["generation"]


# ---------------------------------------------------------------------------
# shipped CEL admission policies
# ---------------------------------------------------------------------------


def test_vap_denies_user_mutation_of_fma_metadata(kube):
    """The deny rule comes from the SHIPPED YAML artifact, evaluated by
    the CEL engine — not a parallel Python re-implementation."""
    ks, ctl = kube["user"], kube["ctl"]
    pod = ctl.create(mk_pod(
        "prov", annotations={C.REQUESTER_ANNOTATION: "u1 req1"},
        labels={C.DUAL_LABEL: "req1"}))

    hacked = ob.deepcopy(pod)
    hacked["metadata"]["annotations"][C.REQUESTER_ANNOTATION] = "u2 evil"
    with pytest.raises(Invalid) as ei:
        ks.update(hacked)
    assert "fma-immutable-fields" in str(ei.value)

    # the controller service account may do the same mutation
    cur = ctl.get("Pod", "prov")
    cur["metadata"]["annotations"][C.REQUESTER_ANNOTATION] = "u2 req2"
    ctl.update(cur)

    # non-protected metadata stays freely editable by users
    cur = ks.get("Pod", "prov")
    cur["metadata"]["annotations"]["my-note"] = "hello"
    ks.update(cur)


def test_vap_denies_isc_switch_on_bound_requester(kube):
    ks, ctl = kube["user"], kube["ctl"]
    ks.create(mk_pod(
        "req", annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc-a"}))
    # unbound: switching the ISC is allowed
    cur = ks.get("Pod", "req")
    cur["metadata"]["annotations"][C.INFERENCE_SERVER_CONFIG_ANNOTATION] = \
        "isc-b"
    ks.update(cur)
    # bind it (as the controller would)
    cur = ctl.get("Pod", "req")
    cur["metadata"]["labels"][C.DUAL_LABEL] = "launcher1"
    ctl.update(cur)
    # bound: switching is denied for users
    cur = ks.get("Pod", "req")
    cur["metadata"]["annotations"][C.INFERENCE_SERVER_CONFIG_ANNOTATION] = \
        "isc-c"
    with pytest.raises(Invalid) as ei:
        ks.update(cur)
    assert "fma-bound-serverreqpod" in str(ei.value)


# ---------------------------------------------------------------------------
# watch
# ---------------------------------------------------------------------------


def test_watch_multiplexes_kinds_and_resumes(kube):
    ks = kube["user"]
    token = ks.list_revision()
    got = []
    stop = threading.Event()

    def consume():
        for ev in ks.watch(since=token, stop=stop):
            got.append((ev.type, ev.kind, ob.name_of(ev.obj)))
            if len(got) >= 3:
                return

    th = threading.Thread(target=consume, daemon=True)
    th.start()
    time.sleep(0.3)
    ks.create(mk_pod("w1"))
    ks.create(ob.new_object("LauncherConfig", "lcw",
                            spec={"maxInstances": 1, "podTemplate": {}}))
    ks.delete("Pod", "w1")
    th.join(timeout=20)
    assert not th.is_alive(), f"watch stalled; got {got}"
    kinds = {(t, k, n) for t, k, n in got}
    assert ("ADDED", "Pod", "w1") in kinds
    assert ("ADDED", "LauncherConfig", "lcw") in kinds
    assert ("DELETED", "Pod", "w1") in kinds
    stop.set()


def test_watch_unknown_token_raises_revision_too_old(kube):
    ks = kube["user"]
    with pytest.raises(RevisionTooOld):
        for _ in ks.watch(since=999999):
            pass


def test_watch_only_requested_kinds(kube):
    ks = kube["user"]
    token = ks.list_revision()
    stop = threading.Event()
    got = []

    def consume():
        for ev in ks.watch(since=token, kinds=["Pod"], stop=stop):
            got.append((ev.kind, ob.name_of(ev.obj)))
            if len(got) >= 1:
                return

    th = threading.Thread(target=consume, daemon=True)
    th.start()
    time.sleep(0.3)
    ks.create(ob.new_object("ConfigMap", "cm-ignored"))
    ks.create(mk_pod("w2"))
    th.join(timeout=20)
    assert got == [("Pod", "w2")]
    stop.set()


def test_crd_schema_rejects_duplicate_lpp_keys(kube):
    """CRD structural validation at the apiserver: an LPP with duplicate
    countForLauncher launcherConfigName keys is rejected at create
    (reference test-cases.sh:266-296, the listMapKey semantics)."""
    ks = kube["user"]
    with pytest.raises(Invalid) as ei:
        ks.create(ob.new_object(
            "LauncherPopulationPolicy", "dup",
            spec={"enhancedNodeSelector": {"labelSelector": {}},
                  "countForLauncher": [
                      {"launcherConfigName": "lc1", "launcherCount": 1},
                      {"launcherConfigName": "lc1", "launcherCount": 2}]}))
    assert "unique" in str(ei.value)
    # malformed ISC spec rejected too
    with pytest.raises(Invalid):
        ks.create(ob.new_object(
            "InferenceServerConfig", "bad",
            spec={"modelServerConfig": {"port": "not-a-port"}}))


def test_watch_survives_apiserver_restart():
    """KubeStore's per-resource watch threads reconnect from their cursor
    after the apiserver drops (reflector behavior): events created after
    the restart still arrive on the same watch iterator."""
    store = MemStore()
    port = free_port()
    srv = ServerThread(create_app(store), port)
    srv.__enter__()
    ks = KubeStore(f"http://127.0.0.1:{port}", actor="user")
    token = ks.list_revision()
    got = []
    stop = threading.Event()

    def consume():
        for ev in ks.watch(since=token, kinds=["Pod"], stop=stop):
            got.append(ob.name_of(ev.obj))
            if len(got) >= 2:
                return

    th = threading.Thread(target=consume, daemon=True)
    th.start()
    time.sleep(0.3)
    ks.create(mk_pod("before-restart"))
    deadline = time.time() + 10
    while time.time() < deadline and len(got) < 1:
        time.sleep(0.05)
    assert got == ["before-restart"]

    srv.__exit__()  # apiserver down
    time.sleep(0.5)
    srv2 = ServerThread(create_app(store), port)  # same store, same port
    srv2.__enter__()
    try:
        ks.create(mk_pod("after-restart"))
        th.join(timeout=20)
        assert not th.is_alive(), f"watch did not resume; got {got}"
        assert got == ["before-restart", "after-restart"]
    finally:
        stop.set()
        srv2.__exit__()


# ---------------------------------------------------------------------------
# in-cluster construction (the Helm chart's --backend=in-cluster path)
# ---------------------------------------------------------------------------


def test_in_cluster_store_from_sa_mount(tmp_path, monkeypatch):
    """KubeStore.in_cluster builds from the standard service-account
    mount the way client-go's rest.InClusterConfig does; the bearer token
    rides on every request (reads and watch streams included)."""
    (tmp_path / "token").write_text("sekrit-token\n")
    (tmp_path / "namespace").write_text("prod-ns")
    monkeypatch.setenv("KUBERNETES_SERVICE_HOST", "10.0.0.1")
    monkeypatch.setenv("KUBERNETES_SERVICE_PORT", "6443")
    ks = KubeStore.in_cluster(actor="dual-pods-controller",
                              sa_dir=str(tmp_path))
    assert ks.base == "https://10.0.0.1:6443"
    assert ks._client.headers["Authorization"] == "Bearer sekrit-token"
    # write-path headers carry the token, not the test-double username
    assert ks._hdr(None) == {"Authorization": "Bearer sekrit-token"}
    assert KubeStore.in_cluster_namespace(str(tmp_path)) == "prod-ns"
    # missing mount falls back to default namespace
    assert KubeStore.in_cluster_namespace(str(tmp_path / "nope")) == \
        "default"


def test_controller_entrypoints_backend_flag():
    """Both controller mains accept --backend {store,kube,in-cluster} and
    build the right store (the chart deploys with --backend=in-cluster)."""
    from fma_amd.controller.dualpods.__main__ import make_store

    class A:
        backend = "kube"
        store_url = "http://127.0.0.1:9"
        namespace = "ns1"

    st, ns = make_store(A(), "dual-pods-controller")
    assert isinstance(st, KubeStore) and ns == "ns1"
    A.backend = "store"
    from fma_amd.store.client import StoreClient
    st, ns = make_store(A(), "dual-pods-controller")
    assert isinstance(st, StoreClient)


# ---------------------------------------------------------------------------
# PATCH verbs (merge + strategic-merge content types)
# ---------------------------------------------------------------------------


def test_patch_merge_and_strategic_over_the_wire(kube):
    ks = kube["user"]
    ks.create(mk_pod("pm", annotations={"keep": "1", "drop": "x"}))
    out = ks.patch("Pod", "pm",
                   {"metadata": {"annotations": {"drop": None, "n": "2"}}})
    assert out["metadata"]["annotations"] == {"keep": "1", "n": "2"}
    # strategic: containers merge by name instead of replacing
    ks.patch("Pod", "pm", {"spec": {"containers": [
        {"name": "c1", "image": "a"}]}}, strategic=True)
    out = ks.patch("Pod", "pm", {"spec": {"containers": [
        {"name": "c2", "image": "b"}]}}, strategic=True)
    assert [c["name"] for c in out["spec"]["containers"]] == ["c1", "c2"]
    # CRD group path works too
    ks.create(ob.new_object(
        "InferenceServerConfig", "iscp",
        spec={"modelServerConfig": {"port": 8000},
              "launcherConfigName": "lc1"}))
    out = ks.patch("InferenceServerConfig", "iscp",
                   {"metadata": {"labels": {"x": "y"}}})
    assert out["metadata"]["labels"] == {"x": "y"}


def test_patch_vap_denies_protected_mutation(kube):
    """VAP admission sees the MERGED object: a user patching an
    FMA-managed annotation is denied exactly like an update."""
    ks, ctl = kube["user"], kube["ctl"]
    ctl.create(mk_pod("pv",
                      annotations={C.REQUESTER_ANNOTATION: "u1 req1"}))
    with pytest.raises(Invalid) as ei:
        ks.patch("Pod", "pv", {"metadata": {"annotations": {
            C.REQUESTER_ANNOTATION: "evil"}}})
    assert "fma-immutable-fields" in str(ei.value)
    ctl.patch("Pod", "pv", {"metadata": {"annotations": {
        C.REQUESTER_ANNOTATION: "u2 req2"}}})


def test_patch_unsupported_content_type_415(kube):
    ks = kube["user"]
    ks.create(mk_pod("p415"))
    r = httpx.patch(kube["base"] + "/api/v1/namespaces/default/pods/p415",
                    content=b'[]',
                    headers={"Content-Type": "application/json-patch+json"})
    assert r.status_code == 415


def test_watch_bookmarks_keep_idle_watchers_resumable():
    """k8s watch bookmarks: an idle Pod watcher's cursor advances past
    history churned by OTHER kinds, so after an apiserver drop it
    reconnects cleanly instead of hitting 410 on evicted revisions."""
    store = MemStore()
    store._history_cap = 50  # tiny buffer: churn evicts old revisions
    port = free_port()
    srv = ServerThread(create_app(store), port)
    srv.__enter__()
    ks = KubeStore(f"http://127.0.0.1:{port}", actor="user")
    token = ks.list_revision()
    got, errors = [], []
    stop = threading.Event()

    def consume():
        try:
            for ev in ks.watch(since=token, kinds=["Pod"], stop=stop):
                got.append(ob.name_of(ev.obj))
                if len(got) >= 1:
                    return
        except RevisionTooOld as e:
            errors.append(e)

    th = threading.Thread(target=consume, daemon=True)
    th.start()
    time.sleep(0.3)
    # other-kind churn way past the history cap while the Pod watcher idles
    for i in range(120):
        store.create(ob.new_object("ConfigMap", f"churn-{i}"))
    time.sleep(1.6)  # > the 1 s poll period: a BOOKMARK advances the cursor

    srv.__exit__()  # drop the apiserver mid-watch
    time.sleep(0.4)
    srv2 = ServerThread(create_app(store), port)
    srv2.__enter__()
    try:
        ks.create(mk_pod("after-bookmark"))
        th.join(timeout=20)
        assert not errors, f"watch 410d despite bookmarks: {errors}"
        assert got == ["after-bookmark"], f"got {got}"
    finally:
        stop.set()
        srv2.__exit__()


def test_watch_threads_do_not_leak():
    """Controller+server lifecycles leave no lingering watch threads:
    bounded read timeouts let abandoned per-kind watchers observe their
    stop event, and ServerThread closes the double's watch pool. (The
    leak — ~6 threads per cycle, hundreds over a full suite — was the
    root cause of load-correlated e2e flakes.)"""
    import gc

    from fma_amd.controller.dualpods.controller import (ControllerConfig,
                                                        DualPodsController)
    from fma_amd.controller.httpadapter import HttpAdapter

    before = threading.active_count()
    for i in range(3):
        port = free_port()
        with ServerThread(create_app(MemStore()), port):
            ctl = DualPodsController(
                KubeStore(f"http://127.0.0.1:{port}",
                          actor="dual-pods-controller"),
                HttpAdapter(), ControllerConfig())
            ctl.start()
            time.sleep(0.4)
            ctl.stop()
    gc.collect()
    deadline = time.time() + 20
    while time.time() < deadline:
        if threading.active_count() <= before + 2:
            break
        time.sleep(0.5)
    assert threading.active_count() <= before + 2, \
        [t.name for t in threading.enumerate()]
