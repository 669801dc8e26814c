"""Differential fuzz: MemStore+python-hooks vs KubeStore+apiserver double.

The in-process admission hooks (fma_amd/store/admission.py) claim to
enforce the same rules as the SHIPPED CEL ValidatingAdmissionPolicy YAML
that the kube apiserver double evaluates (reference
config/validating-admission-policies/*). The conformance suites prove the
controllers behave the same over both backends; this test proves the
STORES themselves agree: deterministic random CRUD sequences — biased
toward the protected annotation/label keys, stale-RV updates and invalid
CRD specs — are applied to both backends in lockstep, asserting the same
accept/deny outcome per op and identical final object state.
"""

import random

import pytest

from fma_amd.api import contracts as C
from fma_amd.store import objects as ob
from fma_amd.store.admission import (PROTECTED_ANNOTATIONS, PROTECTED_LABELS,
                                     install_policies)
from fma_amd.store.indexes import install_pod_indexes
from fma_amd.store.kubeapiserver import create_app
from fma_amd.store.kubestore import KubeStore
from fma_amd.store.memstore import (AlreadyExists, ApiError, Conflict,
                                    Invalid, MemStore, NotFound)

from tests.test_live_servers import ServerThread, free_port

pytestmark = pytest.mark.timeout(240)

NAMES = ["alpha", "beta", "gamma"]
ACTORS = ["user", "dual-pods-controller"]
MUTABLE_KEYS = ["my-note", "team"]
ALL_ANN = list(PROTECTED_ANNOTATIONS) + MUTABLE_KEYS
ALL_LBL = list(PROTECTED_LABELS) + MUTABLE_KEYS


def _mk_pod(rng, name):
    ann = {}
    if rng.random() < 0.5:
        ann[C.INFERENCE_SERVER_CONFIG_ANNOTATION] = "isc-x"
    return ob.new_object("Pod", name, annotations=ann,
                         spec={"nodeName": "node-a", "containers": []})


def _mk_crd(rng, name):
    if rng.random() < 0.5:
        # LPP; sometimes with duplicate countForLauncher keys (invalid)
        lc = [{"launcherConfigName": "lc1", "launcherCount": 1}]
        if rng.random() < 0.3:
            lc.append({"launcherConfigName": "lc1", "launcherCount": 2})
        return ob.new_object(
            "LauncherPopulationPolicy", name,
            spec={"enhancedNodeSelector": {"labelSelector": {}},
                  "countForLauncher": lc})
    # ISC; sometimes with a malformed port (invalid)
    port = 8000 if rng.random() < 0.7 else "not-a-port"
    return ob.new_object(
        "InferenceServerConfig", name,
        spec={"modelServerConfig": {"port": port},
              "launcherConfigName": "lc1"})


class _Step:
    """One randomized operation, applied identically to both stores."""

    def __init__(self, rng):
        self.kind_pod = rng.random() < 0.7
        self.name = rng.choice(NAMES)
        self.actor = rng.choice(ACTORS)
        self.op = rng.choices(
            ["create", "update", "stale_update", "delete", "bad_delete",
             "status", "finalize", "patch"],
            weights=[26, 28, 8, 13, 4, 7, 7, 7])[0]
        self.rng_state = rng.getstate()

    def run(self, store, is_kube):
        rng = random.Random()
        rng.setstate(self.rng_state)
        kind = "Pod" if self.kind_pod else None
        name = self.name

        def call(fn, *a, **kw):
            if is_kube:
                return fn(*a, actor=self.actor, **kw)
            return fn(*a, actor=self.actor, **kw)

        if self.op == "create":
            obj = _mk_pod(rng, name) if self.kind_pod else _mk_crd(rng, name)
            if rng.random() < 0.3:
                obj["metadata"]["finalizers"] = ["dual-pods.llm-d.ai/test"]
            self._kind = obj["kind"]
            return call(store.create, obj)
        # the remaining ops need an existing object of SOME kind
        target = None
        for k in (["Pod"] if self.kind_pod
                  else ["InferenceServerConfig", "LauncherPopulationPolicy"]):
            target = store.try_get(k, name)
            if target is not None:
                kind = k
                break
        if target is None:
            raise NotFound(f"{name} absent")
        if self.op in ("update", "stale_update"):
            cur = ob.deepcopy(target)
            meta = cur["metadata"]
            # mutate one random annotation and/or label
            if rng.random() < 0.7:
                key = rng.choice(ALL_ANN)
                meta.setdefault("annotations", {})[key] = \
                    f"v{rng.randrange(3)}"
            if rng.random() < 0.5:
                key = rng.choice(ALL_LBL)
                meta.setdefault("labels", {})[key] = f"v{rng.randrange(3)}"
            if self.op == "stale_update":
                meta["resourceVersion"] = "1"
            return call(store.update, cur)
        if self.op == "patch":
            key = rng.choice(ALL_ANN)
            val = None if rng.random() < 0.3 else f"v{rng.randrange(3)}"
            body = {"metadata": {"annotations": {key: val}}}
            return call(store.patch, kind, name, body,
                        strategic=rng.random() < 0.5)
        if self.op == "status":
            cur = ob.deepcopy(target)
            cur["status"] = {"phase": rng.choice(["Running", "Pending"]),
                             "marker": rng.randrange(3)}
            return call(store.update, cur, subresource="status")
        if self.op == "finalize":
            # drop finalizers; on a deleting object this completes removal
            cur = ob.deepcopy(target)
            cur["metadata"]["finalizers"] = []
            return call(store.update, cur)
        if self.op == "delete":
            return call(store.delete, kind, name)
        return call(store.delete, kind, name, expect_uid="wrong-uid")


def _strip(obj):
    o = ob.deepcopy(obj)
    m = o.get("metadata", {})
    for k in ("uid", "creationTimestamp", "managedFields"):
        m.pop(k, None)
    if m.get("deletionTimestamp"):
        m["deletionTimestamp"] = "<set>"  # wall-clock; compare presence
    return o


def _snapshot(store):
    snap = {}
    for kind in ("Pod", "InferenceServerConfig", "LauncherPopulationPolicy"):
        for o in store.list(kind):
            snap[(kind, ob.name_of(o))] = _strip(o)
    return snap


@pytest.fixture(scope="module")
def double():
    backing = MemStore()
    port = free_port()
    with ServerThread(create_app(backing), port):
        yield {"base": f"http://127.0.0.1:{port}", "backing": backing}


def _reset(backing):
    from fma_amd.store.admission import crd_schema_policy
    with backing._lock:
        backing._objects.clear()
        backing._history.clear()
        backing._revision = 0
        backing._indexes.clear()
        backing._indexed_keys.clear()
        backing._admission.clear()
    install_pod_indexes(backing)
    backing.add_admission_hook(crd_schema_policy)


@pytest.mark.parametrize("seed", range(12))
def test_memstore_and_kube_double_agree(double, seed):
    _reset(double["backing"])
    mem = MemStore()
    install_pod_indexes(mem)
    install_policies(mem)
    kube = KubeStore(double["base"])

    rng = random.Random(seed)
    divergences = []
    for step_no in range(40):
        step = _Step(rng)
        outcomes = []
        for store, is_kube in ((mem, False), (kube, True)):
            try:
                step.run(store, is_kube)
                outcomes.append("ok")
            except (Invalid, Conflict, NotFound, AlreadyExists) as e:
                outcomes.append(type(e).__name__)
            except ApiError as e:  # pragma: no cover - unexpected class
                outcomes.append(f"ApiError{e.code}")
        if outcomes[0] != outcomes[1]:
            divergences.append((step_no, step.op, step.actor, outcomes))
    assert not divergences, divergences

    mem_snap, kube_snap = _snapshot(mem), _snapshot(kube)
    assert set(mem_snap) == set(kube_snap)
    for key in mem_snap:
        assert mem_snap[key] == kube_snap[key], key

    # the two backends must also have EMITTED the same watch-event
    # sequence (type, kind, name) — the double is a thin wire layer, so
    # its backing history is directly comparable
    backing = double["backing"]
    mem_events = [(e.type, e.kind, ob.name_of(e.obj)) for e in mem._history]
    kube_events = [(e.type, e.kind, ob.name_of(e.obj))
                   for e in backing._history]
    assert mem_events == kube_events


def test_index_get_parity_between_backends(double):
    """Controllers do reconcile lookups through index_get; every standard
    Pod index must return identical result sets from MemStore (true
    store-side indexes) and KubeStore (client-side evaluation over a
    LIST) for the same cluster state."""
    from fma_amd.store.indexes import POD_INDEXES

    _reset(double["backing"])
    mem = MemStore()
    install_pod_indexes(mem)
    kube = KubeStore(double["base"])

    rng = random.Random(99)
    for i in range(25):
        ann, lbl = {}, {}
        if rng.random() < 0.5:
            ann[C.REQUESTER_ANNOTATION] = f"u{rng.randrange(3)} req{i}"
        if rng.random() < 0.5:
            ann[C.INFERENCE_SERVER_CONFIG_ANNOTATION] = \
                f"isc-{rng.randrange(3)}"
        if rng.random() < 0.4:
            ann[C.ACCELERATORS_ANNOTATION] = ",".join(
                f"GPU-{rng.randrange(4)}" for _ in range(rng.randrange(1, 3)))
        if rng.random() < 0.4:
            ann[C.NOMINAL_ANNOTATION] = f"h{rng.randrange(3)}"
        if rng.random() < 0.5:
            lbl[C.DUAL_LABEL] = f"l{rng.randrange(3)}"
        if rng.random() < 0.4:
            lbl[C.LAUNCHER_CONFIG_NAME_LABEL] = f"lc{rng.randrange(2)}"
        pod = ob.new_object(
            "Pod", f"ix-{i}", annotations=ann, labels=lbl,
            spec={"nodeName": f"node-{rng.randrange(3)}",
                  "containers": []})
        mem.create(ob.deepcopy(pod))
        kube.create(pod)

    keys = (["u0 req1", "isc-0", "isc-1", "GPU-0", "GPU-3", "h0", "l1",
             "lc0", "node-0", "node-2", "absent"])
    hits = 0
    for index_name in POD_INDEXES:
        for key in keys:
            m = sorted(ob.name_of(o)
                       for o in mem.index_get("Pod", index_name, key))
            k = sorted(ob.name_of(o)
                       for o in kube.index_get("Pod", index_name, key))
            assert m == k, (index_name, key, m, k)
            hits += len(m)
    assert hits > 10, f"parity checked mostly-empty sets (hits={hits})"
