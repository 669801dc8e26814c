"""Store semantics (preconditions, finalizers, watch) + admission policies."""

import threading

import pytest

from fma_amd.api import contracts as C
from fma_amd.store import objects as ob
from fma_amd.store.admission import install_policies
from fma_amd.store.memstore import (AlreadyExists, Conflict, Invalid,
                                    MemStore, NotFound)


def test_create_get_update_delete():
    s = MemStore()
    pod = s.create(ob.new_object("Pod", "p1"))
    assert ob.uid_of(pod)
    assert ob.rv_of(pod)
    with pytest.raises(AlreadyExists):
        s.create(ob.new_object("Pod", "p1"))
    pod["spec"]["x"] = 1
    pod2 = s.update(pod)
    assert ob.meta(pod2)["generation"] == 2  # spec change bumps generation
    pod2["status"] = {"phase": "Running"}
    pod3 = s.update(pod2)
    assert ob.meta(pod3)["generation"] == 2  # status change does not
    s.delete("Pod", "p1")
    with pytest.raises(NotFound):
        s.get("Pod", "p1")


def test_resource_version_conflict():
    s = MemStore()
    pod = s.create(ob.new_object("Pod", "p"))
    stale = ob.deepcopy(pod)
    s.update(pod)  # bumps RV
    stale["spec"]["y"] = 2
    with pytest.raises(Conflict):
        s.update(stale)


def test_preconditioned_delete():
    s = MemStore()
    pod = s.create(ob.new_object("Pod", "p"))
    with pytest.raises(Conflict):
        s.delete("Pod", "p", expect_uid="wrong")
    s.delete("Pod", "p", expect_uid=ob.uid_of(pod))


def test_finalizer_gated_deletion():
    s = MemStore()
    pod = ob.new_object("Pod", "p")
    pod["metadata"]["finalizers"] = ["x/protect"]
    pod = s.create(pod)
    s.delete("Pod", "p")
    cur = s.get("Pod", "p")  # still there, deleting
    assert ob.is_deleting(cur)
    cur["metadata"]["finalizers"] = []
    s.update(cur)
    with pytest.raises(NotFound):
        s.get("Pod", "p")


def test_owner_reference_gc():
    s = MemStore()
    owner = s.create(ob.new_object("LauncherConfig", "lc",
                                   spec={"maxInstances": 1}))
    child = ob.new_object("Pod", "child")
    ob.meta(child)["ownerReferences"] = [
        {"kind": "LauncherConfig", "name": "lc", "uid": ob.uid_of(owner)}]
    s.create(child)
    s.delete("LauncherConfig", "lc")
    assert s.try_get("Pod", "child") is None


def test_watch_delivers_events():
    s = MemStore()
    stop = threading.Event()
    got = []

    def consume():
        for ev in s.watch(stop=stop):
            got.append((ev.type, ob.name_of(ev.obj)))
            if len(got) >= 3:
                return

    t = threading.Thread(target=consume, daemon=True)
    t.start()
    pod = s.create(ob.new_object("Pod", "w1"))
    pod["spec"]["z"] = 1
    s.update(pod)
    s.delete("Pod", "w1")
    t.join(timeout=5)
    stop.set()
    assert got == [("ADDED", "w1"), ("MODIFIED", "w1"), ("DELETED", "w1")]


def test_watch_replays_history():
    s = MemStore()
    s.create(ob.new_object("Pod", "a"))
    s.create(ob.new_object("Pod", "b"))
    evs = list(s.watch(since=0, timeout=0.05))
    assert [ob.name_of(e.obj) for e in evs] == ["a", "b"]


# -- admission ------------------------------------------------------------

def bound_requester(s):
    pod = ob.new_object(
        "Pod", "req",
        annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"},
        labels={C.DUAL_LABEL: "launcher1"})
    return s.create(pod, actor="user")


def test_policy_blocks_user_touching_managed_metadata():
    s = MemStore()
    install_policies(s)
    pod = s.create(ob.new_object("Pod", "p"), actor="user")
    pod = s.get("Pod", "p")
    ob.annotations_of(pod)[C.REQUESTER_ANNOTATION] = "evil"
    with pytest.raises(Invalid):
        s.update(pod, actor="user")
    # the controller may
    s.update(pod, actor="dual-pods-controller")


def test_policy_blocks_isc_change_on_bound_requester():
    s = MemStore()
    install_policies(s)
    pod = bound_requester(s)
    pod = s.get("Pod", "req")
    ob.annotations_of(pod)[C.INFERENCE_SERVER_CONFIG_ANNOTATION] = "other"
    with pytest.raises(Invalid):
        s.update(pod, actor="user")


def test_policy_allows_isc_change_on_unbound_requester():
    s = MemStore()
    install_policies(s)
    pod = ob.new_object(
        "Pod", "req",
        annotations={C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"})
    s.create(pod, actor="user")
    pod = s.get("Pod", "req")
    ob.annotations_of(pod)[C.INFERENCE_SERVER_CONFIG_ANNOTATION] = "other"
    s.update(pod, actor="user")  # unbound: allowed


def test_store_concurrent_crud_stress():
    """Many threads mutating + watching: no lost updates, no deadlocks."""
    import threading

    s = MemStore()
    base = s.create(ob.new_object("ConfigMap", "counter"))
    base["data"] = {"n": "0"}
    s.update(base)
    increments_per_thread = 50
    nthreads = 8

    def worker(tid):
        for i in range(increments_per_thread):
            while True:
                cur = s.get("ConfigMap", "counter")
                cur["data"]["n"] = str(int(cur["data"]["n"]) + 1)
                try:
                    s.update(cur)
                    break
                except Conflict:
                    continue
            s.create(ob.new_object("Pod", f"p-{tid}-{i}"))

    threads = [threading.Thread(target=worker, args=(t,))
               for t in range(nthreads)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=60)
    cur = s.get("ConfigMap", "counter")
    assert int(cur["data"]["n"]) == nthreads * increments_per_thread
    assert len(s.list("Pod")) == nthreads * increments_per_thread
    # watch history is strictly increasing
    revs = [e.revision for e in s.watch(since=0, timeout=0.05)]
    assert revs == sorted(revs) and len(revs) == len(set(revs))


# -- property fuzz: the CEL-policy analogs hold under arbitrary attack ------

from hypothesis import given, settings, strategies as st  # noqa: E402

from fma_amd.store.admission import (FMA_ACTORS, PROTECTED_ANNOTATIONS,  # noqa: E402
                                     PROTECTED_LABELS)

_texts = st.text(
    alphabet=st.characters(whitelist_categories=("Ll", "Lu", "Nd")),
    min_size=1, max_size=12)


@settings(max_examples=150, deadline=None)
@given(key=st.sampled_from(PROTECTED_ANNOTATIONS + PROTECTED_LABELS),
       value=_texts,
       actor=st.one_of(_texts, st.sampled_from(sorted(FMA_ACTORS))))
def test_protected_fields_fuzz(key, value, actor):
    """Reference CEL policy (fma-immutable-fields.yaml): any change to a
    protected annotation/label by a non-FMA actor is rejected; FMA
    service accounts always pass (property-tested over random keys,
    values and actor names)."""
    import copy

    from fma_amd.store import objects as ob
    from fma_amd.store.admission import install_policies
    from fma_amd.store.memstore import Invalid, MemStore

    store = MemStore()
    install_policies(store)
    pod = store.create(ob.new_object("Pod", "t1", spec={}),
                       actor="dual-pods-controller")
    mutated = copy.deepcopy(pod)
    if key in PROTECTED_LABELS:
        ob.labels_of(mutated)[key] = value
    else:
        ob.annotations_of(mutated)[key] = value
    if actor in FMA_ACTORS:
        store.update(mutated, actor=actor)  # must not raise
    else:
        try:
            store.update(mutated, actor=actor)
            raise AssertionError(
                f"{actor!r} mutated protected {key!r} unchallenged")
        except Invalid:
            pass


@settings(max_examples=200, deadline=None)
@given(n=st.integers(0, 10**12),
       suffix=st.sampled_from([None, "m", "k", "M", "G", "Ki", "Mi", "Gi",
                               "Ti"]))
def test_parse_quantity_properties(n, suffix):
    """k8s quantity parsing (reference resource.Quantity subset): value
    scales by the suffix factor, ordering is preserved, garbage raises."""
    import pytest

    from fma_amd.api.types import parse_quantity

    factors = {None: 1, "m": 1e-3, "k": 1e3, "M": 1e6, "G": 1e9,
               "Ki": 2**10, "Mi": 2**20, "Gi": 2**30, "Ti": 2**40}
    text = f"{n}{suffix or ''}"
    v = parse_quantity(text)
    assert v == n * factors[suffix]
    # ordering across representations
    assert parse_quantity(f"{n + 1}{suffix or ''}") > v or n * factors[
        suffix] == (n + 1) * factors[suffix]
    # ints/floats pass through
    assert parse_quantity(n) == float(n)
    with pytest.raises(ValueError):
        parse_quantity(f"{n}Zz")


def test_memstore_concurrent_updates_one_winner_per_rv():
    """Optimistic concurrency under real thread contention: N threads
    race to update the same object with the SAME expected RV — exactly
    one wins per round, everyone else gets Conflict (the apiserver
    semantic every reconcile loop here leans on)."""
    import threading

    from fma_amd.store import objects as ob
    from fma_amd.store.memstore import Conflict, MemStore

    store = MemStore()
    store.create(ob.new_object("Pod", "contended", spec={}))
    rounds = 20
    threads = 8
    wins = []
    for _ in range(rounds):
        cur = store.get("Pod", "contended")
        rv = ob.rv_of(cur)
        barrier = threading.Barrier(threads)
        results = [None] * threads

        def attempt(i):
            import copy
            mine = copy.deepcopy(cur)
            ob.annotations_of(mine)["winner"] = f"t{i}"
            barrier.wait()
            try:
                store.update(mine, expect_rv=rv)
                results[i] = "win"
            except Conflict:
                results[i] = "lose"

        ts = [threading.Thread(target=attempt, args=(i,))
              for i in range(threads)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        assert results.count("win") == 1, results
        wins.append(results.index("win"))
    # the final object reflects the last winner exactly
    final = store.get("Pod", "contended")
    assert ob.annotations_of(final)["winner"] == f"t{wins[-1]}"


def test_watch_raises_revision_too_old_on_history_eviction():
    """A watch cursor that falls behind the trimmed event history raises
    RevisionTooOld (410) so the consumer re-LISTs, instead of silently
    skipping evicted events (ADVICE round-1, memstore.watch)."""
    from fma_amd.store.memstore import MemStore, RevisionTooOld

    st = MemStore()
    st._history_cap = 50
    first = st.create({"kind": "Pod", "metadata": {"name": "p0"}})
    start_rev = st.list_revision()
    for i in range(100):
        st.create({"kind": "Pod", "metadata": {"name": f"p{i+1}"}})
    # history was trimmed past start_rev
    with pytest.raises(RevisionTooOld):
        for _ in st.watch(since=start_rev, timeout=0.1):
            pass
    del first


def test_watch_current_cursor_survives_eviction():
    """A caught-up watcher is NOT disturbed by eviction of events it
    already consumed."""
    from fma_amd.store.memstore import MemStore

    st = MemStore()
    st._history_cap = 50
    for i in range(100):
        st.create({"kind": "Pod", "metadata": {"name": f"q{i}"}})
    cursor = st.list_revision()
    st.create({"kind": "Pod", "metadata": {"name": "fresh"}})
    evs = list(st.watch(since=cursor, timeout=0.1))
    assert [e.obj["metadata"]["name"] for e in evs] == ["fresh"]


def test_crd_schema_admission_on_memstore():
    """install_policies enforces CRD structural validation in-process
    (duplicate LPP keys, malformed specs) — same rule as the kube
    double's apiserver-level check."""
    from fma_amd.store import objects as ob
    from fma_amd.store.admission import install_policies
    from fma_amd.store.memstore import Invalid, MemStore

    st = MemStore()
    install_policies(st)
    with pytest.raises(Invalid):
        st.create(ob.new_object(
            "LauncherPopulationPolicy", "dup",
            spec={"enhancedNodeSelector": {"labelSelector": {}},
                  "countForLauncher": [
                      {"launcherConfigName": "lc1", "launcherCount": 1},
                      {"launcherConfigName": "lc1", "launcherCount": 2}]}))
    # well-formed objects pass
    st.create(ob.new_object(
        "LauncherPopulationPolicy", "ok",
        spec={"enhancedNodeSelector": {"labelSelector": {}},
              "countForLauncher": [
                  {"launcherConfigName": "lc1", "launcherCount": 1}]}))


# ---------------------------------------------------------------------------
# server-side PATCH (merge + strategic merge)
# ---------------------------------------------------------------------------


def test_patch_merge_semantics():
    st = MemStore()
    st.create(ob.new_object("Pod", "pp", annotations={"keep": "1",
                                                      "drop": "x"},
                            spec={"nodeName": "n", "containers": [
                                {"name": "c1", "image": "a"}]}))
    out = st.patch("Pod", "pp", {
        "metadata": {"annotations": {"drop": None, "new": "2"}}})
    anns = out["metadata"]["annotations"]
    assert anns == {"keep": "1", "new": "2"}
    # RFC 7386: lists replace under plain merge
    out = st.patch("Pod", "pp", {"spec": {"containers": [
        {"name": "c2", "image": "b"}]}})
    assert [c["name"] for c in out["spec"]["containers"]] == ["c2"]


def test_patch_strategic_merges_named_lists():
    st = MemStore()
    st.create(ob.new_object("Pod", "ps", spec={"containers": [
        {"name": "c1", "image": "a", "env": [
            {"name": "E1", "value": "1"}]}]}))
    out = st.patch("Pod", "ps", {"spec": {"containers": [
        {"name": "c1", "env": [{"name": "E2", "value": "2"}]},
        {"name": "c2", "image": "b"}]}}, strategic=True)
    cs = out["spec"]["containers"]
    assert [c["name"] for c in cs] == ["c1", "c2"]
    assert cs[0]["image"] == "a"  # untouched field survives
    assert [e["name"] for e in cs[0]["env"]] == ["E1", "E2"]


def test_patch_runs_admission_on_merged_result():
    st = MemStore()
    install_policies(st)
    st.create(ob.new_object(
        "Pod", "pa",
        annotations={C.REQUESTER_ANNOTATION: "u p"}),
        actor="dual-pods-controller")
    with pytest.raises(Invalid):
        st.patch("Pod", "pa", {"metadata": {"annotations": {
            C.REQUESTER_ANNOTATION: "evil"}}}, actor="user")
    # FMA actor may
    st.patch("Pod", "pa", {"metadata": {"annotations": {
        C.REQUESTER_ANNOTATION: "u2 p2"}}},
        actor="dual-pods-controller")


def test_concurrent_patches_do_not_lose_updates():
    """Two writers patching DIFFERENT annotation keys concurrently: with
    read-modify-write one side's updates would be lost on conflict; with
    server-side patch every write lands."""
    import threading as th
    st = MemStore()
    st.create(ob.new_object("Pod", "pc"))
    n = 30

    def writer(key):
        for i in range(n):
            st.patch("Pod", "pc",
                     {"metadata": {"annotations": {key: str(i)}}})

    t1 = th.Thread(target=writer, args=("a",))
    t2 = th.Thread(target=writer, args=("b",))
    t1.start(); t2.start(); t1.join(); t2.join()
    anns = st.get("Pod", "pc")["metadata"]["annotations"]
    assert anns["a"] == str(n - 1) and anns["b"] == str(n - 1)
