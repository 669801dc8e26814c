"""Metrics registry contents + accelerator-memory sleep budget."""

import pytest

from fma_amd.api import contracts as C
from fma_amd.controller import metrics
from fma_amd.controller.dualpods.controller import (ControllerConfig,
                                                    DualPodsController)
from fma_amd.store import objects as ob

from tests.test_dualpods_controller import (FakeInstanceServer, drive,
                                            infsvr_item, mk_world)


def test_metric_names_match_reference():
    # touching the factories registers them; names per reference
    # controller.go:204-293 / metrics.go:79-95
    metrics.actuation_seconds()
    metrics.launcher_create_seconds()
    metrics.http_latency_seconds()
    metrics.requester_count()
    metrics.isc_count()
    metrics.duality()
    metrics.launcher_pod_count()
    if not metrics.HAVE_PROM:
        pytest.skip("prometheus_client unavailable")
    from prometheus_client import REGISTRY
    names = {m.name for m in REGISTRY.collect()}
    for want in ("fma_actuation_seconds", "fma_launcher_create_seconds",
                 "fma_http_latency_seconds", "fma_requester_count",
                 "fma_isc_count", "fma_duality", "fma_launcher_pod_count"):
        assert want in names, f"{want} not registered"


def test_actuation_histogram_observes_on_relay():
    if not metrics.HAVE_PROM:
        pytest.skip("prometheus_client unavailable")
    from prometheus_client import REGISTRY
    before = REGISTRY.get_sample_value(
        "fma_actuation_seconds_count",
        {"path": "hot", "instancesDeleted": "0", "isc_name": "isc1"}) or 0
    w = mk_world()
    drive(w["ctl"], infsvr_item(w["store"]))
    after = REGISTRY.get_sample_value(
        "fma_actuation_seconds_count",
        {"path": "hot", "instancesDeleted": "0", "isc_name": "isc1"}) or 0
    assert after == before + 1


def test_accel_memory_budget_defers_wake():
    """Wake waits while accelerator memory is above the sleeping budget
    (reference accelMemoryIsLowEnough, inference-server.go:1991-2014)."""
    w = mk_world()
    w["ctl"].cfg.accelerator_sleeping_memory_limit_mib = 1  # 1 MiB budget

    # stub reports 2 GiB in use on the GPU -> wake deferred
    class BusyStub:
        def __call__(self, method, path, json, params):
            if path == "/v1/dual-pods/accelerators":
                return 200, ["GPU-0"]
            if path == "/v1/dual-pods/accelerator-memory-usage":
                return 200, {"GPU-0": 2 << 30}
            if path == "/v1/become-ready":
                return 200, {}
            return 404, {}

    busy = BusyStub()
    w["http"].register("10.0.0.1:8081", busy)
    item = infsvr_item(w["store"])
    for _ in range(5):
        w["ctl"]._process(item)
    assert w["inst_srv"].wakes == 0, "wake should wait for memory budget"

    # memory drains -> wake proceeds
    class FreeStub(BusyStub):
        def __call__(self, method, path, json, params):
            if path == "/v1/dual-pods/accelerator-memory-usage":
                return 200, {"GPU-0": 0}
            return super().__call__(method, path, json, params)

    w["http"].register("10.0.0.1:8081", FreeStub())
    drive(w["ctl"], item)
    assert w["inst_srv"].wakes == 1


def test_launcher_stuck_event_recorded():
    from tests.test_populator import (FakeClock, drain_key, launcher_pods,
                                      mk_lc, mk_lpp, mk_node, mk_pop)
    from fma_amd.store.memstore import MemStore as MS
    store = MS()
    clock = FakeClock(2000.0)
    mk_node(store, "n1", labels={"gpu": "x"})
    mk_lc(store)
    mk_lpp(store, "p1", count=1, match_labels={"gpu": "x"})
    pop = mk_pop(store, clock)
    drain_key(pop, ("n1", "lc1"))
    pod = launcher_pods(store, "n1")[0]
    created = ob.meta(store.get("Pod", ob.name_of(pod)))["creationTimestamp"]
    clock.t = created + 500
    drain_key(pop, ("n1", "lc1"))
    events = store.list("Event")
    assert any(e["reason"] == "LauncherStuck" for e in events)


def test_outdated_routing_event_recorded():
    w = mk_world()
    drive(w["ctl"], infsvr_item(w["store"]))
    isc = w["store"].get("InferenceServerConfig", "isc1")
    isc["spec"]["modelServerConfig"]["labels"] = {"llm-d.ai/model": "v2"}
    w["store"].update(isc)
    # re-reconcile the (still bound) server against the changed ISC
    w["ctl"]._process(infsvr_item(w["store"]))
    events = w["store"].list("Event")
    assert any(e["reason"] == "OutdatedRoutingMetadata" for e in events)


def test_debug_endpoint_serves_thread_stacks():
    """The reference's :8003 debug listener analog
    (pkg/observability/prom-and-debug.go:68-79): /debug/threads dumps
    live stacks, /debug/vars serves counters, unknown paths 404."""
    import json
    import urllib.request

    from fma_amd.controller import metrics as M

    srv = M.serve_debug(0)
    port = srv.server_address[1]
    try:
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/debug/threads", timeout=5).read()
        assert b"--- thread" in body
        assert b"test_debug_endpoint" in body  # this very frame
        v = json.loads(urllib.request.urlopen(
            f"http://127.0.0.1:{port}/debug/vars", timeout=5).read())
        assert v["threads"] >= 1
        try:
            urllib.request.urlopen(
                f"http://127.0.0.1:{port}/nope", timeout=5)
            assert False, "expected 404"
        except urllib.error.HTTPError as e:
            assert e.code == 404
    finally:
        srv.shutdown()


def test_innerqueue_metrics_emitted():
    """The reference exports k8s workqueue metrics for its inner queue
    (docs/metrics.md fma_dpc_innerqueue_*); our RateLimitingQueue emits
    the same family when named."""
    import time

    from fma_amd.controller import metrics as M
    from fma_amd.controller.workqueue import QueueAndWorkers

    if not M.HAVE_PROM:
        import pytest
        pytest.skip("prometheus_client unavailable")
    processed = []

    def proc(item):
        processed.append(item)
        return item == "retry-once" and len(processed) < 3

    q = QueueAndWorkers("t", 1, proc, metrics_name="dpc-test")
    q.start()
    q.queue.add("a")
    q.queue.add("retry-once")
    deadline = time.time() + 10
    while time.time() < deadline and len(processed) < 3:
        time.sleep(0.05)
    q.stop()
    from prometheus_client import REGISTRY
    adds = REGISTRY.get_sample_value(
        "fma_dpc_innerqueue_adds_total", {"name": "dpc-test"})
    retries = REGISTRY.get_sample_value(
        "fma_dpc_innerqueue_retries_total", {"name": "dpc-test"})
    work = REGISTRY.get_sample_value(
        "fma_dpc_innerqueue_work_duration_seconds_count",
        {"name": "dpc-test"})
    assert adds and adds >= 2
    assert retries and retries >= 1
    assert work and work >= 2


def _mk_sleeper(store, name, gpus, ts):
    pod = ob.new_object(
        "Pod", name,
        labels={C.SLEEPING_LABEL: "true"},
        annotations={C.NOMINAL_ANNOTATION: "h-" + name,
                     C.ACCELERATORS_ANNOTATION: ",".join(gpus)},
        spec={"nodeName": "node-a",
              "containers": [{"name": "inference-server"}]})
    pod = store.create(pod)
    ob.meta(pod)["creationTimestamp"] = ts
    return store.update(pod)


def test_sleeper_budget_is_per_gpu():
    """The budget counts sleepers per GPU index, not per node (reference
    enforceSleeperBudget inference-server.go:1354-1428): a new provider on
    GPU-0 must not evict the only sleeper on GPU-1."""
    from fma_amd.controller.dualpods.controller import ServerData
    from fma_amd.store.memstore import MemStore

    store = MemStore()
    ctl = DualPodsController(store, None, ControllerConfig(sleeper_limit=1))
    _mk_sleeper(store, "s-g0-old", ["GPU-0"], "2026-01-01T00:00:00Z")
    _mk_sleeper(store, "s-g0-new", ["GPU-0"], "2026-01-02T00:00:00Z")
    _mk_sleeper(store, "s-g1", ["GPU-1"], "2026-01-01T00:00:00Z")

    sdata = ServerData(uid="u1", requester_name="r1", gpus=["GPU-0"])
    ctl._enforce_sleeper_budget("node-a", sdata)

    names = {ob.name_of(p) for p in store.list("Pod")}
    # GPU-0 had 2 sleepers with limit 1: the oldest goes; GPU-1 untouched
    assert names == {"s-g0-new", "s-g1"}


def test_sleeper_budget_within_limit_no_eviction():
    from fma_amd.controller.dualpods.controller import ServerData
    from fma_amd.store.memstore import MemStore

    store = MemStore()
    ctl = DualPodsController(store, None, ControllerConfig(sleeper_limit=2))
    _mk_sleeper(store, "s1", ["GPU-0"], "2026-01-01T00:00:00Z")
    _mk_sleeper(store, "s2", ["GPU-0"], "2026-01-02T00:00:00Z")
    sdata = ServerData(uid="u1", requester_name="r1", gpus=["GPU-0"])
    ctl._enforce_sleeper_budget("node-a", sdata)
    assert len(store.list("Pod")) == 2


def test_debug_profile_endpoint_samples_stacks():
    """/debug/profile is the pprof-profile analog: folded flamegraph
    lines sampled across live threads."""
    import threading
    import time

    import httpx

    from fma_amd.controller import metrics as M

    stop = threading.Event()

    def busy_loop_marker():
        while not stop.is_set():
            sum(range(200))
            time.sleep(0.001)

    th = threading.Thread(target=busy_loop_marker, name="busy-marker",
                          daemon=True)
    th.start()
    srv = M.serve_debug(0)
    try:
        port = srv.server_address[1]
        r = httpx.get(
            f"http://127.0.0.1:{port}/debug/profile?seconds=0.4&hz=200",
            timeout=10)
        assert r.status_code == 200
        assert "busy_loop_marker" in r.text
        # folded format: "name;frame;frame N"
        line = next(l for l in r.text.splitlines()
                    if "busy_loop_marker" in l)
        assert line.rsplit(" ", 1)[1].isdigit()
    finally:
        stop.set()
        srv.shutdown()
