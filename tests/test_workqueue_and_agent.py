"""Workqueue semantics + node-agent lifecycle units."""

import sys
import threading
import time

import pytest

from fma_amd.controller.workqueue import (InitialSyncTracker,
                                          QueueAndWorkers,
                                          RateLimitingQueue)
from fma_amd.node.agent import NodeAgent
from fma_amd.store import objects as ob
from fma_amd.store.memstore import MemStore

pytestmark = pytest.mark.timeout(90)


def test_queue_no_concurrent_same_item():
    q = RateLimitingQueue()
    q.add("a")
    item = q.get()
    assert item == "a"
    q.add("a")  # re-added while processing: goes to dirty, not queue
    assert len(q) == 0
    q.done("a")
    assert len(q) == 1  # re-queued after done


def test_queue_dedup():
    q = RateLimitingQueue()
    q.add("x")
    q.add("x")
    assert len(q) == 1


def test_rate_limit_backoff_grows_and_forgets():
    q = RateLimitingQueue(max_backoff=0.05)
    for _ in range(10):
        q.add_rate_limited("item")
        got = q.get()
        q.done(got)
    assert q._failures["item"] == 10
    q.forget("item")
    assert "item" not in q._failures


def test_workers_retry_after_float():
    """A float return re-queues after that delay without failure count."""
    seen = []
    done = threading.Event()

    def process(item):
        seen.append(time.perf_counter())
        if len(seen) == 1:
            return 0.2  # scheduled wait
        done.set()
        return False

    w = QueueAndWorkers("t", 1, process)
    w.start()
    w.queue.add("i")
    assert done.wait(5)
    w.stop()
    assert len(seen) == 2
    assert seen[1] - seen[0] >= 0.18
    assert w.queue._failures.get("i") is None


def test_initial_sync_tracker():
    fired = []
    t = InitialSyncTracker(lambda: fired.append(1))
    t.register("a")
    t.register("b")
    t.start()
    assert not fired
    t.mark_processed("a")
    t.mark_processed("b")
    assert fired == [1]
    t.mark_processed("c")  # no double fire
    assert fired == [1]


# -- node agent -----------------------------------------------------------

def sleeper_pod(name, marker):
    return ob.new_object(
        "Pod", name,
        spec={"nodeName": "n1", "containers": [{
            "name": "main",
            "command": [sys.executable, "-c",
                        f"import time; print('{marker}', flush=True); "
                        "time.sleep(60)"],
        }]})


@pytest.fixture()
def agent(tmp_path):
    store = MemStore()
    a = NodeAgent(store, "n1", node_index=11, log_dir=str(tmp_path))
    a.start()
    yield store, a
    a.stop()


def wait(cond, timeout=20):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if cond():
            return True
        time.sleep(0.05)
    return False


def test_agent_runs_and_reports_status(agent):
    store, a = agent
    store.create(sleeper_pod("p1", "alive-p1"))
    assert wait(lambda: "p1" in a.pods)
    assert wait(lambda: ob.pod_is_ready(store.get("Pod", "p1")))
    pod = store.get("Pod", "p1")
    assert pod["status"]["phase"] == "Running"
    assert pod["status"]["podIP"].startswith("127.11.")


def test_agent_kills_on_delete(agent):
    store, a = agent
    store.create(sleeper_pod("p2", "alive-p2"))
    assert wait(lambda: "p2" in a.pods)
    proc = a.pods["p2"].proc
    store.delete("Pod", "p2")
    assert wait(lambda: proc.poll() is not None)
    assert "p2" not in a.pods


def test_agent_restarts_on_uid_change(agent):
    store, a = agent
    store.create(sleeper_pod("p3", "gen1"))
    assert wait(lambda: "p3" in a.pods)
    old_proc = a.pods["p3"].proc
    store.delete("Pod", "p3")
    assert wait(lambda: "p3" not in a.pods)
    store.create(sleeper_pod("p3", "gen2"))
    assert wait(lambda: "p3" in a.pods and a.pods["p3"].proc is not old_proc)


def test_agent_reports_failed_phase(agent):
    store, a = agent
    pod = ob.new_object(
        "Pod", "crash",
        spec={"nodeName": "n1", "restartPolicy": "Never",
              "containers": [{
                  "name": "main",
                  "command": [sys.executable, "-c", "import sys; sys.exit(3)"],
              }]})
    store.create(pod)
    assert wait(lambda: store.get("Pod", "crash")["status"].get("phase")
                == "Failed")


def test_reap_pod_tree_kills_marked_orphans(tmp_path):
    """_reap_pod_tree must kill processes carrying the pod's
    FMA_POD_TREE marker even when they detached into their own process
    group and their parent is gone (the crashed-launcher orphan case:
    an orphan holding the server port blocks every later instance)."""
    import os
    import subprocess
    import time as _time

    from fma_amd.node.agent import NodeAgent
    from fma_amd.store.memstore import MemStore

    agent = NodeAgent(MemStore(), "nX", node_index=9,
                      log_dir=str(tmp_path))
    marker_env = dict(os.environ,
                      FMA_POD_TREE="nX/podX/uid-123")
    orphan = subprocess.Popen(
        [os.sys.executable, "-c", "import time; time.sleep(300)"],
        env=marker_env, start_new_session=True)
    bystander = subprocess.Popen(
        [os.sys.executable, "-c", "import time; time.sleep(300)"],
        start_new_session=True)
    try:
        n = agent._reap_pod_tree("podX", "uid-123")
        assert n == 1
        deadline = _time.time() + 10
        while _time.time() < deadline and orphan.poll() is None:
            _time.sleep(0.1)
        assert orphan.poll() is not None, "marked orphan survived"
        assert bystander.poll() is None, "unmarked process was killed!"
    finally:
        for p in (orphan, bystander):
            if p.poll() is None:
                p.kill()


def test_crashloop_restarts_with_backoff(tmp_path):
    """A crash-looping pod (restartPolicy Always) is restarted with
    growing backoff and its restartCount surfaces in containerStatuses,
    kubelet-style (the reference's PodIsInTrouble,
    utils/pod-helper.go:44, keys off restarts + unready)."""
    import sys
    import time as _time

    from fma_amd.api import contracts as C  # noqa: F401
    from fma_amd.node.agent import NodeAgent
    from fma_amd.store import objects as ob
    from fma_amd.store.memstore import MemStore

    store = MemStore()
    agent = NodeAgent(store, "n1", node_index=11, log_dir=str(tmp_path))
    agent.start()
    try:
        pod = ob.new_object(
            "Pod", "looper",
            spec={"nodeName": "n1", "containers": [{
                "name": "main",
                "command": [sys.executable, "-c", "import sys; sys.exit(5)"],
            }]})
        store.create(pod)
        deadline = _time.time() + 30
        while _time.time() < deadline:
            pp = agent.pods.get("looper")
            if pp is not None and pp.restarts >= 2:
                break
            _time.sleep(0.1)
        assert pp is not None and pp.restarts >= 2, "no restarts happened"
        cur = store.get("Pod", "looper")
        assert cur["status"]["phase"] == "Running"  # not Failed: it loops
        assert cur["status"]["containerStatuses"][0]["restartCount"] >= 1
    finally:
        agent.stop()


# ---------------------------------------------------------------------------
# Two-level per-node queue (reference controller.go:404-424, 1044-1097)
# ---------------------------------------------------------------------------


def test_two_level_queue_oldest_first_drain():
    from fma_amd.controller.workqueue import TwoLevelQueue

    q = TwoLevelQueue(node_of=lambda it: it[0])
    q.add(("n1", "c"))
    q.add(("n1", "a"))
    q.add(("n1", "b"))
    q.add(("n1", "c"))  # re-add keeps original add time (stays first)
    assert q.take_ready("n1") == [("n1", "c"), ("n1", "a"), ("n1", "b")]
    assert q.take_ready("n1") == []


def test_two_level_queue_process_after():
    import time as _t

    from fma_amd.controller.workqueue import TwoLevelQueue

    q = TwoLevelQueue(node_of=lambda it: it[0])
    q.add(("n1", "later"), delay=0.25)
    q.add(("n1", "now"))
    assert q.take_ready("n1") == [("n1", "now")]
    pending = q.earliest_pending("n1")
    assert pending is not None and 0 < pending <= 0.25
    _t.sleep(0.3)
    assert q.take_ready("n1") == [("n1", "later")]
    assert q.earliest_pending("n1") is None


def test_hot_node_cannot_starve_other_nodes():
    """One node whose item always fails (exponential per-node backoff)
    must not delay another node's items, even with a single worker."""
    import threading
    import time as _t

    from fma_amd.controller.workqueue import NodeQueueAndWorkers

    done = threading.Event()
    calls = []

    def process(item):
        calls.append(item)
        if item[0] == "hot":
            return True  # always retry
        done.set()
        return False

    w = NodeQueueAndWorkers("t", 1, process, node_of=lambda it: it[0])
    w.start()
    try:
        w.queue.add(("hot", "spinner"))
        _t.sleep(0.05)  # let the hot node fail a few times first
        w.queue.add(("cold", "one-shot"))
        assert done.wait(timeout=2.0), \
            f"cold item starved; calls={calls[:10]}"
    finally:
        w.stop()


def test_node_serialized_but_nodes_parallel():
    """Items of one node never run concurrently; items of different nodes
    do (two workers, a barrier that only two nodes together can pass)."""
    import threading

    from fma_amd.controller.workqueue import NodeQueueAndWorkers

    barrier = threading.Barrier(2, timeout=5)
    in_flight = {}
    overlap = []
    mu = threading.Lock()

    def process(item):
        node = item[0]
        with mu:
            if in_flight.get(node):
                overlap.append(item)
            in_flight[node] = True
        try:
            barrier.wait()  # needs BOTH nodes in flight at once
        except threading.BrokenBarrierError:
            pass
        with mu:
            in_flight[node] = False
        return False

    w = NodeQueueAndWorkers("t", 2, process, node_of=lambda it: it[0])
    w.start()
    try:
        w.queue.add(("na", 1))
        w.queue.add(("nb", 1))
        import time as _t
        deadline = _t.time() + 3
        while _t.time() < deadline and len(w.queue.outer) + len(w.queue):
            _t.sleep(0.02)
        assert not overlap
    finally:
        w.stop()


def test_store_index_lookup_matches_scan():
    from fma_amd.api import contracts as C
    from fma_amd.store import objects as ob
    from fma_amd.store.indexes import install_pod_indexes
    from fma_amd.store.memstore import MemStore

    st = MemStore()
    install_pod_indexes(st)
    p1 = st.create(ob.new_object(
        "Pod", "prov1",
        annotations={C.REQUESTER_ANNOTATION: "u1 req1",
                     C.ACCELERATORS_ANNOTATION: "GPU-0,GPU-1"},
        spec={"nodeName": "node-a", "containers": []}))
    st.create(ob.new_object(
        "Pod", "req1", annotations={
            C.INFERENCE_SERVER_CONFIG_ANNOTATION: "isc1"},
        spec={"nodeName": "node-a", "containers": []}))

    assert [ob.name_of(p) for p in
            st.index_get("Pod", "requester", "u1 req1")] == ["prov1"]
    assert [ob.name_of(p) for p in
            st.index_get("Pod", "gpu", "GPU-1")] == ["prov1"]
    assert [ob.name_of(p) for p in
            st.index_get("Pod", "inferenceserverconfig", "isc1")] == ["req1"]

    # update re-indexes: annotation change moves the pod between keys
    ob.annotations_of(p1)[C.REQUESTER_ANNOTATION] = "u2 req2"
    p1 = st.update(p1)
    assert st.index_get("Pod", "requester", "u1 req1") == []
    assert [ob.name_of(p) for p in
            st.index_get("Pod", "requester", "u2 req2")] == ["prov1"]

    # delete drops index entries
    st.delete("Pod", "prov1")
    assert st.index_get("Pod", "requester", "u2 req2") == []
    assert st.index_get("Pod", "gpu", "GPU-0") == []


def test_index_registered_late_backfills():
    from fma_amd.api import contracts as C
    from fma_amd.store import objects as ob
    from fma_amd.store.memstore import MemStore

    st = MemStore()
    st.create(ob.new_object(
        "Pod", "early", annotations={C.NOMINAL_ANNOTATION: "h1"},
        spec={"containers": []}))
    st.add_index("Pod", "nominal",
                 lambda p: [ob.annotations_of(p).get(C.NOMINAL_ANNOTATION)]
                 if ob.annotations_of(p).get(C.NOMINAL_ANNOTATION) else [])
    assert [ob.name_of(p) for p in
            st.index_get("Pod", "nominal", "h1")] == ["early"]


def test_two_level_queue_no_item_lost_fuzz():
    """Property: under random add/add_after/add_rate_limited from several
    threads, every item is eventually processed at least once."""
    import random
    import threading
    import time as _t

    from fma_amd.controller.workqueue import NodeQueueAndWorkers

    processed = set()
    mu = threading.Lock()

    def process(item):
        with mu:
            processed.add(item)
        return False

    w = NodeQueueAndWorkers("fuzz", 3, process, node_of=lambda it: it[0])
    w.start()
    all_items = []
    try:
        rng = random.Random(11)

        def feeder(tid):
            r = random.Random(tid)
            for i in range(120):
                item = (f"n{r.randint(0, 5)}", tid, i)
                with mu:
                    all_items.append(item)
                mode = r.random()
                if mode < 0.6:
                    w.queue.add(item)
                elif mode < 0.85:
                    w.queue.add_after(item, r.random() * 0.05)
                else:
                    w.queue.add_rate_limited(item)

        threads = [threading.Thread(target=feeder, args=(t,))
                   for t in range(4)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        deadline = _t.time() + 20
        while _t.time() < deadline:
            with mu:
                if set(all_items) <= processed:
                    break
            _t.sleep(0.05)
        with mu:
            missing = set(all_items) - processed
        assert not missing, f"{len(missing)} items never processed"
    finally:
        w.stop()


def test_store_indexes_consistent_under_random_crud_fuzz():
    """Property: after any random CRUD sequence (with finalizers and
    status writes mixed in), every registered index returns exactly what
    a brute-force scan returns."""
    import random

    from fma_amd.api import contracts as C
    from fma_amd.store.indexes import POD_INDEXES, install_pod_indexes
    from fma_amd.store.memstore import ApiError, MemStore

    rng = random.Random(77)
    st = MemStore()
    install_pod_indexes(st)
    names = [f"p{i}" for i in range(12)]
    anns_keys = [C.REQUESTER_ANNOTATION, C.NOMINAL_ANNOTATION,
                 C.INFERENCE_SERVER_CONFIG_ANNOTATION,
                 C.ACCELERATORS_ANNOTATION]
    for step in range(400):
        name = rng.choice(names)
        op = rng.random()
        try:
            if op < 0.4:
                pod = ob.new_object(
                    "Pod", name,
                    labels={C.COMPONENT_LABEL: C.LAUNCHER_COMPONENT}
                    if rng.random() < 0.4 else {},
                    annotations={k: rng.choice(["", "a b", "h1", "GPU-0,GPU-1"])
                                 for k in rng.sample(anns_keys,
                                                     rng.randint(0, 3))},
                    spec={"nodeName": rng.choice(["n1", "n2", ""]),
                          "containers": []})
                if rng.random() < 0.2:
                    pod["metadata"]["finalizers"] = ["t/f"]
                st.create(pod)
            elif op < 0.7:
                cur = st.try_get("Pod", name)
                if cur is None:
                    continue
                k = rng.choice(anns_keys)
                ob.annotations_of(cur)[k] = rng.choice(
                    ["", "u2 q", "h2", "GPU-2"])
                if rng.random() < 0.3:
                    cur["metadata"]["finalizers"] = []
                st.update(cur)
            elif op < 0.9:
                st.delete("Pod", name)
            else:
                cur = st.try_get("Pod", name)
                if cur is not None:
                    cur["status"] = {"phase": "Running"}
                    st.update(cur, subresource="status")
        except ApiError:
            pass

        if step % 50 == 49:
            pods = st.list("Pod")
            for iname, fn in POD_INDEXES.items():
                expect = {}
                for p in pods:
                    for key in fn(p) or []:
                        expect.setdefault(key, set()).add(ob.name_of(p))
                keys = set(expect)
                for key in keys:
                    got = {ob.name_of(p)
                           for p in st.index_get("Pod", iname, key)}
                    assert got == expect[key], (iname, key, step)
                # spot-check an absent key returns empty
                assert st.index_get("Pod", iname, "no-such-key") == []


def test_pod_pdeathsig_reaps_on_hard_agent_kill(tmp_path):
    """FMA_POD_PDEATHSIG=1 (set by the test harness): pods die when the
    agent PROCESS is SIGKILLed, so a hard-killed pytest run cannot leave
    orphaned servers on fixed ports poisoning later runs (observed when
    two oversubscribed suites aborted mid-e2e)."""
    import os
    import signal
    import subprocess
    import sys
    import time

    script = r"""
import os, sys, time
sys.path.insert(0, os.getcwd())
os.environ["FMA_POD_PDEATHSIG"] = "1"
from fma_amd.node.agent import NodeAgent
from fma_amd.store.memstore import MemStore
from fma_amd.store import objects as ob
st = MemStore()
agent = NodeAgent(st, "node-t", log_dir=sys.argv[1])
st.create(ob.new_object("Pod", "sleepy", spec={
    "nodeName": "node-t",
    "containers": [{"name": "main",
                    "command": [sys.executable, "-c",
                                "import time; time.sleep(120)"]}]}))
agent._sync_all()
pp = agent.pods["sleepy"]
print(f"CHILD={pp.proc.pid}", flush=True)
time.sleep(120)
"""
    proc = subprocess.Popen([sys.executable, "-c", script, str(tmp_path)],
                            stdout=subprocess.PIPE, text=True)
    try:
        line = proc.stdout.readline()
        assert line.startswith("CHILD="), line
        child_pid = int(line.split("=")[1])
        # child alive while the agent process lives
        os.kill(child_pid, 0)
        proc.kill()  # SIGKILL: no teardown runs in the agent process
        proc.wait(timeout=10)
        deadline = time.time() + 5
        gone = False
        while time.time() < deadline:
            try:
                os.kill(child_pid, 0)
            except ProcessLookupError:
                gone = True
                break
            time.sleep(0.1)
        assert gone, f"pod process {child_pid} survived agent SIGKILL"
    finally:
        if proc.poll() is None:
            proc.kill()
