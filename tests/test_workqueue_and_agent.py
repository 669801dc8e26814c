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
