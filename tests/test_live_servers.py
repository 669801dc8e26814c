"""Live-server tests: launcher watch streaming, HTTP store client."""

import json
import socket
import threading
import time

import httpx
import pytest
import uvicorn

from fma_amd.store import objects as ob
from fma_amd.store.client import StoreClient
from fma_amd.store.memstore import Conflict, MemStore
from fma_amd.store.server import create_app as store_app

pytestmark = pytest.mark.timeout(120)


def free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class ServerThread:
    def __init__(self, app, port):
        self.config = uvicorn.Config(app, host="127.0.0.1", port=port,
                                     log_level="warning")
        self.server = uvicorn.Server(self.config)
        self.thread = threading.Thread(target=self.server.run, daemon=True)

    def __enter__(self):
        self.thread.start()
        deadline = time.time() + 15
        while not self.server.started and time.time() < deadline:
            time.sleep(0.05)
        assert self.server.started
        return self

    def __exit__(self, *a):
        self.server.should_exit = True
        self.thread.join(timeout=5)
        close = getattr(self.config.app, "state", None)
        close = getattr(close, "close", None) if close else None
        if close:
            close()  # reap the app's watch pool (thread-leak fix)


def test_store_client_over_live_http():
    store = MemStore()
    port = free_port()
    with ServerThread(store_app(store), port):
        client = StoreClient(f"http://127.0.0.1:{port}", actor="user")
        pod = client.create(ob.new_object("Pod", "live1"))
        assert ob.uid_of(pod)
        assert client.get("Pod", "live1")["metadata"]["name"] == "live1"
        pod["spec"] = {"nodeName": "n"}
        client.update(pod)
        with pytest.raises(Conflict):
            client.update(pod)  # stale RV

        # watch over HTTP: collect events in a thread
        got = []
        stop = threading.Event()

        def consume():
            for ev in client.watch(since=0, kinds=["Pod"], stop=stop):
                got.append((ev.type, ob.name_of(ev.obj)))
                if len(got) >= 3:
                    stop.set()
                    return

        t = threading.Thread(target=consume, daemon=True)
        t.start()
        time.sleep(0.3)
        client.delete("Pod", "live1")
        t.join(timeout=10)
        stop.set()
        assert ("ADDED", "live1") in got
        assert ("DELETED", "live1") in got


def test_launcher_watch_streams_ndjson(monkeypatch, tmp_path):
    import fma_amd.launcher.instance as instance_mod
    from fma_amd.launcher.gputranslator import GpuTranslator
    from fma_amd.launcher.service import InstanceManager, create_app

    from tests.test_launcher_service import _stub_kickoff
    monkeypatch.setenv("FMA_MOCK_GPU_COUNT", "2")
    monkeypatch.setattr(instance_mod, "kickoff", _stub_kickoff)
    mgr = InstanceManager(GpuTranslator("naive"), str(tmp_path))
    port = free_port()
    with ServerThread(create_app(mgr), port):
        base = f"http://127.0.0.1:{port}"
        events = []
        ready = threading.Event()

        def consume():
            with httpx.stream("GET", f"{base}/v2/vllm/instances/watch",
                              timeout=30) as r:
                ready.set()
                for line in r.iter_lines():
                    if line:
                        events.append(json.loads(line))
                    if len(events) >= 2:
                        return

        t = threading.Thread(target=consume, daemon=True)
        t.start()
        assert ready.wait(10)
        time.sleep(0.2)
        r = httpx.put(f"{base}/v2/vllm/instances/w1",
                      json={"options": "--model tiny --port 9400"},
                      timeout=10)
        assert r.status_code == 201
        httpx.delete(f"{base}/v2/vllm/instances/w1", timeout=10)
        t.join(timeout=15)
        types = [e["type"] for e in events]
        assert types[:2] == ["CREATED", "DELETED"]
        assert events[0]["instance_id"] == "w1"
        assert events[0]["revision"] < events[1]["revision"]


def test_dump_launcher_logs_tool(monkeypatch, tmp_path):
    """tools/dump_launcher_logs.py (the dump-launcher-vllm-logs.sh
    analog) lists instances and fetches ranged logs over the live API."""
    import io
    import sys

    import fma_amd.launcher.instance as instance_mod
    from fma_amd.launcher.gputranslator import GpuTranslator
    from fma_amd.launcher.service import InstanceManager, create_app

    from tests.test_launcher_service import _stub_kickoff
    monkeypatch.setenv("FMA_MOCK_GPU_COUNT", "2")
    monkeypatch.setattr(instance_mod, "kickoff", _stub_kickoff)
    sys.path.insert(0, "tools")
    from dump_launcher_logs import dump

    mgr = InstanceManager(GpuTranslator("naive"), str(tmp_path))
    port = free_port()
    with ServerThread(create_app(mgr), port):
        base = f"http://127.0.0.1:{port}"
        r = httpx.put(f"{base}/v2/vllm/instances/d1",
                      json={"options": "--model tiny --port 9401"},
                      timeout=10)
        assert r.status_code == 201
        time.sleep(0.3)  # let the child write some log bytes
        out = io.StringIO()
        n = dump(base, tail=0, out=out)
        assert n == 1
        text = out.getvalue()
        assert "instance d1" in text
        # tail form exercises the Range path (206 or 200 for short logs)
        out2 = io.StringIO()
        dump(base, tail=64, out=out2)
        assert "instance d1" in out2.getvalue()
        httpx.delete(f"{base}/v2/vllm/instances/d1", timeout=10)
