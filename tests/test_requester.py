"""Requester stub tests: probes, SPI, log relay, proxy (CPU)."""

import socket
import threading

from fastapi.testclient import TestClient

from fma_amd.requester.server import (RequesterState, create_probes_app,
                                      create_spi_app)


def make_clients(monkeypatch=None):
    state = RequesterState()
    return (state, TestClient(create_probes_app(state)),
            TestClient(create_spi_app(state)))


def test_ready_flow(monkeypatch):
    state, probes, spi = make_clients()
    assert probes.get("/ready").status_code == 503
    assert spi.post("/v1/become-ready").status_code == 200
    assert probes.get("/ready").status_code == 200
    assert spi.post("/v1/become-unready").status_code == 200
    assert probes.get("/ready").status_code == 503


def test_accelerators_env(monkeypatch):
    monkeypatch.setenv("FMA_ACCELERATORS", "GPU-aa,GPU-bb")
    _, _, spi = make_clients()
    assert spi.get("/v1/dual-pods/accelerators").json() == ["GPU-aa", "GPU-bb"]


def test_accelerator_memory_env(monkeypatch):
    monkeypatch.setenv("FMA_ACCELERATORS", "GPU-aa,GPU-bb")
    monkeypatch.setenv("FMA_ACCEL_MEM_JSON", '{"GPU-aa": 1024}')
    _, _, spi = make_clients()
    body = spi.get("/v1/dual-pods/accelerator-memory-usage").json()
    assert body == {"GPU-aa": 1024, "GPU-bb": 0}


def test_set_log_dedup():
    state, _, spi = make_clients()
    r = spi.post("/v1/set-log", params={"startPos": 0}, content=b"hello")
    assert r.status_code == 200
    # resend overlapping chunk: only new suffix appended
    r = spi.post("/v1/set-log", params={"startPos": 0}, content=b"hello world")
    assert r.status_code == 200
    assert state.log.contents() == b"hello world"
    # beyond-end start position rejected
    r = spi.post("/v1/set-log", params={"startPos": 99}, content=b"x")
    assert r.status_code == 400


def test_proxy_configure_once_and_forward():
    state, _, spi = make_clients()
    # backend echo server
    backend = socket.socket()
    backend.bind(("127.0.0.1", 0))
    backend.listen(1)
    bport = backend.getsockname()[1]

    def echo_once():
        conn, _ = backend.accept()
        data = conn.recv(1024)
        conn.sendall(b"echo:" + data)
        conn.close()

    t = threading.Thread(target=echo_once, daemon=True)
    t.start()

    assert spi.get("/v1/proxy/config").status_code == 404
    r = spi.put("/v1/proxy/config",
                json={"address": "127.0.0.1", "port": bport})
    assert r.status_code == 200
    lport = r.json()["listen_port"]
    assert spi.get("/v1/proxy/config").json()["port"] == bport
    # second PUT -> 409
    assert spi.put("/v1/proxy/config",
                   json={"address": "127.0.0.1", "port": bport}).status_code == 409

    with socket.create_connection(("127.0.0.1", lport), timeout=5) as c:
        c.sendall(b"ping")
        got = c.recv(1024)
    assert got == b"echo:ping"
    state.proxy.close()


def test_proxy_bad_body():
    _, _, spi = make_clients()
    assert spi.put("/v1/proxy/config", json={"address": "x"}).status_code == 400


def test_main_pins_proxy_port(monkeypatch):
    """main() must pin the proxy listener to PROXY_PORT (default 8082,
    reference cmd/requester/main.go --proxy-port): the controller ignores
    the listen_port returned by PUT /v1/proxy/config, so an ephemeral port
    would be unreachable in a real deployment (ADVICE round-1)."""
    import fma_amd.requester.server as rs
    from fma_amd.api import contracts

    captured = {}

    def fake_run(app, **kw):
        captured.setdefault("apps", []).append(app)

    class FakeUvicorn:
        run = staticmethod(fake_run)

    state_holder = {}
    orig_state = rs.RequesterState

    def capture_state(proxy_listen_port=0):
        st = orig_state(proxy_listen_port)
        state_holder["state"] = st
        return st

    monkeypatch.setattr(rs, "RequesterState", capture_state)
    monkeypatch.setitem(__import__("sys").modules, "uvicorn", FakeUvicorn)
    monkeypatch.setenv("PROXY_PORT", "18982")
    rs.main()
    assert state_holder["state"].proxy.listen_port == 18982
    monkeypatch.delenv("PROXY_PORT")
    rs.main()
    assert state_holder["state"].proxy.listen_port == \
        contracts.PROXY_PORT_DEFAULT
