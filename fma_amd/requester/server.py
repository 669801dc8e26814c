"""Requester stub: the in-Pod process of a server-requesting Pod.

Holds the GPU allocation and exposes (reference cmd/requester/main.go:32-85,
pkg/server/requester/{probes,coordination,proxy}):

- probes server (:8080, env PROBES_PORT): GET /ready -> 200/503 from a
  shared atomic readiness bit (probes/server.go:38-87)
- SPI server (:8081, env SPI_PORT; paths pkg/spi/interface.go:34-89):
    GET  /v1/dual-pods/accelerators            JSON array of GPU UUIDs
    GET  /v1/dual-pods/accelerator-memory-usage {uuid: bytes-in-use}
    POST /v1/become-ready | /v1/become-unready
    POST /v1/set-log?startPos=N                append-only log relay
    GET/PUT /v1/proxy/config                   TCP reverse proxy target

MI355X specifics: UUIDs come from amd-smi/rocm-smi (the reference shells
out to nvidia-smi — coordination/server.go:54-73, 95-140); tests inject
FMA_ACCELERATORS / FMA_ACCEL_MEM_JSON.
"""

from __future__ import annotations

import json
import os
import socket
import threading
from typing import Dict, List, Optional

from fastapi import FastAPI, Query, Request
from fastapi.responses import JSONResponse, PlainTextResponse

from fma_amd.api import contracts
from fma_amd.launcher.gputranslator import _rocm_smi_map


def discover_accelerators() -> List[str]:
    env = os.environ.get("FMA_ACCELERATORS")
    if env is not None:
        return [u for u in env.split(",") if u]
    m = _rocm_smi_map()
    if m:
        return sorted(m, key=lambda u: m[u])
    import torch
    if torch.cuda.is_available():
        return [f"GPU-{i}" for i in range(torch.cuda.device_count())]
    return []


def accelerator_memory_usage(uuids: List[str]) -> Dict[str, int]:
    env = os.environ.get("FMA_ACCEL_MEM_JSON")
    if env:
        data = json.loads(env)
        return {u: int(data.get(u, 0)) for u in uuids}
    usage: Dict[str, int] = {}
    try:
        import torch
        if torch.cuda.is_available():
            for i, u in enumerate(uuids):
                free_b, total_b = torch.cuda.mem_get_info(i)
                usage[u] = total_b - free_b
            return usage
    except Exception:
        pass
    return {u: 0 for u in uuids}


class LogSink:
    """Append-only log relay with position dedup (reference
    coordination/server.go:152-209): a chunk at startPos <= size appends
    only its new suffix; startPos > size is a 400."""

    def __init__(self) -> None:
        self.buf = bytearray()
        self.lock = threading.Lock()

    def append(self, start_pos: int, chunk: bytes) -> None:
        with self.lock:
            size = len(self.buf)
            if start_pos > size:
                raise ValueError(f"startPos {start_pos} beyond size {size}")
            new_from = size - start_pos
            if new_from < len(chunk):
                fresh = chunk[new_from:]
                self.buf.extend(fresh)
                # surface relayed server output on the requester's own
                # stdout so the requesting Pod's log shows its server
                import sys
                sys.stdout.write(fresh.decode("utf-8", "replace"))
                sys.stdout.flush()

    def contents(self) -> bytes:
        with self.lock:
            return bytes(self.buf)


class TcpProxy:
    """Configure-once TCP reverse proxy (reference proxy/server.go:39-217):
    PUT delivers the target exactly once (second PUT -> 409) and returns
    only when the listener is up."""

    def __init__(self, listen_port: int = 0):
        self.listen_port = listen_port
        self.target: Optional[Dict[str, object]] = None
        self._server_sock: Optional[socket.socket] = None
        self._lock = threading.Lock()

    def configure(self, address: str, port: int) -> int:
        with self._lock:
            if self.target is not None:
                raise FileExistsError("proxy already configured")
            sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
            sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            sock.bind(("0.0.0.0", self.listen_port))
            sock.listen(64)
            self._server_sock = sock
            self.listen_port = sock.getsockname()[1]
            self.target = {"address": address, "port": port}
            threading.Thread(target=self._accept_loop, daemon=True).start()
            return self.listen_port

    def _accept_loop(self) -> None:
        assert self._server_sock is not None and self.target is not None
        while True:
            try:
                client, _ = self._server_sock.accept()
            except OSError:
                return
            threading.Thread(target=self._pipe_pair, args=(client,),
                             daemon=True).start()

    def _pipe_pair(self, client: socket.socket) -> None:
        try:
            upstream = socket.create_connection(
                (str(self.target["address"]), int(self.target["port"])),
                timeout=30)
        except OSError:
            client.close()
            return

        def pipe(a: socket.socket, b: socket.socket) -> None:
            try:
                while True:
                    data = a.recv(65536)
                    if not data:
                        break
                    b.sendall(data)
            except OSError:
                pass
            finally:
                for s in (a, b):
                    try:
                        s.shutdown(socket.SHUT_RDWR)
                    except OSError:
                        pass

        threading.Thread(target=pipe, args=(client, upstream),
                         daemon=True).start()
        threading.Thread(target=pipe, args=(upstream, client),
                         daemon=True).start()

    def close(self) -> None:
        if self._server_sock is not None:
            self._server_sock.close()


class RequesterState:
    def __init__(self, proxy_listen_port: int = 0):
        self.ready = threading.Event()
        self.log = LogSink()
        self.proxy = TcpProxy(proxy_listen_port)


def create_probes_app(state: RequesterState) -> FastAPI:
    app = FastAPI(title="fma-amd requester probes")

    @app.get(contracts.READY_PATH)
    def ready():
        if state.ready.is_set():
            return PlainTextResponse("ready\n", status_code=200)
        return PlainTextResponse("unready\n", status_code=503)

    return app


def create_spi_app(state: RequesterState) -> FastAPI:
    app = FastAPI(title="fma-amd requester SPI")

    @app.get(contracts.ACCELERATOR_QUERY_PATH)
    def accelerators():
        return discover_accelerators()

    @app.get(contracts.ACCELERATOR_MEMORY_QUERY_PATH)
    def accelerator_memory():
        return accelerator_memory_usage(discover_accelerators())

    @app.post(contracts.BECOME_READY_PATH)
    def become_ready():
        state.ready.set()
        return {"status": "ok"}

    @app.post(contracts.BECOME_UNREADY_PATH)
    def become_unready():
        state.ready.clear()
        return {"status": "ok"}

    @app.post(contracts.SET_LOG_PATH)
    async def set_log(request: Request,
                      startPos: int = Query(default=0, alias="startPos")):
        chunk = await request.body()
        try:
            state.log.append(startPos, chunk)
        except ValueError as e:
            return JSONResponse({"error": str(e)}, status_code=400)
        return {"status": "ok", "size": len(state.log.contents())}

    @app.get(contracts.PROXY_CONFIG_PATH)
    def get_proxy_config():
        if state.proxy.target is None:
            return JSONResponse({"error": "not configured"}, status_code=404)
        return state.proxy.target

    @app.put(contracts.PROXY_CONFIG_PATH)
    def put_proxy_config(body: dict):
        try:
            port = state.proxy.configure(str(body["address"]),
                                         int(body["port"]))
        except FileExistsError:
            return JSONResponse({"error": "already configured"},
                                status_code=409)
        except (KeyError, ValueError):
            return JSONResponse({"error": "need address and port"},
                                status_code=400)
        return {"status": "ok", "listen_port": port}

    return app


def main() -> None:
    import uvicorn

    proxy_port = int(os.environ.get("PROXY_PORT",
                                    contracts.PROXY_PORT_DEFAULT))
    state = RequesterState(proxy_listen_port=proxy_port)
    probes_port = int(os.environ.get("PROBES_PORT",
                                     contracts.PROBES_PORT_DEFAULT))
    spi_port = int(os.environ.get("SPI_PORT", contracts.SPI_PORT_DEFAULT))
    probes = create_probes_app(state)
    spi = create_spi_app(state)

    host = os.environ.get("FMA_BIND_HOST", "0.0.0.0")
    t = threading.Thread(
        target=lambda: uvicorn.run(probes, host=host, port=probes_port,
                                   log_level="warning"),
        daemon=True)
    t.start()
    uvicorn.run(spi, host=host, port=spi_port, log_level="warning")


if __name__ == "__main__":
    main()
